#!/usr/bin/env bash
# Lint gate (reference .dev/pre-commit.sh parity): syntax-compile every Python
# file and run flake8 when available.
set -e
cd "$(dirname "$0")/.."
python -m compileall -q distribuuuu_amd tests tutorial tools *.py
if python -c "import flake8" 2>/dev/null; then
    python -m flake8 --max-line-length 100 distribuuuu_amd
fi
echo "lint OK"
