"""Rank-aware logger.

Mirrors the reference's loguru usage (`/root/reference/distribuuuu/utils.py:71-82`):
rank 0 gets a timestamped file sink under OUT_DIR plus stderr; every rank gets stderr.
Implemented on the stdlib (loguru is not a dependency) with a loguru-like ``logger.info``
surface so call sites read the same.
"""

import logging
import os
import sys
import time

_LOGGER = logging.getLogger("distribuuuu_amd")
_LOGGER.setLevel(logging.INFO)
_LOGGER.propagate = False
_CONFIGURED = False

_FMT = "%(asctime)s | %(levelname)s | %(message)s"
_DATEFMT = "%Y-%m-%d %H:%M:%S"


def setup_logger(rank=0, out_dir=None):
    """Configure sinks. Call once per process after distributed init."""
    global _CONFIGURED
    for h in list(_LOGGER.handlers):
        _LOGGER.removeHandler(h)
    fmt = logging.Formatter(f"[rank {rank}] {_FMT}", datefmt=_DATEFMT)
    sh = logging.StreamHandler(sys.stderr)
    sh.setFormatter(fmt)
    _LOGGER.addHandler(sh)
    if rank == 0 and out_dir:
        os.makedirs(out_dir, exist_ok=True)
        fh = logging.FileHandler(os.path.join(out_dir, f"{time.time()}.log"))
        fh.setFormatter(fmt)
        _LOGGER.addHandler(fh)
    _CONFIGURED = True
    return _LOGGER


class _Proxy:
    """loguru-style proxy: logger.info(...), logger.warning(...), etc."""

    def _ensure(self):
        if not _CONFIGURED:
            setup_logger(rank=int(os.environ.get("RANK", 0)))

    def info(self, msg, *a):
        self._ensure()
        _LOGGER.info(str(msg) if not a else str(msg) % a)

    def warning(self, msg, *a):
        self._ensure()
        _LOGGER.warning(str(msg) if not a else str(msg) % a)

    def error(self, msg, *a):
        self._ensure()
        _LOGGER.error(str(msg) if not a else str(msg) % a)

    def debug(self, msg, *a):
        self._ensure()
        _LOGGER.debug(str(msg) if not a else str(msg) % a)


logger = _Proxy()
