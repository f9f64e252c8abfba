"""Tutorial ladder smoke tests (the reference's teaching scripts are its
de-facto regression suite — SURVEY.md §4)."""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_snsc_runs():
    from tutorial import snsc

    snsc.main(epochs=1)


def test_ddp_launch_tutorial_2proc_gloo():
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    env["MASTER_PORT"] = "29621"
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29621", "tutorial/mnmc_ddp_launch.py"],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "loss" in r.stdout


def test_imagenet_tutorial_single_proc():
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run([sys.executable, "tutorial/imagenet.py"], cwd=REPO,
                       env=env, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "checkpoint round-trip OK" in r.stdout


def test_bench_2proc_gloo():
    """bench.py under torchrun world 2 on CPU (the driver's multi-GPU launch
    shape) — checks the DDP/SyncBN/metric-aggregation path end to end."""
    env = dict(os.environ)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29631", "bench.py", "--gpus", "2", "--steps", "2",
         "--warmup", "1"],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-3000:]
    import json

    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_gpus"] == 2
    assert d["config"]["syncbn"] is True
