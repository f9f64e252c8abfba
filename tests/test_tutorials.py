"""Tutorial ladder smoke tests (the reference's teaching scripts are its
de-facto regression suite — SURVEY.md §4)."""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_snsc_golden_transcript():
    """The docstring transcript is a regression oracle (the reference embeds
    expected stdout in every tutorial docstring, e.g. snsc.py:85-114)."""
    from tutorial import snsc

    losses = snsc.main(epochs=3)
    assert len(losses) == len(snsc.EXPECTED_LOSSES)
    for got, want in zip(losses, snsc.EXPECTED_LOSSES):
        assert abs(got - want) < 5e-3, (losses, snsc.EXPECTED_LOSSES)


def test_ddp_launch_tutorial_2proc_gloo_golden():
    from tutorial.mnmc_ddp_launch import EXPECTED_LOSSES

    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    env["MASTER_PORT"] = "29621"
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29621", "tutorial/mnmc_ddp_launch.py"],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    got = [float(line.rsplit(" ", 1)[1]) for line in r.stdout.splitlines()
           if " loss " in line]
    assert len(got) == len(EXPECTED_LOSSES), r.stdout
    for g, want in zip(got, EXPECTED_LOSSES):
        assert abs(g - want) < 5e-3, (got, EXPECTED_LOSSES)


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
