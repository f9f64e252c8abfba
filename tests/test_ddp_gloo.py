"""Multi-process (gloo, world_size=2) tests for our DDP and SyncBN:
gradients must equal the single-process full-batch reference."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp
import torch.nn as nn


def _init(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)


def _make_model(seed=0):
    """No plain BN here: per-rank BN stats legitimately differ from the
    full-batch reference (that's what SyncBN is for — tested separately)."""
    torch.manual_seed(seed)
    from distribuuuu_amd.ops import Conv2d, Linear, ReLU

    return nn.Sequential(
        Conv2d(3, 8, 3, padding=1), ReLU(),
        nn.Flatten(), Linear(8 * 8 * 8, 5),
    )


def _make_bn_model(seed=0):
    torch.manual_seed(seed)
    from distribuuuu_amd.ops import Conv2d, BatchNorm2d, Linear

    return nn.Sequential(
        Conv2d(3, 8, 3, padding=1), BatchNorm2d(8, act="relu"),
        nn.Flatten(), Linear(8 * 8 * 8, 5),
    )


def _ddp_worker(rank, world, port, q):
    _init(rank, world, port)
    from distribuuuu_amd.parallel import DistributedDataParallel

    torch.manual_seed(123)
    x_full = torch.randn(8, 3, 8, 8)
    y_full = torch.randint(0, 5, (8,))
    net = DistributedDataParallel(_make_model(), bucket_cap_mb=1)
    x = x_full[rank * 4:(rank + 1) * 4]
    y = y_full[rank * 4:(rank + 1) * 4]
    out = net(x)
    loss = nn.functional.cross_entropy(out, y)
    net.zero_grad()
    loss.backward()
    if rank == 0:
        grads = {n: p.grad.clone() for n, p in net.module.named_parameters()}
        q.put(grads)
    dist.barrier()
    dist.destroy_process_group()


def test_ddp_grads_match_full_batch():
    port = 29611
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_ddp_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    grads = q.get()
    for p in procs:
        p.join(120)
        assert p.exitcode == 0

    # single-process full-batch reference (mean CE over the 8 samples)
    torch.manual_seed(123)
    x_full = torch.randn(8, 3, 8, 8)
    y_full = torch.randint(0, 5, (8,))
    ref = _make_model()
    out = ref(x_full)
    loss = nn.functional.cross_entropy(out, y_full)
    loss.backward()
    for n, p in ref.named_parameters():
        assert torch.allclose(grads[n], p.grad, atol=1e-5), n


def _syncbn_worker(rank, world, port, q):
    _init(rank, world, port)
    from distribuuuu_amd.parallel import convert_sync_batchnorm

    net = _make_bn_model(seed=1)
    net = convert_sync_batchnorm(net)
    net.train()
    torch.manual_seed(55)
    x_full = torch.randn(8, 3, 8, 8)
    x = x_full[rank * 4:(rank + 1) * 4]
    out = net(x)
    out.square().mean().backward()
    if rank == 0:
        bn = net[1]
        q.put({
            "running_mean": bn.running_mean.clone(),
            "running_var": bn.running_var.clone(),
            "out0": out.detach().clone(),
            "g_weight": bn.weight.grad.clone(),
        })
    dist.barrier()
    dist.destroy_process_group()


def test_syncbn_stats_match_full_batch():
    port = 29612
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_syncbn_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    got = q.get()
    for p in procs:
        p.join(120)
        assert p.exitcode == 0

    # full-batch reference with plain BN
    net = _make_bn_model(seed=1)
    net.train()
    torch.manual_seed(55)
    x_full = torch.randn(8, 3, 8, 8)
    out = net(x_full)
    bn = net[1]
    assert torch.allclose(got["running_mean"], bn.running_mean, atol=1e-5)
    assert torch.allclose(got["running_var"], bn.running_var, atol=1e-4)
    assert torch.allclose(got["out0"], out[:4].detach(), atol=1e-5)


def _scaled_all_reduce_worker(rank, world, port, q):
    _init(rank, world, port)
    from distribuuuu_amd import utils

    t = [torch.tensor(float(rank + 1)), torch.tensor(float(10 * (rank + 1)))]
    utils.scaled_all_reduce(t)
    if rank == 0:
        q.put([x.item() for x in t])
    dist.barrier()
    dist.destroy_process_group()


def test_scaled_all_reduce_two_ranks():
    port = 29613
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_scaled_all_reduce_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    vals = q.get()
    for p in procs:
        p.join(120)
        assert p.exitcode == 0
    assert vals[0] == pytest.approx(1.5)
    assert vals[1] == pytest.approx(15.0)


class _TwoDtypeNet(nn.Module):
    def __init__(self):
        super().__init__()
        torch.manual_seed(3)
        self.a = nn.Parameter(torch.randn(64, dtype=torch.bfloat16))
        self.b = nn.Parameter(torch.randn(64, dtype=torch.float32))

    def forward(self, x):
        return (x * self.b).sum() + (x.to(torch.bfloat16) * self.a).float().sum()


def _mixed_dtype_worker(rank, world, port, q):
    """bf16 + fp32 parameters in one module (the bf16 training layout where
    BN stays fp32) must split into per-dtype buckets."""
    _init(rank, world, port)
    from distribuuuu_amd.parallel import DistributedDataParallel

    ddp = DistributedDataParallel(_TwoDtypeNet(), bucket_cap_mb=1)
    x = torch.randn(64)
    loss = ddp(x)
    ddp.zero_grad()
    loss.backward()
    if rank == 0:
        dtypes = sorted({str(b.buffer.dtype) for b in ddp._buckets})
        q.put(dtypes)
    dist.barrier()
    dist.destroy_process_group()


def test_ddp_mixed_dtype_buckets():
    port = 29614
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_mixed_dtype_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(120)
    assert all(p.exitcode == 0 for p in procs)
    dtypes = q.get()
    assert "torch.bfloat16" in dtypes and "torch.float32" in dtypes
