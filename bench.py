"""Flagship benchmark: ResNet-50 DDP bf16 training step on N MI355X GPUs.

Driver contract: ``python bench.py --gpus N --steps K --warmup W`` (N>1 is
launched by the driver via torch.distributed.run, one rank per GPU over RCCL).
Measures the BASELINE.json metric — images/sec, whole node, synthetic
3x224x224 data, random-init weights, bf16 compute, per-rank batch 256,
SyncBN on RCCL when world_size > 1 — and prints ONE JSON line from rank 0.
"""

import argparse
import json
import os
import time

import torch
import torch.distributed as dist


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--arch", type=str, default="resnet50")
    p.add_argument("--batch", type=int, default=256, help="per-rank batch size")
    p.add_argument("--dtype", type=str, default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--syncbn", type=int, default=-1,
                   help="-1: auto (on when world>1), 0/1: force")
    p.add_argument("--graph", type=int, default=-1,
                   help="capture the step in a hipGraph: -1 auto "
                        "(on for 1 GPU), 0/1 force")
    return p.parse_args()


def _self_launch(args):
    """`python bench.py --gpus N` without torchrun must NOT silently measure
    1 GPU: spawn torch.distributed.run ourselves, one rank per GPU."""
    import socket
    import subprocess
    import sys

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", f"--nproc-per-node={args.gpus}",
           "--master-addr=127.0.0.1", f"--master-port={port}",
           os.path.abspath(__file__)] + sys.argv[1:]
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    raise SystemExit(subprocess.call(cmd, env=env))


def main():
    args = parse_args()
    if args.gpus > 1 and "WORLD_SIZE" not in os.environ:
        _self_launch(args)
    world_size = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    has_gpu = torch.cuda.is_available()

    if world_size > 1:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29566")
        dist.init_process_group(backend="nccl" if has_gpu else "gloo",
                                rank=rank, world_size=world_size)
    if has_gpu:
        torch.cuda.set_device(local_rank)
        device = torch.device(f"cuda:{local_rank}")
    else:
        device = torch.device("cpu")

    from distribuuuu_amd.config import cfg
    from distribuuuu_amd import trainer as T
    from distribuuuu_amd import utils
    from distribuuuu_amd.data import DeviceSyntheticLoader
    from distribuuuu_amd.parallel import DistributedDataParallel
    from distribuuuu_amd.ops import functional as DF

    use_bf16 = args.dtype == "bf16" and has_gpu
    use_syncbn = (world_size > 1) if args.syncbn == -1 else bool(args.syncbn)
    batch = args.batch if has_gpu else 8  # CPU smoke keeps it tiny

    cfg.defrost()
    cfg.MODEL.ARCH = args.arch
    cfg.MODEL.SYNCBN = use_syncbn
    cfg.TRAIN.DTYPE = "bfloat16" if use_bf16 else "float32"
    cfg.TRAIN.CHANNELS_LAST = bool(has_gpu)
    cfg.OUT_DIR = os.environ.get("BENCH_OUT_DIR", "/tmp/bench_out")
    cfg.freeze()

    dtype = torch.bfloat16 if use_bf16 else torch.float32
    net = T.build_network(device)
    if world_size > 1:
        net = DistributedDataParallel(net, bucket_cap_mb=cfg.TRAIN.BUCKET_CAP_MB)
    optimizer = utils.construct_optimizer(net)
    utils.set_lr(optimizer, 0.2)
    net.train()

    loader = DeviceSyntheticLoader(batch, im_size=224, num_classes=1000,
                                   device=device, dtype=dtype,
                                   channels_last=bool(has_gpu))
    it = iter(loader)

    # world==1: set_to_none avoids 161 per-step grad memsets (no DDP bucket
    # views to preserve); DDP needs stable bucket-view grads.
    zero_none = world_size == 1

    def step():
        inputs, targets = next(it)
        outputs = net(inputs)
        loss = DF.cross_entropy(outputs.float(), targets)
        optimizer.zero_grad(set_to_none=zero_none)
        loss.backward()
        optimizer.step()
        return loss

    for _ in range(args.warmup):
        step()

    # hipGraph capture of the whole train step: the inner loop replays one
    # graph instead of ~900 eager launches. Multi-rank too — RCCL collectives
    # (DDP buckets, SyncBN stats) are capture-legal on ROCm, and an
    # eager-launch-bound rank would cap scaling; capture failure falls back
    # to eager below.
    use_graph = bool(has_gpu) if args.graph == -1 else bool(args.graph)
    graph = None
    if use_graph:
        try:
            torch.cuda.synchronize()
            side = torch.cuda.Stream()
            with torch.cuda.stream(side):
                for _ in range(2):
                    step()
            torch.cuda.current_stream().wait_stream(side)
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                step()
            graph.replay()
            torch.cuda.synchronize()
        except Exception as exc:  # pragma: no cover - fall back to eager
            print(f"[bench] graph capture failed ({exc}); eager mode",
                  flush=True)
            graph = None
        if world_size > 1:
            # every rank must agree (a replaying rank paired with an eager
            # one still matches collective order, but keep it uniform)
            ok = torch.tensor([1.0 if graph is not None else 0.0],
                              device=device)
            dist.all_reduce(ok, op=dist.ReduceOp.MIN)
            if ok.item() < 1.0:
                graph = None

    if world_size > 1:
        dist.barrier()
    if has_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        if graph is not None:
            graph.replay()
        else:
            step()
    if has_gpu:
        torch.cuda.synchronize()
    if world_size > 1:
        dist.barrier()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if world_size > 1:
        t = torch.tensor([elapsed], dtype=torch.float64, device=device
                         if has_gpu else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    global_batch = batch * world_size
    images_per_sec = args.steps * global_batch / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        print(json.dumps({
            "metric": f"images/sec {args.arch} train (whole node)",
            "value": round(images_per_sec, 2),
            "unit": "images/sec",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_bf16 else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.arch,
                "global_batch": global_batch,
                "seq_len": None,
                "im_size": 224,
                "parallelism": f"dp{world_size}",
                "syncbn": use_syncbn,
                "hipgraph": graph is not None,
            },
        }))

    if world_size > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
