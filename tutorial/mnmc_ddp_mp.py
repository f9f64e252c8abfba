"""Tutorial 4 — DDP via mp.spawn with explicit rank math
(reference tutorial/mnmc_ddp_mp.py parity).

Instead of an external launcher, the parent process spawns one worker per
local GPU and computes each worker's GLOBAL rank itself:
    rank = node_rank * gpus_per_node + local_rank
(reference mnmc_ddp_mp.py:56).

Run:  python tutorial/mnmc_ddp_mp.py [--nodes 1 --node-rank 0]
"""

import argparse
import os

import torch
import torch.distributed as dist
import torch.multiprocessing as mp
from torch.utils.data import DataLoader
from torch.utils.data.distributed import DistributedSampler

import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from tutorial.snsc import DummyCifar, build_net  # noqa: E402
from distribuuuu_amd.parallel import DistributedDataParallel  # noqa: E402
from distribuuuu_amd.ops import functional as DF  # noqa: E402


def worker(local_rank, args):
    rank = args.node_rank * args.gpus + local_rank  # global rank math
    world = args.nodes * args.gpus
    os.environ["MASTER_ADDR"] = args.master_addr
    os.environ["MASTER_PORT"] = str(args.master_port)
    has_gpu = torch.cuda.is_available()
    dist.init_process_group("nccl" if has_gpu else "gloo", rank=rank,
                            world_size=world)
    device = torch.device(f"cuda:{local_rank}" if has_gpu else "cpu")
    if has_gpu:
        torch.cuda.set_device(local_rank)
    net = DistributedDataParallel(build_net().to(device))
    ds = DummyCifar()
    sampler = DistributedSampler(ds, num_replicas=world, rank=rank)
    loader = DataLoader(ds, batch_size=64, sampler=sampler)
    opt = torch.optim.SGD(net.parameters(), lr=0.05 * world, momentum=0.9)
    for i, (x, y) in enumerate(loader):
        loss = DF.cross_entropy(net(x.to(device)), y.to(device))
        net.zero_grad()
        loss.backward()
        opt.step()
        if rank == 0 and i % 2 == 0:
            print(f"iter {i} loss {loss.item():.4f}")
    dist.destroy_process_group()


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--nodes", type=int, default=1)
    p.add_argument("--node-rank", type=int, default=0)
    p.add_argument("--gpus", type=int,
                   default=max(torch.cuda.device_count(), 1))
    p.add_argument("--master-addr", default="127.0.0.1")
    p.add_argument("--master-port", type=int, default=29566)
    args = p.parse_args()
    mp.spawn(worker, nprocs=args.gpus, args=(args,))


if __name__ == "__main__":
    main()
