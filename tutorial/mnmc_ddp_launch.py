"""Tutorial 3 — Multi Node, Multi Card DDP via the env launcher
(reference tutorial/mnmc_ddp_launch.py parity).

One process per GPU; torch.distributed.run exports RANK/LOCAL_RANK/WORLD_SIZE
and the rendezvous address; gradients all-reduce on RCCL over xGMI through
our bucketed DDP.

Run (1 node, 8 GPUs):
  python -m torch.distributed.run --nproc-per-node 8 \
      --master-addr 127.0.0.1 tutorial/mnmc_ddp_launch.py
Simulate 2 "nodes" on one host (reference README.md:119-144 recipe) by
splitting GPUs with HIP_VISIBLE_DEVICES and --nnodes 2 --node-rank {0,1}.

Expected transcript (2-process gloo CPU run, seed 0 — asserted by
tests/test_tutorials.py as a regression oracle):

    epoch 0 iter 0 loss 2.4308
    epoch 0 iter 2 loss 2.7715
"""

EXPECTED_LOSSES = [2.4308, 2.7715]

import os

import torch
import torch.distributed as dist
from torch.utils.data import DataLoader
from torch.utils.data.distributed import DistributedSampler

import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from tutorial.snsc import DummyCifar, build_net  # noqa: E402
from distribuuuu_amd.parallel import DistributedDataParallel  # noqa: E402
from distribuuuu_amd.ops import functional as DF  # noqa: E402


def main(epochs=1):
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    has_gpu = torch.cuda.is_available()
    dist.init_process_group(backend="nccl" if has_gpu else "gloo")
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    device = torch.device(f"cuda:{local_rank}" if has_gpu else "cpu")
    if has_gpu:
        torch.cuda.set_device(local_rank)

    torch.manual_seed(0)  # deterministic build (DDP broadcasts rank 0 anyway)
    net = DistributedDataParallel(build_net().to(device))
    ds = DummyCifar()
    sampler = DistributedSampler(ds, shuffle=True)
    loader = DataLoader(ds, batch_size=64, sampler=sampler)
    # linear LR scaling with world size (reference README.md:198-200 recipe)
    opt = torch.optim.SGD(net.parameters(), lr=0.002 * world, momentum=0.9)
    for epoch in range(epochs):
        sampler.set_epoch(epoch)
        for i, (x, y) in enumerate(loader):
            loss = DF.cross_entropy(net(x.to(device)), y.to(device))
            net.zero_grad()
            loss.backward()
            opt.step()
            if rank == 0 and i % 2 == 0:
                print(f"epoch {epoch} iter {i} loss {loss.item():.4f}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
