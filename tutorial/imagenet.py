"""Tutorial 6 — minimal ImageNet-style DDP with the checkpoint hand-off
pattern (reference tutorial/imagenet.py parity).

Demonstrates the save/barrier/load-all-ranks sequence (reference
imagenet.py:146-165): rank 0 serializes, a barrier fences the write, then
EVERY rank loads with a device-specific map_location.

Run:  python -m torch.distributed.run --nproc-per-node 8 \
          --master-addr 127.0.0.1 tutorial/imagenet.py
"""

import os

import torch
import torch.distributed as dist
from torch.utils.data import DataLoader
from torch.utils.data.distributed import DistributedSampler

import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from distribuuuu_amd import models, utils  # noqa: E402
from distribuuuu_amd.data import DummyDataset  # noqa: E402
from distribuuuu_amd.parallel import DistributedDataParallel  # noqa: E402
from distribuuuu_amd.ops import functional as DF  # noqa: E402

CKPT = "/tmp/imagenet_tutorial_ckpt.pth"


def main():
    rank, local_rank = utils.setup_distributed()
    has_gpu = torch.cuda.is_available()
    device = torch.device(f"cuda:{local_rank}" if has_gpu else "cpu")
    net = models.build_model("resnet18", num_classes=1000).to(device)
    net = DistributedDataParallel(net)
    ds = DummyDataset(size=(3, 224, 224), length=256)
    sampler = DistributedSampler(ds)
    loader = DataLoader(ds, batch_size=32, sampler=sampler)
    opt = torch.optim.SGD(net.parameters(), lr=0.1, momentum=0.9)
    for i, (x, y) in enumerate(loader):
        loss = DF.cross_entropy(net(x.to(device)), y.to(device))
        net.zero_grad()
        loss.backward()
        opt.step()
        if rank == 0 and i % 2 == 0:
            print(f"iter {i} loss {loss.item():.4f}")

    # checkpoint hand-off: rank-0 save -> barrier -> all ranks load
    if rank == 0:
        torch.save(utils.unwrap_model(net).state_dict(), CKPT)
    if utils.get_world_size() > 1:
        dist.barrier()
    state = torch.load(
        CKPT, map_location=str(device) if has_gpu else "cpu",
        weights_only=True)
    utils.unwrap_model(net).load_state_dict(state)
    if rank == 0:
        print("checkpoint round-trip OK")
    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
