"""Tutorial 5 — DDP under Slurm (reference tutorial/mnmc_ddp_slurm.py parity).

srun starts one task per GPU; ranks come from SLURM_PROCID/SLURM_NTASKS and
the master address from `scontrol show hostname $SLURM_NODELIST`. This is
the same dual-mode rendezvous the framework's utils.setup_distributed()
implements (reference utils.py:19-51).

Run:  srun -N2 --gres=gpu:8 --ntasks-per-node 8 python tutorial/mnmc_ddp_slurm.py
"""

import os

import torch
from torch.utils.data import DataLoader
from torch.utils.data.distributed import DistributedSampler

import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from tutorial.snsc import DummyCifar, build_net  # noqa: E402
from distribuuuu_amd import utils  # noqa: E402
from distribuuuu_amd.parallel import DistributedDataParallel  # noqa: E402
from distribuuuu_amd.ops import functional as DF  # noqa: E402


def main():
    rank, local_rank = utils.setup_distributed()  # Slurm or launcher env
    world = utils.get_world_size()
    has_gpu = torch.cuda.is_available()
    device = torch.device(f"cuda:{local_rank}" if has_gpu else "cpu")
    net = DistributedDataParallel(build_net().to(device))
    ds = DummyCifar()
    sampler = DistributedSampler(ds)
    loader = DataLoader(ds, batch_size=64, sampler=sampler)
    opt = torch.optim.SGD(net.parameters(), lr=0.05 * world, momentum=0.9)
    for i, (x, y) in enumerate(loader):
        loss = DF.cross_entropy(net(x.to(device)), y.to(device))
        net.zero_grad()
        loss.backward()
        opt.step()
        if rank == 0 and i % 2 == 0:
            print(f"iter {i} loss {loss.item():.4f}")


if __name__ == "__main__":
    main()
