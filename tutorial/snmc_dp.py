"""Tutorial 2 — Single Node, Multi Card with nn.DataParallel
(reference tutorial/snmc_dp.py parity).

DataParallel replicates the module onto every visible GPU each forward and
scatters the batch; simple but single-process (one Python GIL, replicate +
gather every step). Kept for teaching parity — the framework proper uses the
one-process-per-GPU DDP of tutorial 3+.

Run:  python tutorial/snmc_dp.py
"""

import torch
import torch.nn as nn

import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from tutorial.snsc import DummyCifar, build_net  # noqa: E402
from torch.utils.data import DataLoader  # noqa: E402
from distribuuuu_amd.ops import functional as DF  # noqa: E402


def main(epochs=1):
    assert torch.cuda.is_available(), "DataParallel needs GPUs"
    net = build_net().cuda()
    if torch.cuda.device_count() > 1:
        net = nn.DataParallel(net)
    loader = DataLoader(DummyCifar(), batch_size=256, shuffle=True)
    opt = torch.optim.SGD(net.parameters(), lr=0.1, momentum=0.9)
    for epoch in range(epochs):
        for i, (x, y) in enumerate(loader):
            loss = DF.cross_entropy(net(x.cuda()), y.cuda())
            opt.zero_grad()
            loss.backward()
            opt.step()
            print(f"epoch {epoch} iter {i} loss {loss.item():.4f}")


if __name__ == "__main__":
    main()
