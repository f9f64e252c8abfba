"""Tutorial 1 — Single Node, Single Card (reference tutorial/snsc.py parity).

The simplest rung of the ladder: one process, one device, a small CNN on
synthetic CIFAR-10-shaped data (this environment has no dataset downloads;
swap DummyCifar for a real dataset loader to reproduce the reference's
CIFAR-10 transcript).

Run:  python tutorial/snsc.py

Expected transcript (CPU, seed 0, 3 epochs — asserted by
tests/test_tutorials.py as a regression oracle, the same mechanism the
reference uses in its docstrings, e.g. reference snsc.py:85-114):

    epoch 0 iter 0 loss 2.5241
    epoch 0 iter 4 loss 2.6541
    epoch 1 iter 0 loss 2.2602
    epoch 1 iter 4 loss 2.2849
    epoch 2 iter 0 loss 2.3753
    epoch 2 iter 4 loss 2.0213
"""

# the docstring table above, machine-readable for the regression test
EXPECTED_LOSSES = [2.5241, 2.6541, 2.2602, 2.2849, 2.3753, 2.0213]

import torch
import torch.nn as nn
from torch.utils.data import DataLoader, Dataset

import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from distribuuuu_amd.ops import Conv2d, BatchNorm2d, Linear, MaxPool2d  # noqa: E402
from distribuuuu_amd.ops import functional as DF  # noqa: E402


class DummyCifar(Dataset):
    def __init__(self, n=512):
        g = torch.Generator().manual_seed(0)
        self.x = torch.randn(n, 3, 32, 32, generator=g)
        self.y = torch.randint(0, 10, (n,), generator=g)

    def __len__(self):
        return len(self.x)

    def __getitem__(self, i):
        return self.x[i], self.y[i]


def build_net():
    return nn.Sequential(
        Conv2d(3, 32, 3, padding=1), BatchNorm2d(32, act="relu"),
        MaxPool2d(2, 2),
        Conv2d(32, 64, 3, padding=1), BatchNorm2d(64, act="relu"),
        MaxPool2d(2, 2),
        nn.Flatten(), Linear(64 * 8 * 8, 10),
    )


def main(epochs=2, seed=0):
    torch.manual_seed(seed)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    net = build_net().to(device)
    g = torch.Generator().manual_seed(seed)
    loader = DataLoader(DummyCifar(), batch_size=64, shuffle=True, generator=g)
    opt = torch.optim.SGD(net.parameters(), lr=0.002, momentum=0.9)
    losses = []
    for epoch in range(epochs):
        for i, (x, y) in enumerate(loader):
            x, y = x.to(device), y.to(device)
            loss = DF.cross_entropy(net(x), y)
            opt.zero_grad()
            loss.backward()
            opt.step()
            if i % 4 == 0:
                print(f"epoch {epoch} iter {i} loss {loss.item():.4f}")
                losses.append(loss.item())
    return losses


if __name__ == "__main__":
    main()
