"""Run just the 3x3 64->64 conv in a loop for rocprofv3 --pmc counter runs."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch

from distribuuuu_amd.ops.dispatch import ext

e = ext()
cl = torch.channels_last
x = torch.randn(256, 64, 56, 56, dtype=torch.bfloat16, device="cuda").to(memory_format=cl)
w = torch.randn(64, 64, 3, 3, dtype=torch.bfloat16, device="cuda").to(memory_format=cl)
for _ in range(30):
    y = e.conv2d_fwd(x, w, 1, 1, 1, 1, 1, 1, 1)
torch.cuda.synchronize()
print("done")
