"""A/B the stage-1 64-wide conv shapes: auto route (smallk/smallc pipe
kernels, register-direct gathers) vs forced v1 (LDS-staged 128x128)."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch

from distribuuuu_amd.ops.dispatch import ext

e = ext()
cl = torch.channels_last


def bench(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


N = 256
# (label, C, K, R, pad)  — all 56x56 stride 1
CASES = [
    ("1x1 64->64  (conv1 dgrad)", 64, 64, 1, 0),
    ("3x3 64->64  (conv2 fwd/dgrad)", 64, 64, 3, 1),
    ("1x1 256->64 (conv3 dgrad)", 256, 64, 1, 0),
    ("1x1 64->256 (conv3 fwd)", 64, 256, 1, 0),
]
for label, c, k, r, pad in CASES:
    x = torch.randn(N, c, 56, 56, device="cuda",
                    dtype=torch.bfloat16).contiguous(memory_format=cl)
    w = (torch.randn(k, c, r, r, device="cuda", dtype=torch.bfloat16)
         * 0.05).contiguous(memory_format=cl)
    y = torch.empty(N, k, 56, 56, device="cuda",
                    dtype=torch.bfloat16).contiguous(memory_format=cl)
    ta = bench(lambda: e.conv2d_fwd(x, w, 1, 1, pad, pad, 1, 1, 1))
    tb = bench(lambda: e.conv2d_fwd_v1(x, w, y, 56, 56, 1, 1, pad, pad,
                                       1, 1, 1))
    ya = e.conv2d_fwd(x, w, 1, 1, pad, pad, 1, 1, 1).float()
    err = (ya - y.float()).abs().max().item()
    print(f"{label}: auto {ta:7.1f} us   v1 {tb:7.1f} us   "
          f"xerr {err:.2e}")
