"""A/B: grouped deep-3x3 wgrad on the 64x64 tr-staged kernel vs the ring128
route (DISTRIBUUUU_WGRAD_128), on RegNetY-32GF's real shapes at batch 64."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch

from distribuuuu_amd.ops.dispatch import ext

e = ext()
cl = torch.channels_last


def bench(fn, iters=20):
    for _ in range(4):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


# (count, Ct, H, W, Kt, R, stride, groups) — regnety_320 b64 grouped 3x3
SHAPES = [
    (4, 696, 28, 28, 696, 3, 1, 3),
    (1, 696, 56, 56, 696, 3, 2, 3),
    (11, 1392, 14, 14, 1392, 3, 1, 6),
    (1, 1392, 28, 28, 1392, 3, 2, 6),
    (1, 3712, 7, 7, 3712, 3, 2, 16),
]

N = 64
for cnt, c, h, w, k, r, s, g in SHAPES:
    ho = (h + 2 - r) // s + 1
    x = torch.randn(N, c, h, w, device="cuda",
                    dtype=torch.bfloat16).contiguous(memory_format=cl)
    gy = torch.randn(N, k, ho, ho, device="cuda",
                     dtype=torch.bfloat16).contiguous(memory_format=cl)
    res = {}
    for mode in ("0", "1"):
        os.environ["DISTRIBUUUU_WGRAD_128"] = mode
        t = bench(lambda: e.conv2d_wgrad(gy, x, r, r, s, s, 1, 1, 1, 1, g))
        res[mode] = t * 1e6
    # correctness cross-check between the two routes
    os.environ["DISTRIBUUUU_WGRAD_128"] = "0"
    gw0 = e.conv2d_wgrad(gy, x, r, r, s, s, 1, 1, 1, 1, g).float()
    os.environ["DISTRIBUUUU_WGRAD_128"] = "1"
    gw1 = e.conv2d_wgrad(gy, x, r, r, s, s, 1, 1, 1, 1, g).float()
    err = (gw0 - gw1).abs().max().item()
    scl = gw0.abs().max().item()
    print(f"g{g:2d} {c}x{h} s{s} x{cnt}:  64x64 {res['0']:8.1f} us   "
          f"ring128 {res['1']:8.1f} us   xerr {err:.3e}/{scl:.1e}")
os.environ.pop("DISTRIBUUUU_WGRAD_128", None)
