"""Per-layer conv timing map: every distinct ResNet-50 conv shape (with its
per-step multiplicity) through conv2d_fwd / conv2d_dgrad / conv2d_wgrad.
Decomposes the profile's aggregate conv cost so tile-variant work targets the
shapes that actually dominate.
"""
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


# (count, C, H, W, K, R, stride) at batch 256 — ResNet-50
SHAPES = [
    (1, 3, 224, 224, 64, 7, 2),
    (1, 64, 56, 56, 64, 1, 1),
    (2, 256, 56, 56, 64, 1, 1),
    (3, 64, 56, 56, 64, 3, 1),
    (3, 64, 56, 56, 256, 1, 1),
    (1, 64, 56, 56, 256, 1, 1),     # downsample s1 (same shape)
    (1, 256, 56, 56, 128, 1, 1),
    (1, 128, 56, 56, 128, 3, 2),
    (1, 256, 56, 56, 512, 1, 2),    # downsample
    (3, 512, 28, 28, 128, 1, 1),
    (3, 128, 28, 28, 128, 3, 1),
    (4, 128, 28, 28, 512, 1, 1),
    (1, 512, 28, 28, 256, 1, 1),
    (1, 256, 28, 28, 256, 3, 2),
    (1, 512, 28, 28, 1024, 1, 2),   # downsample
    (5, 1024, 14, 14, 256, 1, 1),
    (5, 256, 14, 14, 256, 3, 1),
    (6, 256, 14, 14, 1024, 1, 1),
    (1, 1024, 14, 14, 512, 1, 1),
    (1, 512, 14, 14, 512, 3, 2),
    (1, 1024, 14, 14, 2048, 1, 2),  # downsample
    (2, 2048, 7, 7, 512, 1, 1),
    (3, 512, 7, 7, 512, 3, 1),
    (3, 512, 7, 7, 2048, 1, 1),
]

N = 256
tot = {"fwd": 0.0, "dgrad": 0.0, "wgrad": 0.0}
rows = []
for (cnt, c, h, w, k, r, s) in SHAPES:
    pad = r // 2
    cp = max(c, 8)
    x = torch.randn(N, cp, h, w, dtype=torch.bfloat16, device="cuda").to(
        memory_format=cl)
    wt = torch.randn(k, cp, r, r, dtype=torch.bfloat16, device="cuda").to(
        memory_format=cl)
    ho = (h + 2 * pad - r) // s + 1
    gy = torch.randn(N, k, ho, ho, dtype=torch.bfloat16, device="cuda").to(
        memory_format=cl)
    fl = 2.0 * N * ho * ho * k * r * r * cp
    tf = bench(lambda: e.conv2d_fwd(x, wt, s, s, pad, pad, 1, 1, 1))
    td = bench(lambda: e.conv2d_dgrad(gy, wt, h, w, s, s, pad, pad, 1, 1, 1))
    tw = bench(lambda: e.conv2d_wgrad(gy, x, r, r, s, s, pad, pad, 1, 1, 1))
    tot["fwd"] += cnt * tf
    tot["dgrad"] += cnt * td
    tot["wgrad"] += cnt * tw
    rows.append((cnt * (tf + td + tw), cnt, c, h, k, r, s, tf, td, tw, fl))
    del x, wt, gy
    torch.cuda.empty_cache()

rows.sort(reverse=True)
print(f"{'cost':>8} {'shape':>24} {'fwd':>14} {'dgrad':>14} {'wgrad':>14}")
for (cost, cnt, c, h, k, r, s, tf, td, tw, fl) in rows:
    print(f"{cost*1e3:7.2f}ms {cnt}x C{c:4d}@{h:3d} K{k:4d} "
          f"{r}x{r} s{s}: "
          f"{tf*1e6:6.0f}us {fl/tf/1e12:4.0f}TF "
          f"{td*1e6:6.0f}us {fl/td/1e12:4.0f}TF "
          f"{tw*1e6:6.0f}us {fl/tw/1e12:4.0f}TF")
print(f"totals/step: fwd {tot['fwd']*1e3:.2f} ms  dgrad {tot['dgrad']*1e3:.2f}"
      f" ms  wgrad {tot['wgrad']*1e3:.2f} ms")
