"""A/B: dense 3x3 wgrad 64x64 tr-staged kernel vs the ring128 route
(DISTRIBUUUU_WGRAD_128 force env) on the ResNet-50 b256 3x3 shapes.
With Cg % 128 == 0 a 128-wide n-tile never crosses a tap boundary, so the
round-1 'tap-spanning' objection only applies to the 64-channel shapes."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch

from distribuuuu_amd.ops.dispatch import ext

e = ext()
cl = torch.channels_last


def bench(fn, iters=15):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


# (count/step incl. stride variants, C, H, W, K, R, stride) — rs50 b256 3x3
SHAPES = [
    (3, 64, 56, 56, 64, 3, 1),
    (1, 128, 56, 56, 128, 3, 2),
    (3, 128, 28, 28, 128, 3, 1),
    (1, 256, 28, 28, 256, 3, 2),
    (5, 256, 14, 14, 256, 3, 1),
    (1, 512, 14, 14, 512, 3, 2),
    (2, 512, 7, 7, 512, 3, 1),
]

N = 256
tot = {"0": 0.0, "1": 0.0}
for cnt, c, h, w, k, r, s in SHAPES:
    ho = (h + 2 - r) // s + 1
    x = torch.randn(N, c, h, w, device="cuda",
                    dtype=torch.bfloat16).contiguous(memory_format=cl)
    gy = torch.randn(N, k, ho, ho, device="cuda",
                     dtype=torch.bfloat16).contiguous(memory_format=cl)
    res = {}
    for mode in ("0", "1"):
        os.environ["DISTRIBUUUU_WGRAD_128"] = mode
        t = bench(lambda: e.conv2d_wgrad(gy, x, r, r, s, s, 1, 1, 1, 1, 1))
        res[mode] = t * 1e6
        tot[mode] += cnt * t * 1e3
    os.environ["DISTRIBUUUU_WGRAD_128"] = "0"
    gw0 = e.conv2d_wgrad(gy, x, r, r, s, s, 1, 1, 1, 1, 1).float()
    os.environ["DISTRIBUUUU_WGRAD_128"] = "1"
    gw1 = e.conv2d_wgrad(gy, x, r, r, s, s, 1, 1, 1, 1, 1).float()
    err = (gw0 - gw1).abs().max().item()
    scl = gw0.abs().max().item()
    print(f"{c:4d}x{h:2d} s{s} x{cnt}:  64x64 {res['0']:8.1f} us   "
          f"ring128 {res['1']:8.1f} us   xerr {err:.2e}/{scl:.1e}")
print(f"step totals: 64x64 {tot['0']:.2f} ms   ring128 {tot['1']:.2f} ms")
os.environ.pop("DISTRIBUUUU_WGRAD_128", None)


# --- T128 tr-staged variant A/B (env DISTRIBUUUU_WGRAD_T128) ---
print("\n== 64x64 vs tr-staged 128x128 ==")
tot = {"0": 0.0, "1": 0.0}
for cnt, c, h, w, k, r, s in SHAPES:
    ho = (h + 2 - r) // s + 1
    x = torch.randn(N, c, h, w, device="cuda",
                    dtype=torch.bfloat16).contiguous(memory_format=cl)
    gy = torch.randn(N, k, ho, ho, device="cuda",
                     dtype=torch.bfloat16).contiguous(memory_format=cl)
    res = {}
    for mode in ("0", "1"):
        os.environ["DISTRIBUUUU_WGRAD_T128"] = mode
        t = bench(lambda: e.conv2d_wgrad(gy, x, r, r, s, s, 1, 1, 1, 1, 1))
        res[mode] = t * 1e6
        tot[mode] += cnt * t * 1e3
    os.environ["DISTRIBUUUU_WGRAD_T128"] = "0"
    gw0 = e.conv2d_wgrad(gy, x, r, r, s, s, 1, 1, 1, 1, 1).float()
    os.environ["DISTRIBUUUU_WGRAD_T128"] = "1"
    gw1 = e.conv2d_wgrad(gy, x, r, r, s, s, 1, 1, 1, 1, 1).float()
    err = (gw0 - gw1).abs().max().item()
    scl = gw0.abs().max().item()
    print(f"{c:4d}x{h:2d} s{s} x{cnt}:  64x64 {res['0']:8.1f} us   "
          f"t128 {res['1']:8.1f} us   xerr {err:.2e}/{scl:.1e}")
print(f"step totals: 64x64 {tot['0']:.2f} ms   t128 {tot['1']:.2f} ms")
os.environ.pop("DISTRIBUUUU_WGRAD_T128", None)
