"""A/B the tr128 XCD-aware tile remap (DISTRIBUUUU_WGRAD_XCD=0 disables)
across the rs50 wgrad shapes the t128 route owns (deep 3x3 + deep 1x1)."""
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
    return (time.perf_counter() - t0) / iters


# (count/step, C, H, W, K, R, stride, pad)
SHAPES = [
    (3, 128, 56, 56, 128, 3, 1, 1),   # wait: dense 3x3 list
    (3, 128, 28, 28, 128, 3, 1, 1),
    (5, 256, 14, 14, 256, 3, 1, 1),
    (2, 512, 7, 7, 512, 3, 1, 1),
    (3, 256, 56, 56, 64, 1, 1, 0),    # stage1 conv1 wgrad shapes (gy=64ch)
    (3, 64, 56, 56, 256, 1, 1, 0),    # stage1 conv3
    (4, 512, 28, 28, 128, 1, 1, 0),
    (4, 128, 28, 28, 512, 1, 1, 0),
    (6, 1024, 14, 14, 256, 1, 1, 0),
    (6, 256, 14, 14, 1024, 1, 1, 0),
    (3, 2048, 7, 7, 512, 1, 1, 0),
    (3, 512, 7, 7, 2048, 1, 1, 0),
]

N = 256
tot = {"0": 0.0, "1": 0.0}
for cnt, c, h, w, k, r, s, pad in SHAPES:
    ho = (h + 2 * pad - r) // s + 1
    x = torch.randn(N, c, h, w, device="cuda",
                    dtype=torch.bfloat16).contiguous(memory_format=cl)
    gy = torch.randn(N, k, ho, ho, device="cuda",
                     dtype=torch.bfloat16).contiguous(memory_format=cl)
    res = {}
    for mode in ("0", "1"):
        os.environ["DISTRIBUUUU_WGRAD_XCD"] = mode
        t = bench(lambda: e.conv2d_wgrad(gy, x, r, r, s, s, pad, pad, 1, 1, 1))
        res[mode] = t * 1e6
        tot[mode] += cnt * t * 1e3
    print(f"{c:4d}x{h:2d} {r}x{r} x{cnt}: off {res['0']:8.1f} us   "
          f"xcd {res['1']:8.1f} us   {res['0']/res['1']:.2f}x")
os.environ.pop("DISTRIBUUUU_WGRAD_XCD", None)
print(f"totals (weighted): off {tot['0']:.2f} ms   xcd {tot['1']:.2f} ms")
