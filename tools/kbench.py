"""Per-kernel micro-benchmarks on representative ResNet-50 shapes.

Usage (on a GPU box): python tools/kbench.py [group ...]
Groups: bn conv wgrad pool ce sgd all
Prints achieved GB/s (memory-bound ops) or TF/s (MFMA ops) per shape.
"""

import sys
import time

import torch

sys.path.insert(0, ".")
from distribuuuu_amd.ops.dispatch import require_ext  # noqa: E402

e = require_ext()
DEV = "cuda"


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def _cl(x):
    return x.contiguous(memory_format=torch.channels_last)


BN_SHAPES = [
    (256, 64, 112, 112),
    (256, 256, 56, 56),
    (256, 512, 28, 28),
    (256, 1024, 14, 14),
    (256, 2048, 7, 7),
]

CONV_SHAPES = [
    # (N, C, H, W, K, R, stride) — ResNet-50 hot layers
    (256, 64, 56, 56, 64, 3, 1),
    (256, 64, 56, 56, 256, 1, 1),
    (256, 256, 56, 56, 64, 1, 1),
    (256, 128, 28, 28, 128, 3, 1),
    (256, 256, 14, 14, 256, 3, 1),
    (256, 512, 7, 7, 512, 3, 1),
    (256, 256, 56, 56, 512, 1, 2),
    (256, 128, 56, 56, 128, 3, 2),
    (256, 3, 224, 224, 64, 7, 2),
]


def bench_bn():
    print("== bn kernels (GB/s moved) ==")
    for shp in BN_SHAPES:
        n, c, h, w = shp
        x = _cl(torch.randn(*shp, device=DEV, dtype=torch.bfloat16))
        nbytes = x.numel() * 2
        t = timeit(lambda: e.bn_sums(x))
        print(f"bn_sums      {str(shp):22s} {t*1e6:8.1f} us  {nbytes/t/1e9:7.0f} GB/s")
        scale = torch.randn(c, device=DEV)
        shift = torch.randn(c, device=DEV)
        res = _cl(torch.randn_like(x))
        t = timeit(lambda: e.bn_apply_act(x, scale, shift, 1, res))
        print(f"bn_apply+res {str(shp):22s} {t*1e6:8.1f} us  {3*nbytes/t/1e9:7.0f} GB/s")
        gy = _cl(torch.randn_like(x))
        y = _cl(torch.randn_like(x))
        mean = torch.zeros(c, device=DEV)
        rstd = torch.ones(c, device=DEV)
        g = torch.ones(c, device=DEV)
        b = torch.zeros(c, device=DEV)
        sc = torch.ones(c, device=DEV)
        sh = torch.zeros(c, device=DEV)
        t = timeit(lambda: e.bn_bwd(gy, x, res, mean, rstd, g, sc, sh, 1,
                                    True, True))
        print(f"bn_bwd(all)  {str(shp):22s} {t*1e6:8.1f} us  {8*nbytes/t/1e9:7.0f} GB/s")


def bench_conv():
    print("== conv fwd/dgrad (TF/s) ==")
    for shp in CONV_SHAPES:
        n, c, h, w, k, r, s = shp
        cc = 8 if c < 8 else c
        x = _cl(torch.randn(n, cc, h, w, device=DEV, dtype=torch.bfloat16))
        wt = _cl(torch.randn(k, cc, r, r, device=DEV, dtype=torch.bfloat16))
        p = r // 2
        ho = (h + 2 * p - r) // s + 1
        flops = 2.0 * n * ho * ho * k * cc * r * r
        t = timeit(lambda: e.conv2d_fwd(x, wt, s, s, p, p, 1, 1, 1))
        line = f"fwd   {str(shp):26s} {t*1e6:8.1f} us  {flops/t/1e12:6.1f} TF"
        gy = _cl(torch.randn(n, k, ho, ho, device=DEV, dtype=torch.bfloat16))
        if c >= 8:
            t2 = timeit(lambda: e.conv2d_dgrad(gy, wt, h, w, s, s, p, p, 1, 1, 1))
            line += f" | dgrad {t2*1e6:8.1f} us {flops/t2/1e12:6.1f} TF"
        print(line)


def bench_wgrad():
    print("== conv wgrad (TF/s) ==")
    for shp in CONV_SHAPES:
        n, c, h, w, k, r, s = shp
        cc = 8 if c < 8 else c
        x = _cl(torch.randn(n, cc, h, w, device=DEV, dtype=torch.bfloat16))
        p = r // 2
        ho = (h + 2 * p - r) // s + 1
        gy = _cl(torch.randn(n, k, ho, ho, device=DEV, dtype=torch.bfloat16))
        flops = 2.0 * n * ho * ho * k * cc * r * r
        t = timeit(lambda: e.conv2d_wgrad(gy, x, r, r, s, s, p, p, 1, 1, 1))
        print(f"wgrad {str(shp):26s} {t*1e6:8.1f} us  {flops/t/1e12:6.1f} TF")


def bench_pool():
    print("== pool (GB/s) ==")
    x = _cl(torch.randn(256, 64, 112, 112, device=DEV, dtype=torch.bfloat16))
    nb = x.numel() * 2
    t = timeit(lambda: e.maxpool_fwd(x, 3, 2, 1))
    print(f"maxpool_fwd  {t*1e6:8.1f} us  {nb/t/1e9:7.0f} GB/s (input read)")
    y, idx = e.maxpool_fwd(x, 3, 2, 1)
    gy = _cl(torch.randn_like(y))
    t = timeit(lambda: e.maxpool_bwd(gy, idx, 112, 112, 3, 2, 1))
    print(f"maxpool_bwd  {t*1e6:8.1f} us  {nb/t/1e9:7.0f} GB/s (gx write)")
    x2 = _cl(torch.randn(256, 2048, 7, 7, device=DEV, dtype=torch.bfloat16))
    t = timeit(lambda: e.gap_fwd(x2))
    print(f"gap_fwd      {t*1e6:8.1f} us  {x2.numel()*2/t/1e9:7.0f} GB/s")


def bench_gemm():
    print("== gemm_nt (TF/s) ==")
    for m, n, k in [(256, 1000, 2048), (4096, 4096, 4096), (8192, 8192, 8192)]:
        a = torch.randn(m, k, device=DEV, dtype=torch.bfloat16)
        b = torch.randn(n, k, device=DEV, dtype=torch.bfloat16)
        t = timeit(lambda: e.gemm_nt(a, b))
        print(f"gemm_nt {m}x{n}x{k}: {t*1e6:8.1f} us  {2.0*m*n*k/t/1e12:6.1f} TF")


if __name__ == "__main__":
    groups = sys.argv[1:] or ["all"]
    if "bn" in groups or "all" in groups:
        bench_bn()
    if "conv" in groups or "all" in groups:
        bench_conv()
    if "wgrad" in groups or "all" in groups:
        bench_wgrad()
    if "pool" in groups or "all" in groups:
        bench_pool()
    if "gemm" in groups or "all" in groups:
        bench_gemm()
