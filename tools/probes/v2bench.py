"""Correctness + A/B timing of conv v2 (ring/counted-vmcnt) vs v1."""
import sys, time
import torch
import torch.nn.functional as F

sys.path.insert(0, "/root/repo")
from distribuuuu_amd.ops.dispatch import require_ext

e = require_ext()
cl = lambda t: t.contiguous(memory_format=torch.channels_last)


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


SHAPES = [
    (256, 64, 56, 56, 256, 1, 1, 0),
    (256, 256, 56, 56, 512, 1, 2, 0),
    (256, 512, 28, 28, 256, 1, 1, 0),
    (256, 256, 14, 14, 256, 3, 1, 1),
    (256, 512, 7, 7, 512, 3, 1, 1),
    (256, 2048, 7, 7, 512, 1, 1, 0),
    (8, 8192, 32, 32, 8192, 1, 1, 0),  # gemm-like 8192-deep
]
for (n, c, h, w_, k, r, s, p) in SHAPES:
    torch.manual_seed(0)
    x = cl(torch.randn(n, c, h, w_, device="cuda", dtype=torch.bfloat16))
    w = cl(torch.randn(k, c, r, r, device="cuda", dtype=torch.bfloat16) * 0.05)
    y2 = e.conv2d_fwd_v2(x, w, s, s, p, p, 1, 1)
    ref = F.conv2d(x.float(), w.float(), None, s, p, 1, 1)
    err = (y2.float() - ref).abs().max().item()
    sc = ref.abs().max().item()
    ok = err < 2e-2 * max(sc, 1.0)
    ho = (h + 2 * p - r) // s + 1
    fl = 2.0 * n * ho * ho * k * c * r * r
    t1 = timeit(lambda: e.conv2d_fwd(x, w, s, s, p, p, 1, 1, 1))
    t2 = timeit(lambda: e.conv2d_fwd_v2(x, w, s, s, p, p, 1, 1))
    print(f"{(n,c,h,k,r,s)}: {'OK ' if ok else f'FAIL err={err:.3f} '} "
          f"v1 {t1*1e6:7.1f}us {fl/t1/1e12:6.1f}TF | v2 {t2*1e6:7.1f}us "
          f"{fl/t2/1e12:6.1f}TF")
