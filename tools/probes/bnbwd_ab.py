"""A/B probe: bn_bwd_stats (stage-1 reduce) bandwidth across the ResNet-50
BN shape mix, sweeping the stage-1 grid cap (DISTRIBUUUU_BN_S1GRID).

Context: after the pointer-increment rewrite, the profile still shows
bn_bwd_reduce at ~3.2 TB/s (res variant) / ~1.3 TB/s (non-res small-C
layers) vs ~7 TB/s for an ATen copy. This measures each shape in isolation
to find where the cap is (grid size? C-dependent mapping? stream count?).
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch

from distribuuuu_amd.ops.dispatch import ext

e = ext()
cl = torch.channels_last


def bench(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


# (N, C, H, W, has_res) — the distinct ResNet-50 BN shapes at batch 256
SHAPES = [
    (256, 64, 112, 112, 0),
    (256, 64, 56, 56, 0),
    (256, 256, 56, 56, 1),
    (256, 128, 28, 28, 0),
    (256, 512, 28, 28, 1),
    (256, 256, 14, 14, 0),
    (256, 1024, 14, 14, 1),
    (256, 512, 7, 7, 0),
    (256, 2048, 7, 7, 1),
]


def main():
    for grid in (512, 1024, 2048, 4096):
        os.environ["DISTRIBUUUU_BN_S1GRID"] = str(grid)
        print(f"--- stage-1 grid cap {grid}")
        for (n, c, h, w, res) in SHAPES:
            x = torch.randn(n, c, h, w, dtype=torch.bfloat16,
                            device="cuda").to(memory_format=cl)
            gy = torch.randn_like(x)
            y = torch.relu(torch.randn_like(x))
            scale = torch.rand(c, device="cuda")
            shift = torch.rand(c, device="cuda")
            r = torch.randn_like(x) if res else None
            dt = bench(lambda: e.bn_bwd_stats(gy, x, r, scale, shift, 1))
            gb = 3 * x.numel() * 2 / 1e9  # act=relu reads gy, x, y
            print(f"  C={c:4d} {h:3d}x{w:<3d} res={res}: {dt*1e6:7.1f}us "
                  f"{gb/dt/1e9*1e9:6.0f} GB/s")
        del x, gy, y, r
        torch.cuda.empty_cache()
    a = torch.randn(256, 256, 56, 56, dtype=torch.bfloat16,
                    device="cuda").to(memory_format=cl)
    b = torch.randn_like(a)
    dt = bench(lambda: torch.add(a, b))
    print(f"torch.add 2R1W : {dt*1e6:7.1f}us {3*a.numel()*2/1e9/dt:6.0f} GB/s")
    dt = bench(lambda: a.sum(dtype=torch.float32))
    print(f"torch.sum 1R   : {dt*1e6:7.1f}us {a.numel()*2/1e9/dt:6.0f} GB/s")


if __name__ == "__main__":
    main()
