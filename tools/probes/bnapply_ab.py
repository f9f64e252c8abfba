import sys, time, torch
sys.path.insert(0, "/root/repo")
from distribuuuu_amd.ops.dispatch import require_ext

e = require_ext()
cl = lambda t: t.contiguous(memory_format=torch.channels_last)


def timeit(fn, iters=30, warmup=8):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


for (n, c, h) in [(256, 64, 79), (256, 128, 56), (256, 256, 40),
                  (256, 512, 28), (256, 1024, 20), (256, 2048, 14)]:
    x = cl(torch.randn(n, c, h, h, device="cuda", dtype=torch.bfloat16))
    res = cl(torch.randn_like(x))
    scale = torch.randn(c, device="cuda")
    shift = torch.randn(c, device="cuda")
    nb = x.numel() * 2
    t0 = timeit(lambda: e.bn_apply_act(x, scale, shift, 1, None))
    t1 = timeit(lambda: e.bn_apply_act(x, scale, shift, 1, res))
    print(f"C={c:5d} HW={h}x{h}: nores {t0*1e6:7.1f}us {2*nb/t0/1e9:6.0f} GB/s"
          f" | res {t1*1e6:7.1f}us {3*nb/t1/1e9:6.0f} GB/s")
a = torch.randn(51_000_000, device="cuda", dtype=torch.bfloat16)
b = torch.empty_like(a)
t = timeit(lambda: b.copy_(a))
print(f"ATen copy 102MB: {t*1e6:.1f}us {2*a.numel()*2/t/1e9:.0f} GB/s")
