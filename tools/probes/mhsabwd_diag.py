import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))
import torch
from distribuuuu_amd.ops import attention as A

torch.manual_seed(0)
n, heads, h, w, d = 2, 4, 14, 14, 32
l = h * w
mk = lambda *s: torch.randn(*s, device="cuda", dtype=torch.bfloat16, requires_grad=True)
q, k, v = mk(n, heads, l, d), mk(n, heads, l, d), mk(n, heads, l, d)
rel_h, rel_w = mk(2*h-1, d), mk(2*w-1, d)
out = A._HIPMHSARelPos.apply(q, k, v, rel_h, rel_w, h, w)
g = torch.randn_like(out)
out.backward(g)

refs = [t.detach().float().requires_grad_(True) for t in (q, k, v, rel_h, rel_w)]
ref = A._torch_mhsa(*refs, h, w)
ref.backward(g.float())

# old recompute path in bf16 for comparison
olds = [t.detach().clone().requires_grad_(True) for t in (q, k, v, rel_h, rel_w)]
out2 = A._torch_mhsa(*olds, h, w)
out2.backward(g)

names = ["q","k","v","rel_h","rel_w"]
for nm, got, want, old in zip(names, (q,k,v,rel_h,rel_w), refs, olds):
    e_new = (got.grad.float() - want.grad).abs()
    e_old = (old.grad.float() - want.grad).abs()
    rel = e_new / (want.grad.abs() + 1e-3)
    print(f"{nm:6s} new: max {e_new.max():8.4f} mean {e_new.mean():8.5f} | "
          f"old-bf16: max {e_old.max():8.4f} mean {e_old.mean():8.5f} | "
          f"grad max {want.grad.abs().max():8.3f} relmax {rel.max():.4f}")
