import time, torch
def t(fn, it=30):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(it): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/it
BH, L, D = 512, 196, 32
P = torch.randn(BH, L, L, dtype=torch.bfloat16, device="cuda")
dO = torch.randn(BH, L, D, dtype=torch.bfloat16, device="cuda")
V = torch.randn(BH, L, D, dtype=torch.bfloat16, device="cuda")
print(f"dV = P^T dO  : {t(lambda: torch.bmm(P.transpose(1,2), dO))*1e6:7.1f}us")
print(f"dP = dO V^T  : {t(lambda: torch.bmm(dO, V.transpose(1,2)))*1e6:7.1f}us")
print(f"dQ = S K     : {t(lambda: torch.bmm(P, V))*1e6:7.1f}us")
Pf = P.float()
print(f"rowsum(dP*P) : {t(lambda: (P*P).sum(-1))*1e6:7.1f}us")
q = torch.randn(BH*L, D, dtype=torch.bfloat16, device="cuda")
rw = torch.randn(27, D, dtype=torch.bfloat16, device="cuda")
print(f"RW gemm      : {t(lambda: q @ rw.t())*1e6:7.1f}us")
IW = torch.randint(0, 27, (1, L, L), device="cuda").expand(BH, L, L)
dRW = torch.zeros(BH, L, 27, dtype=torch.float32, device="cuda")
dSf = torch.randn(BH, L, L, dtype=torch.float32, device="cuda")
print(f"scatter_add  : {t(lambda: dRW.scatter_add(2, IW, dSf))*1e6:7.1f}us")
print(f"gather       : {t(lambda: torch.gather(dSf, 2, IW))*1e6:7.1f}us")
