import os, sys, torch
sys.path.insert(0, "/root/repo")
from distribuuuu_amd import models
from distribuuuu_amd.ops import functional as DF
from distribuuuu_amd.ops.optim import HIPSGD


def run(tag):
    torch.manual_seed(0)
    m = models.build_model("efficientnet_b0", num_classes=10).to("cuda").to(torch.bfloat16)
    for mod in m.modules():
        if hasattr(mod, "running_mean"):
            mod.float()
    m = m.to(memory_format=torch.channels_last)
    opt = HIPSGD(m.parameters(), lr=0.01, momentum=0.9, weight_decay=5e-5, nesterov=True)
    x = torch.randn(4, 3, 64, 64, device="cuda", dtype=torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)
    t = torch.randint(0, 10, (4,), device="cuda")
    losses = []
    for i in range(3):
        out = m(x)
        fin = torch.isfinite(out.float()).all().item()
        loss = DF.cross_entropy(out.float(), t)
        opt.zero_grad(set_to_none=False)
        loss.backward()
        badg = sum(0 if torch.isfinite(p.grad.float()).all() else 1
                   for p in m.parameters() if p.grad is not None)
        opt.step()
        losses.append((round(loss.item(), 4), fin, badg))
    print(tag, losses)


run("default(v2 on)")
