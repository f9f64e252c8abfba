"""End-to-end GPU train-step checks: every baseline arch does one bf16 NHWC
forward+backward+fused-SGD step with finite loss and full grad coverage."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _one_step(arch, im=64, batch=4):
    from distribuuuu_amd import models, utils
    from distribuuuu_amd.ops import functional as DF
    from distribuuuu_amd.ops.optim import HIPSGD

    torch.manual_seed(0)
    m = models.build_model(arch, num_classes=10).to("cuda").to(torch.bfloat16)
    for mod in m.modules():
        if hasattr(mod, "running_mean"):
            mod.float()
    m = m.to(memory_format=torch.channels_last)
    opt = HIPSGD(m.parameters(), lr=0.01, momentum=0.9, weight_decay=5e-5,
                 nesterov=True)
    x = torch.randn(batch, 3, im, im, device="cuda", dtype=torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)
    t = torch.randint(0, 10, (batch,), device="cuda")
    losses = []
    for _ in range(3):
        out = m(x)
        loss = DF.cross_entropy(out.float(), t)
        opt.zero_grad(set_to_none=False)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    torch.cuda.synchronize()
    assert all(torch.isfinite(torch.tensor(losses))), losses
    return losses


@pytest.mark.parametrize("arch", ["resnet18", "resnet50", "regnetx_160",
                                  "efficientnet_b0", "densenet121"])
def test_train_step(arch):
    losses = _one_step(arch)
    # loss should move (params update)
    assert losses[0] != losses[-1]


def test_botnet_train_step_224():
    losses = _one_step("botnet50", im=224, batch=2)
    assert losses[0] != losses[-1]


def test_regnety_groupconv_step():
    _one_step("regnety_160")
