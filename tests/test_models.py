"""Model zoo tests: param parity with the reference README table and
forward/backward shape checks on CPU."""

import pytest
import torch

from distribuuuu_amd import models

# (arch, params in M from /root/reference/README.md:208-217 + torchvision refs)
PARAM_TABLE = [
    ("resnet18", 11.690),
    ("resnet50", 25.557),
    ("botnet50", 20.859),
    ("efficientnet_b0", 5.289),
    ("regnetx_160", 54.279),
    ("regnety_160", 83.590),
    ("regnety_320", 145.047),
    ("densenet121", 7.979),
]


@pytest.mark.parametrize("arch,expected_m", PARAM_TABLE)
def test_param_counts(arch, expected_m):
    m = models.build_model(arch)
    n = sum(p.numel() for p in m.parameters() if p.requires_grad) / 1e6
    assert n == pytest.approx(expected_m, abs=5e-3), f"{arch}: {n:.3f}M"


@pytest.mark.parametrize("arch", ["resnet18", "resnet50", "densenet121"])
def test_forward_backward_small(arch):
    m = models.build_model(arch, num_classes=10)
    x = torch.randn(2, 3, 64, 64)
    y = m(x)
    assert y.shape == (2, 10)
    y.sum().backward()
    grads = [p.grad for p in m.parameters() if p.requires_grad]
    assert all(g is not None for g in grads)


def test_botnet_forward_224():
    m = models.build_model("botnet50", num_classes=10)
    x = torch.randn(1, 3, 224, 224)
    y = m(x)
    assert y.shape == (1, 10)


def test_efficientnet_forward():
    m = models.build_model("efficientnet_b0", num_classes=10)
    y = m(torch.randn(2, 3, 64, 64))
    assert y.shape == (2, 10)


def test_regnet_forward():
    m = models.build_model("regnetx_160", num_classes=10)
    y = m(torch.randn(1, 3, 64, 64))
    assert y.shape == (1, 10)


def test_unknown_arch_raises():
    with pytest.raises(KeyError):
        models.build_model("not_an_arch")


def test_registry_has_all_reference_archs():
    """Every arch the reference can train (its own registry + the timm-path
    baselines) exists natively here."""
    for arch in ["resnet18", "resnet34", "resnet50", "resnet101", "resnet152",
                 "resnext50_32x4d", "resnext101_32x8d", "wide_resnet50_2",
                 "wide_resnet101_2", "densenet121", "densenet161",
                 "densenet169", "densenet201", "botnet50", "efficientnet_b0",
                 "regnetx_160", "regnety_160", "regnety_320"]:
        assert callable(getattr(models, arch))


def test_densenet_memory_efficient_matches():
    torch.manual_seed(0)
    from distribuuuu_amd.models.densenet import DenseNet

    a = DenseNet(16, (2, 2), 32, num_classes=10, memory_efficient=False)
    torch.manual_seed(0)
    b = DenseNet(16, (2, 2), 32, num_classes=10, memory_efficient=True)
    b.load_state_dict(a.state_dict())
    a.eval(), b.eval()
    x = torch.randn(2, 3, 64, 64)
    assert torch.allclose(a(x), b(x), atol=1e-6)


def test_build_model_unknown_arch_raises():
    import pytest as _pytest

    from distribuuuu_amd.models import build_model

    with _pytest.raises(KeyError):
        build_model("no_such_arch_anywhere")


def test_fork_cpu_passthrough():
    """fork() must be a no-op off-GPU (the accumulate fusion is a GPU
    epilogue feature); the two handles alias the input."""
    import torch

    from distribuuuu_amd.ops import functional as DF

    x = torch.randn(2, 4, requires_grad=True)
    a, b = DF.fork(x)
    assert a is x and b is x
