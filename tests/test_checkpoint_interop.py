"""Checkpoint interoperability: the model zoo's state_dict layout must match
the torchvision layout the reference uses (reference models/resnet.py,
models/densenet.py are torchvision copies), so reference/torchvision
checkpoints load strict and our checkpoints load there."""

import torch

from distribuuuu_amd import models
from distribuuuu_amd.models.densenet import remap_legacy_densenet_keys


def _torchvision_resnet_keys(layers, block_expansion, basic):
    """Independently generate the torchvision ResNet state_dict key list from
    the documented naming scheme (conv1/bn1/layerL.B.convN/bnN/downsample)."""
    def bn(prefix):
        return [f"{prefix}.{s}" for s in
                ("weight", "bias", "running_mean", "running_var",
                 "num_batches_tracked")]

    keys = ["conv1.weight"] + bn("bn1")
    inplanes = 64
    for li, nblocks in enumerate(layers, start=1):
        planes = 64 * 2 ** (li - 1)
        for b in range(nblocks):
            p = f"layer{li}.{b}"
            nconv = 2 if basic else 3
            for c in range(1, nconv + 1):
                keys.append(f"{p}.conv{c}.weight")
                keys += bn(f"{p}.bn{c}")
            stride = 1 if li == 1 else 2
            if b == 0 and (stride != 1 or inplanes != planes * block_expansion):
                keys.append(f"{p}.downsample.0.weight")
                keys += bn(f"{p}.downsample.1")
        inplanes = planes * block_expansion
    keys += ["fc.weight", "fc.bias"]
    return keys


def test_resnet18_state_dict_layout():
    net = models.build_model("resnet18", num_classes=1000)
    expected = _torchvision_resnet_keys([2, 2, 2, 2], 1, basic=True)
    assert list(net.state_dict().keys()) == expected


def test_resnet50_state_dict_layout():
    net = models.build_model("resnet50", num_classes=1000)
    expected = _torchvision_resnet_keys([3, 4, 6, 3], 4, basic=False)
    assert list(net.state_dict().keys()) == expected


def test_torchvision_checkpoint_roundtrip(tmp_path):
    """A checkpoint written with these key names loads strict into a freshly
    built model (what loading a downloaded torchvision file does)."""
    src = models.build_model("resnet18", num_classes=10)
    path = tmp_path / "tv.pth"
    torch.save(src.state_dict(), path)
    dst = models.build_model("resnet18", num_classes=10)
    sd = torch.load(path, map_location="cpu", weights_only=True)
    dst.load_state_dict(sd)  # strict
    for k, v in dst.state_dict().items():
        assert torch.equal(v, src.state_dict()[k])


def test_densenet_legacy_key_remap():
    sd = {
        "features.denseblock1.denselayer1.norm.1.weight": torch.ones(1),
        "features.denseblock1.denselayer1.conv.2.weight": torch.ones(1),
        "features.norm5.weight": torch.ones(1),  # untouched
    }
    out = remap_legacy_densenet_keys(dict(sd))
    assert "features.denseblock1.denselayer1.norm1.weight" in out
    assert "features.denseblock1.denselayer1.conv2.weight" in out
    assert "features.norm5.weight" in out
    assert "features.denseblock1.denselayer1.norm.1.weight" not in out


def test_densenet121_accepts_remapped_legacy_checkpoint():
    """Forge a legacy-dotted copy of a real state dict (legacy files dot only
    weight/bias/running stats, not num_batches_tracked), remap, strict-load."""
    net = models.build_model("densenet121", num_classes=1000)
    legacy = {}
    for k, v in net.state_dict().items():
        lk = k
        if ".denselayer" in k and not k.endswith("num_batches_tracked"):
            for tag in ("norm1", "norm2", "conv1", "conv2"):
                if f".{tag}." in lk:
                    lk = lk.replace(f".{tag}.", f".{tag[:-1]}.{tag[-1]}.")
                    break
        legacy[lk] = v
    assert any(".norm.1." in k for k in legacy)  # the forge did something
    net2 = models.build_model("densenet121", num_classes=1000)
    net2.load_state_dict(remap_legacy_densenet_keys(legacy))
