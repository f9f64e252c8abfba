"""Config system parity tests (reference config.py surface)."""

import glob
import os

import pytest

from distribuuuu_amd.config import cfg, load_cfg_fom_args, merge_from_file, reset_cfg

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_defaults():
    assert cfg.MODEL.ARCH == "resnet18"
    assert cfg.TRAIN.BATCH_SIZE == 32
    assert cfg.OPTIM.LR_POLICY == "cos"
    assert cfg.RNG_SEED is None


@pytest.mark.parametrize("preset", sorted(glob.glob(os.path.join(REPO, "config", "*.yaml"))))
def test_presets_load(preset):
    merge_from_file(preset)
    assert cfg.MODEL.ARCH in os.path.basename(preset)


def test_reference_presets_load_unmodified():
    """The reference repo's own YAML presets must merge cleanly."""
    ref = "/root/reference/config"
    if not os.path.isdir(ref):
        pytest.skip("reference not mounted")
    for preset in sorted(glob.glob(os.path.join(ref, "*.yaml"))):
        reset_cfg()
        merge_from_file(preset)


def test_cli_overrides():
    load_cfg_fom_args(argv=["OPTIM.BASE_LR", "0.4", "TRAIN.BATCH_SIZE", "64"])
    assert cfg.OPTIM.BASE_LR == 0.4
    assert cfg.TRAIN.BATCH_SIZE == 64


def test_freeze_blocks_writes():
    cfg.freeze()
    with pytest.raises(AttributeError):
        cfg.MODEL.ARCH = "resnet50"
    cfg.defrost()


def test_unknown_key_rejected():
    with pytest.raises(KeyError):
        cfg.merge_from_list(["MODEL.NOPE", "1"])


def test_dump_and_reload(tmp_path):
    cfg.OUT_DIR = str(tmp_path)
    cfg.MODEL.ARCH = "resnet50"
    from distribuuuu_amd.config import dump_cfg

    dump_cfg()
    reset_cfg()
    assert cfg.MODEL.ARCH == "resnet18"
    merge_from_file(str(tmp_path / "config.yaml"))
    assert cfg.MODEL.ARCH == "resnet50"


def test_type_coercion():
    cfg.merge_from_list(["OPTIM.WEIGHT_DECAY", "1e-5"])
    assert cfg.OPTIM.WEIGHT_DECAY == pytest.approx(1e-5)
    with pytest.raises(TypeError):
        cfg.merge_from_list(["TRAIN.BATCH_SIZE", "hello"])
