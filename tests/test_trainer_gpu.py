"""GPU end-to-end drive of trainer.train_epoch/validate — catches integration
breaks the kernel unit tests can't (e.g. round-1's 0-dim accuracy tensors
crashing the meter update at the first PRINT_FREQ iteration)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

from distribuuuu_amd import models, utils  # noqa: E402
from distribuuuu_amd.config import cfg  # noqa: E402
from distribuuuu_amd.ops import functional as DF  # noqa: E402


def _require_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")


class _TinyDeviceLoader:
    """Few fixed device batches; exposes len() like a DataLoader."""

    def __init__(self, n_batches, batch, classes, device):
        g = torch.Generator(device="cpu").manual_seed(0)
        self.batches = []
        for _ in range(n_batches):
            x = torch.randn(batch, 3, 64, 64, generator=g).to(
                device, torch.bfloat16).contiguous(
                memory_format=torch.channels_last)
            y = torch.randint(0, classes, (batch,), generator=g).to(device)
            self.batches.append((x, y))

    def __len__(self):
        return len(self.batches)

    def __iter__(self):
        return iter(self.batches)


def test_train_epoch_and_validate_gpu(tmp_path):
    _require_gpu()
    from distribuuuu_amd import trainer

    cfg.defrost()
    cfg.OUT_DIR = str(tmp_path)
    cfg.MODEL.NUM_CLASSES = 10
    cfg.TRAIN.PRINT_FREQ = 1  # exercise the metric/meter path at idx 0
    cfg.TRAIN.DTYPE = "bfloat16"
    cfg.TRAIN.CHANNELS_LAST = True

    device = torch.device("cuda:0")
    net = models.build_model("resnet18", num_classes=10)
    net = net.to(device).to(torch.bfloat16)
    for m in net.modules():
        if hasattr(m, "running_mean") and m.running_mean is not None:
            m.float()
    net = net.to(memory_format=torch.channels_last)
    opt = utils.construct_optimizer(net)
    loader = _TinyDeviceLoader(3, 8, 10, device)
    trainer.train_epoch(loader, net, DF.cross_entropy, opt, 0, device,
                        torch.bfloat16)
    top1, top5 = trainer.validate(loader, net, DF.cross_entropy, device,
                                  torch.bfloat16)
    assert 0.0 <= top1 <= 100.0
    assert 0.0 <= top5 <= 100.0
