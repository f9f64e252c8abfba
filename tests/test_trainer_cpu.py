"""Single-process CPU integration: BASELINE.json config #1
(ResNet-18 snsc-style on synthetic data, plumbing only)."""

import torch

from distribuuuu_amd import models, utils
from distribuuuu_amd.config import cfg
from distribuuuu_amd.ops import functional as DF


def test_synthetic_train_loss_decreases(tmp_path):
    cfg.OUT_DIR = str(tmp_path)
    cfg.RNG_SEED = 7
    utils.setup_seed(0)
    net = models.build_model("resnet18", num_classes=10)
    opt = utils.construct_optimizer(net)
    utils.set_lr(opt, 0.05)
    net.train()
    x = torch.randn(16, 3, 64, 64)
    y = torch.randint(0, 10, (16,))
    losses = []
    for _ in range(8):
        out = net(x)
        loss = DF.cross_entropy(out, y)
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert all(torch.isfinite(torch.tensor(losses)))
    # should overfit the fixed batch
    assert losses[-1] < losses[0]


def test_train_epoch_runs(tmp_path):
    """Drive trainer.train_epoch end-to-end with the dummy dataset."""
    from distribuuuu_amd import trainer
    from distribuuuu_amd.data import construct_train_loader

    cfg.OUT_DIR = str(tmp_path)
    cfg.MODEL.DUMMY_INPUT = True
    cfg.MODEL.NUM_CLASSES = 10
    cfg.TRAIN.BATCH_SIZE = 4
    cfg.TRAIN.IM_SIZE = 32
    cfg.TRAIN.WORKERS = 0
    cfg.TRAIN.PRINT_FREQ = 2
    # shrink the dummy dataset via monkeypatched length
    import distribuuuu_amd.data as data_mod

    orig = data_mod.DummyDataset
    data_mod.DummyDataset = lambda size=(3, 32, 32), length=1000: orig(size, 8)
    try:
        loader = construct_train_loader()
        net = models.build_model("resnet18", num_classes=10)
        opt = utils.construct_optimizer(net)
        trainer.train_epoch(loader, net, DF.cross_entropy, opt, 0,
                            torch.device("cpu"), torch.float32)
    finally:
        data_mod.DummyDataset = orig


def test_validate_runs(tmp_path):
    from distribuuuu_amd import trainer
    from distribuuuu_amd.data import construct_val_loader
    import distribuuuu_amd.data as data_mod

    cfg.OUT_DIR = str(tmp_path)
    cfg.MODEL.DUMMY_INPUT = True
    cfg.MODEL.NUM_CLASSES = 10
    cfg.TRAIN.IM_SIZE = 32
    cfg.TEST.BATCH_SIZE = 4
    cfg.TRAIN.WORKERS = 0
    orig = data_mod.DummyDataset
    data_mod.DummyDataset = lambda size=(3, 32, 32), length=1000: orig(size, 8)
    try:
        loader = construct_val_loader()
        net = models.build_model("resnet18", num_classes=10)
        top1, top5 = trainer.validate(loader, net, DF.cross_entropy,
                                      torch.device("cpu"), torch.float32)
        assert 0.0 <= top1 <= 100.0
    finally:
        data_mod.DummyDataset = orig


def test_resume_equivalence(tmp_path):
    """train-save-reload-continue must land on the same
    weights as an uninterrupted run (checkpoint carries model + optimizer
    momentum; LR is a pure function of the epoch)."""
    cfg.OUT_DIR = str(tmp_path)
    torch.manual_seed(3)
    x = torch.randn(8, 3, 32, 32)
    y = torch.randint(0, 10, (8,))

    def make():
        torch.manual_seed(11)
        net = models.build_model("resnet18", num_classes=10)
        opt = utils.construct_optimizer(net)
        utils.set_lr(opt, 0.05)
        return net, opt

    def step(net, opt):
        net.train()
        loss = DF.cross_entropy(net(x), y)
        opt.zero_grad()
        loss.backward()
        opt.step()

    # uninterrupted: 4 steps
    net_a, opt_a = make()
    for _ in range(4):
        step(net_a, opt_a)

    # interrupted: 2 steps, checkpoint, fresh objects, resume, 2 steps
    net_b, opt_b = make()
    for _ in range(2):
        step(net_b, opt_b)
    path = utils.save_checkpoint(net_b, opt_b, epoch=1, best_acc1=0.0)
    net_c, opt_c = make()
    start_epoch, _ = utils.load_checkpoint(path, net_c, opt_c)
    assert start_epoch == 2
    for _ in range(2):
        step(net_c, opt_c)

    for pa, pc in zip(net_a.parameters(), net_c.parameters()):
        assert torch.allclose(pa, pc, atol=1e-6), "resume diverged"
