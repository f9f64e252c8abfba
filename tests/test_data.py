"""Data pipeline host logic: crop parameter semantics (torchvision
RandomResizedCrop / Resize+CenterCrop parity), dummy dataset, synthetic
loader, and the raw-batch collate layout the GPU augmentation kernel
consumes."""

import random

import numpy as np
import torch

from distribuuuu_amd import data as D


def test_draw_crop_train_bounds_and_distribution():
    random.seed(0)
    np.random.seed(0)
    h, w = 300, 400
    areas, flips = [], 0
    for _ in range(500):
        cy, cx, ch, cw, flip = D._draw_crop(h, w, True, 224, 256)
        assert 0 <= cy and cy + ch <= h
        assert 0 <= cx and cx + cw <= w
        assert ch > 0 and cw > 0
        areas.append(ch * cw / (h * w))
        flips += flip
    # RandomResizedCrop scale range (0.08, 1.0); allow fallback square crops
    assert min(areas) >= 0.05
    assert max(areas) <= 1.0
    assert 150 < flips < 350  # ~50% horizontal flips


def test_draw_crop_val_center():
    # val: shorter side resized to 256 then 224 center crop — the source box
    # is (s * 224/256) centered
    cy, cx, ch, cw, flip = D._draw_crop(512, 768, False, 224, 256)
    assert flip == 0
    assert ch == cw == round(512 * 224 / 256)
    assert cy == (512 - ch) // 2 and cx == (768 - cw) // 2


def test_draw_crop_train_tiny_image_fallback():
    random.seed(1)
    cy, cx, ch, cw, flip = D._draw_crop(8, 8, True, 224, 256)
    assert ch <= 8 and cw <= 8 and cy + ch <= 8 and cx + cw <= 8


def test_dummy_dataset_reference_semantics():
    ds = D.DummyDataset(size=(3, 32, 32), length=17)
    assert len(ds) == 17
    x, y = ds[3]
    assert x.shape == (3, 32, 32)
    assert y == 0


def test_raw_batch_collate_layout():
    random.seed(0)
    imgs = [(np.random.randint(0, 255, (40, 50, 3), dtype=np.uint8), 7),
            (np.random.randint(0, 255, (60, 30, 3), dtype=np.uint8), 2)]
    raw, meta, labels = D.RawBatchCollate(True, 224)(imgs)
    assert raw.numel() == 40 * 50 * 3 + 60 * 30 * 3
    assert meta.shape == (2, 8) and meta.dtype == torch.int32
    assert meta[0, 0] == 0 and meta[1, 0] == 40 * 50 * 3  # byte offsets
    assert (meta[:, 1] == torch.tensor([40, 60])).all()   # heights
    assert labels.tolist() == [7, 2]


def test_device_synthetic_loader_cpu():
    loader = D.DeviceSyntheticLoader(4, im_size=32, num_classes=10,
                                     device="cpu", dtype=torch.float32,
                                     channels_last=False)
    it = iter(loader)
    x, y = next(it)
    assert x.shape == (4, 3, 32, 32)
    assert y.shape == (4,) and y.dtype == torch.long
    assert int(y.max()) < 10
