"""Data pipeline (SURVEY.md K17 + C6).

Three tiers:

* ``DummyDataset`` — behavior parity with the reference's RAM-resident random
  dataset (`/root/reference/distribuuuu/utils.py:109-118`), selected by
  cfg.MODEL.DUMMY_INPUT.
* ``DeviceSyntheticLoader`` — the MI355X-native benchmark feed: batches are
  generated directly in HBM (no H2D copy at all) on a side stream; this is what
  bench.py uses for the synthetic-data metric (BASELINE.json).
* ``ImageFolderDataset`` — real-data path: PIL-decoded class-per-directory
  layout with RandomResizedCrop/Flip/Normalize (train) or Resize/CenterCrop
  (val), matching the reference's torchvision transforms
  (utils.py:121-184) without the torchvision dependency.

All loaders shard with torch's DistributedSampler (deterministic per-epoch
permutation via set_epoch — reference trainer.py:33).
"""

import os
import random

import numpy as np
import torch
from torch.utils.data import DataLoader, Dataset
from torch.utils.data.distributed import DistributedSampler

from .config import cfg

IMAGENET_MEAN = (0.485, 0.456, 0.406)
IMAGENET_STD = (0.229, 0.224, 0.225)


class DummyDataset(Dataset):
    """Random tensors held in RAM, label always 0 (reference utils.py:109-118)."""

    def __init__(self, size=(3, 224, 224), length=1000):
        self.data = torch.randn([length] + list(size))
        self.length = length

    def __len__(self):
        return self.length

    def __getitem__(self, idx):
        return self.data[idx], 0


class ImageFolderDataset(Dataset):
    """class-per-subdirectory image dataset (torchvision.ImageFolder layout)."""

    EXTS = (".jpg", ".jpeg", ".png", ".bmp", ".webp")

    def __init__(self, root, train=True, im_size=224, resize=256):
        from PIL import Image  # noqa: F401  (validated import)

        self.root = root
        self.train = train
        self.im_size = im_size
        self.resize = resize
        classes = sorted(
            d for d in os.listdir(root) if os.path.isdir(os.path.join(root, d))
        )
        self.class_to_idx = {c: i for i, c in enumerate(classes)}
        self.samples = []
        for c in classes:
            cdir = os.path.join(root, c)
            for fname in sorted(os.listdir(cdir)):
                if fname.lower().endswith(self.EXTS):
                    self.samples.append((os.path.join(cdir, fname),
                                         self.class_to_idx[c]))

    def __len__(self):
        return len(self.samples)

    def _random_resized_crop(self, img):
        from PIL import Image

        w, h = img.size
        area = w * h
        for _ in range(10):
            target_area = random.uniform(0.08, 1.0) * area
            ar = np.exp(random.uniform(np.log(3 / 4), np.log(4 / 3)))
            cw = int(round(np.sqrt(target_area * ar)))
            ch = int(round(np.sqrt(target_area / ar)))
            if cw <= w and ch <= h:
                x0 = random.randint(0, w - cw)
                y0 = random.randint(0, h - ch)
                img = img.crop((x0, y0, x0 + cw, y0 + ch))
                return img.resize((self.im_size, self.im_size), Image.BILINEAR)
        # fallback: center crop
        s = min(w, h)
        img = img.crop(((w - s) // 2, (h - s) // 2, (w + s) // 2, (h + s) // 2))
        return img.resize((self.im_size, self.im_size), Image.BILINEAR)

    def _center_crop(self, img):
        from PIL import Image

        w, h = img.size
        scale = self.resize / min(w, h)
        img = img.resize((int(round(w * scale)), int(round(h * scale))),
                         Image.BILINEAR)
        w, h = img.size
        left = (w - self.im_size) // 2
        top = (h - self.im_size) // 2
        return img.crop((left, top, left + self.im_size, top + self.im_size))

    def __getitem__(self, idx):
        from PIL import Image

        path, label = self.samples[idx]
        with Image.open(path) as img:
            img = img.convert("RGB")
            if self.train:
                img = self._random_resized_crop(img)
                if random.random() < 0.5:
                    img = img.transpose(Image.FLIP_LEFT_RIGHT)
            else:
                img = self._center_crop(img)
        arr = np.asarray(img, dtype=np.float32) / 255.0      # HWC
        arr = (arr - IMAGENET_MEAN) / IMAGENET_STD
        return torch.from_numpy(arr.transpose(2, 0, 1).copy()).float(), label


class RawImageDataset(ImageFolderDataset):
    """Decode-only dataset: returns (uint8 HWC ndarray, label). Crop/flip/
    normalize run on the GPU (K17) — CPU workers only decode."""

    def __getitem__(self, idx):
        from PIL import Image

        path, label = self.samples[idx]
        with Image.open(path) as img:
            arr = np.asarray(img.convert("RGB"), dtype=np.uint8)
        return arr, label


def _draw_crop(h, w, train, im_size, resize):
    """RandomResizedCrop parameters (train) or Resize+CenterCrop box (val),
    matching the reference's torchvision transforms."""
    if train:
        area = h * w
        for _ in range(10):
            target_area = random.uniform(0.08, 1.0) * area
            ar = np.exp(random.uniform(np.log(3 / 4), np.log(4 / 3)))
            cw = int(round(np.sqrt(target_area * ar)))
            ch = int(round(np.sqrt(target_area / ar)))
            if cw <= w and ch <= h:
                return (random.randint(0, h - ch), random.randint(0, w - cw),
                        ch, cw, int(random.random() < 0.5))
        s = min(h, w)
        return ((h - s) // 2, (w - s) // 2, s, s, 0)
    # val: shorter side -> resize, center crop im_size: equivalent crop box
    s = min(h, w)
    box = int(round(s * im_size / resize))
    return ((h - box) // 2, (w - box) // 2, box, box, 0)


class RawBatchCollate:
    """Concatenate decoded uint8 images + per-image crop metadata for the GPU
    augmentation kernel."""

    def __init__(self, train, im_size, resize=256):
        self.train = train
        self.im_size = im_size
        self.resize = resize

    def __call__(self, batch):
        metas = []
        offset = 0
        bufs = []
        labels = []
        for arr, label in batch:
            h, w = arr.shape[0], arr.shape[1]
            cy, cx, ch, cw, flip = _draw_crop(h, w, self.train, self.im_size,
                                              self.resize)
            metas.append([offset, h, w, cy, cx, ch, cw, flip])
            bufs.append(arr.reshape(-1))
            offset += arr.size
            labels.append(label)
        raw = torch.from_numpy(np.concatenate(bufs))
        meta = torch.tensor(metas, dtype=torch.int32)
        return raw, meta, torch.tensor(labels, dtype=torch.long)


class GPUAugLoader:
    """Wraps a raw-byte DataLoader: pinned hipMemcpyAsync H2D on a side
    stream + the fused crop/flip/normalize kernel, one batch prefetched ahead
    of compute (SURVEY.md K17 / BASELINE.json data pipeline)."""

    def __init__(self, loader, im_size, dtype, device="cuda"):
        self.loader = loader
        self.im_size = im_size
        self.dtype = dtype
        self.device = device
        self.stream = torch.cuda.Stream(device)
        self.sampler = loader.sampler

    def __len__(self):
        return len(self.loader)

    def _issue(self, cpu_batch):
        from .ops.dispatch import require_ext

        raw, meta, labels = cpu_batch
        with torch.cuda.stream(self.stream):
            raw_d = raw.to(self.device, non_blocking=True)
            meta_d = meta.to(self.device, non_blocking=True)
            labels_d = labels.to(self.device, non_blocking=True)
            x = require_ext().aug_crop_flip_norm(
                raw_d, meta_d, self.im_size, list(IMAGENET_MEAN),
                list(IMAGENET_STD), self.dtype)
        ev = torch.cuda.Event()
        ev.record(self.stream)
        return x, labels_d, ev

    @staticmethod
    def _hand_off(pending):
        x, y, ev = pending
        cur = torch.cuda.current_stream()
        cur.wait_event(ev)
        # the consumer stream now owns these side-stream allocations: without
        # record_stream the caching allocator could recycle the block for a
        # later side-stream aug batch while main-stream backward still reads
        # the saved inputs (wgrad) — intermittent silent batch corruption
        x.record_stream(cur)
        y.record_stream(cur)
        return x, y

    def __iter__(self):
        it = iter(self.loader)
        pending = None
        for cpu_batch in it:
            issued = self._issue(cpu_batch)
            if pending is not None:
                yield self._hand_off(pending)
            pending = issued
        if pending is not None:
            yield self._hand_off(pending)


def construct_train_loader():
    """Per-rank train loader (reference utils.py:121-152): per-rank batch size,
    DistributedSampler(shuffle=True), drop_last=True, pinned memory. On a GPU
    with the HIP extension, augmentation runs on-device (GPUAugLoader)."""
    if cfg.MODEL.DUMMY_INPUT:
        ds = DummyDataset(size=[3, cfg.TRAIN.IM_SIZE, cfg.TRAIN.IM_SIZE])
    else:
        root = os.path.join(cfg.TRAIN.DATASET, cfg.TRAIN.SPLIT)
        if _gpu_aug_available():
            ds = RawImageDataset(root, train=True, im_size=cfg.TRAIN.IM_SIZE)
            loader = DataLoader(
                ds,
                batch_size=cfg.TRAIN.BATCH_SIZE,
                shuffle=not _dist(),
                sampler=DistributedSampler(ds, shuffle=True) if _dist() else None,
                num_workers=cfg.TRAIN.WORKERS,
                pin_memory=cfg.TRAIN.PIN_MEMORY,
                drop_last=True,
                collate_fn=RawBatchCollate(True, cfg.TRAIN.IM_SIZE),
            )
            dtype = (torch.bfloat16 if cfg.TRAIN.DTYPE == "bfloat16"
                     else torch.float32)
            return GPUAugLoader(loader, cfg.TRAIN.IM_SIZE, dtype)
        ds = ImageFolderDataset(root, train=True, im_size=cfg.TRAIN.IM_SIZE)
    sampler = DistributedSampler(ds, shuffle=True) if _dist() else None
    return DataLoader(
        ds,
        batch_size=cfg.TRAIN.BATCH_SIZE,
        shuffle=(sampler is None),
        sampler=sampler,
        num_workers=cfg.TRAIN.WORKERS,
        pin_memory=cfg.TRAIN.PIN_MEMORY,
        drop_last=True,
    )


def _gpu_aug_available():
    if not torch.cuda.is_available():
        return False
    from .ops.dispatch import hip_op_available

    return hip_op_available("aug_crop_flip_norm")


def construct_val_loader():
    """Per-rank val loader (reference utils.py:155-184): no shuffle, keep last."""
    if cfg.MODEL.DUMMY_INPUT:
        ds = DummyDataset(size=[3, cfg.TRAIN.IM_SIZE, cfg.TRAIN.IM_SIZE])
    else:
        root = os.path.join(cfg.TEST.DATASET, cfg.TEST.SPLIT)
        ds = ImageFolderDataset(root, train=False, im_size=cfg.TRAIN.IM_SIZE,
                                resize=cfg.TEST.IM_SIZE)
    sampler = DistributedSampler(ds, shuffle=False) if _dist() else None
    return DataLoader(
        ds,
        batch_size=cfg.TEST.BATCH_SIZE,
        shuffle=False,
        sampler=sampler,
        num_workers=cfg.TRAIN.WORKERS,
        pin_memory=cfg.TRAIN.PIN_MEMORY,
        drop_last=False,
    )


def _dist():
    import torch.distributed as dist

    return dist.is_initialized() and dist.get_world_size() > 1


class DeviceSyntheticLoader:
    """Device-resident synthetic batches for benchmarking: one fixed batch is
    created in HBM at init (288 GB per GPU — keep it resident)
    and yielded every step; zero H2D traffic, zero CPU worker cost. The
    compute per step is identical to a fresh batch (bench.py reports
    ``data: synthetic``). Yields (inputs, targets) like a DataLoader."""

    def __init__(self, batch_size, im_size=224, num_classes=1000, length=10 ** 9,
                 device="cuda", dtype=torch.float32, channels_last=False):
        self.length = length
        g = torch.Generator(device="cpu").manual_seed(1234)
        x = torch.randn(batch_size, 3, im_size, im_size, generator=g)
        self.x = x.to(device=device, dtype=dtype)
        if channels_last:
            self.x = self.x.contiguous(memory_format=torch.channels_last)
        self.y = torch.randint(0, num_classes, (batch_size,), generator=g).to(device)

    def __len__(self):
        return self.length

    def __iter__(self):
        for _ in range(self.length):
            yield self.x, self.y
