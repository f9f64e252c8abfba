"""Training / evaluation drivers.

Behavior parity with the reference trainer (`/root/reference/distribuuuu/trainer.py`):
``train_model`` (init dist -> seed -> logger -> build -> [SyncBN] -> device -> DDP ->
loaders -> loss/optimizer -> auto-resume -> epoch loop), ``train_epoch``/``validate``
hot loops with the same metric/all-reduce/print cadence, ``test_model``.

MI355X-first deltas:
* our own bucketed-RCCL DDP (parallel.ddp) and RCCL SyncBN (parallel.syncbn);
* optional bf16 compute (cfg.TRAIN.DTYPE) with fp32 master weights in the fused
  SGD step, and NHWC (cfg.TRAIN.CHANNELS_LAST) end-to-end;
* metric readout (.item() D2H sync) only at PRINT_FREQ boundaries instead of
  every iteration (the reference syncs each iter — SURVEY.md §3.2 note).
"""

import time

import torch

from . import models
from .config import cfg
from .data import construct_train_loader, construct_val_loader
from .logger import logger, setup_logger
from .ops import functional as DF
from .parallel import DistributedDataParallel, convert_sync_batchnorm
from . import utils


def _compute_dtype():
    return torch.bfloat16 if cfg.TRAIN.DTYPE == "bfloat16" else torch.float32


def _prepare_batch(inputs, targets, device, dtype):
    inputs = inputs.to(device, non_blocking=True)
    targets = targets.to(device, non_blocking=True)
    if dtype != torch.float32:
        inputs = inputs.to(dtype)
    if cfg.TRAIN.CHANNELS_LAST and inputs.dim() == 4:
        inputs = inputs.contiguous(memory_format=torch.channels_last)
    return inputs, targets


def build_network(device):
    """Registry build (reference trainer.py:117-128; no timm fallback — every
    baseline arch is native), optional SyncBN convert, device + layout moves."""
    net = models.build_model(cfg.MODEL.ARCH, num_classes=cfg.MODEL.NUM_CLASSES)
    if cfg.MODEL.SYNCBN:
        net = convert_sync_batchnorm(net)
    net = net.to(device)
    if _compute_dtype() == torch.bfloat16:
        net = net.to(torch.bfloat16)
        # keep BN params/buffers fp32 for stat accuracy
        for m in net.modules():
            if hasattr(m, "running_mean") and m.running_mean is not None:
                m.float()
    if cfg.TRAIN.CHANNELS_LAST:
        net = net.to(memory_format=torch.channels_last)
    return net


def train_epoch(train_loader, net, criterion, optimizer, epoch, device, dtype):
    """One epoch of the hot loop (reference trainer.py:14-64)."""
    topk = cfg.TRAIN.TOPK
    (batch_time, data_time, losses, top1, topk_m, ips,
     progress) = utils.construct_meters(
        len(train_loader), f"Epoch[{epoch + 1}/{cfg.OPTIM.MAX_EPOCH}]", topk,
        batch_size=cfg.TRAIN.BATCH_SIZE * utils.get_world_size())
    lr = utils.get_epoch_lr(epoch)
    utils.set_lr(optimizer, lr)
    if utils.get_rank() == 0:
        logger.info(f"Epoch {epoch + 1}, LR {lr:.6f}")
    if hasattr(train_loader, "sampler") and hasattr(train_loader.sampler,
                                                    "set_epoch"):
        train_loader.sampler.set_epoch(epoch)
    net.train()
    # world==1: set_to_none skips the per-step grad memsets (161 for ResNet-50);
    # under DDP the bucket grad-views must stay installed, so keep memsets there
    zero_none = utils.get_world_size() == 1
    end = time.time()
    for idx, (inputs, targets) in enumerate(train_loader):
        data_time.update(time.time() - end)
        inputs, targets = _prepare_batch(inputs, targets, device, dtype)
        outputs = net(inputs)
        loss = criterion(outputs.float(), targets)
        optimizer.zero_grad(set_to_none=zero_none)
        loss.backward()
        optimizer.step()
        acc1, acck = utils.accuracy(outputs, targets, topk=(1, topk))
        if (idx % cfg.TRAIN.PRINT_FREQ == 0) or (idx + 1 == len(train_loader)):
            metrics = [loss.detach(), acc1[0], acck[0]]
            utils.scaled_all_reduce(metrics)
            losses.update(metrics[0].item(), inputs.size(0))
            top1.update(metrics[1].item(), inputs.size(0))
            topk_m.update(metrics[2].item(), inputs.size(0))
        step_t = time.time() - end
        batch_time.update(step_t)
        ips.update(cfg.TRAIN.BATCH_SIZE * utils.get_world_size()
                   / max(step_t, 1e-9))
        end = time.time()
        if idx % cfg.TRAIN.PRINT_FREQ == 0 and utils.get_rank() == 0:
            progress.display(idx)


@torch.no_grad()
def validate(val_loader, net, criterion, device, dtype):
    """Validation loop (reference trainer.py:67-103); returns (top1, topk)."""
    topk = cfg.TRAIN.TOPK
    batch_time, data_time, losses, top1, topk_m, progress = utils.construct_meters(
        len(val_loader), "Test: ", topk)
    net.eval()
    end = time.time()
    for idx, (inputs, targets) in enumerate(val_loader):
        data_time.update(time.time() - end)
        inputs, targets = _prepare_batch(inputs, targets, device, dtype)
        outputs = net(inputs)
        loss = criterion(outputs.float(), targets)
        acc1, acck = utils.accuracy(outputs, targets, topk=(1, topk))
        metrics = [loss.detach(), acc1[0], acck[0]]
        utils.scaled_all_reduce(metrics)
        losses.update(metrics[0].item(), inputs.size(0))
        top1.update(metrics[1].item(), inputs.size(0))
        topk_m.update(metrics[2].item(), inputs.size(0))
        batch_time.update(time.time() - end)
        end = time.time()
        if idx % cfg.TEST.PRINT_FREQ == 0 and utils.get_rank() == 0:
            progress.display(idx)
    return top1.avg, topk_m.avg


def train_model():
    """Full training driver (reference trainer.py:106-173)."""
    rank, local_rank = utils.setup_distributed()
    device = torch.device(f"cuda:{local_rank}" if torch.cuda.is_available()
                          else "cpu")
    utils.setup_seed(rank)
    setup_logger(rank, cfg.OUT_DIR)
    dtype = _compute_dtype()

    net = build_network(device)
    if utils.get_world_size() > 1:
        net = DistributedDataParallel(net, bucket_cap_mb=cfg.TRAIN.BUCKET_CAP_MB)
    train_loader = construct_train_loader()
    val_loader = construct_val_loader()
    criterion = DF.cross_entropy
    optimizer = utils.construct_optimizer(net)

    n_params, mb = utils.count_parameters(net)
    if rank == 0:
        logger.info(f"Model {cfg.MODEL.ARCH}: {n_params / 1e6:.3f}M params "
                    f"({mb:.1f} MB fp32)")

    start_epoch, best_acc1 = 0, 0.0
    if cfg.TRAIN.AUTO_RESUME and utils.has_checkpoint():
        ckpt = utils.get_last_checkpoint()
        start_epoch, best_acc1 = utils.load_checkpoint(ckpt, net, optimizer)
        if rank == 0:
            logger.info(f"Auto-resumed from {ckpt} at epoch {start_epoch}")
    elif cfg.MODEL.WEIGHTS:
        # reference trainer.py:147-149 also resumes epoch/best from the
        # weights file; load_checkpoint only advances the epoch when the
        # optimizer state was actually restored
        start_epoch, best_acc1 = utils.load_checkpoint(
            cfg.MODEL.WEIGHTS, net,
            optimizer if cfg.TRAIN.LOAD_OPT else None)
        if rank == 0:
            logger.info(f"Loaded initial weights from {cfg.MODEL.WEIGHTS}")

    for epoch in range(start_epoch, cfg.OPTIM.MAX_EPOCH):
        train_epoch(train_loader, net, criterion, optimizer, epoch, device, dtype)
        acc1, acck = validate(val_loader, net, criterion, device, dtype)
        best = acc1 > best_acc1
        best_acc1 = max(acc1, best_acc1)
        path = utils.save_checkpoint(net, optimizer, epoch, best_acc1, best)
        if rank == 0:
            logger.info(
                f"ACCURACY: TOP1 {acc1:.3f}(BEST {best_acc1:.3f}) | "
                f"TOP{cfg.TRAIN.TOPK} {acck:.3f} | SAVED {path}")


def test_model():
    """Eval-only driver (reference trainer.py:176-209)."""
    rank, local_rank = utils.setup_distributed()
    device = torch.device(f"cuda:{local_rank}" if torch.cuda.is_available()
                          else "cpu")
    setup_logger(rank, cfg.OUT_DIR)
    dtype = _compute_dtype()
    net = models.build_model(cfg.MODEL.ARCH, num_classes=cfg.MODEL.NUM_CLASSES)
    net = net.to(device)
    if dtype == torch.bfloat16:
        net = net.to(torch.bfloat16)
    if utils.get_world_size() > 1:
        net = DistributedDataParallel(net, bucket_cap_mb=cfg.TRAIN.BUCKET_CAP_MB)
    val_loader = construct_val_loader()
    if cfg.MODEL.WEIGHTS:
        utils.load_checkpoint(cfg.MODEL.WEIGHTS, net)
        if rank == 0:
            logger.info(f"Loaded weights from {cfg.MODEL.WEIGHTS}")
    acc1, acck = validate(val_loader, net, DF.cross_entropy, device, dtype)
    if rank == 0:
        logger.info(f"ACCURACY: TOP1 {acc1:.3f} | TOP{cfg.TRAIN.TOPK} {acck:.3f}")
    return acc1, acck
