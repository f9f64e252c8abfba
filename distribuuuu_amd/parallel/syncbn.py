"""SyncBatchNorm: cross-rank batch statistics on RCCL.

Replaces torch.nn.SyncBatchNorm (reference `trainer.py:131`,
`nn.SyncBatchNorm.convert_sync_batchnorm`). Forward computes local per-channel
sum / sum-of-squares (on GPU via the bn_stats HIP kernel), all-reduces ONE
coalesced [2C+1] fp32 tensor (the reference's implementation all-reduces
mean & var separately per layer), normalizes with the fused apply(+residual)
(+act) kernel, and backward all-reduces the two grad-stat vectors in one
message. Keeps the convert_sync_batchnorm(module) toggle semantics of
cfg.MODEL.SYNCBN.
"""

import torch
import torch.distributed as dist
import torch.nn.functional as F

from ..ops.modules import BatchNorm2d
from ..ops import functional as DF
from ..ops.dispatch import use_hip, ext


class _SyncBNFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var, momentum, eps,
                act, residual, process_group):
        world_size = dist.get_world_size(process_group)
        c = x.shape[1]
        if use_hip(x, "bn_stats"):
            x = DF._cl(x)
            s, ss = ext().bn_sums(x)  # fp32 [C] sum, sumsq
        else:
            xf = x.float()
            s = xf.sum(dim=(0, 2, 3))
            ss = (xf * xf).sum(dim=(0, 2, 3))
        count = torch.full((1,), x.numel() / c, dtype=torch.float32,
                           device=x.device)
        packed = torch.cat([s, ss, count])          # ONE [2C+1] collective
        dist.all_reduce(packed, group=process_group)
        s, ss, total = packed[:c], packed[c:2 * c], packed[2 * c].item()
        mean = s / total
        var = ss / total - mean * mean
        if running_mean is not None:
            with torch.no_grad():
                unbiased = var * (total / max(total - 1, 1))
                running_mean.mul_(1 - momentum).add_(mean, alpha=momentum)
                running_var.mul_(1 - momentum).add_(unbiased, alpha=momentum)
        rstd = (var + eps).rsqrt()
        if use_hip(x, "bn_apply_act"):
            res = DF._cl(residual) if residual is not None else None
            scale = weight.float() * rstd
            shift = bias.float() - mean * scale
            y = ext().bn_apply_act(x, scale, shift, DF._ACTS[act], res)
        else:
            y = (x.float() - mean.reshape(1, -1, 1, 1)) * rstd.reshape(1, -1, 1, 1)
            y = y * weight.float().reshape(1, -1, 1, 1) + bias.float().reshape(1, -1, 1, 1)
            if residual is not None:
                y = y + residual.float()
            y = DF._apply_act(y, act).to(x.dtype)
        ctx.save_for_backward(x, weight, mean, rstd, y)
        ctx.meta = (act, total, residual is not None, process_group)
        return y

    @staticmethod
    def backward(ctx, gy):
        x, weight, mean, rstd, y = ctx.saved_tensors
        act, total, has_res, group = ctx.meta
        gy = gy.contiguous()
        # activation backward through the saved post-act output
        if act == "relu":
            gy = gy * (y > 0).to(gy.dtype)
        elif act != "none":
            raise NotImplementedError(f"SyncBN act backward: {act}")
        gres = gy if has_res else None
        xf = x.float()
        gyf = gy.float()
        xhat = (xf - mean.reshape(1, -1, 1, 1)) * rstd.reshape(1, -1, 1, 1)
        sum_gy = gyf.sum(dim=(0, 2, 3))
        sum_gy_xhat = (gyf * xhat).sum(dim=(0, 2, 3))
        packed = torch.cat([sum_gy, sum_gy_xhat])   # ONE [2C] collective
        dist.all_reduce(packed, group=group)
        c = x.shape[1]
        sum_gy, sum_gy_xhat = packed[:c], packed[c:]
        gw = sum_gy_xhat
        gb = sum_gy
        w_rstd = (weight.float() * rstd).reshape(1, -1, 1, 1)
        gx = w_rstd * (gyf - (sum_gy.reshape(1, -1, 1, 1)
                              + xhat * sum_gy_xhat.reshape(1, -1, 1, 1)) / total)
        return (gx.to(x.dtype), gw.to(weight.dtype), gb.to(weight.dtype),
                None, None, None, None, None, gres, None)


class SyncBatchNorm(BatchNorm2d):
    """Drop-in for ops.BatchNorm2d with cross-rank statistics."""

    def __init__(self, num_features, eps=1e-5, momentum=0.1, act="none",
                 process_group=None):
        super().__init__(num_features, eps, momentum, act)
        self.process_group = process_group

    def forward(self, x, residual=None):
        if (not self.training or not dist.is_initialized()
                or dist.get_world_size(self.process_group) == 1):
            return super().forward(x, residual)
        self.num_batches_tracked += 1
        return _SyncBNFunction.apply(
            x, self.weight, self.bias, self.running_mean, self.running_var,
            self.momentum, self.eps, self.act, residual, self.process_group)

    @classmethod
    def convert_sync_batchnorm(cls, module, process_group=None):
        """Recursively replace ops.BatchNorm2d with SyncBatchNorm
        (torch.nn.SyncBatchNorm.convert_sync_batchnorm semantics)."""
        if isinstance(module, BatchNorm2d) and not isinstance(module, cls):
            out = cls(module.num_features, module.eps, module.momentum,
                      module.act, process_group)
            out.weight = module.weight
            out.bias = module.bias
            out.running_mean = module.running_mean
            out.running_var = module.running_var
            out.num_batches_tracked = module.num_batches_tracked
            out.training = module.training
            return out
        for name, child in module.named_children():
            new_child = cls.convert_sync_batchnorm(child, process_group)
            if new_child is not child:
                setattr(module, name, new_child)
        return module


def convert_sync_batchnorm(module, process_group=None):
    return SyncBatchNorm.convert_sync_batchnorm(module, process_group)
