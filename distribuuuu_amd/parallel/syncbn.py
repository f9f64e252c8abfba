"""SyncBatchNorm: cross-rank batch statistics on RCCL.

Replaces torch.nn.SyncBatchNorm (reference `trainer.py:131`,
`nn.SyncBatchNorm.convert_sync_batchnorm`). Forward computes local per-channel
sum / sum-of-squares with the bn_sums HIP kernel, all-reduces ONE coalesced
[2C] fp32 tensor (the reference's implementation all-reduces mean & var
separately per layer), and normalizes with the fused apply(+residual)(+act)
kernel. Backward computes local RAW grad-stat sums with bn_bwd_stats,
all-reduces the [2C] vector, and finishes with the same fused finalize+dx
kernels the single-rank path uses (bn_bwd_apply) — no ATen fp32
materializations on the hot path. A plain-torch composition backs the CPU /
gloo test tier.
"""

import torch
import torch.distributed as dist

from ..ops.modules import BatchNorm2d
from ..ops import functional as DF
from ..ops.dispatch import use_hip, ext


class _SyncBNFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var, momentum, eps,
                act, residual, process_group):
        c = x.shape[1]
        gamma = weight.float().contiguous()
        beta = bias.float().contiguous()
        hip = use_hip(x, "bn_sums")
        if hip:
            part = getattr(x, "_bn_partials", None)
            srcw = getattr(x, "_bn_src_weight", None)
            x = DF._cl(x)
            if part is not None:
                # conv-epilogue partials (F1): no pass over x at all
                both = ext().bn_reduce_partials(part)
                s, ss = both[:c], both[c:]
            else:
                if srcw is not None:
                    srcw._emit_bn_partials = True
                s, ss = ext().bn_sums(x)  # fp32 [C] views of one [2C] tensor
        else:
            xf = x.float()
            s = xf.sum(dim=(0, 2, 3))
            ss = (xf * xf).sum(dim=(0, 2, 3))
        # equal per-rank counts in training (DistributedSampler, drop_last):
        # total is computed host-side so no .item() device sync per layer
        world = dist.get_world_size(process_group)
        total = (x.numel() / c) * world
        packed = torch.cat([s, ss])                 # ONE [2C] collective
        dist.all_reduce(packed, group=process_group)
        s, ss = packed[:c], packed[c:]
        mean = s / total
        var = (ss / total - mean * mean).clamp_(min=0)
        if running_mean is not None:
            with torch.no_grad():
                unbiased = var * (total / max(total - 1, 1))
                running_mean.mul_(1 - momentum).add_(mean, alpha=momentum)
                running_var.mul_(1 - momentum).add_(unbiased, alpha=momentum)
        rstd = (var + eps).rsqrt()
        scale = gamma * rstd
        shift = beta - mean * scale
        res = None
        if residual is not None:
            res = DF._cl(residual) if hip else residual
        if hip:
            y = ext().bn_apply_act(x, scale.contiguous(), shift.contiguous(),
                                   DF._ACTS[act], res)
        else:
            y = x.float() * scale.reshape(1, -1, 1, 1) + shift.reshape(1, -1, 1, 1)
            if res is not None:
                y = y + res.float()
            y = DF._apply_act(y, act).to(x.dtype)
        # hip backward recomputes the act mask from x (F3a): y not saved
        ctx.save_for_backward(x, gamma, scale.contiguous(), shift.contiguous(),
                              mean.contiguous(), rstd.contiguous(),
                              y if not hip else x.new_empty(0),
                              res if res is not None else x.new_empty(0))
        ctx.meta = (act, total, residual is not None, process_group, hip,
                    weight.dtype)
        return y

    @staticmethod
    def backward(ctx, gy):
        x, gamma, scale, shift, mean, rstd, y, res = ctx.saved_tensors
        act, total, has_res, group, hip, w_dtype = ctx.meta
        act_id = DF._ACTS[act]
        if hip:
            gy = DF._cl(gy)
            e = ext()
            sums = e.bn_bwd_stats(gy, x, res if has_res else None,
                                  scale, shift, act_id)      # raw [2C]
            dist.all_reduce(sums, group=group)
            gx, gw, gb, gres = e.bn_bwd_apply(
                gy, x, res if has_res else None, mean, rstd, gamma,
                scale, shift, sums, total, act_id, True, has_res)
            return (gx, gw.to(w_dtype), gb.to(w_dtype), None, None, None,
                    None, None, gres if has_res else None, None)
        # CPU / test-tier composition
        gyf = gy.float()
        if act == "relu":
            gyf = gyf * (y > 0).float()
        elif act != "none":
            raise NotImplementedError(f"SyncBN CPU act backward: {act}")
        gres = gyf.to(gy.dtype) if has_res else None
        xf = x.float()
        xhat = (xf - mean.reshape(1, -1, 1, 1)) * rstd.reshape(1, -1, 1, 1)
        sum_gy = gyf.sum(dim=(0, 2, 3))
        sum_gy_xhat = (gyf * xhat).sum(dim=(0, 2, 3))
        packed = torch.cat([sum_gy, sum_gy_xhat])   # ONE [2C] collective
        dist.all_reduce(packed, group=group)
        c = x.shape[1]
        sum_gy, sum_gy_xhat = packed[:c], packed[c:]
        w_rstd = (gamma * rstd).reshape(1, -1, 1, 1)
        gx = w_rstd * (gyf - (sum_gy.reshape(1, -1, 1, 1)
                              + xhat * sum_gy_xhat.reshape(1, -1, 1, 1)) / total)
        return (gx.to(x.dtype), sum_gy_xhat.to(w_dtype), sum_gy.to(w_dtype),
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
        self._nbt_pending += 1
        rm, rv = self.running_mean, self.running_var
        return _SyncBNFunction.apply(
            x, self.weight, self.bias, rm, rv, self.momentum, self.eps,
            self.act, residual, self.process_group)

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
