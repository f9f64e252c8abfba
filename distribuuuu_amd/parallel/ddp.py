"""Data-parallel module wrapper: bucketed gradient all-reduce on RCCL over xGMI.

Replaces the reference's use of torch.nn.parallel.DistributedDataParallel
(`/root/reference/distribuuuu/trainer.py:134`) with our own implementation so the
communication schedule is xGMI-native rather than NVSwitch-tuned:

* Parameters are packed into flat buckets in REVERSE registration order (autograd
  produces last-layer grads first), so all-reduce overlaps the remaining backward.
* Default bucket cap is 8 MB (cfg.TRAIN.BUCKET_CAP_MB) — small enough that a
  ResNet-50 bf16 step (~51 MB of grads) yields ≥6 in-flight buckets to spread over
  RCCL's rings across the 7 xGMI point-to-point links; the CUDA-era 25 MB default
  would serialize on one or two rings.
* Gradients are bucket views (zero-copy); the end-of-backward autograd callback
  waits on the async works and applies the 1/world_size scaling.
* Rank-0 parameter/buffer broadcast at construction (collective C2).
"""

import torch
import torch.distributed as dist
import torch.nn as nn


class _Bucket:
    __slots__ = ("params", "buffer", "views", "pending", "work")

    def __init__(self):
        self.params = []
        self.buffer = None
        self.views = {}
        self.pending = 0
        self.work = None


class DistributedDataParallel(nn.Module):
    def __init__(self, module, device_ids=None, output_device=None,
                 bucket_cap_mb=8, broadcast_buffers=True, process_group=None):
        super().__init__()
        self.module = module
        self.process_group = process_group
        self.broadcast_buffers = broadcast_buffers
        self.world_size = (dist.get_world_size(process_group)
                           if dist.is_initialized() else 1)
        self._bucket_cap = int(bucket_cap_mb * 1024 * 1024)
        self._buckets = []
        self._param_to_bucket = {}
        self._callback_queued = False
        if self.world_size > 1:
            self._broadcast_state()
            self._build_buckets()
            self._register_hooks()

    # -- setup -------------------------------------------------------------
    def _broadcast_state(self):
        """One-shot rank-0 -> all param+buffer broadcast at construction
        (per tensor; init-only, so latency is irrelevant — the per-step
        buffer sync below is the coalesced one)."""
        tensors = [p.data for p in self.module.parameters()]
        tensors += [b.data for b in self.module.buffers()]
        for t in tensors:
            dist.broadcast(t, src=0, group=self.process_group)

    def _build_buckets(self):
        params = [p for p in self.module.parameters() if p.requires_grad]
        bucket = _Bucket()
        size = 0
        for p in reversed(params):
            nbytes = p.numel() * p.element_size()
            # a bucket is one flat tensor: split on size cap AND dtype change
            # (bf16 models keep BN parameters in fp32)
            if bucket.params and (size + nbytes > self._bucket_cap
                                  or p.dtype != bucket.params[0].dtype):
                self._buckets.append(bucket)
                bucket = _Bucket()
                size = 0
            bucket.params.append(p)
            size += nbytes
        if bucket.params:
            self._buckets.append(bucket)
        for b in self._buckets:
            total = sum(p.numel() for p in b.params)
            dtype = b.params[0].dtype
            device = b.params[0].device
            b.buffer = torch.zeros(total, dtype=dtype, device=device)
            off = 0
            for p in b.params:
                b.views[p] = b.buffer[off:off + p.numel()].view_as(p)
                off += p.numel()
            b.pending = len(b.params)
        for b in self._buckets:
            for p in b.params:
                self._param_to_bucket[p] = b
        self._install_grad_views()

    def _install_grad_views(self):
        for b in self._buckets:
            for p in b.params:
                p.grad = b.views[p]

    def _register_hooks(self):
        for b in self._buckets:
            for p in b.params:
                p.register_post_accumulate_grad_hook(self._make_hook(b))

    def _make_hook(self, bucket):
        def hook(p):
            view = bucket.views[p]
            if p.grad is None or p.grad.data_ptr() != view.data_ptr():
                # grad was re-allocated (zero_grad(set_to_none=True)); fold it in
                if p.grad is not None:
                    view.copy_(p.grad)
                    p.grad = view
            bucket.pending -= 1
            if bucket.pending == 0:
                bucket.work = dist.all_reduce(
                    bucket.buffer, op=dist.ReduceOp.SUM,
                    group=self.process_group, async_op=True)
            if not self._callback_queued:
                self._callback_queued = True
                torch.autograd.Variable._execution_engine.queue_callback(
                    self._finalize_backward)
        return hook

    def _finalize_backward(self):
        self._callback_queued = False
        inv = 1.0 / self.world_size
        for b in self._buckets:
            if b.work is not None:
                b.work.wait()
                b.work = None
            elif b.pending != len(b.params):
                # partial bucket (unused params) — reduce what we have
                dist.all_reduce(b.buffer, op=dist.ReduceOp.SUM,
                                group=self.process_group)
            b.buffer.mul_(inv)
            b.pending = len(b.params)

    # -- forward -----------------------------------------------------------
    def _sync_buffers(self):
        """Coalesced rank-0 buffer broadcast: ONE collective instead of one per
        buffer (the reference's DDP issues ~160 tiny broadcasts for ResNet-50's
        BN stats; on xGMI latency dominates small messages)."""
        bufs = [b for b in self.module.buffers() if b.dtype.is_floating_point]
        if not bufs:
            return
        flat = torch.cat([b.reshape(-1).float() for b in bufs])
        dist.broadcast(flat, src=0, group=self.process_group)
        off = 0
        with torch.no_grad():
            for b in bufs:
                n = b.numel()
                b.copy_(flat[off:off + n].view_as(b))
                off += n

    def forward(self, *args, **kwargs):
        if (self.world_size > 1 and self.broadcast_buffers
                and self.module.training):
            self._sync_buffers()
        return self.module(*args, **kwargs)

    def zero_grad(self, set_to_none=False):
        """Zero bucket buffers in one memset per bucket; grad views stay installed."""
        if self.world_size <= 1:
            return self.module.zero_grad(set_to_none=set_to_none)
        for b in self._buckets:
            b.buffer.zero_()
        self._install_grad_views()
