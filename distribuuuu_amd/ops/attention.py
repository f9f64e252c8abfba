"""BoTNet MHSA with decomposed 2-D relative position logits.

CPU / reference path: plain torch composition (einsum QK^T + rel-pos pad-shift
+ softmax + PV). GPU path: one fused CDNA4 HIP kernel (`mhsa_relpos_fwd`) — at
L = H*W = 196 the whole score tile fits in LDS, so no sequence partitioning is
needed.
"""

import torch
import torch.nn.functional as F

from .dispatch import use_hip, ext, fallback_warn


def rel_to_abs(x):
    """[B, L, 2L-1] relative logits -> [B, L, L] absolute, pad-shift trick
    (reference botnet.py:25-40, without its hard .cuda() temporaries)."""
    b, l, _ = x.shape
    col_pad = x.new_zeros(b, l, 1)
    x = torch.cat([x, col_pad], dim=2)                       # [b, l, 2l]
    flat = x.reshape(b, l * 2 * l)
    flat = torch.cat([flat, x.new_zeros(b, l - 1)], dim=1)   # [b, 2l^2 + l - 1]
    final = flat.reshape(b, l + 1, 2 * l - 1)
    return final[:, :l, l - 1:]


def relative_logits_1d(q, rel_k):
    """q: [B, heads, H, W, d]; rel_k: [2W-1, d] -> [B, heads, H, W, H', W']-ready
    per-axis logits [B, heads, H, W, W] expanded over the other axis."""
    b, heads, h, w, _ = q.shape
    logits = torch.einsum("bhxyd,md->bhxym", q, rel_k)       # [b, heads, H, W, 2W-1]
    logits = logits.reshape(b * heads * h, w, 2 * w - 1)
    logits = rel_to_abs(logits)                              # [b*heads*H, W, W]
    logits = logits.reshape(b, heads, h, w, w)
    return logits.unsqueeze(3).expand(-1, -1, -1, h, -1, -1)  # [b,heads,H,H',W,W']


def _rel_pos_logits(q, rel_h, rel_w, h, w):
    """Full 2-D decomposed relative logits: sum of W-axis and H-axis terms.
    q: [N, heads, L, d] -> [N, heads, L, L]."""
    n, heads, l, d = q.shape
    q2 = q.reshape(n, heads, h, w, d)
    # W-axis: indexes differ along width
    logits_w = relative_logits_1d(q2, rel_w)                 # [n,h,H,H',W,W']
    logits_w = logits_w.permute(0, 1, 2, 4, 3, 5)            # [n,h,H,W,H',W']
    # H-axis: transpose H/W, apply, transpose back
    q2t = q2.permute(0, 1, 3, 2, 4)                          # [n,heads,W,H,d]
    logits_h = relative_logits_1d(q2t, rel_h)                # [n,h,W,W',H,H']
    logits_h = logits_h.permute(0, 1, 4, 2, 5, 3)            # [n,h,H,W,H',W']
    out = logits_w + logits_h
    return out.reshape(n, heads, l, l)


def _torch_mhsa(q, k, v, rel_h, rel_w, h, w):
    logits = torch.einsum("bhxd,bhyd->bhxy", q, k)
    logits = logits + _rel_pos_logits(q, rel_h, rel_w, h, w)
    attn = F.softmax(logits, dim=-1)
    return torch.einsum("bhxy,bhyd->bhxd", attn, v)


_REL_IDX = {}


def _rel_bwd_idx(h, w, device):
    """Gather maps for the rel-pos backward: dRW[x, m] sums dS over y with
    yw = m + xw - (W-1) (the inverse of the forward's RW[x, yw-xw+W-1]
    lookup), read from a (W-1)-padded row; likewise for the H axis."""
    key = (h, w, str(device))
    if key not in _REL_IDX:
        xs = torch.arange(h * w, device=device)
        iw = (torch.arange(2 * w - 1, device=device).unsqueeze(0)
              + (xs % w).unsqueeze(1))                    # [l, 2w-1]
        ih = (torch.arange(2 * h - 1, device=device).unsqueeze(0)
              + (xs // w).unsqueeze(1))                   # [l, 2h-1]
        _REL_IDX[key] = (iw.unsqueeze(0), ih.unsqueeze(0))
    return _REL_IDX[key]


class _HIPMHSARelPos(torch.autograd.Function):
    """Fused CDNA4 forward (one kernel: QK^T + decomposed rel-pos + softmax
    + PV, score tile LDS-resident; the per-row rel-logit tables come from
    the MFMA NT GEMM). The kernel also writes the softmax probs P
    (39 MB/layer at L=196), and backward is three HIP kernels (mhsa_bwd):
    per-q-tile dP -> dS -> dQ with the rel-pos fold, a dK/dV accumulation
    pass, and the rel-table grad reduction — the previous ATen bmm/
    elementwise chain cost ~3 ms/step on BoTNet-50."""

    @staticmethod
    def forward(ctx, q, k, v, rel_h, rel_w, h, w):
        n, heads, l, d = q.shape
        e = ext()
        qf = q.reshape(n * heads, l, d).contiguous()
        kf = k.reshape(n * heads, l, d).contiguous()
        vt = v.reshape(n * heads, l, d).transpose(1, 2).contiguous()
        # per-row relative-logit tables RW = q@rel_w^T, RH = q@rel_h^T —
        # dedicated kernel with fp32 accumulate AND fp32 output (a bf16
        # GEMM output rounds the tables and shifts large-logit softmaxes)
        rows = n * heads * l
        rw = e.mhsa_rel_tables(qf, rel_w.contiguous(), rows, 1, 1, d, 1.0)
        rh = e.mhsa_rel_tables(qf, rel_h.contiguous(), rows, 1, 1, d, 1.0)
        rw = rw.reshape(n * heads, l, -1)
        rh = rh.reshape(n * heads, l, -1)
        need_grad = any(t.requires_grad for t in (q, k, v, rel_h, rel_w))
        pout = (torch.empty(n * heads, l, l, dtype=q.dtype, device=q.device)
                if need_grad else None)
        out = e.mhsa_fwd(qf, kf, vt, rw, rh, h, w, pout, 1, d, d, 1.0)
        ctx.save_for_backward(q, k, v, rel_h, rel_w,
                              pout if pout is not None else q.new_empty(0))
        ctx.hw = (h, w)
        return out.reshape(n, heads, l, d)

    @staticmethod
    def backward(ctx, gout):
        q, k, v, rel_h, rel_w, P = ctx.saved_tensors
        h, w = ctx.hw
        n, heads, l, d = q.shape
        bh = n * heads
        qf = q.reshape(bh, l, d).contiguous()
        kt = k.reshape(bh, l, d).transpose(1, 2).contiguous()
        vf = v.reshape(bh, l, d).contiguous()
        dO = gout.reshape(bh, l, d).contiguous()
        # dense layout: dq/dk share one [2, bh, l, d] buffer (kqoff splits it)
        dqk = torch.empty(2, bh, l, d, dtype=q.dtype, device=q.device)
        dvb = torch.empty(bh, l, d, dtype=q.dtype, device=q.device)
        dqk, dvb, grw, grh = ext().mhsa_bwd(
            dO, P, qf, kt, vf, rel_w.contiguous(), rel_h.contiguous(), h, w,
            1, d, bh * l * d, d, d, 1.0, dqk, dvb)
        return (dqk[0].reshape(n, heads, l, d), dqk[1].reshape(n, heads, l, d),
                dvb.reshape(n, heads, l, d), grh.to(rel_h.dtype),
                grw.to(rel_w.dtype), None, None)


def mhsa_relpos(q, k, v, rel_h, rel_w, h, w):
    """q (pre-scaled), k, v: [N, heads, L, d]; returns [N, heads, L, d_v]."""
    if (use_hip(q, "mhsa_fwd") and q.dtype == torch.bfloat16
            and q.shape[-1] % 32 == 0 and q.shape[-1] <= 128):
        return _HIPMHSARelPos.apply(q, k, v, rel_h, rel_w, h, w)
    if q.is_cuda:
        fallback_warn("mhsa_relpos",
                      f"dtype {q.dtype} head_dim {q.shape[-1]}")
    return _torch_mhsa(q, k, v, rel_h, rel_w, h, w)


class _HIPMHSARelPosNHWC(torch.autograd.Function):
    """NHWC in-place attention I/O: consumes the qkv convs' channels_last
    outputs DIRECTLY (physical [N, L, heads*D]) via per-pixel-strided kernel
    addressing — no chunk/transpose/contiguous copies on q, k, dO, dq, dk,
    dv, no separate q-scale pass (the scale folds into the logits/tables),
    and the output/grads come back as channels_last 4-D, exactly what the
    surrounding 1x1 convs produce and consume. Only the transposed v/k
    operands (vt, kt — MFMA B-operands must be l-contiguous) remain ATen
    permute-copies."""

    @staticmethod
    def forward(ctx, qk, v, rel_h, rel_w, heads, dqk, dv, h, w, scale):
        e = ext()
        n = qk.shape[0]
        l = h * w
        qkc = qk.contiguous(memory_format=torch.channels_last)
        vc = v.contiguous(memory_format=torch.channels_last)
        qpix = 2 * heads * dqk
        vpix = heads * dv
        # vt [B, dv, L] from physical [N, L, heads, dv] (one copy)
        vt = (vc.permute(0, 2, 3, 1).reshape(n, l, heads, dv)
              .permute(0, 2, 3, 1).reshape(n * heads, dv, l).contiguous())
        rows = n * heads * l
        rw = e.mhsa_rel_tables(qkc, rel_w.contiguous(), rows, heads, l, qpix,
                               scale)
        rh = e.mhsa_rel_tables(qkc, rel_h.contiguous(), rows, heads, l, qpix,
                               scale)
        need_grad = any(t.requires_grad for t in (qk, v, rel_h, rel_w))
        pout = (torch.empty(n * heads, l, l, dtype=qk.dtype, device=qk.device)
                if need_grad else None)
        kview = (qkc.permute(0, 2, 3, 1).reshape(n * l, qpix)
                 [:, heads * dqk:])  # data_ptr offset only; kernel strides
        out = e.mhsa_fwd(qkc, kview, vt, rw, rh, h, w, pout, heads, qpix,
                         qpix, scale)
        ctx.save_for_backward(qkc, vc, rel_h, rel_w,
                              pout if pout is not None else qk.new_empty(0))
        ctx.conf = (heads, dqk, dv, h, w, scale)
        return out  # [N, heads*dv, H, W] channels_last

    @staticmethod
    def backward(ctx, gout):
        qkc, vc, rel_h, rel_w, P = ctx.saved_tensors
        heads, dqk, dv, h, w, scale = ctx.conf
        n = qkc.shape[0]
        l = h * w
        qpix = 2 * heads * dqk
        vpix = heads * dv
        gc = gout.contiguous(memory_format=torch.channels_last)
        # kt [B, dqk, L] from the k half of qk (one copy)
        kt = (qkc.permute(0, 2, 3, 1).reshape(n, l, 2 * heads, dqk)
              [:, :, heads:, :].permute(0, 2, 3, 1)
              .reshape(n * heads, dqk, l).contiguous())
        dqk_out = torch.empty_like(qkc)
        dv_out = torch.empty_like(vc)
        dqk_out, dv_out, grw, grh = ext().mhsa_bwd(
            gc, P, qkc, kt, vc, rel_w.contiguous(), rel_h.contiguous(), h, w,
            heads, qpix, heads * dqk, vpix, vpix, scale, dqk_out, dv_out)
        return (dqk_out, dv_out, grh.to(rel_h.dtype), grw.to(rel_w.dtype),
                None, None, None, None, None, None)


def mhsa_relpos_nhwc(qk, v, rel_h, rel_w, heads, dqk, dv, h, w, scale):
    """qk: [N, 2*heads*dqk, H, W] cl; v: [N, heads*dv, H, W] cl ->
    [N, heads*dv, H, W] cl."""
    return _HIPMHSARelPosNHWC.apply(qk, v, rel_h, rel_w, heads, dqk, dv, h,
                                    w, scale)
