"""BoTNet MHSA with decomposed 2-D relative position logits.

CPU / reference path: plain torch composition (einsum QK^T + rel-pos pad-shift
+ softmax + PV). GPU path: one fused CDNA4 HIP kernel (`mhsa_relpos_fwd`) — at
L = H*W = 196 the whole score tile fits in LDS, so no sequence partitioning is
needed.
"""

import torch
import torch.nn.functional as F

from .dispatch import use_hip, ext


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


class _HIPMHSARelPos(torch.autograd.Function):
    """Fused CDNA4 forward (one kernel: QK^T + decomposed rel-pos + softmax
    + PV, score tile LDS-resident); backward recomputes through the torch
    composition flash-style (the L x L attention matrix is never saved)."""

    @staticmethod
    def forward(ctx, q, k, v, rel_h, rel_w, h, w):
        n, heads, l, d = q.shape
        qf = q.reshape(n * heads, l, d).contiguous()
        kf = k.reshape(n * heads, l, d).contiguous()
        vt = v.reshape(n * heads, l, d).transpose(1, 2).contiguous()
        # per-row relative-logit tables (fp32): RW = q@rel_w^T, RH = q@rel_h^T
        rw = torch.matmul(qf.float(), rel_w.float().t()).contiguous()
        rh = torch.matmul(qf.float(), rel_h.float().t()).contiguous()
        out = ext().mhsa_fwd(qf, kf, vt, rw, rh, h, w)
        ctx.save_for_backward(q, k, v, rel_h, rel_w)
        ctx.hw = (h, w)
        return out.reshape(n, heads, l, d)

    @staticmethod
    def backward(ctx, gout):
        q, k, v, rel_h, rel_w = ctx.saved_tensors
        h, w = ctx.hw
        with torch.enable_grad():
            q_ = q.detach().requires_grad_(True)
            k_ = k.detach().requires_grad_(True)
            v_ = v.detach().requires_grad_(True)
            rh_ = rel_h.detach().requires_grad_(True)
            rw_ = rel_w.detach().requires_grad_(True)
            out = _torch_mhsa(q_, k_, v_, rh_, rw_, h, w)
            gq, gk, gv, grh, grw = torch.autograd.grad(
                out, [q_, k_, v_, rh_, rw_], gout)
        return gq, gk, gv, grh, grw, None, None


def mhsa_relpos(q, k, v, rel_h, rel_w, h, w):
    """q (pre-scaled), k, v: [N, heads, L, d]; returns [N, heads, L, d_v]."""
    if (use_hip(q, "mhsa_fwd") and q.dtype == torch.bfloat16
            and q.shape[-1] % 32 == 0 and q.shape[-1] <= 128):
        return _HIPMHSARelPos.apply(q, k, v, rel_h, rel_w, h, w)
    return _torch_mhsa(q, k, v, rel_h, rel_w, h, w)
