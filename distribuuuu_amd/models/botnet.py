"""BoTNet-50: ResNet-50 with stage 4 replaced by Bottleneck-Transformer blocks.

Parity with the reference (`/root/reference/distribuuuu/models/botnet.py`): three
BoTBlocks at 14x14 (stride into 2 on the first when the stack downsamples),
multi-head self-attention with 2-D decomposed relative position embeddings
(rel_to_abs pad-shift trick, reference botnet.py:25-57), 4 heads, d_qk=d_v=128,
zero-init of each block's last BN gamma. The reference's `rel_to_abs` allocates
`.cuda()` zeros directly (botnet.py:33,36 — hard device assumption); here all
temporaries follow the input's device.

On GPU the attention body (QK^T + rel-pos + softmax + PV, L=196) runs as one
fused CDNA4 HIP kernel (ops.attention.mhsa_relpos); CPU composes the same math
in plain torch for the test-tier reference.
"""

import torch
import torch.nn as nn

from ..ops import AdaptiveAvgPool2d, AvgPool2d, BatchNorm2d, Conv2d, Linear
from ..ops import functional as DF
from ..ops.attention import mhsa_relpos, mhsa_relpos_nhwc
from .resnet import resnet50


class MHSA(nn.Module):
    """Multi-head self-attention over an HxW feature map with decomposed
    relative position embeddings (reference botnet.py:163-215), or absolute
    position embeddings when rel_pos_emb=False (reference AbsPosEmb,
    botnet.py:60-74)."""

    def __init__(self, dim, fmap_size, heads=4, dim_qk=128, dim_v=128,
                 rel_pos_emb=True):
        super().__init__()
        self.heads = heads
        self.dim_qk = dim_qk
        self.dim_v = dim_v
        self.scale = dim_qk ** -0.5
        self.fmap_size = fmap_size
        self.rel_pos_emb = rel_pos_emb
        out_qk = heads * dim_qk
        out_v = heads * dim_v
        self.to_qk = Conv2d(dim, 2 * out_qk, 1, bias=False)
        self.to_v = Conv2d(dim, out_v, 1, bias=False)
        h, w = fmap_size
        if rel_pos_emb:
            self.rel_h = nn.Parameter(
                torch.randn(2 * h - 1, dim_qk) * dim_qk ** -0.5)
            self.rel_w = nn.Parameter(
                torch.randn(2 * w - 1, dim_qk) * dim_qk ** -0.5)
        else:
            self.abs_h = nn.Parameter(torch.randn(h, dim_qk) * dim_qk ** -0.5)
            self.abs_w = nn.Parameter(torch.randn(w, dim_qk) * dim_qk ** -0.5)

    def forward(self, x):
        n, _, hh, ww = x.shape
        qk = self.to_qk(x)
        v = self.to_v(x)
        if (self.rel_pos_emb and self.dim_qk == self.dim_v
                and qk.is_cuda and qk.dtype == torch.bfloat16
                and self.dim_qk % 32 == 0 and self.dim_qk <= 128):
            from ..ops.dispatch import use_hip
            if use_hip(qk, "mhsa_fwd"):
                # NHWC in-place path: the qkv convs' channels_last outputs
                # feed the kernels directly (no chunk/permute copies)
                return mhsa_relpos_nhwc(qk, v, self.rel_h, self.rel_w,
                                        self.heads, self.dim_qk, self.dim_v,
                                        hh, ww, self.scale)
        q, k = qk.chunk(2, dim=1)
        # [N, heads, L, d]
        q = q.reshape(n, self.heads, self.dim_qk, hh * ww).transpose(2, 3)
        k = k.reshape(n, self.heads, self.dim_qk, hh * ww).transpose(2, 3)
        v = v.reshape(n, self.heads, self.dim_v, hh * ww).transpose(2, 3)
        if self.rel_pos_emb:
            out = mhsa_relpos(q * self.scale, k, v, self.rel_h, self.rel_w,
                              hh, ww)
        else:
            qs = q * self.scale
            emb = (self.abs_h.unsqueeze(1) + self.abs_w.unsqueeze(0)).reshape(
                hh * ww, self.dim_qk)
            logits = torch.einsum("bhxd,bhyd->bhxy", qs, k)
            logits = logits + torch.einsum("bhxd,yd->bhxy", qs, emb)
            attn = torch.softmax(logits, dim=-1)
            out = torch.einsum("bhxy,bhyd->bhxd", attn, v)
        return out.transpose(2, 3).reshape(n, self.heads * self.dim_v, hh, ww)


class BoTBlock(nn.Module):
    """1x1 down -> BN/ReLU -> MHSA -> (avgpool2 if stride 2) -> BN/ReLU ->
    1x1 up -> BN (zero-init gamma) + residual + ReLU (reference botnet.py:101-160)."""

    def __init__(self, dim, fmap_size, dim_out, stride=1, heads=4, proj_factor=4,
                 dim_qk=128, dim_v=128, rel_pos_emb=True):
        super().__init__()
        self.shortcut = None
        if dim != dim_out or stride != 1:
            self.shortcut = nn.Sequential(
                Conv2d(dim, dim_out, 1, stride=stride, bias=False),
                BatchNorm2d(dim_out, act="relu"),
            )
        bottleneck_dim = dim_out // proj_factor
        attn_out = heads * dim_v
        self.conv1 = Conv2d(dim, bottleneck_dim, 1, bias=False)
        self.bn1 = BatchNorm2d(bottleneck_dim, act="relu")
        self.mhsa = MHSA(bottleneck_dim, fmap_size, heads, dim_qk, dim_v,
                         rel_pos_emb)
        self.pool = AvgPool2d(2) if stride == 2 else None
        self.bn2 = BatchNorm2d(attn_out, act="relu")
        self.conv3 = Conv2d(attn_out, dim_out, 1, bias=False)
        self.bn3 = BatchNorm2d(dim_out, act="relu")
        nn.init.zeros_(self.bn3.weight)

    def forward(self, x):
        xm, xs = DF.fork(x)
        identity = xs if self.shortcut is None else self.shortcut(xs)
        out = self.bn1(self.conv1(xm))
        out = self.mhsa(out)
        if self.pool is not None:
            out = self.pool(out)
        out = self.bn2(out)
        out = self.conv3(out)
        return self.bn3(out, residual=identity)


class BoTStack(nn.Module):
    """Stack of 3 BoTBlocks replacing ResNet c5 (reference botnet.py:218-290)."""

    def __init__(self, dim=1024, fmap_size=(14, 14), dim_out=2048, heads=4,
                 proj_factor=4, num_layers=3, stride=2, rel_pos_emb=True):
        super().__init__()
        blocks = []
        fm = fmap_size
        for i in range(num_layers):
            is_first = i == 0
            s = stride if is_first else 1
            blocks.append(
                BoTBlock(dim if is_first else dim_out, fm, dim_out, stride=s,
                         heads=heads, proj_factor=proj_factor,
                         rel_pos_emb=rel_pos_emb)
            )
            if is_first and stride == 2:
                fm = (fm[0] // 2, fm[1] // 2)
        self.net = nn.Sequential(*blocks)

    def forward(self, x):
        return self.net(x)


def botnet50(num_classes=1000, fmap_size=(14, 14), **kwargs):
    """ResNet-50 backbone with layer4 -> BoTStack: 3 blocks all at 14x14,
    stride 1 (reference botnet.py:275-290)."""
    kwargs.pop("pretrained", None)
    backbone = resnet50(num_classes=num_classes)
    model = nn.Sequential(
        backbone.conv1,
        backbone.bn1,
        backbone.maxpool,
        backbone.layer1,
        backbone.layer2,
        backbone.layer3,
        BoTStack(dim=1024, fmap_size=fmap_size, dim_out=2048, stride=1),
        AdaptiveAvgPool2d(1),
        nn.Flatten(1),
        Linear(2048, num_classes),
    )
    return model
