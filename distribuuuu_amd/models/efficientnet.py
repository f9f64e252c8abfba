"""EfficientNet-B0, MI355X-native.

The reference reaches this arch through timm (`/root/reference/distribuuuu/trainer.py:123-128`,
`config/efficientnet_b0.yaml`). Implemented natively: MBConv (1x1 expand -> kxk depthwise
-> SE -> 1x1 project) with SiLU, BN eps 1e-3, stochastic-depth-free (matching the
reference recipe which only uses RandomResizedCrop/Flip augmentation). 5.289M params.

The depthwise conv + small-GEMM SE are SURVEY.md K5 — dedicated HIP kernels
(per-channel depthwise, no MFMA) on the GPU path.
"""

import math

import torch.nn as nn

from ..ops import functional as DF

from ..ops import AdaptiveAvgPool2d, BatchNorm2d, Conv2d, Dropout, Linear, SiLU, Sigmoid


class SqueezeExcite(nn.Module):
    def __init__(self, channels, se_channels):
        super().__init__()
        self.avg_pool = AdaptiveAvgPool2d(1)
        self.conv_reduce = Conv2d(channels, se_channels, 1, bias=True)
        self.act = SiLU()
        self.conv_expand = Conv2d(se_channels, channels, 1, bias=True)
        self.gate = Sigmoid()

    def forward(self, x):
        s = self.avg_pool(x)
        s = self.gate(self.conv_expand(self.act(self.conv_reduce(s))))
        return DF.se_scale(x, s)


class MBConv(nn.Module):
    def __init__(self, c_in, c_out, kernel, stride, expand_ratio, se_ratio=0.25):
        super().__init__()
        c_mid = c_in * expand_ratio
        self.use_residual = stride == 1 and c_in == c_out
        self.expand = None
        if expand_ratio != 1:
            self.expand = nn.Sequential(
                Conv2d(c_in, c_mid, 1, bias=False),
                BatchNorm2d(c_mid, eps=1e-3, act="silu"),
            )
        self.dw = Conv2d(c_mid, c_mid, kernel, stride=stride,
                         padding=kernel // 2, groups=c_mid, bias=False)
        self.dw_bn = BatchNorm2d(c_mid, eps=1e-3, act="silu")
        self.se = SqueezeExcite(c_mid, max(1, int(c_in * se_ratio)))
        self.project = Conv2d(c_mid, c_out, 1, bias=False)
        self.project_bn = BatchNorm2d(c_out, eps=1e-3)

    def forward(self, x):
        if not self.use_residual:
            out = x if self.expand is None else self.expand(x)
            out = self.dw_bn(self.dw(out))
            out = self.se(out)
            return self.project_bn(self.project(out))
        # residual block: fuse the add into project_bn (same output) and
        # fork the input so the expand conv's dgrad can accumulate into
        # the shortcut gradient (see ops.functional.fork)
        xm, xs = DF.fork(x)
        out = xm if self.expand is None else self.expand(xm)
        out = self.dw_bn(self.dw(out))
        out = self.se(out)
        return self.project_bn(self.project(out), residual=xs)


# (expand_ratio, channels, repeats, stride, kernel) — B0 table
_B0_CFG = [
    (1, 16, 1, 1, 3),
    (6, 24, 2, 2, 3),
    (6, 40, 2, 2, 5),
    (6, 80, 3, 2, 3),
    (6, 112, 3, 1, 5),
    (6, 192, 4, 2, 5),
    (6, 320, 1, 1, 3),
]


class EfficientNet(nn.Module):
    def __init__(self, cfg_table=_B0_CFG, width_mult=1.0, depth_mult=1.0,
                 dropout=0.2, num_classes=1000):
        super().__init__()

        def _round_ch(c):
            c = c * width_mult
            new_c = max(8, int(c + 4) // 8 * 8)
            if new_c < 0.9 * c:
                new_c += 8
            return new_c

        def _round_rep(r):
            return int(math.ceil(depth_mult * r))

        stem_c = _round_ch(32)
        self.stem = nn.Sequential(
            Conv2d(3, stem_c, 3, stride=2, padding=1, bias=False),
            BatchNorm2d(stem_c, eps=1e-3, act="silu"),
        )
        blocks = []
        c_in = stem_c
        for expand, c, reps, stride, k in cfg_table:
            c_out = _round_ch(c)
            for i in range(_round_rep(reps)):
                blocks.append(MBConv(c_in, c_out, k, stride if i == 0 else 1, expand))
                c_in = c_out
        self.blocks = nn.Sequential(*blocks)
        head_c = _round_ch(1280)
        self.head = nn.Sequential(
            Conv2d(c_in, head_c, 1, bias=False),
            BatchNorm2d(head_c, eps=1e-3, act="silu"),
        )
        self.avgpool = AdaptiveAvgPool2d(1)
        self.dropout = Dropout(dropout)
        self.classifier = Linear(head_c, num_classes)

    def forward(self, x):
        x = self.head(self.blocks(self.stem(x)))
        x = self.dropout(self.avgpool(x).flatten(1))
        return self.classifier(x)


def efficientnet_b0(**kw):
    kw.pop("pretrained", None)
    return EfficientNet(**kw)
