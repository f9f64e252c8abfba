"""RegNetX / RegNetY family, MI355X-native.

The reference reaches these archs through timm (`/root/reference/distribuuuu/trainer.py:123-128`;
configs `config/regnetx_160.yaml`, `regnety_160.yaml`, `regnety_320.yaml`). Implemented here
natively from the RegNet design-space parameterization (w0, wa, wm, depth, group width,
optional SE for the Y variants). Parameter counts match the reference README table
(regnetx_160 54.28M, regnety_160 83.59M, regnety_320 145.05M).

The 3x3 grouped conv is the flagship group-conv MFMA path of BASELINE.json config #4.
"""

import numpy as np
import torch.nn as nn

from ..ops import functional as DF  # noqa: E501
from ..ops import AdaptiveAvgPool2d, BatchNorm2d, Conv2d, Linear, ReLU, Sigmoid


def _generate_widths(w0, wa, wm, depth, q=8):
    """Quantized linear width schedule -> per-stage (widths, depths)."""
    ws_cont = w0 + wa * np.arange(depth)
    ks = np.round(np.log(ws_cont / w0) / np.log(wm))
    ws = w0 * np.power(wm, ks)
    ws = np.round(ws / q).astype(int) * q
    widths, depths = np.unique(ws, return_counts=True)
    order = np.argsort(widths)
    return list(widths[order]), list(depths[order])


def _adjust_ws_gs(widths, group_w):
    """Group-compatibility adjustment (pycls semantics, bottleneck ratio 1):
    per-stage group width g = min(group_w, w); stage width rounded to a
    multiple of g. The adjusted width IS the stage width everywhere."""
    gs = [min(group_w, w) for w in widths]
    ws = [int(round(w / g) * g) for w, g in zip(widths, gs)]
    return ws, gs


class SE(nn.Module):
    def __init__(self, channels, se_channels):
        super().__init__()
        self.avg_pool = AdaptiveAvgPool2d(1)
        self.fc1 = Conv2d(channels, se_channels, 1, bias=True)
        self.act = ReLU()
        self.fc2 = Conv2d(se_channels, channels, 1, bias=True)
        self.gate = Sigmoid()

    def forward(self, x):
        s = self.gate(self.fc2(self.act(self.fc1(self.avg_pool(x)))))
        return DF.se_scale(x, s)


class RegBlock(nn.Module):
    """X/Y bottleneck block (bottleneck ratio 1): 1x1 -> 3x3 group (stride) ->
    [SE] -> 1x1, residual join fused into the last BN."""

    def __init__(self, w_in, w_out, stride, group_width, se_ratio=None):
        super().__init__()
        w_b, groups = w_out, w_out // group_width
        self.proj = None
        if w_in != w_out or stride != 1:
            self.proj = nn.Sequential(
                Conv2d(w_in, w_out, 1, stride=stride, bias=False),
                BatchNorm2d(w_out),
            )
        self.a = Conv2d(w_in, w_b, 1, bias=False)
        self.a_bn = BatchNorm2d(w_b, act="relu")
        self.b = Conv2d(w_b, w_b, 3, stride=stride, padding=1,
                        groups=groups, bias=False)
        self.b_bn = BatchNorm2d(w_b, act="relu")
        self.se = None
        if se_ratio:
            se_channels = max(1, int(round(w_in * se_ratio)))
            self.se = SE(w_b, se_channels)
        self.c = Conv2d(w_b, w_out, 1, bias=False)
        self.c_bn = BatchNorm2d(w_out, act="relu")
        nn.init.zeros_(self.c_bn.weight)

    def forward(self, x):
        xm, xs = DF.fork(x)
        identity = xs if self.proj is None else self.proj(xs)
        out = self.a_bn(self.a(xm))
        out = self.b_bn(self.b(out))
        if self.se is not None:
            out = self.se(out)
        out = self.c(out)
        return self.c_bn(out, residual=identity)


class RegNet(nn.Module):
    def __init__(self, w0, wa, wm, depth, group_w, se_ratio=None,
                 stem_w=32, num_classes=1000):
        super().__init__()
        widths, depths = _generate_widths(w0, wa, wm, depth)
        widths, group_ws = _adjust_ws_gs(widths, group_w)
        self.stem = nn.Sequential(
            Conv2d(3, stem_w, 3, stride=2, padding=1, bias=False),
            BatchNorm2d(stem_w, act="relu"),
        )
        stages = []
        w_in = stem_w
        for w, d, g in zip(widths, depths, group_ws):
            blocks = []
            for i in range(d):
                stride = 2 if i == 0 else 1
                blocks.append(RegBlock(w_in, w, stride, g, se_ratio))
                w_in = w
            stages.append(nn.Sequential(*blocks))
        self.stages = nn.Sequential(*stages)
        self.head_pool = AdaptiveAvgPool2d(1)
        self.head_fc = Linear(w_in, num_classes)

    def forward(self, x):
        x = self.stages(self.stem(x))
        return self.head_fc(self.head_pool(x).flatten(1))


def _regnet(w0, wa, wm, depth, group_w, se_ratio=None, **kwargs):
    kwargs.pop("pretrained", None)
    return RegNet(w0, wa, wm, depth, group_w, se_ratio, **kwargs)


def regnetx_160(**kw):
    """RegNetX-16GF."""
    return _regnet(216, 55.59, 2.1, 22, 128, None, **kw)


def regnety_160(**kw):
    """RegNetY-16GF (SE 0.25)."""
    return _regnet(200, 106.23, 2.48, 18, 112, 0.25, **kw)


def regnety_320(**kw):
    """RegNetY-32GF (SE 0.25) — the 145M-param baseline row."""
    return _regnet(232, 115.89, 2.53, 20, 232, 0.25, **kw)
