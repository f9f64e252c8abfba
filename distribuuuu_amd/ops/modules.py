"""nn.Module wrappers over the functional layer.

Models in distribuuuu_amd/models/ are built exclusively from these modules so
every FLOP routes through the MI355X kernel path on GPU (and through ATen on
CPU for the no-GPU test tier). Parameter/buffer names match torch.nn so
checkpoints interoperate with the reference's state_dict layout.
"""

import math

import torch
import torch.nn as nn

from . import functional as DF


def _pair(v):
    return tuple(v) if isinstance(v, (tuple, list)) else (v, v)


class Conv2d(nn.Module):
    def __init__(self, in_channels, out_channels, kernel_size, stride=1,
                 padding=0, dilation=1, groups=1, bias=False):
        super().__init__()
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.kernel_size = _pair(kernel_size)
        self.stride = _pair(stride)
        self.padding = _pair(padding)
        self.dilation = _pair(dilation)
        self.groups = groups
        self.weight = nn.Parameter(
            torch.empty(out_channels, in_channels // groups, *self.kernel_size)
        )
        if bias:
            self.bias = nn.Parameter(torch.zeros(out_channels))
        else:
            self.register_parameter("bias", None)
        nn.init.kaiming_normal_(self.weight, mode="fan_out", nonlinearity="relu")

    def forward(self, x):
        return DF.conv2d(x, self.weight, self.bias, self.stride, self.padding,
                         self.dilation, self.groups)

    def extra_repr(self):
        return (f"{self.in_channels}, {self.out_channels}, "
                f"kernel_size={self.kernel_size}, stride={self.stride}, "
                f"padding={self.padding}, groups={self.groups}")


class BatchNorm2d(nn.Module):
    """BN with an optional fused activation / residual-add epilogue.

    ``act`` ∈ {none, relu, silu, sigmoid}; ``forward(x, residual=...)`` fuses the
    skip-connection add before the activation (SURVEY.md K6+K8+K20 fusion).
    """

    def __init__(self, num_features, eps=1e-5, momentum=0.1, act="none"):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.act = act
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer("num_batches_tracked", torch.tensor(0, dtype=torch.long))
        # batch counter kept as a host int and folded into the buffer only
        # when the state dict is read: the per-step `buffer += 1` was a
        # 4.7 us GPU launch per BN layer (53/step on ResNet-50, ~0.25
        # ms/step inside the captured graph) for a value nothing reads
        # during training
        self._nbt_pending = 0

    def _save_to_state_dict(self, destination, prefix, keep_vars):
        self._flush_nbt()
        super()._save_to_state_dict(destination, prefix, keep_vars)

    def _load_from_state_dict(self, *args, **kwargs):
        self._nbt_pending = 0
        super()._load_from_state_dict(*args, **kwargs)

    def _flush_nbt(self):
        if self._nbt_pending:
            self.num_batches_tracked += self._nbt_pending
            self._nbt_pending = 0

    def forward(self, x, residual=None):
        if self.training:
            self._nbt_pending += 1
        return DF.batch_norm_act(
            x, self.weight, self.bias, self.running_mean, self.running_var,
            self.training, self.momentum, self.eps, self.act, residual,
        )

    def extra_repr(self):
        return f"{self.num_features}, eps={self.eps}, act={self.act}"


class ReLU(nn.Module):
    def __init__(self, inplace=True):
        super().__init__()

    def forward(self, x):
        return DF.relu(x)


class SiLU(nn.Module):
    def forward(self, x):
        return DF._apply_act(x, "silu")


class Sigmoid(nn.Module):
    def forward(self, x):
        return DF._apply_act(x, "sigmoid")


class MaxPool2d(nn.Module):
    def __init__(self, kernel_size, stride, padding=0):
        super().__init__()
        self.kernel_size = kernel_size
        self.stride = stride
        self.padding = padding

    def forward(self, x):
        return DF.max_pool2d(x, self.kernel_size, self.stride, self.padding)


class AvgPool2d(nn.Module):
    def __init__(self, kernel_size, stride=None):
        super().__init__()
        self.kernel_size = kernel_size
        self.stride = stride or kernel_size

    def forward(self, x):
        return DF.avg_pool2d(x, self.kernel_size, self.stride)


class AdaptiveAvgPool2d(nn.Module):
    def __init__(self, output_size=1):
        super().__init__()
        self.output_size = output_size

    def forward(self, x):
        return DF.adaptive_avg_pool2d(x, self.output_size)


class Linear(nn.Module):
    def __init__(self, in_features, out_features, bias=True):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        if bias:
            self.bias = nn.Parameter(torch.zeros(out_features))
        else:
            self.register_parameter("bias", None)
        nn.init.normal_(self.weight, 0, 0.01)
        bound = 1 / math.sqrt(in_features)
        if self.bias is not None:
            nn.init.uniform_(self.bias, -bound, bound)

    def forward(self, x):
        return DF.linear(x, self.weight, self.bias)

    def extra_repr(self):
        return f"{self.in_features}, {self.out_features}"


class Dropout(nn.Module):
    def __init__(self, p=0.0):
        super().__init__()
        self.p = p

    def forward(self, x):
        return DF.dropout(x, self.p, self.training)
