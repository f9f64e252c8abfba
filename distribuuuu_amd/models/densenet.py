"""DenseNet family, MI355X-native.

Parity with the reference (`/root/reference/distribuuuu/models/densenet.py`):
BN-ReLU-1x1 -> BN-ReLU-3x3 dense layers with concat growth, transition
BN-ReLU-1x1-avgpool2 blocks, densenet121/161/169/201 factories, optional
dropout and gradient-checkpointed (memory_efficient) bottleneck recompute
(reference densenet.py:81-110) via torch.utils.checkpoint.
"""

import re

import torch
import torch.nn as nn
import torch.utils.checkpoint as cp

from ..ops import AvgPool2d, BatchNorm2d, Conv2d, Dropout, Linear, MaxPool2d
from ..ops import functional as DF


class _DenseLayer(nn.Module):
    def __init__(self, num_input_features, growth_rate, bn_size, drop_rate,
                 memory_efficient=False):
        super().__init__()
        self.norm1 = BatchNorm2d(num_input_features, act="relu")
        self.conv1 = Conv2d(num_input_features, bn_size * growth_rate, 1, bias=False)
        self.norm2 = BatchNorm2d(bn_size * growth_rate, act="relu")
        self.conv2 = Conv2d(bn_size * growth_rate, growth_rate, 3, padding=1,
                            bias=False)
        self.drop = Dropout(drop_rate)
        self.memory_efficient = memory_efficient

    def _bottleneck(self, *inputs):
        concat = torch.cat(inputs, 1)
        return self.conv1(self.norm1(concat))

    def forward(self, *prev_features):
        if self.memory_efficient and any(p.requires_grad for p in prev_features):
            bottleneck = cp.checkpoint(self._bottleneck, *prev_features,
                                       use_reentrant=False)
        else:
            bottleneck = self._bottleneck(*prev_features)
        out = self.conv2(self.norm2(bottleneck))
        return self.drop(out)


class _DenseBlock(nn.ModuleDict):
    def __init__(self, num_layers, num_input_features, bn_size, growth_rate,
                 drop_rate, memory_efficient=False):
        super().__init__()
        for i in range(num_layers):
            self.add_module(
                f"denselayer{i + 1}",
                _DenseLayer(num_input_features + i * growth_rate, growth_rate,
                            bn_size, drop_rate, memory_efficient),
            )

    def forward(self, init_features):
        features = [init_features]
        for layer in self.values():
            features.append(layer(*features))
        return torch.cat(features, 1)


class _Transition(nn.Sequential):
    def __init__(self, num_input_features, num_output_features):
        super().__init__()
        self.norm = BatchNorm2d(num_input_features, act="relu")
        self.conv = Conv2d(num_input_features, num_output_features, 1, bias=False)
        self.pool = AvgPool2d(kernel_size=2, stride=2)


class DenseNet(nn.Module):
    def __init__(self, growth_rate=32, block_config=(6, 12, 24, 16),
                 num_init_features=64, bn_size=4, drop_rate=0.0,
                 num_classes=1000, memory_efficient=False):
        super().__init__()
        self.features = nn.Sequential()
        self.features.add_module("conv0", Conv2d(3, num_init_features, 7, stride=2,
                                                 padding=3, bias=False))
        self.features.add_module("norm0", BatchNorm2d(num_init_features, act="relu"))
        self.features.add_module("pool0", MaxPool2d(kernel_size=3, stride=2,
                                                    padding=1))
        num_features = num_init_features
        for i, num_layers in enumerate(block_config):
            block = _DenseBlock(num_layers, num_features, bn_size, growth_rate,
                                drop_rate, memory_efficient)
            self.features.add_module(f"denseblock{i + 1}", block)
            num_features += num_layers * growth_rate
            if i != len(block_config) - 1:
                trans = _Transition(num_features, num_features // 2)
                self.features.add_module(f"transition{i + 1}", trans)
                num_features //= 2
        self.features.add_module("norm5", BatchNorm2d(num_features, act="relu"))
        self.classifier = Linear(num_features, num_classes)

    def forward(self, x):
        out = self.features(x)
        out = DF.adaptive_avg_pool2d(out, 1).flatten(1)
        return self.classifier(out)


model_urls = {
    "densenet121": "https://download.pytorch.org/models/densenet121-a639ec97.pth",
    "densenet169": "https://download.pytorch.org/models/densenet169-b2777c0a.pth",
    "densenet201": "https://download.pytorch.org/models/densenet201-c1103571.pth",
    "densenet161": "https://download.pytorch.org/models/densenet161-8d451a50.pth",
}

# legacy torchvision DenseNet checkpoints use dotted sub-layer names
# ('denselayer1.norm.1.weight'); current module names drop the dot
_LEGACY_KEY = re.compile(
    r"^(.*denselayer\d+\.(?:norm|relu|conv))\."
    r"((?:[12])\.(?:weight|bias|running_mean|running_var))$")


def remap_legacy_densenet_keys(state_dict):
    """norm.1 -> norm1 etc., matching the reference's pretrained-key remap
    (reference densenet.py:266-282)."""
    for key in list(state_dict.keys()):
        res = _LEGACY_KEY.match(key)
        if res:
            state_dict[res.group(1) + res.group(2)] = state_dict.pop(key)
    return state_dict


def _densenet(arch, growth_rate, block_config, num_init_features,
              pretrained=False, progress=True, **kwargs):
    model = DenseNet(growth_rate, block_config, num_init_features, **kwargs)
    if pretrained:
        from torch.hub import load_state_dict_from_url

        state_dict = load_state_dict_from_url(model_urls[arch],
                                              progress=progress,
                                              map_location="cpu")
        model.load_state_dict(remap_legacy_densenet_keys(state_dict))
    return model


def densenet121(**kw):
    return _densenet("densenet121", 32, (6, 12, 24, 16), 64, **kw)


def densenet161(**kw):
    return _densenet("densenet161", 48, (6, 12, 36, 24), 96, **kw)


def densenet169(**kw):
    return _densenet("densenet169", 32, (6, 12, 32, 32), 64, **kw)


def densenet201(**kw):
    return _densenet("densenet201", 32, (6, 12, 48, 32), 64, **kw)
