"""ResNet / ResNeXt / Wide-ResNet family, MI355X-native.

Capability parity with the reference zoo (`/root/reference/distribuuuu/models/resnet.py`):
ResNet v1.5 (stride on the 3x3 of the bottleneck, resnet.py:106-161 there), Kaiming init,
optional zero-init of the last BN gamma per block. Rebuilt here on the fused op layer:
conv->bn->relu chains run BN+ReLU in the conv's consumer kernel and the residual join is
fused into the last BN of each block (BatchNorm2d(act=..., residual=...)).

State_dict key names match the torchvision layout the reference uses, so reference
checkpoints load directly.
"""

import torch.nn as nn

from ..ops import functional as DF
from ..ops import AdaptiveAvgPool2d, BatchNorm2d, Conv2d, Linear, MaxPool2d


def conv3x3(cin, cout, stride=1, groups=1, dilation=1):
    return Conv2d(cin, cout, 3, stride=stride, padding=dilation, dilation=dilation,
                  groups=groups, bias=False)


def conv1x1(cin, cout, stride=1):
    return Conv2d(cin, cout, 1, stride=stride, bias=False)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes, planes, stride=1, downsample=None, groups=1,
                 base_width=64, dilation=1):
        super().__init__()
        if groups != 1 or base_width != 64:
            raise ValueError("BasicBlock only supports groups=1, base_width=64")
        self.conv1 = conv3x3(inplanes, planes, stride)
        self.bn1 = BatchNorm2d(planes, act="relu")
        self.conv2 = conv3x3(planes, planes)
        self.bn2 = BatchNorm2d(planes, act="relu")  # residual joins before relu
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        xm, xs = DF.fork(x)
        identity = xs if self.downsample is None else self.downsample(xs)
        out = self.bn1(self.conv1(xm))
        out = self.conv2(out)
        return self.bn2(out, residual=identity)


class Bottleneck(nn.Module):
    # v1.5: stride lives on the 3x3, not the first 1x1
    expansion = 4

    def __init__(self, inplanes, planes, stride=1, downsample=None, groups=1,
                 base_width=64, dilation=1):
        super().__init__()
        width = int(planes * (base_width / 64.0)) * groups
        self.conv1 = conv1x1(inplanes, width)
        self.bn1 = BatchNorm2d(width, act="relu")
        self.conv2 = conv3x3(width, width, stride, groups, dilation)
        self.bn2 = BatchNorm2d(width, act="relu")
        self.conv3 = conv1x1(width, planes * self.expansion)
        self.bn3 = BatchNorm2d(planes * self.expansion, act="relu")
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        xm, xs = DF.fork(x)
        identity = xs if self.downsample is None else self.downsample(xs)
        out = self.bn1(self.conv1(xm))
        out = self.bn2(self.conv2(out))
        out = self.conv3(out)
        return self.bn3(out, residual=identity)


class ResNet(nn.Module):
    def __init__(self, block, layers, num_classes=1000, zero_init_residual=False,
                 groups=1, width_per_group=64):
        super().__init__()
        self.inplanes = 64
        self.groups = groups
        self.base_width = width_per_group
        self.conv1 = Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = BatchNorm2d(64, act="relu")
        self.maxpool = MaxPool2d(kernel_size=3, stride=2, padding=1)
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2)
        self.avgpool = AdaptiveAvgPool2d(1)
        self.fc = Linear(512 * block.expansion, num_classes)

        for m in self.modules():
            if isinstance(m, BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)
        if zero_init_residual:
            for m in self.modules():
                if isinstance(m, Bottleneck):
                    nn.init.zeros_(m.bn3.weight)
                elif isinstance(m, BasicBlock):
                    nn.init.zeros_(m.bn2.weight)

    def _make_layer(self, block, planes, blocks, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                conv1x1(self.inplanes, planes * block.expansion, stride),
                BatchNorm2d(planes * block.expansion),
            )
        layers = [block(self.inplanes, planes, stride, downsample, self.groups,
                        self.base_width)]
        self.inplanes = planes * block.expansion
        layers += [
            block(self.inplanes, planes, groups=self.groups,
                  base_width=self.base_width)
            for _ in range(1, blocks)
        ]
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.bn1(self.conv1(x)))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x)
        return self.fc(x.flatten(1))


# Published torchvision weight URLs (the reference loads the same files,
# resnet.py:23-33 there); the state_dict layout of these models matches
# torchvision key-for-key, so the checkpoints load strict.
model_urls = {
    "resnet18": "https://download.pytorch.org/models/resnet18-5c106cde.pth",
    "resnet34": "https://download.pytorch.org/models/resnet34-333f7ec4.pth",
    "resnet50": "https://download.pytorch.org/models/resnet50-19c8e357.pth",
    "resnet101": "https://download.pytorch.org/models/resnet101-5d3b4d8f.pth",
    "resnet152": "https://download.pytorch.org/models/resnet152-b121ed2d.pth",
    "resnext50_32x4d":
        "https://download.pytorch.org/models/resnext50_32x4d-7cdf4587.pth",
    "resnext101_32x8d":
        "https://download.pytorch.org/models/resnext101_32x8d-8ba56ff5.pth",
    "wide_resnet50_2":
        "https://download.pytorch.org/models/wide_resnet50_2-95faca4d.pth",
    "wide_resnet101_2":
        "https://download.pytorch.org/models/wide_resnet101_2-32ee1156.pth",
}


def _resnet(arch, block, layers, pretrained=False, progress=True, **kwargs):
    model = ResNet(block, layers, **kwargs)
    if pretrained:
        from torch.hub import load_state_dict_from_url

        state_dict = load_state_dict_from_url(model_urls[arch],
                                              progress=progress,
                                              map_location="cpu")
        model.load_state_dict(state_dict)
    return model


def resnet18(**kw):
    return _resnet("resnet18", BasicBlock, [2, 2, 2, 2], **kw)


def resnet34(**kw):
    return _resnet("resnet34", BasicBlock, [3, 4, 6, 3], **kw)


def resnet50(**kw):
    return _resnet("resnet50", Bottleneck, [3, 4, 6, 3], **kw)


def resnet101(**kw):
    return _resnet("resnet101", Bottleneck, [3, 4, 23, 3], **kw)


def resnet152(**kw):
    return _resnet("resnet152", Bottleneck, [3, 8, 36, 3], **kw)


def resnext50_32x4d(**kw):
    kw.setdefault("groups", 32)
    kw.setdefault("width_per_group", 4)
    return _resnet("resnext50_32x4d", Bottleneck, [3, 4, 6, 3], **kw)


def resnext101_32x8d(**kw):
    kw.setdefault("groups", 32)
    kw.setdefault("width_per_group", 8)
    return _resnet("resnext101_32x8d", Bottleneck, [3, 4, 23, 3], **kw)


def wide_resnet50_2(**kw):
    kw.setdefault("width_per_group", 128)
    return _resnet("wide_resnet50_2", Bottleneck, [3, 4, 6, 3], **kw)


def wide_resnet101_2(**kw):
    kw.setdefault("width_per_group", 128)
    return _resnet("wide_resnet101_2", Bottleneck, [3, 4, 23, 3], **kw)
