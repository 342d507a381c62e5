"""ResNet v1.5 (bottleneck) — the reference's benchmark vehicle trains
torchvision ResNet-50 on synthetic data (reference
example/pytorch/benchmark_byteps.py:60-82); torchvision is not a
dependency here, so the architecture is implemented directly.

MI355X notes: runs channels_last (NHWC) bf16; every BatchNorm is the
fused BN(+residual)(+ReLU) gfx950 kernel set (byteps_amd.torch.fused_bn
→ ops/csrc/bn.hip) — one memory pass instead of MIOpen's 3 kernels plus
separate add/relu.  On CPU the same modules fall back to native torch
ops, so numerics tests run anywhere.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..torch.fused_bn import FusedBNReLU


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch: int, width: int, stride: int = 1,
                 downsample: nn.Module = None):
        super().__init__()
        out_ch = width * self.expansion
        self.conv1 = nn.Conv2d(in_ch, width, 1, bias=False)
        self.bn1 = FusedBNReLU(width, relu=True)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = FusedBNReLU(width, relu=True)
        self.conv3 = nn.Conv2d(width, out_ch, 1, bias=False)
        # bn3 fuses the residual add + final relu into the same pass
        self.bn3 = FusedBNReLU(out_ch, relu=True)
        self.downsample = downsample

    def forward(self, x):
        identity = x
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return self.bn3(self.conv3(out), residual=identity)


class Downsample(nn.Module):
    def __init__(self, in_ch: int, out_ch: int, stride: int):
        super().__init__()
        self.conv = nn.Conv2d(in_ch, out_ch, 1, stride=stride, bias=False)
        self.bn = FusedBNReLU(out_ch, relu=False)

    def forward(self, x):
        return self.bn(self.conv(x))


class ResNet(nn.Module):
    def __init__(self, layers, num_classes: int = 1000):
        super().__init__()
        self.in_ch = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = FusedBNReLU(64, relu=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(64, layers[0])
        self.layer2 = self._make_layer(128, layers[1], stride=2)
        self.layer3 = self._make_layer(256, layers[2], stride=2)
        self.layer4 = self._make_layer(512, layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * Bottleneck.expansion, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, FusedBNReLU):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)
        # zero-init the last BN of each block (standard v1.5 recipe)
        for m in self.modules():
            if isinstance(m, Bottleneck):
                nn.init.zeros_(m.bn3.weight)

    def _make_layer(self, width: int, blocks: int, stride: int = 1):
        downsample = None
        out_ch = width * Bottleneck.expansion
        if stride != 1 or self.in_ch != out_ch:
            downsample = Downsample(self.in_ch, out_ch, stride)
        layers = [Bottleneck(self.in_ch, width, stride, downsample)]
        self.in_ch = out_ch
        for _ in range(1, blocks):
            layers.append(Bottleneck(self.in_ch, width))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.bn1(self.conv1(x)))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = torch.flatten(self.avgpool(x), 1)
        return self.fc(x)


def resnet50(num_classes: int = 1000) -> ResNet:
    return ResNet([3, 4, 6, 3], num_classes)


def resnet101(num_classes: int = 1000) -> ResNet:
    return ResNet([3, 4, 23, 3], num_classes)
