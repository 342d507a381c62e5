"""VGG-16 — the reference's communication-bound benchmark model
(docs/performance.md: VGG-16 shows the largest BytePS gains because its
138M params dominate the step time)."""

from __future__ import annotations

import torch.nn as nn

_CFG16 = [64, 64, "M", 128, 128, "M", 256, 256, 256, "M",
          512, 512, 512, "M", 512, 512, 512, "M"]


class VGG(nn.Module):
    def __init__(self, cfg, num_classes: int = 1000, bn: bool = False):
        super().__init__()
        layers = []
        in_ch = 3
        for v in cfg:
            if v == "M":
                layers.append(nn.MaxPool2d(2, 2))
            else:
                layers.append(nn.Conv2d(in_ch, v, 3, padding=1))
                if bn:
                    layers.append(nn.BatchNorm2d(v))
                layers.append(nn.ReLU(inplace=True))
                in_ch = v
        self.features = nn.Sequential(*layers)
        self.avgpool = nn.AdaptiveAvgPool2d(7)
        self.classifier = nn.Sequential(
            nn.Linear(512 * 7 * 7, 4096), nn.ReLU(True), nn.Dropout(),
            nn.Linear(4096, 4096), nn.ReLU(True), nn.Dropout(),
            nn.Linear(4096, num_classes))

    def forward(self, x):
        x = self.avgpool(self.features(x)).flatten(1)
        return self.classifier(x)


def vgg16(num_classes: int = 1000, bn: bool = False) -> VGG:
    return VGG(_CFG16, num_classes, bn)
