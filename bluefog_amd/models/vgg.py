# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""VGG family (the reference benchmark accepts torchvision model names —
examples/pytorch_benchmark.py `--model vgg16`; torchvision is not a
dependency here, so this is the standard self-contained implementation
with BatchNorm variants)."""

from typing import List, Union

import torch.nn as nn

__all__ = ["VGG", "vgg11", "vgg13", "vgg16", "vgg19"]

_CFGS = {
    "A": [64, "M", 128, "M", 256, 256, "M", 512, 512, "M", 512, 512, "M"],
    "B": [64, 64, "M", 128, 128, "M", 256, 256, "M", 512, 512, "M", 512, 512, "M"],
    "D": [64, 64, "M", 128, 128, "M", 256, 256, 256, "M", 512, 512, 512, "M",
          512, 512, 512, "M"],
    "E": [64, 64, "M", 128, 128, "M", 256, 256, 256, 256, "M", 512, 512, 512,
          512, "M", 512, 512, 512, 512, "M"],
}


class VGG(nn.Module):
    def __init__(self, cfg: List[Union[int, str]], num_classes: int = 1000,
                 batch_norm: bool = False, dropout: float = 0.5):
        super().__init__()
        layers: List[nn.Module] = []
        in_ch = 3
        for v in cfg:
            if v == "M":
                layers.append(nn.MaxPool2d(2, 2))
                continue
            layers.append(nn.Conv2d(in_ch, v, 3, padding=1))
            if batch_norm:
                layers.append(nn.BatchNorm2d(v))
            layers.append(nn.ReLU(inplace=True))
            in_ch = v
        self.features = nn.Sequential(*layers)
        self.avgpool = nn.AdaptiveAvgPool2d(7)
        self.classifier = nn.Sequential(
            nn.Linear(512 * 7 * 7, 4096), nn.ReLU(inplace=True), nn.Dropout(dropout),
            nn.Linear(4096, 4096), nn.ReLU(inplace=True), nn.Dropout(dropout),
            nn.Linear(4096, num_classes),
        )
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, 0, 0.01)
                nn.init.zeros_(m.bias)

    def forward(self, x):
        x = self.avgpool(self.features(x)).flatten(1)
        return self.classifier(x)


def vgg11(num_classes=1000, batch_norm=False):
    return VGG(_CFGS["A"], num_classes, batch_norm)


def vgg13(num_classes=1000, batch_norm=False):
    return VGG(_CFGS["B"], num_classes, batch_norm)


def vgg16(num_classes=1000, batch_norm=False):
    return VGG(_CFGS["D"], num_classes, batch_norm)


def vgg19(num_classes=1000, batch_norm=False):
    return VGG(_CFGS["E"], num_classes, batch_norm)
