"""VGG for CIFAR-shape inputs (reference: VGG/models/vgg.py:14, config 'D' =
VGG-16 with batch-norm, 10-class head for CIFAR-10)."""
from __future__ import annotations

import torch
import torch.nn as nn

_CFG = {
    "vgg11": [64, "M", 128, "M", 256, 256, "M", 512, 512, "M", 512, 512, "M"],
    "vgg13": [64, 64, "M", 128, 128, "M", 256, 256, "M", 512, 512, "M", 512, 512, "M"],
    "vgg16": [64, 64, "M", 128, 128, "M", 256, 256, 256, "M", 512, 512, 512, "M",
              512, 512, 512, "M"],
    "vgg19": [64, 64, "M", 128, 128, "M", 256, 256, 256, 256, "M", 512, 512, 512, 512,
              "M", 512, 512, 512, 512, "M"],
}


class VGG(nn.Module):
    def __init__(self, arch: str = "vgg16", num_classes: int = 10, batch_norm: bool = True):
        super().__init__()
        layers = []
        in_ch = 3
        for v in _CFG[arch]:
            if v == "M":
                layers.append(nn.MaxPool2d(2, 2))
            else:
                layers.append(nn.Conv2d(in_ch, v, 3, padding=1, bias=not batch_norm))
                if batch_norm:
                    layers.append(nn.BatchNorm2d(v))
                layers.append(nn.ReLU(inplace=True))
                in_ch = v
        self.features = nn.Sequential(*layers)
        self.classifier = nn.Sequential(
            nn.Linear(512, 512),
            nn.ReLU(inplace=True),
            nn.Dropout(0.5),
            nn.Linear(512, num_classes),
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.features(x)
        x = torch.flatten(x, 1)
        return self.classifier(x)


def vgg16(num_classes: int = 10, **kw) -> VGG:
    return VGG("vgg16", num_classes=num_classes, **kw)
