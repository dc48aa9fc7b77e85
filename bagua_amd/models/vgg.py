"""VGG (Simonyan & Zisserman, 2014) — the flagship benchmark model.

Standard configurations; torchvision-compatible layer structure so
parameter counts match the reference's benchmark exactly
(reference benchmark: examples/benchmark/synthetic_benchmark.py uses
torchvision.models vgg16).
"""

from typing import List, Union

import torch
import torch.nn as nn

_CFGS = {
    "A": [64, "M", 128, "M", 256, 256, "M", 512, 512, "M", 512, 512, "M"],
    "B": [64, 64, "M", 128, 128, "M", 256, 256, "M", 512, 512, "M",
          512, 512, "M"],
    "D": [64, 64, "M", 128, 128, "M", 256, 256, 256, "M", 512, 512, 512,
          "M", 512, 512, 512, "M"],
    "E": [64, 64, "M", 128, 128, "M", 256, 256, 256, 256, "M",
          512, 512, 512, 512, "M", 512, 512, 512, 512, "M"],
}


class VGG(nn.Module):
    def __init__(self, cfg: List[Union[int, str]], num_classes: int = 1000,
                 batch_norm: bool = False, dropout: float = 0.5):
        super().__init__()
        layers: List[nn.Module] = []
        in_ch = 3
        for v in cfg:
            if v == "M":
                layers.append(nn.MaxPool2d(kernel_size=2, stride=2))
            else:
                layers.append(nn.Conv2d(in_ch, v, kernel_size=3, padding=1))
                if batch_norm:
                    layers.append(nn.BatchNorm2d(v))
                layers.append(nn.ReLU(inplace=True))
                in_ch = v
        self.features = nn.Sequential(*layers)
        self.avgpool = nn.AdaptiveAvgPool2d((7, 7))
        self.classifier = nn.Sequential(
            nn.Linear(512 * 7 * 7, 4096),
            nn.ReLU(True),
            nn.Dropout(p=dropout),
            nn.Linear(4096, 4096),
            nn.ReLU(True),
            nn.Dropout(p=dropout),
            nn.Linear(4096, num_classes),
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.features(x)
        x = self.avgpool(x)
        x = torch.flatten(x, 1)
        return self.classifier(x)


def vgg11(num_classes=1000, batch_norm=False):
    return VGG(_CFGS["A"], num_classes, batch_norm)


def vgg13(num_classes=1000, batch_norm=False):
    return VGG(_CFGS["B"], num_classes, batch_norm)


def vgg16(num_classes=1000, batch_norm=False):
    return VGG(_CFGS["D"], num_classes, batch_norm)


def vgg19(num_classes=1000, batch_norm=False):
    return VGG(_CFGS["E"], num_classes, batch_norm)
