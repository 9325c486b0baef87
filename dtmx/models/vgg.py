"""VGG (reference example/image-classification/symbols/vgg.py — BASELINE.md
inference rows)."""
from __future__ import annotations

import torch.nn as nn

from ..ops.layers import Conv2dNHWC, Dropout, LinearBF16, MaxPool2dNHWC, ReLU

_CFG = {
    11: [64, "M", 128, "M", 256, 256, "M", 512, 512, "M", 512, 512, "M"],
    16: [64, 64, "M", 128, 128, "M", 256, 256, 256, "M", 512, 512, 512, "M",
         512, 512, 512, "M"],
    19: [64, 64, "M", 128, 128, "M", 256, 256, 256, 256, "M", 512, 512, 512, 512,
         "M", 512, 512, 512, 512, "M"],
}


class VGG(nn.Module):
    def __init__(self, num_layers=16, num_classes=1000):
        super().__init__()
        self.spec = {"network": "vgg", "num_layers": num_layers, "num_classes": num_classes}
        layers = []
        in_ch = 3
        for v in _CFG[num_layers]:
            if v == "M":
                layers.append(MaxPool2dNHWC(2, 2))
            else:
                layers += [Conv2dNHWC(in_ch, v, 3, 1, 1, bias=True), ReLU()]
                in_ch = v
        self.features = nn.Sequential(*layers)
        self.classifier = nn.Sequential(
            LinearBF16(512 * 7 * 7, 4096), ReLU(), Dropout(0.5),
            LinearBF16(4096, 4096), ReLU(), Dropout(0.5),
            LinearBF16(4096, num_classes),
        )

    def forward(self, x):
        x = self.features(x)
        x = x.reshape(x.shape[0], -1)
        return self.classifier(x)


def get_symbol(num_classes=1000, num_layers=16, **kwargs):
    return VGG(num_layers=num_layers, num_classes=num_classes)
