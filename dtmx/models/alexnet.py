"""AlexNet (reference example/image-classification/symbols/alexnet.py —
a BASELINE.md benchmark row)."""
from __future__ import annotations

import torch.nn as nn

from ..ops.layers import (LRN, Conv2dNHWC, Dropout, LinearBF16,
                          MaxPool2dNHWC, ReLU)


class AlexNet(nn.Module):
    def __init__(self, num_classes=1000):
        super().__init__()
        self.spec = {"network": "alexnet", "num_classes": num_classes}
        self.features = nn.Sequential(
            # LRN after the first two relus per the reference symbol
            # (alexnet.py:34,41 — mx.sym.LRN alpha=1e-4 beta=0.75 knorm=2
            # nsize=5); kept in the benchmarked graph
            Conv2dNHWC(3, 64, 11, 4, 2, bias=True), ReLU(), LRN(),
            MaxPool2dNHWC(3, 2),
            Conv2dNHWC(64, 192, 5, 1, 2, bias=True), ReLU(), LRN(),
            MaxPool2dNHWC(3, 2),
            Conv2dNHWC(192, 384, 3, 1, 1, bias=True), ReLU(),
            Conv2dNHWC(384, 256, 3, 1, 1, bias=True), ReLU(),
            Conv2dNHWC(256, 256, 3, 1, 1, bias=True), ReLU(), MaxPool2dNHWC(3, 2),
        )
        self.classifier = nn.Sequential(
            LinearBF16(256 * 6 * 6, 4096), ReLU(), Dropout(0.5),
            LinearBF16(4096, 4096), ReLU(), Dropout(0.5),
            LinearBF16(4096, num_classes),
        )

    def forward(self, x):
        x = self.features(x)
        x = x.reshape(x.shape[0], -1)
        return self.classifier(x)


def get_symbol(num_classes=1000, **kwargs):
    return AlexNet(num_classes=num_classes)
