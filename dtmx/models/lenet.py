"""LeNet (reference example/image-classification/symbols/lenet.py,
tests/nightly/dist_lenet.py)."""
from __future__ import annotations

import torch.nn as nn

from ..ops.layers import Conv2dNHWC, LinearBF16, MaxPool2dNHWC, ReLU


class LeNet(nn.Module):
    def __init__(self, num_classes=10, in_channels=1, image_hw=28):
        super().__init__()
        self.spec = {"network": "lenet", "num_classes": num_classes}
        self.features = nn.Sequential(
            Conv2dNHWC(in_channels, 20, 5, 1, 0, bias=True), ReLU(),
            MaxPool2dNHWC(2, 2),
            Conv2dNHWC(20, 50, 5, 1, 0, bias=True), ReLU(),
            MaxPool2dNHWC(2, 2),
        )
        feat_hw = ((image_hw - 4) // 2 - 4) // 2
        self.classifier = nn.Sequential(
            LinearBF16(50 * feat_hw * feat_hw, 500), ReLU(),
            LinearBF16(500, num_classes),
        )

    def forward(self, x):
        x = self.features(x)
        x = x.reshape(x.shape[0], -1)
        return self.classifier(x)


def get_symbol(num_classes=10, image_shape="1,28,28", **kwargs):
    c, h, _ = (int(x) for x in image_shape.split(","))
    return LeNet(num_classes=num_classes, in_channels=c, image_hw=h)
