"""GoogLeNet v1 (reference example/image-classification/symbols/googlenet.py:
ConvFactory conv+relu, InceptionFactory 4-branch concat, no BN) built
MI355X-native on NHWC convs with the fused relu elementwise op."""
from __future__ import annotations

import torch
import torch.nn as nn

from ..ops.layers import (Conv2dNHWC, Dropout, GlobalAvgPool, LinearBF16,
                          MaxPool2dNHWC, ReLU)


class ConvRelu(nn.Module):
    def __init__(self, cin, cout, k, stride=1, pad=0):
        super().__init__()
        self.conv = Conv2dNHWC(cin, cout, k, stride=stride, padding=pad,
                               bias=True)
        self.act = ReLU()

    def forward(self, x):
        return self.act(self.conv(x))


class Inception(nn.Module):
    """1x1 | 1x1->3x3 | 1x1->5x5 | maxpool->1x1 (reference
    InceptionFactory)."""

    def __init__(self, cin, n1, n3r, n3, n5r, n5, proj):
        super().__init__()
        self.b1 = ConvRelu(cin, n1, 1)
        self.b2 = nn.Sequential(ConvRelu(cin, n3r, 1), ConvRelu(n3r, n3, 3, 1, 1))
        self.b3 = nn.Sequential(ConvRelu(cin, n5r, 1), ConvRelu(n5r, n5, 5, 1, 2))
        self.b4 = nn.Sequential(MaxPool2dNHWC(3, 1, 1), ConvRelu(cin, proj, 1))
        self.out_channels = n1 + n3 + n5 + proj

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x), self.b4(x)], dim=1)


class GoogLeNet(nn.Module):
    def __init__(self, num_classes=1000, image_shape="3,224,224"):
        super().__init__()
        c = int(image_shape.split(",")[0])
        self.features = nn.Sequential(
            ConvRelu(c, 64, 7, 2, 3), MaxPool2dNHWC(3, 2, 1),
            ConvRelu(64, 64, 1), ConvRelu(64, 192, 3, 1, 1),
            MaxPool2dNHWC(3, 2, 1),
            Inception(192, 64, 96, 128, 16, 32, 32),    # 3a -> 256
            Inception(256, 128, 128, 192, 32, 96, 64),  # 3b -> 480
            MaxPool2dNHWC(3, 2, 1),
            Inception(480, 192, 96, 208, 16, 48, 64),   # 4a -> 512
            Inception(512, 160, 112, 224, 24, 64, 64),  # 4b
            Inception(512, 128, 128, 256, 24, 64, 64),  # 4c
            Inception(512, 112, 144, 288, 32, 64, 64),  # 4d -> 528
            Inception(528, 256, 160, 320, 32, 128, 128),  # 4e -> 832
            MaxPool2dNHWC(3, 2, 1),
            Inception(832, 256, 160, 320, 32, 128, 128),  # 5a
            Inception(832, 384, 192, 384, 48, 128, 128),  # 5b -> 1024
        )
        self.gap = GlobalAvgPool()
        self.drop = Dropout(0.4)
        self.fc = LinearBF16(1024, num_classes)

    def forward(self, x):
        return self.fc(self.drop(self.gap(self.features(x))))


def get_symbol(num_classes=1000, image_shape="3,224,224", **kwargs):
    return GoogLeNet(num_classes=num_classes, image_shape=image_shape)
