"""Inception-ResNet-v2 (reference example/image-classification/symbols/
inception-resnet-v2.py): inception branches whose concat projects back to
the trunk width and adds residually with a scale. Square-3x3 substitution
for the paper's asymmetric factorizations (see inception_v3.py note)."""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.layers import (BatchNorm2dNHWC, Conv2dNHWC, Dropout,
                          GlobalAvgPool, LinearBF16, MaxPool2dNHWC, ReLU)


class ConvBN(nn.Module):
    def __init__(self, cin, cout, k, stride=1, pad=0, relu=True):
        super().__init__()
        self.conv = Conv2dNHWC(cin, cout, k, stride=stride, padding=pad)
        self.bn = BatchNorm2dNHWC(cout, fuse_relu=relu)

    def forward(self, x):
        return self.bn(self.conv(x))


class Stem(nn.Module):
    """299 -> 35x35x320 (the v4 stem shape family)."""

    def __init__(self, cin):
        super().__init__()
        self.seq = nn.Sequential(
            ConvBN(cin, 32, 3, 2, 0), ConvBN(32, 32, 3, 1, 0),
            ConvBN(32, 64, 3, 1, 1), MaxPool2dNHWC(3, 2, 0),
            ConvBN(64, 80, 1), ConvBN(80, 192, 3, 1, 0),
            MaxPool2dNHWC(3, 2, 0))
        self.b1 = ConvBN(192, 96, 1)
        self.b2 = nn.Sequential(ConvBN(192, 48, 1), ConvBN(48, 64, 5, 1, 2))
        self.b3 = nn.Sequential(ConvBN(192, 64, 1), ConvBN(64, 96, 3, 1, 1),
                                ConvBN(96, 96, 3, 1, 1))
        self.b4 = ConvBN(192, 64, 1)

    def forward(self, x):
        x = self.seq(x)
        p = F.avg_pool2d(x, 3, 1, 1, count_include_pad=False)
        return torch.cat([self.b1(x), self.b2(x), self.b3(x), self.b4(p)], dim=1)


class Block35(nn.Module):
    """x + scale * proj(concat(1x1, 1x1->3x3, 1x1->3x3->3x3)); trunk 320."""

    def __init__(self, cin=320, scale=0.17):
        super().__init__()
        self.scale = scale
        self.b1 = ConvBN(cin, 32, 1)
        self.b2 = nn.Sequential(ConvBN(cin, 32, 1), ConvBN(32, 32, 3, 1, 1))
        self.b3 = nn.Sequential(ConvBN(cin, 32, 1), ConvBN(32, 48, 3, 1, 1),
                                ConvBN(48, 64, 3, 1, 1))
        self.proj = ConvBN(128, cin, 1, relu=False)
        self.act = ReLU()

    def forward(self, x):
        y = torch.cat([self.b1(x), self.b2(x), self.b3(x)], dim=1)
        return self.act(x + self.scale * self.proj(y))


class RedA(nn.Module):  # 320 -> 1088
    def __init__(self, cin=320):
        super().__init__()
        self.b1 = ConvBN(cin, 384, 3, 2, 0)
        self.b2 = nn.Sequential(ConvBN(cin, 256, 1), ConvBN(256, 256, 3, 1, 1),
                                ConvBN(256, 384, 3, 2, 0))
        self.pool = MaxPool2dNHWC(3, 2, 0)

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.pool(x)], dim=1)


class Block17(nn.Module):
    """trunk 1088; 7x7 factorization as a 3x3 pair."""

    def __init__(self, cin=1088, scale=0.10):
        super().__init__()
        self.scale = scale
        self.b1 = ConvBN(cin, 192, 1)
        self.b2 = nn.Sequential(ConvBN(cin, 128, 1), ConvBN(128, 160, 3, 1, 1),
                                ConvBN(160, 192, 3, 1, 1))
        self.proj = ConvBN(384, cin, 1, relu=False)
        self.act = ReLU()

    def forward(self, x):
        y = torch.cat([self.b1(x), self.b2(x)], dim=1)
        return self.act(x + self.scale * self.proj(y))


class RedB(nn.Module):  # 1088 -> 2080
    def __init__(self, cin=1088):
        super().__init__()
        self.b1 = nn.Sequential(ConvBN(cin, 256, 1), ConvBN(256, 384, 3, 2, 0))
        self.b2 = nn.Sequential(ConvBN(cin, 256, 1), ConvBN(256, 288, 3, 2, 0))
        self.b3 = nn.Sequential(ConvBN(cin, 256, 1), ConvBN(256, 288, 3, 1, 1),
                                ConvBN(288, 320, 3, 2, 0))
        self.pool = MaxPool2dNHWC(3, 2, 0)

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x), self.pool(x)],
                         dim=1)


class Block8(nn.Module):
    """trunk 2080; 3x3 pair for the 1x3/3x1 factorization."""

    def __init__(self, cin=2080, scale=0.20, relu=True):
        super().__init__()
        self.scale = scale
        self.b1 = ConvBN(cin, 192, 1)
        self.b2 = nn.Sequential(ConvBN(cin, 192, 1), ConvBN(192, 224, 3, 1, 1),
                                ConvBN(224, 256, 3, 1, 1))
        self.proj = ConvBN(448, cin, 1, relu=False)
        self.act = ReLU() if relu else None

    def forward(self, x):
        y = torch.cat([self.b1(x), self.b2(x)], dim=1)
        out = x + self.scale * self.proj(y)
        return self.act(out) if self.act is not None else out


class InceptionResNetV2(nn.Module):
    def __init__(self, num_classes=1000, image_shape="3,299,299"):
        super().__init__()
        c = int(image_shape.split(",")[0])
        layers = [Stem(c)]
        layers += [Block35() for _ in range(5)]
        layers += [RedA()]
        layers += [Block17() for _ in range(10)]
        layers += [RedB()]
        layers += [Block8() for _ in range(5)]
        layers += [Block8(relu=False), ConvBN(2080, 1536, 1)]
        self.features = nn.Sequential(*layers)
        self.gap = GlobalAvgPool()
        self.drop = Dropout(0.2)
        self.fc = LinearBF16(1536, num_classes)

    def forward(self, x):
        return self.fc(self.drop(self.gap(self.features(x))))


def get_symbol(num_classes=1000, image_shape="3,299,299", **kwargs):
    return InceptionResNetV2(num_classes=num_classes, image_shape=image_shape)
