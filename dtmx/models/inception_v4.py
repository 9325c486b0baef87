"""Inception-v4 (reference example/image-classification/symbols/
inception-v4.py). MI355X-native: NHWC conv + fused-relu BN; the paper's
asymmetric 1x7/7x1 (and 1x3/3x1) factorizations use the same square-3x3
substitution as dtmx's inception-v3 (square kernels keep the hand HIP conv
path; topology and FLOP class stay comparable — see inception_v3.py)."""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.layers import (BatchNorm2dNHWC, Conv2dNHWC, Dropout,
                          GlobalAvgPool, LinearBF16, MaxPool2dNHWC)


class ConvBN(nn.Module):
    def __init__(self, cin, cout, k, stride=1, pad=0):
        super().__init__()
        self.conv = Conv2dNHWC(cin, cout, k, stride=stride, padding=pad)
        self.bn = BatchNorm2dNHWC(cout, fuse_relu=True)

    def forward(self, x):
        return self.bn(self.conv(x))


class _AvgPool3s1(nn.Module):
    def forward(self, x):
        return F.avg_pool2d(x, 3, 1, 1, count_include_pad=False)


class Stem(nn.Module):
    """299 -> 35x35x384 (reference inception-v4 stem)."""

    def __init__(self, cin):
        super().__init__()
        self.a = nn.Sequential(ConvBN(cin, 32, 3, 2, 0), ConvBN(32, 32, 3, 1, 0),
                               ConvBN(32, 64, 3, 1, 1))
        self.b1 = MaxPool2dNHWC(3, 2, 0)
        self.b2 = ConvBN(64, 96, 3, 2, 0)
        self.c1 = nn.Sequential(ConvBN(160, 64, 1), ConvBN(64, 96, 3, 1, 0))
        self.c2 = nn.Sequential(ConvBN(160, 64, 1), ConvBN(64, 64, 3, 1, 1),
                                ConvBN(64, 64, 3, 1, 1), ConvBN(64, 96, 3, 1, 0))
        self.d1 = ConvBN(192, 192, 3, 2, 0)
        self.d2 = MaxPool2dNHWC(3, 2, 0)

    def forward(self, x):
        x = self.a(x)
        x = torch.cat([self.b1(x), self.b2(x)], dim=1)
        x = torch.cat([self.c1(x), self.c2(x)], dim=1)
        return torch.cat([self.d1(x), self.d2(x)], dim=1)


class IncA(nn.Module):  # in/out 384
    def __init__(self, cin=384):
        super().__init__()
        self.b1 = ConvBN(cin, 96, 1)
        self.b2 = nn.Sequential(ConvBN(cin, 64, 1), ConvBN(64, 96, 3, 1, 1))
        self.b3 = nn.Sequential(ConvBN(cin, 64, 1), ConvBN(64, 96, 3, 1, 1),
                                ConvBN(96, 96, 3, 1, 1))
        self.b4 = nn.Sequential(_AvgPool3s1(), ConvBN(cin, 96, 1))

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x), self.b4(x)], dim=1)


class RedA(nn.Module):  # 384 -> 1024
    def __init__(self, cin=384):
        super().__init__()
        self.b1 = ConvBN(cin, 384, 3, 2, 0)
        self.b2 = nn.Sequential(ConvBN(cin, 192, 1), ConvBN(192, 224, 3, 1, 1),
                                ConvBN(224, 256, 3, 2, 0))
        self.b3 = MaxPool2dNHWC(3, 2, 0)

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x)], dim=1)


class IncB(nn.Module):  # in/out 1024; 7x7 factorizations as 3x3 stacks
    def __init__(self, cin=1024):
        super().__init__()
        self.b1 = ConvBN(cin, 384, 1)
        self.b2 = nn.Sequential(ConvBN(cin, 192, 1), ConvBN(192, 224, 3, 1, 1),
                                ConvBN(224, 256, 3, 1, 1))
        self.b3 = nn.Sequential(ConvBN(cin, 192, 1), ConvBN(192, 192, 3, 1, 1),
                                ConvBN(192, 224, 3, 1, 1), ConvBN(224, 224, 3, 1, 1),
                                ConvBN(224, 256, 3, 1, 1))
        self.b4 = nn.Sequential(_AvgPool3s1(), ConvBN(cin, 128, 1))

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x), self.b4(x)], dim=1)


class RedB(nn.Module):  # 1024 -> 1536
    def __init__(self, cin=1024):
        super().__init__()
        self.b1 = nn.Sequential(ConvBN(cin, 192, 1), ConvBN(192, 192, 3, 2, 0))
        self.b2 = nn.Sequential(ConvBN(cin, 256, 1), ConvBN(256, 256, 3, 1, 1),
                                ConvBN(256, 320, 3, 1, 1), ConvBN(320, 320, 3, 2, 0))
        self.b3 = MaxPool2dNHWC(3, 2, 0)

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x)], dim=1)


class IncC(nn.Module):  # in/out 1536; 1x3/3x1 splits as 3x3 pairs
    def __init__(self, cin=1536):
        super().__init__()
        self.b1 = ConvBN(cin, 256, 1)
        self.b2_stem = ConvBN(cin, 384, 1)
        self.b2a = ConvBN(384, 256, 3, 1, 1)
        self.b2b = ConvBN(384, 256, 3, 1, 1)
        self.b3_stem = nn.Sequential(ConvBN(cin, 384, 1), ConvBN(384, 448, 3, 1, 1),
                                     ConvBN(448, 512, 3, 1, 1))
        self.b3a = ConvBN(512, 256, 3, 1, 1)
        self.b3b = ConvBN(512, 256, 3, 1, 1)
        self.b4 = nn.Sequential(_AvgPool3s1(), ConvBN(cin, 256, 1))

    def forward(self, x):
        s2 = self.b2_stem(x)
        s3 = self.b3_stem(x)
        return torch.cat([self.b1(x), self.b2a(s2), self.b2b(s2),
                          self.b3a(s3), self.b3b(s3), self.b4(x)], dim=1)


class InceptionV4(nn.Module):
    def __init__(self, num_classes=1000, image_shape="3,299,299"):
        super().__init__()
        c = int(image_shape.split(",")[0])
        layers = [Stem(c)]
        layers += [IncA() for _ in range(4)]
        layers += [RedA()]
        layers += [IncB() for _ in range(7)]
        layers += [RedB()]
        layers += [IncC() for _ in range(3)]
        self.features = nn.Sequential(*layers)
        self.gap = GlobalAvgPool()
        self.drop = Dropout(0.2)
        self.fc = LinearBF16(1536, num_classes)

    def forward(self, x):
        return self.fc(self.drop(self.gap(self.features(x))))


def get_symbol(num_classes=1000, image_shape="3,299,299", **kwargs):
    return InceptionV4(num_classes=num_classes, image_shape=image_shape)
