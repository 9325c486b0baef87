"""Inception-BN (GoogLeNet v2; reference
example/image-classification/symbols/inception-bn.py:36-145) built
MI355X-native: NHWC Conv + fused-ReLU BatchNorm HIP kernels, channel concat,
7x7 global average pool. Includes the reference's <=28px "simple" variant
(SimpleFactory/DownsampleFactory) for CIFAR shapes."""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.layers import (BatchNorm2dNHWC, Conv2dNHWC, GlobalAvgPool,
                          LinearBF16, MaxPool2dNHWC)


class ConvBNRelu(nn.Module):
    """ConvFactory: conv -> BN -> relu (relu fused into the BN apply pass)."""

    def __init__(self, cin, cout, k, stride=1, pad=0):
        super().__init__()
        self.conv = Conv2dNHWC(cin, cout, k, stride=stride, padding=pad, bias=False)
        self.bn = BatchNorm2dNHWC(cout, eps=1e-10 + 1e-5, momentum=0.9,
                                  fuse_relu=True)

    def forward(self, x):
        return self.bn(self.conv(x))


class _AvgPool3s1(nn.Module):
    def forward(self, x):
        return F.avg_pool2d(x, 3, 1, 1, count_include_pad=True)


class IncA(nn.Module):
    """InceptionFactoryA: 1x1 | 1x1->3x3 | 1x1->3x3->3x3 | pool->1x1."""

    def __init__(self, cin, n1, n3r, n3, d3r, d3, pool, proj):
        super().__init__()
        self.b1 = ConvBNRelu(cin, n1, 1)
        self.b2 = nn.Sequential(ConvBNRelu(cin, n3r, 1), ConvBNRelu(n3r, n3, 3, 1, 1))
        self.b3 = nn.Sequential(ConvBNRelu(cin, d3r, 1), ConvBNRelu(d3r, d3, 3, 1, 1),
                                ConvBNRelu(d3, d3, 3, 1, 1))
        pool_l = _AvgPool3s1() if pool == "avg" else MaxPool2dNHWC(3, 1, 1)
        self.b4 = nn.Sequential(pool_l, ConvBNRelu(cin, proj, 1))
        self.out_channels = n1 + n3 + d3 + proj

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x), self.b4(x)], dim=1)


class IncB(nn.Module):
    """InceptionFactoryB (grid reduction): 1x1->3x3/2 | 1x1->3x3->3x3/2 | maxpool/2."""

    def __init__(self, cin, n3r, n3, d3r, d3):
        super().__init__()
        self.b1 = nn.Sequential(ConvBNRelu(cin, n3r, 1), ConvBNRelu(n3r, n3, 3, 2, 1))
        self.b2 = nn.Sequential(ConvBNRelu(cin, d3r, 1), ConvBNRelu(d3r, d3, 3, 1, 1),
                                ConvBNRelu(d3, d3, 3, 2, 1))
        self.pool = MaxPool2dNHWC(3, 2, 1)
        self.out_channels = n3 + d3 + cin

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.pool(x)], dim=1)


class Simple(nn.Module):
    """SimpleFactory (<=28px variant): 1x1 | 3x3."""

    def __init__(self, cin, c1, c3):
        super().__init__()
        self.b1 = ConvBNRelu(cin, c1, 1)
        self.b2 = ConvBNRelu(cin, c3, 3, 1, 1)
        self.out_channels = c1 + c3

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x)], dim=1)


class Down(nn.Module):
    """DownsampleFactory (<=28px variant): 3x3/2 | maxpool/2."""

    def __init__(self, cin, c3):
        super().__init__()
        self.b1 = ConvBNRelu(cin, c3, 3, 2, 1)
        self.pool = MaxPool2dNHWC(3, 2, 1)
        self.out_channels = c3 + cin

    def forward(self, x):
        return torch.cat([self.b1(x), self.pool(x)], dim=1)


class InceptionBN(nn.Module):
    def __init__(self, num_classes=1000, image_shape="3,224,224"):
        super().__init__()
        c, h, w = (int(v) for v in image_shape.split(","))
        if h <= 28:  # reference "simpler version"
            blocks = [ConvBNRelu(c, 96, 3, 1, 1),
                      Simple(96, 32, 32), Simple(64, 32, 48), Down(80, 80),
                      Simple(160, 112, 48), Simple(160, 96, 64),
                      Simple(160, 80, 80), Simple(160, 48, 96), Down(144, 96),
                      Simple(240, 176, 160), Simple(336, 176, 160)]
            feat = 336
        else:
            blocks = [ConvBNRelu(c, 64, 7, 2, 3), MaxPool2dNHWC(3, 2, 0),
                      ConvBNRelu(64, 64, 1), ConvBNRelu(64, 192, 3, 1, 1),
                      MaxPool2dNHWC(3, 2, 0),
                      IncA(192, 64, 64, 64, 64, 96, "avg", 32),
                      IncA(256, 64, 64, 96, 64, 96, "avg", 64),
                      IncB(320, 128, 160, 64, 96),
                      IncA(576, 224, 64, 96, 96, 128, "avg", 128),
                      IncA(576, 192, 96, 128, 96, 128, "avg", 128),
                      IncA(576, 160, 128, 160, 128, 160, "avg", 128),
                      IncA(608, 96, 128, 192, 160, 192, "avg", 128),
                      IncB(608, 128, 192, 192, 256),
                      IncA(1056, 352, 192, 320, 160, 224, "avg", 128),
                      IncA(1024, 352, 192, 320, 192, 224, "max", 128)]
            feat = 1024
        self.features = nn.Sequential(*blocks)
        self.gap = GlobalAvgPool()
        self.fc = LinearBF16(feat, num_classes)

    def forward(self, x):
        return self.fc(self.gap(self.features(x)))


def get_symbol(num_classes=1000, image_shape="3,224,224", **kwargs):
    return InceptionBN(num_classes=num_classes, image_shape=image_shape)
