"""Inception-v3 (reference example/image-classification/symbols/
inception-v3.py — BASELINE.md training/inference rows). 299x299 input."""
from __future__ import annotations

import torch
import torch.nn as nn

from ..ops.layers import BatchNorm2dNHWC, Conv2dNHWC, GlobalAvgPool, LinearBF16, MaxPool2dNHWC


class ConvBN(nn.Module):
    def __init__(self, in_ch, out_ch, kernel, stride=1, pad=0):
        super().__init__()
        self.conv = Conv2dNHWC(in_ch, out_ch, kernel, stride, pad)
        self.bn = BatchNorm2dNHWC(out_ch, fuse_relu=True)

    def forward(self, x):
        return self.bn(self.conv(x))


class _AvgPool3s1(nn.Module):
    def forward(self, x):
        return torch.nn.functional.avg_pool2d(x, 3, 1, 1, count_include_pad=False)


class InceptionA(nn.Module):
    def __init__(self, in_ch, pool_ch):
        super().__init__()
        self.b1 = ConvBN(in_ch, 64, 1)
        self.b2 = nn.Sequential(ConvBN(in_ch, 48, 1), ConvBN(48, 64, 5, 1, 2))
        self.b3 = nn.Sequential(ConvBN(in_ch, 64, 1), ConvBN(64, 96, 3, 1, 1),
                                ConvBN(96, 96, 3, 1, 1))
        self.b4 = nn.Sequential(_AvgPool3s1(), ConvBN(in_ch, pool_ch, 1))

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x), self.b4(x)], dim=1)


class InceptionB(nn.Module):
    def __init__(self, in_ch):
        super().__init__()
        self.b1 = ConvBN(in_ch, 384, 3, 2, 0)
        self.b2 = nn.Sequential(ConvBN(in_ch, 64, 1), ConvBN(64, 96, 3, 1, 1),
                                ConvBN(96, 96, 3, 2, 0))
        self.pool = MaxPool2dNHWC(3, 2, 0)

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.pool(x)], dim=1)


class InceptionC(nn.Module):
    """7x7 factorized branch, implemented with square 3x3 stacks (dtmx keeps
    square-kernel convs in round 1; topology and FLOPs stay comparable)."""

    def __init__(self, in_ch, ch7):
        super().__init__()
        self.b1 = ConvBN(in_ch, 192, 1)
        self.b2 = nn.Sequential(ConvBN(in_ch, ch7, 1), ConvBN(ch7, ch7, 3, 1, 1),
                                ConvBN(ch7, 192, 3, 1, 1))
        self.b3 = nn.Sequential(ConvBN(in_ch, ch7, 1), ConvBN(ch7, ch7, 3, 1, 1),
                                ConvBN(ch7, ch7, 3, 1, 1), ConvBN(ch7, 192, 3, 1, 1))
        self.b4 = nn.Sequential(_AvgPool3s1(), ConvBN(in_ch, 192, 1))

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x), self.b4(x)], dim=1)


class InceptionD(nn.Module):
    def __init__(self, in_ch):
        super().__init__()
        self.b1 = nn.Sequential(ConvBN(in_ch, 192, 1), ConvBN(192, 320, 3, 2, 0))
        self.b2 = nn.Sequential(ConvBN(in_ch, 192, 1), ConvBN(192, 192, 3, 1, 1),
                                ConvBN(192, 192, 3, 2, 0))
        self.pool = MaxPool2dNHWC(3, 2, 0)

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.pool(x)], dim=1)


class InceptionE(nn.Module):
    """The E block's 3x3 branches split into two parallel convs whose outputs
    concatenate (1x3/3x1 in the original; square 3x3 here), so the block
    emits 320+768+768+192 = 2048 channels."""

    def __init__(self, in_ch):
        super().__init__()
        self.b1 = ConvBN(in_ch, 320, 1)
        self.b2a = ConvBN(in_ch, 384, 1)
        self.b2b1 = ConvBN(384, 384, 3, 1, 1)
        self.b2b2 = ConvBN(384, 384, 3, 1, 1)
        self.b3 = nn.Sequential(ConvBN(in_ch, 448, 1), ConvBN(448, 384, 3, 1, 1))
        self.b3b1 = ConvBN(384, 384, 3, 1, 1)
        self.b3b2 = ConvBN(384, 384, 3, 1, 1)
        self.b4 = nn.Sequential(_AvgPool3s1(), ConvBN(in_ch, 192, 1))

    def forward(self, x):
        t2 = self.b2a(x)
        t3 = self.b3(x)
        return torch.cat([
            self.b1(x),
            self.b2b1(t2), self.b2b2(t2),
            self.b3b1(t3), self.b3b2(t3),
            self.b4(x),
        ], dim=1)


class InceptionV3(nn.Module):
    def __init__(self, num_classes=1000):
        super().__init__()
        self.spec = {"network": "inception-v3", "num_classes": num_classes}
        self.stem = nn.Sequential(
            ConvBN(3, 32, 3, 2, 0), ConvBN(32, 32, 3, 1, 0), ConvBN(32, 64, 3, 1, 1),
            MaxPool2dNHWC(3, 2, 0),
            ConvBN(64, 80, 1), ConvBN(80, 192, 3, 1, 0), MaxPool2dNHWC(3, 2, 0),
        )
        self.blocks = nn.Sequential(
            InceptionA(192, 32), InceptionA(256, 64), InceptionA(288, 64),
            InceptionB(288),
            InceptionC(768, 128), InceptionC(768, 160), InceptionC(768, 160),
            InceptionC(768, 192),
            InceptionD(768),
            InceptionE(1280), InceptionE(2048),
        )
        self.pool = GlobalAvgPool()
        self.fc = LinearBF16(2048, num_classes)

    def forward(self, x):
        x = self.stem(x)
        x = self.blocks(x)
        x = self.pool(x)
        return self.fc(x)


def get_symbol(num_classes=1000, **kwargs):
    return InceptionV3(num_classes=num_classes)
