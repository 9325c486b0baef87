"""MobileNet v1 and v2 (reference example/image-classification/symbols/
mobilenet.py, mobilenetv2.py). Depthwise convs route through
GroupedConv2dNHWC (torch/MIOpen substrate — the hand HIP convs are dense
gather kernels); pointwise 1x1s stay on the native path."""
from __future__ import annotations

import torch.nn as nn

from ..ops.layers import (BatchNorm2dNHWC, Conv2dNHWC, GlobalAvgPool,
                          GroupedConv2dNHWC, LinearBF16)


class ConvBN(nn.Module):
    def __init__(self, cin, cout, k, stride=1, pad=0, groups=1, relu=True):
        super().__init__()
        if groups == 1:
            self.conv = Conv2dNHWC(cin, cout, k, stride=stride, padding=pad)
        else:
            self.conv = GroupedConv2dNHWC(cin, cout, k, stride=stride,
                                          padding=pad, groups=groups)
        self.bn = BatchNorm2dNHWC(cout, fuse_relu=relu)

    def forward(self, x):
        return self.bn(self.conv(x))


class DepthwiseSep(nn.Module):
    """3x3 depthwise + 1x1 pointwise (reference mobilenet.py conv pairs)."""

    def __init__(self, cin, cout, stride):
        super().__init__()
        self.dw = ConvBN(cin, cin, 3, stride, 1, groups=cin)
        self.pw = ConvBN(cin, cout, 1)

    def forward(self, x):
        return self.pw(self.dw(x))


class MobileNet(nn.Module):
    def __init__(self, num_classes=1000, image_shape="3,224,224",
                 multiplier=1.0):
        super().__init__()
        c = int(image_shape.split(",")[0])
        def ch(n):
            return max(8, int(n * multiplier))
        cfg = [(64, 1), (128, 2), (128, 1), (256, 2), (256, 1), (512, 2),
               (512, 1), (512, 1), (512, 1), (512, 1), (512, 1), (1024, 2),
               (1024, 1)]
        layers = [ConvBN(c, ch(32), 3, 2, 1)]
        cin = ch(32)
        for cout, s in cfg:
            layers.append(DepthwiseSep(cin, ch(cout), s))
            cin = ch(cout)
        self.features = nn.Sequential(*layers)
        self.gap = GlobalAvgPool()
        self.fc = LinearBF16(cin, num_classes)

    def forward(self, x):
        return self.fc(self.gap(self.features(x)))


class InvertedResidual(nn.Module):
    """MobileNet-v2 bottleneck: 1x1 expand -> 3x3 dw -> 1x1 project (linear),
    residual when stride 1 and cin == cout."""

    def __init__(self, cin, cout, stride, expand):
        super().__init__()
        mid = cin * expand
        self.use_res = stride == 1 and cin == cout
        blocks = []
        if expand != 1:
            blocks.append(ConvBN(cin, mid, 1))
        blocks += [ConvBN(mid, mid, 3, stride, 1, groups=mid),
                   ConvBN(mid, cout, 1, relu=False)]
        self.body = nn.Sequential(*blocks)

    def forward(self, x):
        out = self.body(x)
        return x + out if self.use_res else out


class MobileNetV2(nn.Module):
    def __init__(self, num_classes=1000, image_shape="3,224,224",
                 multiplier=1.0):
        super().__init__()
        c = int(image_shape.split(",")[0])
        def ch(n):
            return max(8, int(n * multiplier))
        # (expand, cout, repeats, stride) — reference mobilenetv2.py table
        cfg = [(1, 16, 1, 1), (6, 24, 2, 2), (6, 32, 3, 2), (6, 64, 4, 2),
               (6, 96, 3, 1), (6, 160, 3, 2), (6, 320, 1, 1)]
        layers = [ConvBN(c, ch(32), 3, 2, 1)]
        cin = ch(32)
        for t, cout, n, s in cfg:
            for i in range(n):
                layers.append(InvertedResidual(cin, ch(cout), s if i == 0 else 1, t))
                cin = ch(cout)
        layers.append(ConvBN(cin, 1280, 1))
        self.features = nn.Sequential(*layers)
        self.gap = GlobalAvgPool()
        self.fc = LinearBF16(1280, num_classes)

    def forward(self, x):
        return self.fc(self.gap(self.features(x)))


def get_symbol(num_classes=1000, image_shape="3,224,224", multiplier=1.0, **kwargs):
    return MobileNet(num_classes=num_classes, image_shape=image_shape,
                     multiplier=multiplier)


def get_symbol_v2(num_classes=1000, image_shape="3,224,224", multiplier=1.0, **kwargs):
    return MobileNetV2(num_classes=num_classes, image_shape=image_shape,
                       multiplier=multiplier)
