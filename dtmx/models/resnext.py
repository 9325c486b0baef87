"""ResNeXt (reference example/image-classification/symbols/resnext.py):
bottleneck blocks whose 3x3 is a grouped conv (cardinality 32, width 4).
Grouped 3x3s route through GroupedConv2dNHWC (torch/MIOpen substrate); the
1x1s and BNs stay on the native HIP path with fused relu/residual."""
from __future__ import annotations

import torch.nn as nn

from ..ops.layers import (BatchNorm2dNHWC, Conv2dNHWC, GlobalAvgPool,
                          GroupedConv2dNHWC, LinearBF16, MaxPool2dNHWC)


class ResNeXtBlock(nn.Module):
    def __init__(self, cin, ch, stride=1, cardinality=32, base_width=4):
        super().__init__()
        mid = cardinality * base_width * (ch // 64)
        out = ch * 4
        self.conv1 = Conv2dNHWC(cin, mid, 1)
        self.bn1 = BatchNorm2dNHWC(mid, fuse_relu=True)
        self.conv2 = GroupedConv2dNHWC(mid, mid, 3, stride, 1,
                                       groups=cardinality)
        self.bn2 = BatchNorm2dNHWC(mid, fuse_relu=True)
        self.conv3 = Conv2dNHWC(mid, out, 1)
        self.bn3 = BatchNorm2dNHWC(out, fuse_relu=True)  # +residual +relu
        if stride != 1 or cin != out:
            self.down = Conv2dNHWC(cin, out, 1, stride=stride)
            self.down_bn = BatchNorm2dNHWC(out)
        else:
            self.down = None

    def forward(self, x):
        sc = self.down_bn(self.down(x)) if self.down is not None else x
        y = self.bn1(self.conv1(x))
        y = self.bn2(self.conv2(y))
        return self.bn3(self.conv3(y), residual=sc)


class ResNeXt(nn.Module):
    def __init__(self, num_layers=50, num_classes=1000,
                 image_shape="3,224,224", cardinality=32, base_width=4):
        super().__init__()
        units = {50: [3, 4, 6, 3], 101: [3, 4, 23, 3], 152: [3, 8, 36, 3]}
        if num_layers not in units:
            raise ValueError(f"resnext: unsupported depth {num_layers}")
        c = int(image_shape.split(",")[0])
        layers = [Conv2dNHWC(c, 64, 7, 2, 3), BatchNorm2dNHWC(64, fuse_relu=True),
                  MaxPool2dNHWC(3, 2, 1)]
        cin = 64
        for stage, n in enumerate(units[num_layers]):
            ch = 64 * (2 ** stage)
            for i in range(n):
                layers.append(ResNeXtBlock(cin, ch, 2 if (i == 0 and stage > 0) else 1,
                                           cardinality, base_width))
                cin = ch * 4
        self.features = nn.Sequential(*layers)
        self.gap = GlobalAvgPool()
        self.fc = LinearBF16(cin, num_classes)

    def forward(self, x):
        return self.fc(self.gap(self.features(x)))


def get_symbol(num_classes=1000, num_layers=50, image_shape="3,224,224",
               **kwargs):
    return ResNeXt(num_layers=num_layers, num_classes=num_classes,
                   image_shape=image_shape)
