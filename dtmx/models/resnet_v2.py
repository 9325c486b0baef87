"""Pre-activation ResNet v2 (the reference's headline resnet topology:
example/image-classification/symbols/resnet.py, "Identity Mappings in Deep
Residual Networks" order bn->relu->conv). MI355X-native: the bn->relu pairs
use the fused-relu BN apply kernel; the residual join is a plain add (v2
has no post-join activation), and the network ends with a final bn->relu
before pooling."""
from __future__ import annotations

import torch.nn as nn

from ..ops.layers import (BatchNorm2dNHWC, Conv2dNHWC, GlobalAvgPool,
                          LinearBF16, MaxPool2dNHWC)


class PreactBasic(nn.Module):
    expansion = 1

    def __init__(self, in_ch, ch, stride=1):
        super().__init__()
        self.bn1 = BatchNorm2dNHWC(in_ch, fuse_relu=True)
        self.conv1 = Conv2dNHWC(in_ch, ch, 3, stride, 1)
        self.bn2 = BatchNorm2dNHWC(ch, fuse_relu=True)
        self.conv2 = Conv2dNHWC(ch, ch, 3, 1, 1)
        # v2 shortcut: 1x1 conv on the PRE-ACTIVATED input when shape changes
        self.down = (Conv2dNHWC(in_ch, ch, 1, stride)
                     if stride != 1 or in_ch != ch else None)

    def forward(self, x):
        a = self.bn1(x)
        sc = self.down(a) if self.down is not None else x
        y = self.conv1(a)
        y = self.conv2(self.bn2(y))
        return y + sc


class PreactBottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch, ch, stride=1):
        super().__init__()
        out = ch * 4
        self.bn1 = BatchNorm2dNHWC(in_ch, fuse_relu=True)
        self.conv1 = Conv2dNHWC(in_ch, ch, 1)
        self.bn2 = BatchNorm2dNHWC(ch, fuse_relu=True)
        self.conv2 = Conv2dNHWC(ch, ch, 3, stride, 1)
        self.bn3 = BatchNorm2dNHWC(ch, fuse_relu=True)
        self.conv3 = Conv2dNHWC(ch, out, 1)
        self.down = (Conv2dNHWC(in_ch, out, 1, stride)
                     if stride != 1 or in_ch != out else None)

    def forward(self, x):
        a = self.bn1(x)
        sc = self.down(a) if self.down is not None else x
        y = self.conv1(a)
        y = self.conv2(self.bn2(y))
        y = self.conv3(self.bn3(y))
        return y + sc


_CONFIGS = {
    18: (PreactBasic, [2, 2, 2, 2]),
    34: (PreactBasic, [3, 4, 6, 3]),
    50: (PreactBottleneck, [3, 4, 6, 3]),
    101: (PreactBottleneck, [3, 4, 23, 3]),
    152: (PreactBottleneck, [3, 8, 36, 3]),
}


class ResNetV2(nn.Module):
    def __init__(self, num_layers=50, num_classes=1000,
                 image_shape=(3, 224, 224)):
        super().__init__()
        if num_layers not in _CONFIGS:
            raise ValueError(f"unsupported resnet-v2 depth {num_layers}")
        block, layers = _CONFIGS[num_layers]
        small = image_shape[-1] <= 64
        ch = 64
        if small:
            self.stem = nn.Sequential(Conv2dNHWC(image_shape[0], ch, 3, 1, 1))
        else:
            self.stem = nn.Sequential(
                Conv2dNHWC(image_shape[0], ch, 7, 2, 3),
                BatchNorm2dNHWC(ch, fuse_relu=True),
                MaxPool2dNHWC(3, 2, 1),
            )
        stages = []
        in_ch = ch
        for i, n in enumerate(layers):
            stride = 1 if i == 0 else 2
            blocks = [block(in_ch, ch * (2 ** i), stride)]
            in_ch = ch * (2 ** i) * block.expansion
            for _ in range(1, n):
                blocks.append(block(in_ch, ch * (2 ** i)))
            stages.append(nn.Sequential(*blocks))
        self.stages = nn.Sequential(*stages)
        self.final_bn = BatchNorm2dNHWC(in_ch, fuse_relu=True)
        self.pool = GlobalAvgPool()
        self.fc = LinearBF16(in_ch, num_classes)

    def forward(self, x):
        x = self.stages(self.stem(x))
        return self.fc(self.pool(self.final_bn(x)))


def get_symbol(num_classes=1000, num_layers=50, image_shape="3,224,224",
               **kwargs):
    if isinstance(image_shape, str):
        image_shape = tuple(int(v) for v in image_shape.split(","))
    return ResNetV2(num_layers=num_layers, num_classes=num_classes,
                    image_shape=image_shape)
