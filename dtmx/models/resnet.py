"""ResNet (reference example/image-classification/symbols/resnet.py:29-196).

NHWC/bf16-first build from dtmx layers. Supports 18/34/50/101/152 layers,
ImageNet (224x224) and CIFAR (32x32) variants. Uses fused BN+ReLU and fused
residual add+ReLU (one HBM round trip each — the MI355X BN/elementwise ops
are bandwidth-bound, SURVEY.md §7 hard part 2).
"""
from __future__ import annotations

import os

import torch
import torch.nn as nn

from ..ops.layers import (
    AddRelu,
    BatchNorm2dNHWC,
    Conv2dNHWC,
    GlobalAvgPool,
    LinearBF16,
    MaxPool2dNHWC,
)


def _use_fused_block(module, x):
    # whole-block manual backward with residual-join grad fusion
    # (ops/fusedblock.py); DTMX_FUSED_BLOCK=0 restores the layer-by-layer path
    return (module.training and x.is_cuda
            and os.environ.get("DTMX_FUSED_BLOCK", "1") == "1")


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_ch, ch, stride=1):
        super().__init__()
        self.conv1 = Conv2dNHWC(in_ch, ch, 3, stride, 1)
        self.bn1 = BatchNorm2dNHWC(ch, fuse_relu=True)
        self.conv2 = Conv2dNHWC(ch, ch, 3, 1, 1)
        self.bn2 = BatchNorm2dNHWC(ch, fuse_relu=True)  # fused +residual +relu
        self.downsample = None
        if stride != 1 or in_ch != ch:
            self.downsample = nn.Sequential(
                Conv2dNHWC(in_ch, ch, 1, stride, 0), BatchNorm2dNHWC(ch)
            )

    def forward(self, x):
        if _use_fused_block(self, x):
            from ..ops.fusedblock import fused_basic_block
            return fused_basic_block(x, self)
        sc = x if self.downsample is None else self.downsample(x)
        y = self.bn1(self.conv1(x))
        return self.bn2(self.conv2(y), sc)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch, ch, stride=1):
        super().__init__()
        out_ch = ch * self.expansion
        self.conv1 = Conv2dNHWC(in_ch, ch, 1, 1, 0)
        self.bn1 = BatchNorm2dNHWC(ch, fuse_relu=True)
        self.conv2 = Conv2dNHWC(ch, ch, 3, stride, 1)  # v1.5: stride on the 3x3
        self.bn2 = BatchNorm2dNHWC(ch, fuse_relu=True)
        self.conv3 = Conv2dNHWC(ch, out_ch, 1, 1, 0)
        self.bn3 = BatchNorm2dNHWC(out_ch, fuse_relu=True)  # fused +residual +relu
        self.downsample = None
        if stride != 1 or in_ch != out_ch:
            self.downsample = nn.Sequential(
                Conv2dNHWC(in_ch, out_ch, 1, stride, 0), BatchNorm2dNHWC(out_ch)
            )

    def forward(self, x):
        if _use_fused_block(self, x):
            from ..ops.fusedblock import fused_bottleneck
            return fused_bottleneck(x, self)
        sc = x if self.downsample is None else self.downsample(x)
        y = self.bn1(self.conv1(x))
        y = self.bn2(self.conv2(y))
        return self.bn3(self.conv3(y), sc)


_CONFIGS = {
    18: (BasicBlock, [2, 2, 2, 2]),
    34: (BasicBlock, [3, 4, 6, 3]),
    50: (Bottleneck, [3, 4, 6, 3]),
    101: (Bottleneck, [3, 4, 23, 3]),
    152: (Bottleneck, [3, 8, 36, 3]),
}


class ResNet(nn.Module):
    def __init__(self, num_layers=50, num_classes=1000, image_shape=(3, 224, 224)):
        super().__init__()
        if num_layers not in _CONFIGS:
            raise ValueError(f"unsupported resnet depth {num_layers}")
        block, layers = _CONFIGS[num_layers]
        self.spec = {
            "network": "resnet",
            "num_layers": num_layers,
            "num_classes": num_classes,
            "image_shape": list(image_shape),
        }
        small = image_shape[-1] <= 64  # CIFAR variant
        ch = 64
        if small:
            self.stem = nn.Sequential(
                Conv2dNHWC(image_shape[0], ch, 3, 1, 1),
                BatchNorm2dNHWC(ch, fuse_relu=True),
            )
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
        self.pool = GlobalAvgPool()
        self.fc = LinearBF16(in_ch, num_classes)

    def forward(self, x):
        x = self.stem(x)
        x = self.stages(x)
        x = self.pool(x)
        return self.fc(x)


def get_symbol(num_classes=1000, num_layers=50, image_shape="3,224,224", **kwargs):
    if isinstance(image_shape, str):
        image_shape = tuple(int(v) for v in image_shape.split(","))
    return ResNet(num_layers=num_layers, num_classes=num_classes, image_shape=image_shape)
