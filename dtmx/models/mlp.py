"""MLP (reference example/image-classification/symbols/mlp.py): the CPU
plumbing-check model (BASELINE.json config 1)."""
from __future__ import annotations

import torch.nn as nn

from ..ops.layers import LinearBF16, ReLU


class MLP(nn.Module):
    def __init__(self, num_classes=10, input_dim=784, hidden=(128, 64)):
        super().__init__()
        self.spec = {"network": "mlp", "num_classes": num_classes,
                     "input_dim": input_dim, "hidden": list(hidden)}
        dims = [input_dim] + list(hidden)
        layers = []
        for i in range(len(hidden)):
            layers += [LinearBF16(dims[i], dims[i + 1]), ReLU()]
        layers.append(LinearBF16(dims[-1], num_classes))
        self.net = nn.Sequential(*layers)

    def forward(self, x):
        if x.dim() > 2:
            x = x.reshape(x.shape[0], -1)
        return self.net(x)


def get_symbol(num_classes=10, input_dim=784, **kwargs):
    return MLP(num_classes=num_classes, input_dim=input_dim)
