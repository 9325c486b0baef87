from . import functional  # noqa: F401
from .layers import (  # noqa: F401
    Conv2dNHWC,
    BatchNorm2dNHWC,
    ReLU,
    MaxPool2dNHWC,
    GlobalAvgPool,
    LinearBF16,
    AddRelu,
)
