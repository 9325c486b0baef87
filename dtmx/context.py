"""Device context, mirroring mxnet.context (reference: include/mxnet/base.h:142-210,
python/mxnet/context.py) on top of torch devices.

MXNet serializes a context as (int32 dev_type, int32 dev_id) with dev_type
1=cpu, 2=gpu, 3=cpu_pinned, 5=cpu_shared (base.h Context::Save). We keep the
same numbering so `.params` files round-trip byte-identically.
"""
from __future__ import annotations

import torch


class Context:
    devtype2str = {1: "cpu", 2: "gpu", 3: "cpu_pinned", 5: "cpu_shared"}
    devstr2type = {"cpu": 1, "gpu": 2, "cpu_pinned": 3, "cpu_shared": 5}

    def __init__(self, device_type: str, device_id: int = 0):
        if device_type not in self.devstr2type:
            raise ValueError(f"unknown device type {device_type}")
        self.device_type = device_type
        self.device_id = device_id

    @property
    def device_typeid(self) -> int:
        return self.devstr2type[self.device_type]

    def torch_device(self) -> torch.device:
        if self.device_type == "gpu":
            return torch.device("cuda", self.device_id)  # "cuda" IS ROCm/HIP on torch-rocm
        return torch.device("cpu")

    def __eq__(self, other):
        return (
            isinstance(other, Context)
            and self.device_type == other.device_type
            and self.device_id == other.device_id
        )

    def __hash__(self):
        return hash((self.device_type, self.device_id))

    def __repr__(self):
        return f"{self.device_type}({self.device_id})"


def cpu(device_id: int = 0) -> Context:
    return Context("cpu", device_id)


def gpu(device_id: int = 0) -> Context:
    return Context("gpu", device_id)


def current_context() -> Context:
    return cpu()


def num_gpus() -> int:
    return torch.cuda.device_count() if torch.cuda.is_available() else 0
