"""Random API (reference python/mxnet/random.py / src/operator/random/
sample_op.cu): seeding and the sampler set the examples use. Samplers ride
the torch-ROCm generators (rocRAND underneath); `seed()` seeds torch, numpy
and the HIP dropout op's counter base so a run is reproducible end to end."""
from __future__ import annotations

import os

import numpy as np
import torch


def seed(seed_state: int, ctx: str = "all"):
    """Global RNG seed (reference mx.random.seed)."""
    torch.manual_seed(seed_state)
    np.random.seed(seed_state & 0x7FFFFFFF)
    if torch.cuda.is_available() and ctx in ("all", "gpu"):
        torch.cuda.manual_seed_all(seed_state)
    os.environ["DTMX_SEED"] = str(seed_state)


def uniform(low=0.0, high=1.0, shape=(1,), dtype=torch.float32, ctx=None,
            out=None):
    dev = ctx.torch_device() if ctx is not None else "cpu"
    t = torch.empty(shape, dtype=dtype, device=dev).uniform_(low, high)
    if out is not None:
        out.copy_(t)
        return out
    return t


def normal(loc=0.0, scale=1.0, shape=(1,), dtype=torch.float32, ctx=None,
           out=None):
    dev = ctx.torch_device() if ctx is not None else "cpu"
    t = torch.empty(shape, dtype=dtype, device=dev).normal_(loc, scale)
    if out is not None:
        out.copy_(t)
        return out
    return t


def randint(low, high, shape=(1,), dtype=torch.int64, ctx=None):
    dev = ctx.torch_device() if ctx is not None else "cpu"
    return torch.randint(low, high, shape, dtype=dtype, device=dev)
