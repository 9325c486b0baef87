"""`dtmx.nd` — the mx.nd-style array namespace (reference
python/mxnet/ndarray/ndarray.py op surface). dtmx's NDArray IS the torch
tensor (north-star substrate), so these are thin, mxnet-signature wrappers:
the ops a reference user reaches for first (`nd.array`, `nd.zeros`,
`nd.dot`, `nd.concat`, `nd.one_hot`, `nd.save/load`, ...), returning plain
torch tensors that interoperate with the rest of dtmx."""
from __future__ import annotations

from typing import Optional

import torch

# byte-compatible .params serialization lives in dtmx.ndarray
from .ndarray import array, load, save, zeros  # noqa: F401

NDArray = torch.Tensor


def _dev(ctx):
    if ctx is None:
        return None
    return ctx.torch_device() if hasattr(ctx, "torch_device") else ctx


def ones(shape, ctx=None, dtype=torch.float32):
    return torch.ones(shape, dtype=dtype, device=_dev(ctx))


def full(shape, val, ctx=None, dtype=torch.float32):
    return torch.full(shape if isinstance(shape, (tuple, list)) else (shape,),
                      val, dtype=dtype, device=_dev(ctx))


def arange(start, stop=None, step=1.0, ctx=None, dtype=torch.float32):
    if stop is None:
        start, stop = 0, start
    return torch.arange(start, stop, step, dtype=dtype, device=_dev(ctx))


def zeros_like(t):
    return torch.zeros_like(t)


def ones_like(t):
    return torch.ones_like(t)


def dot(a, b):
    return a @ b


def concat(*args, dim: int = 1, **kwargs):
    # mxnet spells the axis kwarg `dim` (ndarray.concat)
    return torch.cat(list(args), dim=kwargs.get("axis", dim))


def stack(*args, axis: int = 0):
    return torch.stack(list(args), dim=axis)


def split(t, num_outputs: int, axis: int = 1, squeeze_axis: bool = False):
    outs = torch.chunk(t, num_outputs, dim=axis)
    if squeeze_axis:
        outs = [o.squeeze(axis) for o in outs]
    return list(outs)


def one_hot(indices, depth: int, on_value: float = 1.0, off_value: float = 0.0):
    oh = torch.nn.functional.one_hot(indices.long(), depth).float()
    return oh * (on_value - off_value) + off_value


def clip(t, a_min: float, a_max: float):
    return torch.clamp(t, a_min, a_max)


def mean(t, axis=None, keepdims: bool = False):
    if axis is None:
        return t.mean()
    return t.mean(dim=axis, keepdim=keepdims)


def sum(t, axis=None, keepdims: bool = False):  # noqa: A001 (mx.nd name)
    if axis is None:
        return t.sum()
    return t.sum(dim=axis, keepdim=keepdims)


def max(t, axis=None, keepdims: bool = False):  # noqa: A001
    if axis is None:
        return t.max()
    return t.max(dim=axis, keepdim=keepdims).values


def argmax(t, axis: Optional[int] = None):
    return t.argmax() if axis is None else t.argmax(dim=axis)


def softmax(t, axis: int = -1):
    return torch.softmax(t.float(), dim=axis).to(t.dtype)


def log_softmax(t, axis: int = -1):
    return torch.log_softmax(t.float(), dim=axis).to(t.dtype)


def relu(t):
    from .ops import functional as DF
    return DF.relu(t)


def sigmoid(t):
    return torch.sigmoid(t)


def tanh(t):
    return torch.tanh(t)


def exp(t):
    return torch.exp(t)


def log(t):
    return torch.log(t)


def sqrt(t):
    return torch.sqrt(t)


def square(t):
    return t * t


def abs(t):  # noqa: A001
    return torch.abs(t)


def transpose(t, axes=None):
    # no axes = reverse all dims (mx.nd.transpose semantics)
    if axes is None:
        axes = tuple(range(t.dim() - 1, -1, -1))
    return t.permute(axes)


def reshape(t, shape):
    return t.reshape(shape)


def broadcast_to(t, shape):
    return t.expand(shape)


def where(cond, a, b):
    return torch.where(cond.bool(), a, b)


def topk(t, k: int = 1, axis: int = -1, ret_typ: str = "indices"):
    vals, idx = torch.topk(t, k, dim=axis)
    if ret_typ == "value":
        return vals
    if ret_typ == "both":
        return vals, idx
    return idx


def norm(t):
    return t.float().norm()


def waitall():
    """Engine drain (reference mx.nd.waitall): synchronize the GPU stream."""
    if torch.cuda.is_available():
        torch.cuda.synchronize()


def expand_dims(t, axis: int):
    return t.unsqueeze(axis)


def flip(t, axis):
    return torch.flip(t, [axis] if isinstance(axis, int) else list(axis))


def tile(t, reps):
    reps = tuple(reps) if isinstance(reps, (tuple, list)) else (reps,)
    if len(reps) < t.dim():  # mx.nd.tile pads reps with 1s on the left
        reps = (1,) * (t.dim() - len(reps)) + reps
    return t.repeat(reps)


def repeat(t, repeats: int, axis=None):
    if axis is None:
        return t.reshape(-1).repeat_interleave(repeats)
    return t.repeat_interleave(repeats, dim=axis)


def maximum(a, b):
    return torch.maximum(a, torch.as_tensor(b, dtype=a.dtype, device=a.device))


def minimum(a, b):
    return torch.minimum(a, torch.as_tensor(b, dtype=a.dtype, device=a.device))


def flatten(t):
    return t.reshape(t.shape[0], -1)


def squeeze(t, axis=None):
    return t.squeeze() if axis is None else t.squeeze(axis)


def take(t, indices, axis: int = 0):
    from .ops import functional as DF
    if axis == 0 and t.dim() == 2:
        return DF.take(t, indices)
    return torch.index_select(t, axis, indices.long().reshape(-1))
