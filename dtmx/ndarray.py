"""NDArray layer: torch tensors + MXNet `.params` (V2 NDArray) binary serialization.

dtmx rides on PyTorch-ROCm tensors for its array substrate (the reference's
NDArray/autograd layer, include/mxnet/ndarray.h:82-1053). What this module
keeps from the reference is the *on-disk* format, byte-for-byte:

file container (reference src/ndarray/ndarray.cc:1774-1801):
    uint64  0x112  (kMXAPINDArrayListMagic)
    uint64  0      (reserved)
    uint64  N      (number of arrays; dmlc vector serializer length prefix)
    N x NDArray   (see below)
    uint64  K      (number of names)
    K x { uint64 len; bytes }  (dmlc string serializer)

per-array V2 format (reference src/ndarray/ndarray.cc:1574-1640):
    uint32  0xF993fac9  (NDARRAY_V2_MAGIC)
    int32   stype        (0 = default/dense; sparse not emitted by dtmx)
    uint32  ndim; int64 dims[ndim]          (nnvm::TShape::Save, int64_t dim_t)
    int32   dev_type; int32 dev_id          (Context::Save, base.h:188-191)
    int32   type_flag                        (mshadow: 0=f32 1=f64 2=f16 3=u8 4=i32 5=i8 6=i64)
    raw data bytes (C-contiguous)

bf16 has no mshadow type id in the reference era; bf16 tensors are saved as
float32 so reference tooling can read dtmx checkpoints.
"""
from __future__ import annotations

import struct
from typing import Dict, List, Optional, Sequence, Tuple, Union

import numpy as np
import torch

_MAGIC_FILE = 0x112
_MAGIC_ND_V2 = 0xF993FAC9
_MAGIC_ND_V1 = 0xF993FAC8

# mshadow type flags <-> torch dtypes
_TYPE_FLAG_TO_TORCH = {
    0: torch.float32,
    1: torch.float64,
    2: torch.float16,
    3: torch.uint8,
    4: torch.int32,
    5: torch.int8,
    6: torch.int64,
}
_TORCH_TO_TYPE_FLAG = {v: k for k, v in _TYPE_FLAG_TO_TORCH.items()}


def _to_saveable(t: torch.Tensor) -> torch.Tensor:
    t = t.detach()
    if t.is_cuda:
        t = t.cpu()
    if t.dtype == torch.bfloat16:  # no mshadow id for bf16 — widen (see module docstring)
        t = t.float()
    if t.dtype not in _TORCH_TO_TYPE_FLAG:
        raise TypeError(f"dtype {t.dtype} not serializable to .params")
    return t.contiguous()


def _write_ndarray(out: List[bytes], t: torch.Tensor) -> None:
    t = _to_saveable(t)
    out.append(struct.pack("<I", _MAGIC_ND_V2))
    out.append(struct.pack("<i", 0))  # stype = kDefaultStorage
    shape = tuple(t.shape)
    out.append(struct.pack("<I", len(shape)))
    if shape:
        out.append(struct.pack(f"<{len(shape)}q", *shape))
    out.append(struct.pack("<ii", 1, 0))  # Context: cpu(0)
    out.append(struct.pack("<i", _TORCH_TO_TYPE_FLAG[t.dtype]))
    out.append(t.numpy().tobytes())


class _Reader:
    def __init__(self, buf: bytes):
        self.buf = buf
        self.pos = 0

    def read(self, fmt: str):
        size = struct.calcsize(fmt)
        vals = struct.unpack_from(fmt, self.buf, self.pos)
        self.pos += size
        return vals if len(vals) > 1 else vals[0]

    def read_bytes(self, n: int) -> bytes:
        b = self.buf[self.pos : self.pos + n]
        if len(b) != n:
            raise ValueError("truncated .params file")
        self.pos += n
        return b


def _read_ndarray(r: _Reader) -> torch.Tensor:
    magic = r.read("<I")
    if magic == _MAGIC_ND_V2:
        stype = r.read("<i")
        if stype != 0:
            raise NotImplementedError("sparse .params entries are not supported")
        ndim = r.read("<I")
    elif magic == _MAGIC_ND_V1:
        ndim = r.read("<I")
    else:
        # legacy: magic IS ndim, uint32 dims (ndarray.cc LegacyTShapeLoad)
        ndim = magic
        dims = [r.read("<I") for _ in range(ndim)] if ndim else []
        return _read_body(r, dims, legacy=True)
    dims = list(r.read(f"<{ndim}q")) if ndim > 1 else ([r.read("<q")] if ndim == 1 else [])
    return _read_body(r, dims, legacy=False)


def _read_body(r: _Reader, dims: Sequence[int], legacy: bool) -> torch.Tensor:
    r.read("<ii")  # context (ignored; tensors load to cpu)
    type_flag = r.read("<i")
    dtype = _TYPE_FLAG_TO_TORCH.get(type_flag)
    if dtype is None:
        raise ValueError(f"unknown mshadow type flag {type_flag}")
    count = 1
    for d in dims:
        count *= d
    raw = r.read_bytes(count * torch.tensor([], dtype=dtype).element_size())
    arr = np.frombuffer(bytearray(raw), dtype=_np_dtype(dtype)).reshape(dims)
    return torch.from_numpy(arr)


def _np_dtype(dtype: torch.dtype):
    return {
        torch.float32: np.float32,
        torch.float64: np.float64,
        torch.float16: np.float16,
        torch.uint8: np.uint8,
        torch.int32: np.int32,
        torch.int8: np.int8,
        torch.int64: np.int64,
    }[dtype]


def save(fname: str, data: Union[Dict[str, torch.Tensor], List[torch.Tensor], torch.Tensor]):
    """Save tensors to an MXNet-compatible `.params` file (nd.save parity)."""
    if isinstance(data, torch.Tensor):
        names: List[str] = []
        arrays = [data]
    elif isinstance(data, dict):
        names = list(data.keys())
        arrays = [data[k] for k in names]
    else:
        names = []
        arrays = list(data)
    out: List[bytes] = [struct.pack("<QQ", _MAGIC_FILE, 0), struct.pack("<Q", len(arrays))]
    for t in arrays:
        _write_ndarray(out, t)
    out.append(struct.pack("<Q", len(names)))
    for n in names:
        nb = n.encode("utf-8")
        out.append(struct.pack("<Q", len(nb)))
        out.append(nb)
    with open(fname, "wb") as f:
        f.write(b"".join(out))


def load(fname: str) -> Union[Dict[str, torch.Tensor], List[torch.Tensor]]:
    """Load a `.params` file. Returns a dict if names are present, else a list."""
    with open(fname, "rb") as f:
        r = _Reader(f.read())
    magic, _reserved = r.read("<QQ")
    if magic != _MAGIC_FILE:
        raise ValueError(f"not an NDArray file (magic {magic:#x})")
    n = r.read("<Q")
    arrays = [_read_ndarray(r) for _ in range(n)]
    k = r.read("<Q")
    if k == 0:
        return arrays
    if k != n:
        raise ValueError("invalid .params: name/array count mismatch")
    names = []
    for _ in range(k):
        ln = r.read("<Q")
        names.append(r.read_bytes(ln).decode("utf-8"))
    return dict(zip(names, arrays))


# -- small mx.nd-like conveniences used by examples/tests -------------------

def array(obj, ctx=None, dtype=None) -> torch.Tensor:
    t = torch.as_tensor(np.asarray(obj))
    if dtype is not None:
        t = t.to(dtype)
    if ctx is not None:
        t = t.to(ctx.torch_device())
    return t


def zeros(shape, ctx=None, dtype=torch.float32) -> torch.Tensor:
    dev = ctx.torch_device() if ctx is not None else None
    return torch.zeros(shape, dtype=dtype, device=dev)


def ones(shape, ctx=None, dtype=torch.float32) -> torch.Tensor:
    dev = ctx.torch_device() if ctx is not None else None
    return torch.ones(shape, dtype=dtype, device=dev)
