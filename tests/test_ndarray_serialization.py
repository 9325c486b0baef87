"""Byte-level checks of the .params (V2 NDArray) format
(reference src/ndarray/ndarray.cc:1569-1801)."""
import struct

import numpy as np
import pytest
import torch

from dtmx import ndarray as nd


def test_roundtrip_dict(tmp_path):
    data = {
        "arg:fc1_weight": torch.randn(3, 4),
        "arg:fc1_bias": torch.randn(4),
        "aux:bn_running_mean": torch.randn(8),
        "scalar": torch.tensor(3.5),
    }
    f = str(tmp_path / "x.params")
    nd.save(f, data)
    loaded = nd.load(f)
    assert set(loaded.keys()) == set(data.keys())
    for k in data:
        assert torch.allclose(loaded[k], data[k])
        assert loaded[k].shape == data[k].shape


def test_roundtrip_list(tmp_path):
    f = str(tmp_path / "x.params")
    nd.save(f, [torch.arange(6, dtype=torch.float32).reshape(2, 3)])
    loaded = nd.load(f)
    assert isinstance(loaded, list) and len(loaded) == 1
    assert loaded[0][1, 2] == 5


def test_exact_bytes_match_reference_layout(tmp_path):
    """Hand-assemble the reference byte stream for one fp32 2x2 array and
    compare against nd.save output."""
    t = torch.tensor([[1.0, 2.0], [3.0, 4.0]])
    f = str(tmp_path / "x.params")
    nd.save(f, {"w": t})
    got = open(f, "rb").read()
    expect = b"".join([
        struct.pack("<QQ", 0x112, 0),            # file magic + reserved
        struct.pack("<Q", 1),                     # ndarray count
        struct.pack("<I", 0xF993FAC9),            # NDARRAY_V2_MAGIC
        struct.pack("<i", 0),                     # stype = default
        struct.pack("<I", 2), struct.pack("<qq", 2, 2),  # shape
        struct.pack("<ii", 1, 0),                 # context cpu(0)
        struct.pack("<i", 0),                     # type_flag f32
        np.array([[1, 2], [3, 4]], np.float32).tobytes(),
        struct.pack("<Q", 1),                     # name count
        struct.pack("<Q", 1), b"w",
    ])
    assert got == expect


def test_bf16_saved_as_f32(tmp_path):
    t = torch.randn(4, dtype=torch.bfloat16)
    f = str(tmp_path / "x.params")
    nd.save(f, {"w": t})
    loaded = nd.load(f)
    assert loaded["w"].dtype == torch.float32
    assert torch.allclose(loaded["w"], t.float())


def test_dtypes(tmp_path):
    for dtype in [torch.float64, torch.float16, torch.uint8, torch.int32, torch.int8, torch.int64]:
        t = (torch.arange(5) % 3).to(dtype)
        f = str(tmp_path / "x.params")
        nd.save(f, {"w": t})
        loaded = nd.load(f)
        assert loaded["w"].dtype == dtype
        assert torch.equal(loaded["w"], t)
