"""2-bit gradient compression (reference tests/python/unittest/test_kvstore.py
compute_expected_2bit_quantization + dist_sync_kvstore.py compression checks)."""
import numpy as np
import pytest
import torch

from dtmx.parallel.compression import TwoBitCompression


def reference_2bit(grad, residual, threshold):
    g = grad + residual
    q = np.where(g >= threshold, threshold, np.where(g <= -threshold, -threshold, 0.0))
    return q, g - q


def test_cpu_quantize_matches_reference():
    rng = np.random.RandomState(0)
    g = rng.randn(1000).astype(np.float32)
    comp = TwoBitCompression(0.5)
    res = np.zeros(1000, dtype=np.float32)
    out = comp.compress_decompress(torch.from_numpy(g.copy()))
    q, res = reference_2bit(g, np.zeros_like(g), 0.5)
    assert np.allclose(out.numpy(), q)
    # second round uses the residual
    out2 = comp.compress_decompress(torch.from_numpy(g.copy()))
    q2, _ = reference_2bit(g, res, 0.5)
    assert np.allclose(out2.numpy(), q2)


def test_residuals_keyed_per_kvstore_key():
    """Two SAME-SHAPED keys pushed alternately must keep independent
    error-feedback residuals (advisor finding: metadata-keyed residuals
    aliased same-shaped params; reference keeps residual_[key] per key,
    kvstore_dist.h:778). Repro: 0.3-gradients quantize to 0 with residual
    0.3; an aliased residual would make the SECOND key emit 0.5."""
    comp = TwoBitCompression(0.5)
    ga = torch.full((8,), 0.3)
    gb = torch.full((8,), 0.3)
    out_a = comp.compress_decompress(ga.clone(), key=1)
    out_b = comp.compress_decompress(gb.clone(), key=2)
    assert out_a.eq(0).all()
    assert out_b.eq(0).all(), "key 2 consumed key 1's residual"
    # round 2: each key's own residual (0.3) + 0.3 crosses the threshold
    assert comp.compress_decompress(ga.clone(), key=1).eq(0.5).all()
    assert comp.compress_decompress(gb.clone(), key=2).eq(0.5).all()


def test_pack_unpack_roundtrip():
    comp = TwoBitCompression(0.25)
    g = torch.randn(100)
    res = torch.zeros(100)
    packed = comp.quantize(g, res)
    assert packed.numel() == (100 + 15) // 16
    out = comp.dequantize(packed, 100)
    expect = torch.where(g >= 0.25, torch.tensor(0.25),
                         torch.where(g <= -0.25, torch.tensor(-0.25),
                                     torch.tensor(0.0)))
    assert torch.allclose(out, expect)


@pytest.mark.gpu
def test_hip_quantize_matches_cpu():
    comp_gpu = TwoBitCompression(0.5)
    comp_cpu = TwoBitCompression(0.5)
    g = (torch.randn(4096) * 0.8).bfloat16()
    for _ in range(3):  # residual evolves across rounds
        out_gpu = comp_gpu.compress_decompress(g.to("cuda:0"))
        out_cpu = comp_cpu.compress_decompress(g.clone())
        assert torch.allclose(out_gpu.cpu().float(), out_cpu.float(), atol=1e-3)
