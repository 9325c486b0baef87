"""Numeric test harness (reference python/mxnet/test_utils.py:470,790,1207):
assert_almost_equal, finite-difference gradient checking, and cross-device
consistency checks — the operator oracle the dtmx test suite uses."""
from __future__ import annotations

from typing import Callable, Sequence

import numpy as np
import torch


def assert_almost_equal(a, b, rtol: float = 1e-5, atol: float = 1e-8, names=("a", "b")):
    a = a.detach().cpu().float() if isinstance(a, torch.Tensor) else torch.as_tensor(a)
    b = b.detach().cpu().float() if isinstance(b, torch.Tensor) else torch.as_tensor(b)
    torch.testing.assert_close(a, b, rtol=rtol, atol=atol,
                               msg=lambda m: f"{names[0]} vs {names[1]}: {m}")


def check_numeric_gradient(fn: Callable, inputs: Sequence[torch.Tensor],
                           eps: float = 1e-3, rtol: float = 0.02, atol: float = 1e-3):
    """Central-difference check of fn's analytic gradients (reference
    test_utils.py:790). fn maps tensors -> scalar; inputs are fp64 leaves."""
    leaves = [t.detach().double().requires_grad_(True) for t in inputs]
    out = fn(*leaves)
    out.backward()
    for li, leaf in enumerate(leaves):
        flat = leaf.detach().reshape(-1)
        num = torch.zeros_like(flat)
        for i in range(flat.numel()):
            orig = flat[i].item()
            flat[i] = orig + eps
            hi = fn(*leaves).item()
            flat[i] = orig - eps
            lo = fn(*leaves).item()
            flat[i] = orig
            num[i] = (hi - lo) / (2 * eps)
        assert_almost_equal(leaf.grad.reshape(-1), num, rtol=rtol, atol=atol,
                            names=(f"analytic[{li}]", f"numeric[{li}]"))


def check_consistency(fn: Callable, inputs: Sequence[torch.Tensor],
                      devices=("cpu", "cuda:0"), rtol=0.02, atol=1e-2):
    """Run fn on each device and compare outputs (reference
    test_utils.py:1207 check_consistency cpu-vs-gpu)."""
    outs = []
    for dev in devices:
        moved = [t.to(dev) for t in inputs]
        o = fn(*moved)
        outs.append(o.detach().float().cpu())
    for o in outs[1:]:
        assert_almost_equal(outs[0], o, rtol=rtol, atol=atol,
                            names=(devices[0], "other"))
    return outs


def rand_ndarray(shape, dtype=torch.float32, scale=1.0, seed=None) -> torch.Tensor:
    g = torch.Generator()
    if seed is not None:
        g.manual_seed(seed)
    return (torch.randn(*shape, generator=g) * scale).to(dtype)


def same(a, b) -> bool:
    """Exact equality (reference test_utils.py:454 same)."""
    a = a.detach().cpu() if isinstance(a, torch.Tensor) else torch.as_tensor(a)
    b = b.detach().cpu() if isinstance(b, torch.Tensor) else torch.as_tensor(b)
    return bool(torch.equal(a, b))


def assert_almost_equal_ignore_nan(a, b, rtol: float = 1e-5,
                                   atol: float = 1e-8, names=("a", "b")):
    """Compare ignoring positions where EITHER side is NaN (reference
    test_utils.py:519)."""
    a = a.detach().cpu().float() if isinstance(a, torch.Tensor) else torch.as_tensor(a, dtype=torch.float32)
    b = b.detach().cpu().float() if isinstance(b, torch.Tensor) else torch.as_tensor(b, dtype=torch.float32)
    mask = ~(torch.isnan(a) | torch.isnan(b))
    assert_almost_equal(a[mask], b[mask], rtol=rtol, atol=atol, names=names)


def default_context() -> torch.device:
    """The default test device (reference test_utils.py:53 default_context):
    cuda:0 when a GPU is visible, else cpu."""
    return torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
