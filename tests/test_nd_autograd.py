"""dtmx.nd / dtmx.autograd surface (reference mx.nd + mx.autograd idioms a
porting user hits first)."""
import math

import pytest
import torch

import dtmx
from dtmx import autograd, nd


def test_nd_creation_and_ops():
    a = nd.array([[1.0, 2.0], [3.0, 4.0]])
    assert isinstance(a, torch.Tensor) and a.shape == (2, 2)
    z = nd.zeros((2, 3))
    o = nd.ones((2, 3))
    f = nd.full((2,), 7.0)
    r = nd.arange(5)
    assert z.sum() == 0 and o.sum() == 6 and f[0] == 7 and r[-1] == 4
    assert nd.dot(a, a).shape == (2, 2)
    c = nd.concat(z, o, dim=1)
    assert c.shape == (2, 6)
    s = nd.stack(z, o, axis=0)
    assert s.shape == (2, 2, 3)
    parts = nd.split(c, 2, axis=1)
    assert len(parts) == 2 and parts[0].shape == (2, 3)


def test_nd_math_matches_closed_forms():
    t = nd.array([[0.0, 1.0, 2.0]])
    oh = nd.one_hot(nd.array([1]).long(), 3)
    assert oh.tolist() == [[0.0, 1.0, 0.0]]
    assert nd.clip(t, 0.5, 1.5).tolist() == [[0.5, 1.0, 1.5]]
    assert nd.mean(t).item() == 1.0
    assert nd.sum(t, axis=1).tolist() == [3.0]
    assert nd.argmax(t, axis=1).tolist() == [2]
    sm = nd.softmax(t)
    assert abs(sm.sum().item() - 1.0) < 1e-5
    assert nd.topk(t, k=2).tolist() == [[2, 1]]
    assert nd.transpose(t).shape == (3, 1)
    assert nd.where(t > 0.5, nd.ones_like(t), nd.zeros_like(t)).tolist() == [[0.0, 1.0, 1.0]]
    assert abs(nd.norm(t).item() - math.sqrt(5.0)) < 1e-6
    nd.waitall()  # no-op on CPU, must not raise


def test_nd_shape_ops():
    t = torch.arange(6.0).reshape(2, 3)
    assert nd.expand_dims(t, 0).shape == (1, 2, 3)
    assert nd.flip(t, 1)[0].tolist() == [2.0, 1.0, 0.0]
    assert nd.tile(torch.ones(2), 3).shape == (6,)
    assert nd.repeat(torch.tensor([1.0, 2.0]), 2).tolist() == [1.0, 1.0, 2.0, 2.0]
    assert nd.maximum(t, 2.0).max() == 5.0 and nd.maximum(t, 2.0).min() == 2.0
    assert nd.minimum(t, 2.0).max() == 2.0
    assert nd.flatten(torch.ones(2, 3, 4)).shape == (2, 12)
    assert nd.squeeze(torch.ones(1, 3, 1)).shape == (3,)
    assert nd.squeeze(torch.ones(1, 3, 1), 0).shape == (3, 1)


def test_nd_save_load_roundtrip(tmp_path):
    f = str(tmp_path / "x.params")
    nd.save(f, {"w": torch.randn(3, 3)})
    back = nd.load(f)
    assert set(back) == {"w"} and back["w"].shape == (3, 3)


def test_autograd_record_backward_grad():
    x = torch.randn(4, requires_grad=True)
    with autograd.record():
        y = (x * x).sum()
    autograd.backward(y)
    torch.testing.assert_close(x.grad, 2 * x.detach())

    x2 = torch.randn(3, requires_grad=True)
    with autograd.record():
        y2 = (x2 ** 3).sum()
    (g,) = autograd.grad(y2, [x2])
    torch.testing.assert_close(g, 3 * x2.detach() ** 2)

    with autograd.pause():
        assert not autograd.is_recording()
    assert autograd.is_recording()  # default torch state


def test_autograd_mark_variables():
    v = torch.randn(2)
    buf = torch.zeros(2)
    autograd.mark_variables([v], [buf])
    with autograd.record():
        (v * 5).sum().backward()
    assert v.grad is buf or torch.equal(v.grad, torch.full((2,), 5.0))
