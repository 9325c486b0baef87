"""Module training-loop tests (reference tests/python/unittest/test_module.py,
tests/python/train/test_mlp.py)."""
import numpy as np
import pytest
import torch

import dtmx
from dtmx import Module, cpu
from dtmx.io import MNISTIter, NDArrayIter, SyntheticDataIter
from dtmx.models import get_symbol


def _mlp_module():
    net = get_symbol("mlp", num_classes=10, input_dim=784)
    mod = Module(net, context=cpu())
    mod.bind(data_shapes=[("data", (32, 784))], label_shapes=[("softmax_label", (32,))])
    return mod


def test_mlp_learns_synthetic_mnist():
    torch.manual_seed(0)
    np.random.seed(0)
    it = MNISTIter(batch_size=32, flat=True, num_examples=1024, seed=1)
    mod = _mlp_module()
    mod.fit(it, num_epoch=12, optimizer="sgd",
            optimizer_params=(("learning_rate", 0.1), ("momentum", 0.9)))
    score = mod.score(it, "acc")
    assert dict(score)["accuracy"] > 0.9, score


def test_forward_backward_update_changes_params():
    mod = _mlp_module()
    mod.init_params()
    mod.init_optimizer(optimizer_params=(("learning_rate", 0.1),))
    it = MNISTIter(batch_size=32, flat=True, num_examples=64)
    batch = it.next()
    before = [p.detach().clone() for p in mod.symbol.parameters()]
    mod.forward_backward(batch)
    mod.update()
    after = list(mod.symbol.parameters())
    assert any(not torch.allclose(b, a) for b, a in zip(before, after))


def test_checkpoint_roundtrip(tmp_path):
    mod = _mlp_module()
    mod.init_params()
    prefix = str(tmp_path / "mlp")
    mod.save_checkpoint(prefix, 3)
    sym, arg, aux = dtmx.model.load_checkpoint(prefix, 3)
    assert sym["network"] == "mlp"
    mod2 = _mlp_module()
    mod2.init_params(arg_params=arg, aux_params=aux, force_init=True)
    a1, _ = mod.get_params()
    a2, _ = mod2.get_params()
    for k in a1:
        assert torch.allclose(a1[k], a2[k])


def test_resnet_cpu_forward_backward():
    net = get_symbol("resnet", num_layers=18, num_classes=10, image_shape="3,32,32")
    mod = Module(net, context=cpu())
    mod.bind(data_shapes=[("data", (4, 3, 32, 32))],
             label_shapes=[("softmax_label", (4,))])
    mod.init_params()
    mod.init_optimizer(optimizer_params=(("learning_rate", 0.1),))
    it = SyntheticDataIter(10, (4, 3, 32, 32), max_iter=2)
    batch = it.next()
    mod.forward_backward(batch)
    mod.update()
    assert mod._loss.item() > 0


def test_score_and_predict():
    mod = _mlp_module()
    mod.init_params()
    it = MNISTIter(batch_size=32, flat=True, num_examples=64, shuffle=False)
    out = mod.predict(it)
    assert out.shape == (64, 10)
    res = mod.score(it, "acc")
    assert 0.0 <= dict(res)["accuracy"] <= 1.0
