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


def test_checkpoint_resume_equivalence(tmp_path):
    """Train 3 epochs -> checkpoint (+optimizer states) -> 2 more epochs
    must equal fresh-process load(checkpoint) -> 2 epochs (reference
    --load-epoch resume semantics, fit.py:74-84 / module.py:131-190)."""
    import numpy as np

    import dtmx
    from dtmx.io import NDArrayIter
    from dtmx.models import get_symbol

    rng = np.random.RandomState(0)
    X = rng.randn(64, 16).astype(np.float32)
    Y = rng.randint(0, 10, 64).astype(np.float32)

    def make():
        torch.manual_seed(0)
        net = get_symbol("mlp", num_classes=10, input_dim=16)
        mod = dtmx.Module(net, context=dtmx.cpu())
        mod.bind(data_shapes=[("data", (8, 16))],
                 label_shapes=[("softmax_label", (8,))])
        return mod

    it = NDArrayIter({"data": X}, {"softmax_label": Y}, 8)
    opt = (("learning_rate", 0.1), ("momentum", 0.9))

    # continuous run: 3 + 2 epochs with a checkpoint at 3
    mod = make()
    mod.fit(it, num_epoch=3, kvstore="local", optimizer_params=opt)
    prefix = str(tmp_path / "ck")
    mod.save_checkpoint(prefix, 3, save_optimizer_states=True)
    mod.fit(it, num_epoch=5, begin_epoch=3, kvstore="local",
            optimizer_params=opt, force_init=False)
    arg_cont, _ = mod.get_params()

    # resumed run: load the checkpoint into a fresh module
    mod2 = dtmx.Module.load(prefix, 3, get_symbol("mlp", num_classes=10,
                                                  input_dim=16))
    mod2.bind(data_shapes=[("data", (8, 16))],
              label_shapes=[("softmax_label", (8,))])
    mod2.init_params()
    mod2.init_optimizer(kvstore="local", optimizer_params=opt)
    mod2.load_optimizer_states(prefix + "-0003.states")
    it2 = NDArrayIter({"data": X}, {"softmax_label": Y}, 8)
    mod2.fit(it2, num_epoch=5, begin_epoch=3, kvstore="local",
             optimizer_params=opt, force_init=False)
    arg_res, _ = mod2.get_params()

    for k in arg_cont:
        torch.testing.assert_close(arg_res[k], arg_cont[k], rtol=1e-5,
                                    atol=1e-6)
