"""One training step of every model family on the HIP path (zoo coverage;
reference example/image-classification symbols + train tests)."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

# (name, kwargs, shape, lr) — no-BN nets (alexnet/vgg) take a small lr:
# they diverge at 0.01 from random init regardless of backend (verified on
# CPU fp32), so divergence there is not a kernel property.
CASES = [
    ("resnet", {"num_layers": 50}, (4, 3, 224, 224), 0.01),
    ("resnet", {"num_layers": 101}, (2, 3, 224, 224), 0.01),
    ("alexnet", {}, (8, 3, 224, 224), 1e-4),
    ("vgg", {"num_layers": 16}, (4, 3, 224, 224), 1e-4),
    # modest lr: the zoo test checks kernel plumbing, not convergence tuning
    ("inception-v3", {}, (4, 3, 299, 299), 0.002),
    ("inception-bn", {}, (4, 3, 224, 224), 0.002),
    ("lenet", {}, (16, 1, 28, 28), 0.01),
    ("mlp", {}, (32, 784), 0.01),
    # round-2 zoo additions (reference symbols/ tail)
    ("googlenet", {}, (4, 3, 224, 224), 0.002),
    ("inception-v4", {}, (2, 3, 299, 299), 0.002),
    ("inception-resnet-v2", {}, (2, 3, 299, 299), 0.002),
    ("mobilenet", {}, (8, 3, 224, 224), 0.002),
    ("mobilenetv2", {}, (8, 3, 224, 224), 0.002),
    ("resnext", {"num_layers": 50}, (4, 3, 224, 224), 0.01),
    ("resnet-v2", {"num_layers": 50}, (4, 3, 224, 224), 0.01),
]


@pytest.mark.parametrize("name,kwargs,shape,lr", CASES, ids=[c[0] + str(c[1]) for c in CASES])
def test_model_step(name, kwargs, shape, lr):
    import dtmx
    from dtmx.io import DataBatch
    from dtmx.models import get_symbol

    torch.manual_seed(0)
    if len(shape) == 4:
        kwargs.setdefault("image_shape", ",".join(map(str, shape[1:])))
    try:
        net = get_symbol(name, num_classes=100, **kwargs)
    except TypeError:
        kwargs.pop("image_shape", None)
        net = get_symbol(name, num_classes=100, **kwargs)
    mod = dtmx.Module(net, context=dtmx.gpu(0))
    mod.bind(data_shapes=[("data", shape)], label_shapes=[("softmax_label", (shape[0],))],
             dtype=torch.bfloat16)
    mod.init_params()
    mod.init_optimizer(optimizer_params=(("learning_rate", lr), ("momentum", 0.9)))
    data = torch.randn(shape, dtype=torch.bfloat16, device="cuda:0")
    if data.dim() == 4:
        data = data.contiguous(memory_format=torch.channels_last)
    label = torch.randint(0, 100, (shape[0],), device="cuda:0").float()
    batch = DataBatch(data=[data], label=[label])
    losses = []
    for _ in range(3):
        mod.forward_backward(batch)
        mod.update()
        losses.append(mod._loss.item())
    assert all(np.isfinite(losses)), (name, losses)
    assert losses[-1] < losses[0] * 2.0, (name, losses)  # not exploding


@pytest.mark.parametrize("dtype", [torch.float16])
def test_resnet_fp16_step(dtype):
    """fp16 compute path (reference fp16 multi-precision training,
    BASELINE config 5): same kernels templated on _Float16."""
    import dtmx
    from dtmx.io import DataBatch
    from dtmx.models import get_symbol

    torch.manual_seed(0)
    net = get_symbol("resnet", num_layers=18, num_classes=100, image_shape="3,64,64")
    mod = dtmx.Module(net, context=dtmx.gpu(0))
    mod.bind(data_shapes=[("data", (8, 3, 64, 64))],
             label_shapes=[("softmax_label", (8,))], dtype=dtype)
    mod.init_params()
    mod.init_optimizer(optimizer_params=(("learning_rate", 0.05), ("momentum", 0.9)))
    data = torch.randn(8, 3, 64, 64, dtype=dtype, device="cuda:0").contiguous(
        memory_format=torch.channels_last)
    label = torch.randint(0, 100, (8,), device="cuda:0").float()
    batch = DataBatch(data=[data], label=[label])
    losses = []
    for _ in range(8):
        mod.forward_backward(batch)
        mod.update()
        losses.append(mod._loss.item())
    assert all(np.isfinite(losses)), losses
    assert losses[-1] < losses[0], losses


@pytest.mark.parametrize("dtype", [torch.float16])
def test_fp16_conv_numerics(dtype):
    from dtmx.ops.hip import require_ext
    ext = require_ext()
    x = (torch.randn(4, 64, 14, 14) * 0.5).to(dtype)
    w = (torch.randn(128, 64, 3, 3) * 0.1).to(dtype)
    xd = x.to("cuda:0").contiguous(memory_format=torch.channels_last)
    wd = w.to("cuda:0").contiguous(memory_format=torch.channels_last)
    y = ext.conv_fwd(xd, wd, 1, 1)
    ref = torch.nn.functional.conv2d(x.float(), w.float(), None, 1, 1)
    torch.testing.assert_close(y.contiguous().float().cpu(), ref, rtol=0.02,
                               atol=0.02 * ref.abs().mean().item() + 1e-3)
