"""Zoo wiring on CPU: every registered family constructs and one tiny
no-grad forward produces (B, num_classes) (reference symbols/* import +
get_symbol smoke)."""
import pytest
import torch

from dtmx.models import get_symbol

CASES = [
    ("googlenet", {}, (1, 3, 224, 224)),
    ("inception-v4", {}, (1, 3, 299, 299)),
    ("inception-resnet-v2", {}, (1, 3, 299, 299)),
    ("mobilenet", {}, (1, 3, 224, 224)),
    ("mobilenetv2", {}, (1, 3, 224, 224)),
    ("resnext", {"num_layers": 50}, (1, 3, 224, 224)),
    ("resnet-v2", {"num_layers": 50}, (1, 3, 224, 224)),
    ("resnet-v2", {"num_layers": 18}, (1, 3, 32, 32)),
]


@pytest.mark.parametrize("name,kwargs,shape", CASES, ids=[c[0] for c in CASES])
def test_zoo_forward_shape(name, kwargs, shape):
    torch.manual_seed(0)
    net = get_symbol(name, num_classes=17, **kwargs).eval()
    x = torch.randn(*shape)
    with torch.no_grad():
        y = net(x)
    assert y.shape == (shape[0], 17)
    assert torch.isfinite(y.float()).all()


def test_registry_lists_reference_tail():
    from dtmx import models
    for name in ("googlenet", "inception-v4", "inception-resnet-v2",
                 "mobilenet", "mobilenetv2", "resnext", "resnet-v1"):
        assert name in models._REGISTRY
