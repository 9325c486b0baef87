"""Bisect the inception-after-alexnet corruption by op family (DTMX_FALLBACK)."""
import os
import sys

import torch

sys.path.insert(0, ".")
import dtmx  # noqa: E402
from dtmx.io import DataBatch  # noqa: E402
from dtmx.models import get_symbol  # noqa: E402


def make(name, shape, lr, **kwargs):
    net = get_symbol(name, num_classes=100, **kwargs)
    mod = dtmx.Module(net, context=dtmx.gpu(0))
    mod.bind(data_shapes=[("data", shape)],
             label_shapes=[("softmax_label", (shape[0],))], dtype=torch.bfloat16)
    mod.init_params()
    mod.init_optimizer(optimizer_params=(("learning_rate", lr), ("momentum", 0.9)))
    data = torch.randn(shape, dtype=torch.bfloat16, device="cuda:0").contiguous(
        memory_format=torch.channels_last)
    label = torch.randint(0, 100, (shape[0],), device="cuda:0").float()
    return mod, DataBatch(data=[data], label=[label])


torch.manual_seed(0)
amod, abatch = make("alexnet", (8, 3, 224, 224), 1e-4, image_shape="3,224,224")
for _ in range(3):
    amod.forward_backward(abatch)
    amod.update()
del amod, abatch

torch.manual_seed(0)
mod, batch = make("inception-v3", (4, 3, 299, 299), 0.002, image_shape="3,299,299")
losses = []
for step in range(3):
    mod.forward_backward(batch)
    mod.update()
    losses.append(round(mod._loss.item(), 2))
print(f"FALLBACK={os.environ.get('DTMX_FALLBACK','')!r} losses={losses} "
      f"{'EXPLODED' if max(losses) > 100 else 'ok'}")
