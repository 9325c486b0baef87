"""Per-layer output magnitude scan of one inception forward (run under
PYTORCH_NO_CUDA_MEMORY_CACHING=1 where the corruption hits step 0)."""
import os
import sys

import torch

sys.path.insert(0, ".")
import dtmx  # noqa: E402
from dtmx.io import DataBatch  # noqa: E402
from dtmx.models import get_symbol  # noqa: E402

torch.manual_seed(0)
net = get_symbol("inception-v3", num_classes=100, image_shape="3,299,299")
mod = dtmx.Module(net, context=dtmx.gpu(0))
mod.bind(data_shapes=[("data", (4, 3, 299, 299))],
         label_shapes=[("softmax_label", (4,))], dtype=torch.bfloat16)
mod.init_params()
mod.init_optimizer(optimizer_params=(("learning_rate", 0.002), ("momentum", 0.9)))
data = torch.randn(4, 3, 299, 299, dtype=torch.bfloat16, device="cuda:0").contiguous(
    memory_format=torch.channels_last)
label = torch.randint(0, 100, (4,), device="cuda:0").float()
batch = DataBatch(data=[data], label=[label])

rows = []
def mk(name, m):
    def h(_m, _i, o):
        if isinstance(o, torch.Tensor):
            rows.append((name, type(_m).__name__, tuple(o.shape),
                         o.detach().float().abs().max().item()))
    return h
for n, m in net.named_modules():
    if len(list(m.children())) == 0:
        m.register_forward_hook(mk(n, m))

nsteps = int(os.environ.get("STEPS", "1"))
for s in range(nsteps):
    rows.clear()
    mod.forward_backward(batch)
    print(f"--- step {s} loss={mod._loss.item():.3f} ---")
    for name, ty, shp, mx in rows:
        print(f"  {name}|{ty}|{mx:.6g}")
    mod.update()
