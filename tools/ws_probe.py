"""One-off probe: run a model's training steps, then dump the wgrad split-K
workspace cache — every entry must read back all-zero after its drain pass."""
import sys

import torch

sys.path.insert(0, ".")
import dtmx  # noqa: E402
from dtmx.io import DataBatch  # noqa: E402
from dtmx.models import get_symbol  # noqa: E402
from dtmx.ops.hip import require_ext  # noqa: E402


def run(name, shape, **kwargs):
    torch.manual_seed(0)
    net = get_symbol(name, num_classes=100, **kwargs)
    mod = dtmx.Module(net, context=dtmx.gpu(0))
    mod.bind(data_shapes=[("data", shape)],
             label_shapes=[("softmax_label", (shape[0],))], dtype=torch.bfloat16)
    mod.init_params()
    mod.init_optimizer(optimizer_params=(("learning_rate", 1e-4), ("momentum", 0.9)))
    data = torch.randn(shape, dtype=torch.bfloat16, device="cuda:0").contiguous(
        memory_format=torch.channels_last)
    label = torch.randint(0, 100, (shape[0],), device="cuda:0").float()
    batch = DataBatch(data=[data], label=[label])
    for _ in range(3):
        mod.forward_backward(batch)
        mod.update()
    torch.cuda.synchronize()
    ext = require_ext()
    print(f"--- after {name} ---")
    for Ko, RSC, mx in ext.wgrad_ws_stats():
        flag = "  <-- DIRTY" if mx != 0.0 else ""
        print(f"ws[{Ko:5d},{RSC:6d}] maxabs={mx:.6g}{flag}")


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "vgg"
    if which == "vgg":
        run("vgg", (4, 3, 224, 224), num_layers=16, image_shape="3,224,224")
    elif which == "alexnet":
        run("alexnet", (8, 3, 224, 224), image_shape="3,224,224")
    elif which == "resnet":
        run("resnet", (4, 3, 224, 224), num_layers=50, image_shape="3,224,224")
    elif which == "inception":
        run("inception-v3", (4, 3, 299, 299), image_shape="3,299,299")
