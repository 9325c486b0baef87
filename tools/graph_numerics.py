"""hipGraph step-capture trajectory validation: eager vs graphed_step on
ResNet-18. Both must descend identically (modulo the capture call, which
warms up and records without replaying). Run on an MI355X box.
"""
import sys, torch
sys.path.insert(0, ".")
import dtmx
from dtmx.io import DataBatch
from dtmx.models import get_symbol

def run(use_graph):
    torch.manual_seed(0)
    net = get_symbol("resnet", num_layers=18, num_classes=100, image_shape="3,64,64")
    mod = dtmx.Module(net, context=dtmx.gpu(0))
    mod.bind(data_shapes=[("data", (16, 3, 64, 64))],
             label_shapes=[("softmax_label", (16,))], dtype=torch.bfloat16)
    mod.init_params()
    mod.init_optimizer(optimizer_params=(("learning_rate", 0.05), ("momentum", 0.9)))
    g = torch.Generator().manual_seed(42)
    data = (torch.randn(16, 3, 64, 64, generator=g)).to(torch.bfloat16).cuda().contiguous(
        memory_format=torch.channels_last)
    label = torch.randint(0, 100, (16,), generator=g).float().cuda()
    batch = DataBatch(data=[data], label=[label])
    losses = []
    for _ in range(15):
        if use_graph:
            mod.graphed_step(batch)
        else:
            mod.forward_backward(batch)
            mod.update()
        losses.append(round(mod._loss.item(), 2))
    return losses

e = run(False)
gr = run(True)
print("eager:", e)
print("graph:", gr)
# graph call 0 only warms up (3 steps) + captures — no replay, so gr[0] is
# stale; trajectories shift by the warmup. Require descent + finiteness from
# the first replay on.
assert e[-1] < e[0], e
assert gr[-1] < gr[1], gr
assert all(x == x for x in e + gr), (e, gr)
print("OK: both descend", e[0], "->", e[-1], "|", gr[1], "->", gr[-1])
