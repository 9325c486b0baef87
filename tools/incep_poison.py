"""Poison-the-pool repro: fill all free GPU memory with NaN bytes, then run
inception with per-layer NaN checks — the first NaN output is the op reading
memory it never wrote."""
import os
import sys

import torch

sys.path.insert(0, ".")
import dtmx  # noqa: E402
from dtmx.io import DataBatch  # noqa: E402
from dtmx.models import get_symbol  # noqa: E402


def poison():
    # grab the allocator's cached blocks AND most of free VRAM, fill with
    # 0xFF (bf16 NaN), release
    torch.cuda.synchronize()
    free, _ = torch.cuda.mem_get_info()
    bufs = []
    # allocate in 1 GiB chunks until ~6 GiB left
    while free > 6 * 2**30:
        try:
            bufs.append(torch.empty(2**30, dtype=torch.uint8, device="cuda:0"))
            bufs[-1].fill_(255)
        except RuntimeError:
            break
        free, _ = torch.cuda.mem_get_info()
    torch.cuda.synchronize()
    n = len(bufs)
    del bufs
    print(f"poisoned ~{n} GiB")


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

poison()

hooks_on = os.environ.get("HOOKS", "1") == "1"
state = {"bad": None}
if hooks_on:
    def mk(name):
        def h(m, i, o):
            if state["bad"] is None and isinstance(o, torch.Tensor):
                if not torch.isfinite(o.detach().float()).all().item():
                    state["bad"] = name
                    print(f"FIRST NON-FINITE OUTPUT: {name} ({type(m).__name__})")
        return h
    for n, m in mod.symbol.named_modules():
        if len(list(m.children())) == 0:
            m.register_forward_hook(mk(n))

for step in range(3):
    mod.forward_backward(batch)
    mod.update()
    print(f"step {step} loss={mod._loss.item():.3f} first_bad={state['bad']}")
    # check grads for non-finite too
    nf = [n for n, p in mod.symbol.named_parameters()
          if p.grad is not None and not torch.isfinite(p.grad.float()).all().item()]
    if nf:
        print(f"  non-finite grads: {nf[:6]}")
