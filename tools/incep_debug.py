"""Localize the inception-after-alexnet corruption: run the alexnet test body
(allocator preconditioning), then inception, scanning params/grads/buffers
each step for the first exploding tensor."""
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


def scan(mod, tag):
    worst = []
    net = mod.symbol
    for n, p in net.named_parameters():
        m = p.detach().abs().max().item()
        if p.grad is not None:
            g = p.grad.detach().abs().max().item()
        else:
            g = 0.0
        worst.append((max(m, 0), n, m, g))
    for n, b in net.named_buffers():
        worst.append((b.detach().float().abs().max().item(), "buf:" + n,
                      b.detach().float().abs().max().item(), 0.0))
    worst.sort(reverse=True)
    print(f"[{tag}] top tensors:")
    for _, n, m, g in worst[:5]:
        print(f"    {n}: |w|max={m:.4g} |g|max={g:.4g}")
    bad = [(n, m, g) for _, n, m, g in worst if m > 50 or g > 1e4]
    if bad:
        print(f"[{tag}] SUSPECT: {bad[:10]}")


torch.manual_seed(0)
amod, abatch = make("alexnet", (8, 3, 224, 224), 1e-4, image_shape="3,224,224")
for _ in range(3):
    amod.forward_backward(abatch)
    amod.update()
del amod, abatch

torch.manual_seed(0)
mod, batch = make("inception-v3", (4, 3, 299, 299), 0.002, image_shape="3,299,299")
for step in range(3):
    mod.forward_backward(batch)
    print(f"step {step} loss={mod._loss.item():.3f}")
    scan(mod, f"after bwd {step}")
    mod.update()
    scan(mod, f"after upd {step}")


def hook_scan(mod, batch):
    acts = []
    def mk(name):
        def h(m, i, o):
            if isinstance(o, torch.Tensor):
                acts.append((name, o.detach().float().abs().max().item()))
        return h
    hs = [m.register_forward_hook(mk(n)) for n, m in mod.symbol.named_modules()
          if len(list(m.children())) == 0]
    mod.forward_backward(batch)
    for h in hs:
        h.remove()
    print(f"loss={mod._loss.item():.3f}")
    first_bad = None
    for name, mx in acts:
        flag = ""
        if mx > 100 and first_bad is None:
            first_bad = name
            flag = "  <<< FIRST BAD"
        if mx > 100:
            print(f"    {name}: {mx:.4g}{flag}")
    if first_bad is None:
        print("    all activations sane")


print("=== step 3 with activation hooks ===")
hook_scan(mod, batch)
print("=== step 4 with activation hooks ===")
hook_scan(mod, batch)
