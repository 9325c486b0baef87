"""Model summaries (reference python/mxnet/visualization.py
print_summary): per-layer output shapes and parameter counts."""
from __future__ import annotations

import torch
import torch.nn as nn


def print_summary(net: nn.Module, shape=None, dtype=torch.float32) -> str:
    """Markdown-ish layer table; pass `shape` = input shape (with batch) to
    also trace output shapes through a forward."""
    rows = []
    shapes = {}
    hooks = []
    if shape is not None:
        def mk_hook(name):
            def hook(mod, inp, out):
                if isinstance(out, torch.Tensor):
                    shapes[name] = tuple(out.shape)
            return hook

        for name, m in net.named_modules():
            if name:
                hooks.append(m.register_forward_hook(mk_hook(name)))
        with torch.no_grad():
            net(torch.zeros(shape, dtype=dtype))
        for h in hooks:
            h.remove()
    total = 0
    for name, m in net.named_modules():
        if not name or len(list(m.children())):
            continue  # leaves only
        n = sum(p.numel() for p in m.parameters(recurse=False))
        total += n
        rows.append((name, m.__class__.__name__,
                     str(shapes.get(name, "")), n))
    width = max([len(r[0]) for r in rows] + [10])
    out = [f"{'Layer':{width}s} {'Type':18s} {'Output':20s} {'Params':>12s}"]
    out += [f"{r[0]:{width}s} {r[1]:18s} {r[2]:20s} {r[3]:12,d}" for r in rows]
    out.append(f"Total params: {total:,d}")
    text = "\n".join(out)
    print(text)
    return text
