"""`dtmx.autograd` — the mx.autograd surface (reference
python/mxnet/autograd.py) over torch autograd. The substrate records by
default (torch), so `record()` is an enable-grad scope, `pause()` a no-grad
scope; `backward`/`grad` mirror the mxnet call forms."""
from __future__ import annotations

from typing import List, Optional, Sequence

import torch


def record(train_mode: bool = True):
    """`with autograd.record():` — gradient-recording scope (reference
    autograd.py record)."""
    return torch.enable_grad()


def pause(train_mode: bool = False):
    return torch.no_grad()


def is_recording() -> bool:
    return torch.is_grad_enabled()


def is_training() -> bool:
    return torch.is_grad_enabled()


def mark_variables(variables: Sequence[torch.Tensor],
                   gradients: Sequence[torch.Tensor],
                   grad_reqs="write") -> None:
    """Attach gradient buffers to leaves (reference autograd.mark_variables):
    torch tracks leaves automatically; this sets requires_grad and seeds
    `.grad` with the provided buffers."""
    for v, g in zip(variables, gradients):
        v.requires_grad_(True)
        v.grad = g


def backward(heads, head_grads=None, retain_graph: bool = False,
             train_mode: bool = True) -> None:
    heads = heads if isinstance(heads, (list, tuple)) else [heads]
    if head_grads is None:
        head_grads = [torch.ones_like(h) for h in heads]
    elif not isinstance(head_grads, (list, tuple)):
        head_grads = [head_grads]
    torch.autograd.backward(list(heads), list(head_grads),
                            retain_graph=retain_graph)


def grad(heads, variables, head_grads=None, retain_graph: Optional[bool] = None,
         create_graph: bool = False, train_mode: bool = True) -> List[torch.Tensor]:
    heads = heads if isinstance(heads, (list, tuple)) else [heads]
    variables = list(variables) if isinstance(variables, (list, tuple)) else [variables]
    if head_grads is None:
        head_grads = [torch.ones_like(h) for h in heads]
    elif not isinstance(head_grads, (list, tuple)):
        head_grads = [head_grads]
    return list(torch.autograd.grad(
        heads, variables, head_grads,
        retain_graph=bool(retain_graph) or create_graph,
        create_graph=create_graph))
