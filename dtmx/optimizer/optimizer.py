"""Optimizers (reference python/mxnet/optimizer/optimizer.py subset).

Semantics preserved from the reference:

- `rescale_grad` defaults to 1 and Module sets it to 1/(batch_size *
  num_workers) (reference module.py:515-518). Pushed gradients are SUMMED
  across workers (server merge / all-reduce SUM), so the effective update is
  lr * mean-over-global-batch. When the worker count changes at an epoch
  boundary, Module recomputes rescale_grad — the dynamic-minibatch rescale
  (fixing the reference's stale-rescale gap).
- weight decay is applied on the (rescaled, clipped) gradient:
  grad = rescale*grad (+ clip) + wd*weight  (reference optimizer.py:374-400).
- multi-precision: for fp16/bf16 weights keep an fp32 master copy in the
  state; update the master in fp32, write back narrowed
  (reference MP_SGDMomKernel, src/operator/optimizer_op-inl.h:430).
- num_update bookkeeping per index drives the lr scheduler
  (reference optimizer.py:85-113).

On MI355X the fused multi-tensor SGD step runs in the HIP extension
(dtmx/csrc/optimizer.hip); this module is the per-tensor semantic reference
and the CPU path.
"""
from __future__ import annotations

import math
from typing import Dict, Optional, Tuple

import torch

from ..lr_scheduler import LRScheduler


class Optimizer:
    opt_registry: Dict[str, type] = {}

    def __init__(
        self,
        rescale_grad: float = 1.0,
        learning_rate: float = 0.01,
        lr_scheduler: Optional[LRScheduler] = None,
        wd: float = 0.0,
        clip_gradient: Optional[float] = None,
        param_idx2name: Optional[Dict[int, str]] = None,
        multi_precision: bool = False,
        begin_num_update: int = 0,
    ):
        self.rescale_grad = rescale_grad
        self.lr = learning_rate
        self.lr_scheduler = lr_scheduler
        if lr_scheduler is not None:
            self.lr_scheduler.base_lr = learning_rate
        self.wd = wd
        self.clip_gradient = clip_gradient
        self.multi_precision = multi_precision
        self.num_update = begin_num_update
        self.begin_num_update = begin_num_update
        self._index_update_count: Dict[int, int] = {}
        self.idx2name = dict(param_idx2name or {})
        self.lr_mult: Dict[str, float] = {}
        self.wd_mult: Dict[str, float] = {}

    # -- registry ----------------------------------------------------------
    @classmethod
    def register(cls, klass):
        cls.opt_registry[klass.__name__.lower()] = klass
        return klass

    @classmethod
    def create_optimizer(cls, name: str, **kwargs) -> "Optimizer":
        return cls.opt_registry[name.lower()](**kwargs)

    # -- bookkeeping -------------------------------------------------------
    def _update_count(self, index: int):
        self._index_update_count.setdefault(index, self.begin_num_update)
        self._index_update_count[index] += 1
        self.num_update = max(self._index_update_count[index], self.num_update)

    def _get_lr(self, index: int) -> float:
        lr = self.lr_scheduler(self.num_update) if self.lr_scheduler else self.lr
        name = self.idx2name.get(index)
        if name is not None:
            lr *= self.lr_mult.get(name, 1.0)
            if name.endswith("bias") or name.endswith("beta") or name.endswith("gamma"):
                pass
        return lr

    def _get_wd(self, index: int) -> float:
        wd = self.wd
        name = self.idx2name.get(index)
        if name is not None:
            wd *= self.wd_mult.get(name, self._default_wd_mult(name))
        return wd

    @staticmethod
    def _default_wd_mult(name: str) -> float:
        # reference Optimizer.set_wd_mult: no decay on bias/gamma/beta by
        # convention of fit.py-style trainers (mxnet keeps 1.0 by default;
        # we preserve mxnet's default of 1.0 here).
        return 1.0

    # -- interface ---------------------------------------------------------
    def create_state(self, index: int, weight: torch.Tensor):
        return None

    def create_state_multi_precision(self, index: int, weight: torch.Tensor):
        if self.multi_precision and weight.dtype in (torch.float16, torch.bfloat16):
            master = weight.detach().float().clone()
            return (master, self.create_state(index, master))
        return self.create_state(index, weight)

    def update(self, index, weight, grad, state):
        raise NotImplementedError

    def update_multi_precision(self, index, weight, grad, state):
        if self.multi_precision and weight.dtype in (torch.float16, torch.bfloat16):
            master, base_state = state
            grad32 = grad.detach().float()
            self.update(index, master, grad32, base_state)
            with torch.no_grad():
                weight.copy_(master.to(weight.dtype))
        else:
            self.update(index, weight, grad, state)

    # -- shared grad preprocessing ----------------------------------------
    def _preprocess(self, index, grad):
        g = grad.detach() * self.rescale_grad
        if self.clip_gradient is not None:
            g = g.clamp_(-self.clip_gradient, self.clip_gradient)
        return g


@Optimizer.register
class SGD(Optimizer):
    """SGD with momentum (reference SGDMomKernel optimizer_op-inl.h:305):

        mom = momentum*mom - lr*(rescale*grad + wd*weight)
        weight += mom
    """

    def __init__(self, momentum: float = 0.0, **kwargs):
        super().__init__(**kwargs)
        self.momentum = momentum

    def create_state(self, index, weight):
        if self.momentum != 0.0:
            return torch.zeros_like(weight, dtype=torch.float32, device=weight.device)
        return None

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr = self._get_lr(index)
        wd = self._get_wd(index)
        g = self._preprocess(index, grad).to(torch.float32)
        with torch.no_grad():
            w32 = weight if weight.dtype == torch.float32 else weight.float()
            g = g.add(w32, alpha=wd)
            if state is not None:
                state.mul_(self.momentum).add_(g, alpha=-lr)
                w32.add_(state)
            else:
                w32.add_(g, alpha=-lr)
            if w32 is not weight:
                weight.copy_(w32.to(weight.dtype))


@Optimizer.register
class Adam(Optimizer):
    def __init__(self, beta1: float = 0.9, beta2: float = 0.999, epsilon: float = 1e-8, **kwargs):
        kwargs.setdefault("learning_rate", 0.001)
        super().__init__(**kwargs)
        self.beta1 = beta1
        self.beta2 = beta2
        self.epsilon = epsilon

    def create_state(self, index, weight):
        return (
            torch.zeros_like(weight, dtype=torch.float32),
            torch.zeros_like(weight, dtype=torch.float32),
        )

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr = self._get_lr(index)
        wd = self._get_wd(index)
        t = self._index_update_count[index]
        lr *= math.sqrt(1.0 - self.beta2 ** t) / (1.0 - self.beta1 ** t)
        g = self._preprocess(index, grad).to(torch.float32)
        mean, var = state
        with torch.no_grad():
            w32 = weight if weight.dtype == torch.float32 else weight.float()
            g = g.add(w32, alpha=wd)
            mean.mul_(self.beta1).add_(g, alpha=1 - self.beta1)
            var.mul_(self.beta2).addcmul_(g, g, value=1 - self.beta2)
            w32.addcdiv_(mean, var.sqrt().add_(self.epsilon), value=-lr)
            if w32 is not weight:
                weight.copy_(w32.to(weight.dtype))


@Optimizer.register
class LBSGD(SGD):
    """Large-batch SGD with LARS-style layer-wise trust ratio + warmup
    (reference optimizer.py LBSGD)."""

    def __init__(self, momentum: float = 0.0, warmup_strategy: str = "linear",
                 warmup_epochs: int = 5, batch_scale: float = 1.0,
                 updates_per_epoch: int = 32, begin_epoch: int = 0,
                 num_epochs: int = 60, eta: float = 0.001, **kwargs):
        super().__init__(momentum=momentum, **kwargs)
        self.warmup_strategy = warmup_strategy
        self.warmup_updates = max(1, int(warmup_epochs * updates_per_epoch))
        self.batch_scale = batch_scale
        self.eta = eta

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr = self._get_lr(index)
        wd = self._get_wd(index)
        # warmup ramp
        t = self.num_update
        if t < self.warmup_updates:
            if self.warmup_strategy == "linear":
                lr = lr * (t + 1) / self.warmup_updates
            elif self.warmup_strategy == "sqrt":
                lr = lr * math.sqrt((t + 1) / self.warmup_updates)
        g = self._preprocess(index, grad).to(torch.float32)
        with torch.no_grad():
            w32 = weight if weight.dtype == torch.float32 else weight.float()
            wnorm = w32.norm().item()
            gnorm = g.norm().item()
            if wnorm > 0 and gnorm > 0:
                trust = self.eta * wnorm / (gnorm + wd * wnorm + 1e-12)
                lr = min(lr * trust, lr)
            g = g.add(w32, alpha=wd)
            if state is not None:
                state.mul_(self.momentum).add_(g, alpha=-lr)
                w32.add_(state)
            else:
                w32.add_(g, alpha=-lr)
            if w32 is not weight:
                weight.copy_(w32.to(weight.dtype))


class Updater:
    """Callable (index, grad, weight) updater with per-index state
    (reference optimizer.py get_updater / Updater class). This is what runs
    'on the kvstore' in update_on_kvstore mode."""

    def __init__(self, optimizer: Optimizer):
        self.optimizer = optimizer
        self.states: Dict[int, object] = {}

    def __call__(self, index, grad, weight):
        if index not in self.states:
            self.states[index] = self.optimizer.create_state_multi_precision(index, weight)
        self.optimizer.update_multi_precision(index, weight, grad, self.states[index])

    def get_states(self):
        return self.states

    def set_states(self, states):
        self.states = states


def get_updater(optimizer: Optimizer) -> Updater:
    return Updater(optimizer)


def create(name: str, **kwargs) -> Optimizer:
    return Optimizer.create_optimizer(name, **kwargs)
