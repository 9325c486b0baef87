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


@Optimizer.register
class NAG(Optimizer):
    """Nesterov accelerated SGD (reference optimizer_op-inl.h NAGMomKernel /
    python optimizer.py NAG):

        g    = rescale*grad (clipped) + wd*w
        mom  = momentum*mom + g
        w   -= lr*(g + momentum*mom)
    """

    def __init__(self, momentum: float = 0.0, **kwargs):
        super().__init__(**kwargs)
        self.momentum = momentum

    def create_state(self, index, weight):
        if self.momentum != 0.0:
            return torch.zeros_like(weight, dtype=torch.float32)
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
                state.mul_(self.momentum).add_(g)
                w32.add_(g.add(state, alpha=self.momentum), alpha=-lr)
            else:
                w32.add_(g, alpha=-lr)
            if w32 is not weight:
                weight.copy_(w32.to(weight.dtype))


@Optimizer.register
class Signum(Optimizer):
    """Sign-momentum SGD (reference optimizer_op-inl.h SignumKernel /
    python optimizer.py Signum):

        mom = momentum*mom - (1-momentum)*(rescale*grad, clipped)
        w   = (1 - lr*wd_lh)*w + lr*sign(mom)

    With momentum == 0 this is SignSGD: w -= lr*sign(grad)."""

    def __init__(self, momentum: float = 0.9, wd_lh: float = 0.0, **kwargs):
        super().__init__(**kwargs)
        self.momentum = momentum
        self.wd_lh = wd_lh

    def create_state(self, index, weight):
        if self.momentum != 0.0:
            return torch.zeros_like(weight, dtype=torch.float32)
        return None

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr = self._get_lr(index)
        wd = self._get_wd(index)
        g = self._preprocess(index, grad).to(torch.float32)
        with torch.no_grad():
            w32 = weight if weight.dtype == torch.float32 else weight.float()
            if state is not None:
                # reference applies wd inside the momentum term
                g = g.add(w32, alpha=wd)
                state.mul_(self.momentum).add_(g, alpha=-(1.0 - self.momentum))
                w32.mul_(1.0 - lr * self.wd_lh).add_(torch.sign(state), alpha=lr)
            else:
                w32.mul_(1.0 - lr * self.wd_lh).add_(torch.sign(g), alpha=-lr)
            if w32 is not weight:
                weight.copy_(w32.to(weight.dtype))


@Optimizer.register
class RMSProp(Optimizer):
    """RMSProp (reference optimizer_op-inl.h RMSProp{,Alex}Update / python
    optimizer.py RMSProp). centered=False: Tieleman&Hinton; centered=True:
    Graves' variant with gradient-mean centering and momentum gamma2."""

    def __init__(self, gamma1: float = 0.9, gamma2: float = 0.9,
                 epsilon: float = 1e-8, centered: bool = False,
                 clip_weights: Optional[float] = None, **kwargs):
        kwargs.setdefault("learning_rate", 0.001)
        super().__init__(**kwargs)
        self.gamma1 = gamma1
        self.gamma2 = gamma2
        self.epsilon = epsilon
        self.centered = centered
        self.clip_weights = clip_weights

    def create_state(self, index, weight):
        n = torch.zeros_like(weight, dtype=torch.float32)
        if self.centered:
            return (n, torch.zeros_like(n), torch.zeros_like(n))  # n, g, delta
        return (n,)

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr = self._get_lr(index)
        wd = self._get_wd(index)
        g = self._preprocess(index, grad).to(torch.float32)
        with torch.no_grad():
            w32 = weight if weight.dtype == torch.float32 else weight.float()
            g = g.add(w32, alpha=wd)
            if self.centered:
                n, gm, delta = state
                n.mul_(self.gamma1).addcmul_(g, g, value=1 - self.gamma1)
                gm.mul_(self.gamma1).add_(g, alpha=1 - self.gamma1)
                denom = (n - gm * gm).add_(self.epsilon).sqrt_()
                delta.mul_(self.gamma2).addcdiv_(g, denom, value=-lr)
                w32.add_(delta)
            else:
                (n,) = state
                n.mul_(self.gamma1).addcmul_(g, g, value=1 - self.gamma1)
                w32.addcdiv_(g, n.sqrt().add_(self.epsilon), value=-lr)
            if self.clip_weights:
                w32.clamp_(-self.clip_weights, self.clip_weights)
            if w32 is not weight:
                weight.copy_(w32.to(weight.dtype))


@Optimizer.register
class FTRL(Optimizer):
    """FTRL-proximal (reference optimizer_op-inl.h FTRLKernel / python
    optimizer.py Ftrl):

        g  = rescale*grad (clipped)
        z += g - (sqrt(n + g^2) - sqrt(n))/lr * w
        n += g^2
        w  = (sign(z)*lamda1 - z) / ((beta + sqrt(n))/lr + wd) * 1{|z|>lamda1}
    """

    def __init__(self, lamda1: float = 0.01, beta: float = 1.0, **kwargs):
        kwargs.setdefault("learning_rate", 0.1)
        super().__init__(**kwargs)
        self.lamda1 = lamda1
        self.beta = beta

    def create_state(self, index, weight):
        return (
            torch.zeros_like(weight, dtype=torch.float32),  # z
            torch.zeros_like(weight, dtype=torch.float32),  # n
        )

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr = self._get_lr(index)
        wd = self._get_wd(index)
        g = self._preprocess(index, grad).to(torch.float32)
        z, n = state
        with torch.no_grad():
            w32 = weight if weight.dtype == torch.float32 else weight.float()
            new_n = n + g * g
            z.add_(g - (new_n.sqrt() - n.sqrt()) / lr * w32)
            n.copy_(new_n)
            denom = (self.beta + n.sqrt()) / lr + wd
            w32.copy_((torch.sign(z) * self.lamda1 - z) / denom
                      * (z.abs() > self.lamda1))
            if w32 is not weight:
                weight.copy_(w32.to(weight.dtype))


@Optimizer.register
class AdaGrad(Optimizer):
    """AdaGrad (reference python optimizer.py AdaGrad):
        hist += g^2;  w -= lr * (g / (sqrt(hist) + eps) + wd*w)
    """

    def __init__(self, eps: float = 1e-7, **kwargs):
        super().__init__(**kwargs)
        self.float_stable_eps = eps

    def create_state(self, index, weight):
        return torch.zeros_like(weight, dtype=torch.float32)

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr = self._get_lr(index)
        wd = self._get_wd(index)
        g = self._preprocess(index, grad).to(torch.float32)
        with torch.no_grad():
            w32 = weight if weight.dtype == torch.float32 else weight.float()
            state.addcmul_(g, g, value=1.0)
            adj = g / (state.sqrt() + self.float_stable_eps)
            if wd > 0:
                adj = adj.add(w32, alpha=wd)
            w32.add_(adj, alpha=-lr)
            if w32 is not weight:
                weight.copy_(w32.to(weight.dtype))


@Optimizer.register
class GroupAdagrad(Optimizer):
    """Row-wise AdaGrad (reference src/operator/contrib/optimizer_op.cu
    GroupAdagrad / python contrib optimizer): one shared history entry per
    ROW of a 2-D weight (the embedding-table optimizer — pairs with
    row-sparse gradients where only touched rows update):

        hist[row] += mean_j(g[row,j]^2)
        w[row]    -= lr * g[row] / sqrt(hist[row] + eps)

    wd is not supported (reference asserts wd == 0)."""

    def __init__(self, eps: float = 1e-5, **kwargs):
        super().__init__(**kwargs)
        self.epsilon = eps

    def create_state(self, index, weight):
        assert weight.dim() == 2, "GroupAdagrad expects 2-D (row) weights"
        return torch.zeros(weight.shape[0], dtype=torch.float32,
                           device=weight.device)

    def update(self, index, weight, grad, state):
        self._update_count(index)
        lr = self._get_lr(index)
        g = self._preprocess(index, grad).to(torch.float32)
        with torch.no_grad():
            w32 = weight if weight.dtype == torch.float32 else weight.float()
            state.add_((g * g).mean(dim=1))
            w32.add_(g / (state + self.epsilon).sqrt().unsqueeze(1), alpha=-lr)
            if w32 is not weight:
                weight.copy_(w32.to(weight.dtype))

    def update_rows(self, index, weight, grad_rows: torch.Tensor,
                    grad_vals: torch.Tensor, state):
        """Row-sparse update: only `grad_rows` change (reference
        GroupAdagrad FComputeEx on row_sparse grads)."""
        self._update_count(index)
        lr = self._get_lr(index)
        with torch.no_grad():
            g = (grad_vals.float() * self.rescale_grad)
            if self.clip_gradient is not None:
                g = g.clamp_(-self.clip_gradient, self.clip_gradient)
            state[grad_rows] += (g * g).mean(dim=1)
            upd = g / (state[grad_rows] + self.epsilon).sqrt().unsqueeze(1)
            weight[grad_rows] -= (lr * upd).to(weight.dtype)


@Optimizer.register
class AdaDelta(Optimizer):
    """AdaDelta (reference python optimizer.py AdaDelta):
        acc_g = rho*acc_g + (1-rho)*g^2
        d     = sqrt(acc_d + eps)/sqrt(acc_g + eps) * g
        acc_d = rho*acc_d + (1-rho)*d^2
        w    -= d  (wd applied to g)
    """

    def __init__(self, rho: float = 0.90, epsilon: float = 1e-5, **kwargs):
        kwargs.setdefault("learning_rate", 1.0)
        super().__init__(**kwargs)
        self.rho = rho
        self.epsilon = epsilon

    def create_state(self, index, weight):
        return (
            torch.zeros_like(weight, dtype=torch.float32),
            torch.zeros_like(weight, dtype=torch.float32),
        )

    def update(self, index, weight, grad, state):
        self._update_count(index)
        wd = self._get_wd(index)
        g = self._preprocess(index, grad).to(torch.float32)
        acc_g, acc_d = state
        with torch.no_grad():
            w32 = weight if weight.dtype == torch.float32 else weight.float()
            g = g.add(w32, alpha=wd)
            acc_g.mul_(self.rho).addcmul_(g, g, value=1 - self.rho)
            d = (acc_d + self.epsilon).sqrt() / (acc_g + self.epsilon).sqrt() * g
            acc_d.mul_(self.rho).addcmul_(d, d, value=1 - self.rho)
            w32.sub_(d)
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
