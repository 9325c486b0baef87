"""Thin gluon-style imperative API (reference python/mxnet/gluon/: Block,
Trainer, loss). dtmx's substrate is already imperative (torch autograd), so
Block == nn.Module; Trainer wires parameters to a kvstore + optimizer the way
gluon.Trainer does (reference gluon/trainer.py:158+).

Note: the reference's gluon was NOT wired to elastic DT;
dtmx.Module.fit is the elastic path. Trainer here covers the imperative
single/static-distributed workflow.
"""
from __future__ import annotations

from typing import Dict, Optional, Sequence

import torch
import torch.nn as nn

from . import kvstore as kvs
from .optimizer import Optimizer, create as opt_create, get_updater
from .parallel.bucketer import GradBucketer

Block = nn.Module
HybridBlock = nn.Module


class Trainer:
    def __init__(self, params, optimizer, optimizer_params=None, kvstore="device"):
        if isinstance(params, dict):
            self._params = list(params.values())
        else:
            self._params = list(params)
        optimizer_params = dict(optimizer_params or {})
        if isinstance(optimizer, str):
            optimizer = opt_create(optimizer, **optimizer_params)
        self._optimizer: Optimizer = optimizer
        self._updater = get_updater(optimizer)
        self._kv = kvs.create(kvstore) if isinstance(kvstore, str) else kvstore
        self._bucketer: Optional[GradBucketer] = None
        if isinstance(self._kv, kvs.DistKVStore):
            self._bucketer = GradBucketer(self._params)

    @property
    def learning_rate(self):
        return self._optimizer.lr

    def set_learning_rate(self, lr):
        self._optimizer.lr = lr

    def zero_grad(self):
        if self._bucketer is not None:
            self._bucketer.zero_grad()
        else:
            for p in self._params:
                if p.grad is not None:
                    p.grad.zero_()

    def step(self, batch_size: int, ignore_stale_grad=False):
        """Apply one optimizer step; gradients are summed across workers by
        the bucketer's all-reduce (gluon semantics: rescale by 1/batch)."""
        world = self._kv.num_workers if self._kv else 1
        self._optimizer.rescale_grad = 1.0 / (batch_size * world)
        if self._bucketer is not None:
            self._bucketer.finish()
        for i, p in enumerate(self._params):
            if p.grad is None:
                continue
            self._updater(i, p.grad, p.data)

    def allreduce_grads(self):
        if self._bucketer is not None:
            self._bucketer.finish()

    def save_states(self, fname: str):
        """Optimizer-state checkpoint (reference gluon trainer.save_states)."""
        import pickle

        with open(fname, "wb") as f:
            pickle.dump({k: _cpu_state(v) for k, v in
                         self._updater.get_states().items()}, f)

    def load_states(self, fname: str):
        import pickle

        with open(fname, "rb") as f:
            self._updater.set_states(pickle.load(f))


def _cpu_state(s):
    import torch as _t

    if isinstance(s, _t.Tensor):
        return s.detach().cpu()
    if isinstance(s, tuple):
        return tuple(_cpu_state(x) for x in s)
    return s


class L2Loss(nn.Module):
    def forward(self, pred, label):
        return 0.5 * (pred - label.reshape(pred.shape)) ** 2


class SoftmaxCrossEntropyLoss(nn.Module):
    def forward(self, pred, label):
        return torch.nn.functional.cross_entropy(
            pred.float(), label.reshape(-1).long(), reduction="none"
        )
