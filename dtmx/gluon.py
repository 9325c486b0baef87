"""Thin gluon-style imperative API (reference python/mxnet/gluon/: Block,
Trainer, loss). dtmx's substrate is already imperative (torch autograd), so
Block == nn.Module; Trainer wires parameters to a kvstore + optimizer the way
gluon.Trainer does (reference gluon/trainer.py:158+).

Note: the reference's gluon was NOT wired to elastic DT;
dtmx.Module.fit is the elastic path. Trainer here covers the imperative
single/static-distributed workflow.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn as _tnn

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


class L1Loss(nn.Module):
    def forward(self, pred, label):
        return (pred - label.reshape(pred.shape)).abs()


class _GluonNN:
    """`gluon.nn` namespace (reference gluon/nn/): gluon layer names mapped
    onto the substrate's modules with gluon-style ctor args. Lazy in_units /
    in_channels (gluon's deferred shape init) is NOT mirrored — pass the
    input size explicitly, as dtmx layers do."""

    Sequential = _tnn.Sequential
    HybridSequential = _tnn.Sequential
    Flatten = _tnn.Flatten
    Dropout = _tnn.Dropout

    @staticmethod
    def Dense(units, in_units, activation=None, use_bias=True):
        lin = _tnn.Linear(in_units, units, bias=use_bias)
        if activation is None:
            return lin
        acts = {"relu": _tnn.ReLU(), "sigmoid": _tnn.Sigmoid(), "tanh": _tnn.Tanh()}
        return _tnn.Sequential(lin, acts[activation])

    @staticmethod
    def Conv2D(channels, kernel_size, in_channels, strides=1, padding=0,
               use_bias=True, activation=None):
        conv = _tnn.Conv2d(in_channels, channels, kernel_size, strides, padding,
                         bias=use_bias)
        if activation is None:
            return conv
        return _tnn.Sequential(conv, {"relu": _tnn.ReLU()}[activation])

    @staticmethod
    def MaxPool2D(pool_size=2, strides=None, padding=0):
        return _tnn.MaxPool2d(pool_size, strides or pool_size, padding)

    @staticmethod
    def AvgPool2D(pool_size=2, strides=None, padding=0):
        return _tnn.AvgPool2d(pool_size, strides or pool_size, padding)

    @staticmethod
    def BatchNorm(in_channels, momentum=0.9, epsilon=1e-5):
        return _tnn.BatchNorm2d(in_channels, eps=epsilon, momentum=1 - momentum)

    @staticmethod
    def Activation(activation):
        return {"relu": _tnn.ReLU(), "sigmoid": _tnn.Sigmoid(),
                "tanh": _tnn.Tanh(), "softrelu": _tnn.Softplus()}[activation]

    @staticmethod
    def GlobalAvgPool2D():
        return _tnn.AdaptiveAvgPool2d(1)


class _GluonLoss:
    L2Loss = L2Loss
    L1Loss = L1Loss
    SoftmaxCrossEntropyLoss = SoftmaxCrossEntropyLoss
    SoftmaxCELoss = SoftmaxCrossEntropyLoss


class _GluonData:
    """`gluon.data` essentials (reference gluon/data/): torch's dataset/
    loader are the substrate equivalents."""
    from torch.utils.data import DataLoader, Dataset, TensorDataset
    ArrayDataset = TensorDataset


nn_ = _GluonNN  # gluon.nn / gluon.loss / gluon.data namespaces
loss = _GluonLoss
data = _GluonData
globals()["nn"] = _GluonNN  # keep `from dtmx.gluon import nn` working
                            # (the torch.nn import above is module-internal)
