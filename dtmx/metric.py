"""Evaluation metrics (reference python/mxnet/metric.py subset used by
Module.fit / example/image-classification)."""
from __future__ import annotations

from typing import List, Sequence

import torch


class EvalMetric:
    def __init__(self, name: str):
        self.name = name
        self.reset()

    def reset(self):
        self.num_inst = 0
        self.sum_metric = 0.0

    def update(self, labels, preds):
        raise NotImplementedError

    def get(self):
        if self.num_inst == 0:
            return (self.name, float("nan"))
        return (self.name, self.sum_metric / self.num_inst)

    def get_name_value(self):
        name, value = self.get()
        if not isinstance(name, list):
            name = [name]
        if not isinstance(value, list):
            value = [value]
        return list(zip(name, value))


def _as_list(x):
    return x if isinstance(x, (list, tuple)) else [x]


class Accuracy(EvalMetric):
    def __init__(self, name: str = "accuracy"):
        super().__init__(name)

    def update(self, labels, preds):
        for label, pred in zip(_as_list(labels), _as_list(preds)):
            if pred.dim() > 1 and pred.shape[-1] > 1:
                pred_label = pred.argmax(dim=-1)
            else:
                pred_label = pred.round().to(torch.int64)
            label = label.reshape(-1).to(pred_label.device, torch.int64)
            pred_label = pred_label.reshape(-1)
            self.sum_metric += (pred_label == label).sum().item()
            self.num_inst += label.numel()


class TopKAccuracy(EvalMetric):
    def __init__(self, top_k: int = 1):
        super().__init__(f"top_k_accuracy_{top_k}")
        self.top_k = top_k

    def update(self, labels, preds):
        for label, pred in zip(_as_list(labels), _as_list(preds)):
            label = label.reshape(-1).to(torch.int64)
            topk = pred.reshape(label.numel(), -1).topk(self.top_k, dim=-1).indices
            self.sum_metric += (topk == label.to(topk.device).unsqueeze(-1)).any(-1).sum().item()
            self.num_inst += label.numel()


class CrossEntropy(EvalMetric):
    def __init__(self, eps: float = 1e-12, name: str = "cross-entropy"):
        super().__init__(name)
        self.eps = eps

    def update(self, labels, preds):
        for label, pred in zip(_as_list(labels), _as_list(preds)):
            label = label.reshape(-1).to(torch.int64)
            prob = pred.reshape(label.numel(), -1).float()
            p = prob[torch.arange(label.numel(), device=prob.device), label.to(prob.device)]
            self.sum_metric += (-(p + self.eps).log()).sum().item()
            self.num_inst += label.numel()


class Loss(EvalMetric):
    """Running mean of a scalar loss (dtmx extension used by Module.fit)."""

    def __init__(self, name: str = "loss"):
        super().__init__(name)

    def update_loss(self, loss_value: float, count: int = 1):
        self.sum_metric += loss_value * count
        self.num_inst += count


class CompositeEvalMetric(EvalMetric):
    def __init__(self, metrics: Sequence[EvalMetric] = None, name: str = "composite"):
        super().__init__(name)
        self.metrics: List[EvalMetric] = list(metrics) if metrics else []

    def add(self, metric: EvalMetric):
        self.metrics.append(metric)

    def reset(self):
        for m in getattr(self, "metrics", []):
            m.reset()

    def update(self, labels, preds):
        for m in self.metrics:
            m.update(labels, preds)

    def get(self):
        names, values = [], []
        for m in self.metrics:
            n, v = m.get()
            names.append(n)
            values.append(v)
        return (names, values)


def create(metric, **kwargs) -> EvalMetric:
    if isinstance(metric, EvalMetric):
        return metric
    if isinstance(metric, (list, tuple)):
        composite = CompositeEvalMetric()
        for m in metric:
            composite.add(create(m, **kwargs))
        return composite
    metric = metric.lower()
    if metric in ("acc", "accuracy"):
        return Accuracy()
    if metric.startswith("top_k") or metric == "topk":
        return TopKAccuracy(kwargs.get("top_k", 5))
    if metric in ("ce", "cross-entropy"):
        return CrossEntropy()
    raise ValueError(f"unknown metric {metric}")
