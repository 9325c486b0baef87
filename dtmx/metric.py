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


class _BinaryCounts:
    """Shared TP/FP/TN/FN accumulator (reference metric.py
    _BinaryClassificationMetrics)."""

    def __init__(self):
        self.reset()

    def reset(self):
        self.tp = self.fp = self.tn = self.fn = 0

    def update(self, label, pred):
        label = label.reshape(-1).to(torch.int64)
        if pred.dim() > 1 and pred.shape[-1] > 1:
            pred_label = pred.reshape(label.numel(), -1).argmax(-1)
        else:
            pred_label = (pred.reshape(-1).float() > 0.5).to(torch.int64)
        pred_label = pred_label.to(label.device)
        self.tp += int(((pred_label == 1) & (label == 1)).sum())
        self.fp += int(((pred_label == 1) & (label == 0)).sum())
        self.tn += int(((pred_label == 0) & (label == 0)).sum())
        self.fn += int(((pred_label == 0) & (label == 1)).sum())


class F1(EvalMetric):
    """Binary F1 = 2PR/(P+R) (reference metric.py F1)."""

    def __init__(self, name: str = "f1"):
        self._counts = _BinaryCounts()
        super().__init__(name)

    def reset(self):
        super().reset()
        if hasattr(self, "_counts"):
            self._counts.reset()

    def update(self, labels, preds):
        for label, pred in zip(_as_list(labels), _as_list(preds)):
            self._counts.update(label, pred)
        c = self._counts
        prec = c.tp / (c.tp + c.fp) if c.tp + c.fp else 0.0
        rec = c.tp / (c.tp + c.fn) if c.tp + c.fn else 0.0
        f1 = 2 * prec * rec / (prec + rec) if prec + rec else 0.0
        self.sum_metric = f1
        self.num_inst = 1


class MCC(EvalMetric):
    """Matthews correlation coefficient (reference metric.py MCC)."""

    def __init__(self, name: str = "mcc"):
        self._counts = _BinaryCounts()
        super().__init__(name)

    def reset(self):
        super().reset()
        if hasattr(self, "_counts"):
            self._counts.reset()

    def update(self, labels, preds):
        import math
        for label, pred in zip(_as_list(labels), _as_list(preds)):
            self._counts.update(label, pred)
        c = self._counts
        den = math.sqrt(float(c.tp + c.fp) * (c.tp + c.fn) *
                        (c.tn + c.fp) * (c.tn + c.fn))
        self.sum_metric = ((c.tp * c.tn - c.fp * c.fn) / den) if den else 0.0
        self.num_inst = 1


class Perplexity(EvalMetric):
    """exp(mean NLL) with optional ignore_label (reference metric.py
    Perplexity)."""

    def __init__(self, ignore_label=None, eps: float = 1e-12,
                 name: str = "perplexity"):
        super().__init__(name)
        self.ignore_label = ignore_label
        self.eps = eps

    def update(self, labels, preds):
        for label, pred in zip(_as_list(labels), _as_list(preds)):
            label = label.reshape(-1).to(torch.int64)
            prob = pred.reshape(label.numel(), -1).float()
            label = label.to(prob.device)
            p = prob[torch.arange(label.numel(), device=prob.device), label]
            nll = -(p + self.eps).log()
            if self.ignore_label is not None:
                keep = label != self.ignore_label
                nll = nll[keep]
            self.sum_metric += nll.sum().item()
            self.num_inst += nll.numel()

    def get(self):
        import math
        if self.num_inst == 0:
            return (self.name, float("nan"))
        return (self.name, math.exp(self.sum_metric / self.num_inst))


class MAE(EvalMetric):
    def __init__(self, name: str = "mae"):
        super().__init__(name)

    def update(self, labels, preds):
        for label, pred in zip(_as_list(labels), _as_list(preds)):
            label = label.float().reshape(pred.shape)
            self.sum_metric += (label - pred.float()).abs().mean().item()
            self.num_inst += 1


class MSE(EvalMetric):
    def __init__(self, name: str = "mse"):
        super().__init__(name)

    def update(self, labels, preds):
        for label, pred in zip(_as_list(labels), _as_list(preds)):
            label = label.float().reshape(pred.shape)
            self.sum_metric += ((label - pred.float()) ** 2).mean().item()
            self.num_inst += 1


class RMSE(MSE):
    def __init__(self, name: str = "rmse"):
        super().__init__(name)

    def get(self):
        import math
        if self.num_inst == 0:
            return (self.name, float("nan"))
        return (self.name, math.sqrt(self.sum_metric / self.num_inst))


class NegativeLogLikelihood(CrossEntropy):
    def __init__(self, eps: float = 1e-12, name: str = "nll-loss"):
        super().__init__(eps, name)


class PearsonCorrelation(EvalMetric):
    """Mean per-batch Pearson r of flattened pred vs label (reference
    metric.py PearsonCorrelation)."""

    def __init__(self, name: str = "pearsonr"):
        super().__init__(name)

    def update(self, labels, preds):
        for label, pred in zip(_as_list(labels), _as_list(preds)):
            x = label.float().reshape(-1)
            y = pred.float().reshape(-1).to(x.device)
            xc = x - x.mean()
            yc = y - y.mean()
            den = xc.norm() * yc.norm()
            self.sum_metric += ((xc * yc).sum() / den).item() if den > 0 else 0.0
            self.num_inst += 1


class CustomMetric(EvalMetric):
    """Wrap a feval(label_np, pred_np) -> float (reference metric.py
    CustomMetric / metric.np)."""

    def __init__(self, feval, name=None, allow_extra_outputs=False):
        name = name or getattr(feval, "__name__", "custom")
        super().__init__(f"custom({name})")
        self._feval = feval
        self._allow_extra_outputs = allow_extra_outputs

    def update(self, labels, preds):
        labels, preds = _as_list(labels), _as_list(preds)
        if not self._allow_extra_outputs and len(labels) != len(preds):
            raise ValueError("labels/preds length mismatch")
        for label, pred in zip(labels, preds):
            v = self._feval(label.cpu().numpy(), pred.cpu().numpy())
            if isinstance(v, tuple):
                s, n = v
                self.sum_metric += s
                self.num_inst += n
            else:
                self.sum_metric += v
                self.num_inst += 1


def np(numpy_feval, name=None, allow_extra_outputs=False):
    """Decorator-style factory: numpy feval -> CustomMetric (reference
    metric.np)."""
    def factory():
        return CustomMetric(numpy_feval, name, allow_extra_outputs)
    factory.__name__ = name or getattr(numpy_feval, "__name__", "custom")
    return factory()


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
    table = {"f1": F1, "mcc": MCC, "perplexity": Perplexity, "mae": MAE,
             "mse": MSE, "rmse": RMSE, "nll_loss": NegativeLogLikelihood,
             "pearsonr": PearsonCorrelation, "loss": Loss}
    if metric in table:
        return table[metric](**kwargs)
    raise ValueError(f"unknown metric {metric}")
