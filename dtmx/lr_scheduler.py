"""Learning-rate schedulers (reference python/mxnet/lr_scheduler.py).

Schedulers map a global update count -> learning rate; they are consulted by
the optimizer on every update, so the dynamic-minibatch LR rescale (worker
count changing mid-training) composes with them in the optimizer layer.
"""
from __future__ import annotations

import math


class LRScheduler:
    def __init__(self, base_lr: float = 0.01):
        self.base_lr = base_lr

    def __call__(self, num_update: int) -> float:
        raise NotImplementedError


class FactorScheduler(LRScheduler):
    def __init__(self, step: int, factor: float = 1.0, stop_factor_lr: float = 1e-8):
        super().__init__()
        if step < 1:
            raise ValueError("step must be >= 1")
        self.step = step
        self.factor = factor
        self.stop_factor_lr = stop_factor_lr
        self.count = 0

    def __call__(self, num_update: int) -> float:
        while num_update > self.count + self.step:
            self.count += self.step
            self.base_lr *= self.factor
            if self.base_lr < self.stop_factor_lr:
                self.base_lr = self.stop_factor_lr
        return self.base_lr


class MultiFactorScheduler(LRScheduler):
    def __init__(self, step, factor: float = 1.0):
        super().__init__()
        if not all(step[i] < step[i + 1] for i in range(len(step) - 1)):
            raise ValueError("step must be increasing")
        self.step = list(step)
        self.cur_step_ind = 0
        self.factor = factor

    def __call__(self, num_update: int) -> float:
        while self.cur_step_ind <= len(self.step) - 1:
            if num_update > self.step[self.cur_step_ind]:
                self.cur_step_ind += 1
                self.base_lr *= self.factor
            else:
                return self.base_lr
        return self.base_lr


class PolyScheduler(LRScheduler):
    def __init__(self, max_update: int, base_lr: float = 0.01, pwr: int = 2):
        super().__init__(base_lr)
        self.max_update = max_update
        self.power = pwr
        self.base_lr_orig = base_lr

    def __call__(self, num_update: int) -> float:
        if num_update <= self.max_update:
            self.base_lr = self.base_lr_orig * pow(
                1.0 - float(num_update) / float(self.max_update), self.power
            )
        return self.base_lr


class CosineScheduler(LRScheduler):
    """Cosine decay from base_lr to final_lr over max_update (reference
    lr_scheduler.py CosineScheduler)."""

    def __init__(self, max_update: int, base_lr: float = 0.01,
                 final_lr: float = 0.0):
        super().__init__(base_lr)
        if max_update < 1:
            raise ValueError("max_update must be >= 1")
        self.max_update = max_update
        self.final_lr = final_lr
        self.base_lr_orig = base_lr

    def __call__(self, num_update: int) -> float:
        if num_update <= self.max_update:
            self.base_lr = self.final_lr + (self.base_lr_orig - self.final_lr) * (
                1 + math.cos(math.pi * num_update / self.max_update)) / 2
        return self.base_lr


class WarmupScheduler(LRScheduler):
    """Linear warmup to base_lr over warmup_steps, then delegate.

    This carries the dynamic-minibatch-SGD paper's warmup-after-join recipe:
    Module re-arms it when workers join (SURVEY.md §5.3 / arXiv:1904.12043).
    """

    def __init__(self, base_lr: float, warmup_steps: int, after: LRScheduler = None,
                 start_lr: float = 0.0):
        super().__init__(base_lr)
        self.warmup_steps = max(1, warmup_steps)
        self.after = after
        self.start_lr = start_lr
        self.offset = 0  # update count at which the current warmup started

    def rearm(self, num_update: int) -> None:
        self.offset = num_update

    def __call__(self, num_update: int) -> float:
        rel = num_update - self.offset
        target = self.after(num_update) if self.after is not None else self.base_lr
        if rel < self.warmup_steps:
            return self.start_lr + (target - self.start_lr) * rel / self.warmup_steps
        return target
