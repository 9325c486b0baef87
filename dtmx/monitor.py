"""Parameter/gradient monitor (reference python/mxnet/monitor.py): periodic
norm/stat logging of weights and grads, installed via Module."""
from __future__ import annotations

import logging
import re
from typing import Callable, List, Optional, Tuple

import torch


def _default_stat(x: torch.Tensor) -> float:
    return (x.detach().float().norm() / max(1, x.numel()) ** 0.5).item()


class Monitor:
    def __init__(self, interval: int, stat_func: Optional[Callable] = None,
                 pattern: str = ".*", sort: bool = False):
        self.interval = interval
        self.stat_func = stat_func or _default_stat
        self.pattern = re.compile(pattern)
        self.sort = sort
        self.step = 0
        self.activated = False
        self._module = None

    def install(self, module) -> None:
        self._module = module

    def tic(self):
        self.activated = self.step % self.interval == 0
        self.step += 1

    def toc(self) -> List[Tuple[int, str, float]]:
        if not self.activated or self._module is None:
            return []
        out = []
        for name, p in self._module.symbol.named_parameters():
            if self.pattern.match(name):
                out.append((self.step, name, self.stat_func(p.data)))
            if p.grad is not None and self.pattern.match(name + "_grad"):
                out.append((self.step, name + "_grad", self.stat_func(p.grad)))
        if self.sort:
            out.sort(key=lambda r: r[1])
        return out

    def toc_print(self):
        for step, name, value in self.toc():
            logging.info("Batch: %7d %30s %.5e", step, name, value)
