"""Weight initializers, mirroring mxnet.initializer (reference
python/mxnet/initializer.py) on torch tensors.

Dispatch follows the reference's name-based rules (InitDesc/arg-name
suffixes): *_bias -> zero, *_gamma -> one, *_beta -> zero,
*_running_mean/var ("aux") -> zero/one, everything else -> the chosen
weight initializer.
"""
from __future__ import annotations

import math

import torch


class Initializer:
    def __call__(self, name: str, arr: torch.Tensor) -> None:
        with torch.no_grad():
            if name.endswith("bias") or name.endswith("beta"):
                arr.zero_()
            elif name.endswith("gamma") or name.endswith("running_var") or name.endswith("moving_var"):
                arr.fill_(1.0)
            elif name.endswith("running_mean") or name.endswith("moving_mean"):
                arr.zero_()
            else:
                self._init_weight(name, arr)

    def _init_weight(self, name: str, arr: torch.Tensor) -> None:
        raise NotImplementedError


class Uniform(Initializer):
    def __init__(self, scale: float = 0.07):
        self.scale = scale

    def _init_weight(self, name, arr):
        arr.uniform_(-self.scale, self.scale)


class Normal(Initializer):
    def __init__(self, sigma: float = 0.01):
        self.sigma = sigma

    def _init_weight(self, name, arr):
        arr.normal_(0.0, self.sigma)


class Zero(Initializer):
    def _init_weight(self, name, arr):
        arr.zero_()


class One(Initializer):
    def _init_weight(self, name, arr):
        arr.fill_(1.0)


def _fans(arr: torch.Tensor):
    shape = arr.shape
    hw = 1
    if arr.dim() > 2:
        for d in shape[2:]:
            hw *= d
    if arr.dim() >= 2:
        fan_in = shape[1] * hw
        fan_out = shape[0] * hw
    else:
        fan_in = fan_out = arr.numel()
    return fan_in, fan_out


class Xavier(Initializer):
    """Reference initializer.py Xavier: factor by rnd_type/magnitude."""

    def __init__(self, rnd_type: str = "uniform", factor_type: str = "avg", magnitude: float = 3):
        self.rnd_type = rnd_type
        self.factor_type = factor_type
        self.magnitude = float(magnitude)

    def _init_weight(self, name, arr):
        fan_in, fan_out = _fans(arr)
        if self.factor_type == "avg":
            factor = (fan_in + fan_out) / 2.0
        elif self.factor_type == "in":
            factor = fan_in
        elif self.factor_type == "out":
            factor = fan_out
        else:
            raise ValueError("invalid factor_type")
        scale = math.sqrt(self.magnitude / factor)
        if self.rnd_type == "uniform":
            arr.uniform_(-scale, scale)
        elif self.rnd_type == "gaussian":
            arr.normal_(0, scale)
        else:
            raise ValueError("invalid rnd_type")


class MSRAPrelu(Xavier):
    def __init__(self, factor_type: str = "avg", slope: float = 0.25):
        super().__init__("gaussian", factor_type, 2.0 / (1 + slope ** 2))


class Constant(Initializer):
    def __init__(self, value: float = 0.0):
        self.value = float(value)

    def _init_weight(self, name, arr):
        arr.fill_(self.value)


class Orthogonal(Initializer):
    """Orthogonal init via QR of a gaussian (reference initializer.py
    Orthogonal; rand_type='uniform' uses a uniform base instead)."""

    def __init__(self, scale: float = 1.414, rand_type: str = "uniform"):
        self.scale = scale
        self.rand_type = rand_type

    def _init_weight(self, name, arr):
        rows = arr.shape[0]
        cols = arr.numel() // rows
        hi, lo = max(rows, cols), min(rows, cols)
        if self.rand_type == "uniform":
            base = torch.rand(hi, lo, dtype=torch.float32) * 2 - 1
        else:
            base = torch.randn(hi, lo, dtype=torch.float32)
        q, r = torch.linalg.qr(base)  # q: hi x lo, orthonormal columns
        q = q * torch.sign(torch.diagonal(r, 0)).reshape(1, lo)
        mat = q if rows >= cols else q.T
        arr.copy_((self.scale * mat).reshape(arr.shape).to(arr.dtype))


class Bilinear(Initializer):
    """Bilinear upsampling kernel for deconvolution weights (reference
    initializer.py Bilinear)."""

    def _init_weight(self, name, arr):
        w = torch.zeros(arr.numel(), dtype=torch.float32)
        shape = arr.shape
        f = math.ceil(shape[3] / 2.0)
        c = (2 * f - 1 - f % 2) / (2.0 * f)
        for i in range(arr.numel()):
            x = i % shape[3]
            y = (i // shape[3]) % shape[2]
            w[i] = (1 - abs(x / f - c)) * (1 - abs(y / f - c))
        arr.copy_(w.reshape(shape).to(arr.dtype))


class LSTMBias(Initializer):
    """Zero bias with the forget gate set to 1 (reference initializer.py
    LSTMBias _init_bias; gate order i,f,g,o — forget = second quarter).
    Overrides __call__: the base class would zero *bias-named* params before
    this initializer ever saw them."""

    def __init__(self, forget_bias: float = 1.0):
        self.forget_bias = forget_bias

    def __call__(self, name, arr):
        with torch.no_grad():
            self._init_weight(name, arr)

    def _init_weight(self, name, arr):
        arr.zero_()
        n = arr.shape[0] // 4
        arr[n:2 * n] = self.forget_bias


class Mixed:
    """Dispatch by name-pattern to member initializers (reference
    initializer.py Mixed): patterns are regexes tried in order."""

    def __init__(self, patterns, initializers):
        import re
        assert len(patterns) == len(initializers)
        self.map = [(re.compile(p), i) for p, i in zip(patterns, initializers)]

    def __call__(self, name: str, arr: torch.Tensor) -> None:
        for pat, init in self.map:
            if pat.match(name):
                init(name, arr)
                return
        raise ValueError(f"Mixed: no pattern matches parameter {name} — "
                         "add a '.*' catch-all")


_REGISTRY = {
    "default": lambda: Xavier(rnd_type="gaussian", factor_type="in", magnitude=2),
    "xavier": Xavier,
    "msra": MSRAPrelu,
    "uniform": Uniform,
    "normal": Normal,
    "zero": Zero,
    "one": One,
    "constant": Constant,
    "orthogonal": Orthogonal,
    "bilinear": Bilinear,
    "lstmbias": LSTMBias,
}


def create(name: str) -> Initializer:
    return _REGISTRY[name.lower()]()
