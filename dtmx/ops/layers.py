"""nn.Module layers over dtmx ops (the building blocks dtmx.models use).

These hold torch Parameters (so autograd/bucketing/checkpointing work
unchanged) and route compute through dtmx.ops.functional — HIP kernels on
MI355X, torch reference on CPU.
"""
from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn

from . import functional as DF


class Conv2dNHWC(nn.Module):
    def __init__(self, in_channels, out_channels, kernel_size, stride=1, padding=0,
                 bias=False):
        super().__init__()
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.kernel_size = kernel_size
        self.stride = stride
        self.padding = padding
        w = torch.empty(out_channels, in_channels, kernel_size, kernel_size)
        nn.init.kaiming_normal_(w, mode="fan_in", nonlinearity="relu")
        self.weight = nn.Parameter(w)
        self.bias = nn.Parameter(torch.zeros(out_channels)) if bias else None

    def forward(self, x):
        # bias-free training convs can emit fused BN-stat slabs for the
        # (conventionally following) BatchNorm; carried as a tensor attribute
        # so unrelated consumers simply ignore them. Round 1's atomicAdd slab
        # scheme measured -3.5% and was off by default; the rebuilt no-atomic
        # epilogue (EpiBF16FwdStats) is on by default — DTMX_FUSE_BN_STATS=0
        # restores the standalone stats pass.
        import os as _os
        if (self.bias is None and x.is_cuda and self.training
                and _os.environ.get("DTMX_FUSE_BN_STATS", "1") == "1"):
            y, ps, pss = DF.conv2d(x, self.weight, self.stride, self.padding,
                                   want_stats=True)
            if ps.numel():
                y._dtmx_bn_stats = (ps, pss)
            return y
        y = DF.conv2d(x, self.weight, self.stride, self.padding)
        if self.bias is not None:
            y = y + self.bias.reshape(1, -1, 1, 1).to(y.dtype)
        return y

    def extra_repr(self):
        return (f"{self.in_channels}, {self.out_channels}, k={self.kernel_size}, "
                f"s={self.stride}, p={self.padding}")


class BatchNorm2dNHWC(nn.Module):
    """mxnet-momentum BN (moving = m*moving + (1-m)*batch, m=0.9 default,
    reference batch_norm-inl.h); running stats tracked in fp32 regardless of
    compute dtype. Optional fused ReLU (the ResNet hot path)."""

    def __init__(self, num_features, eps=1e-5, momentum=0.9, fuse_relu=False):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.fuse_relu = fuse_relu
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))

    def _apply(self, fn, recurse=True):
        # keep running stats fp32 when the module is cast to bf16/fp16
        super()._apply(fn, recurse)
        self.running_mean.data = self.running_mean.data.float()
        self.running_var.data = self.running_var.data.float()
        return self

    def forward(self, x, residual=None):
        pre_stats = getattr(x, "_dtmx_bn_stats", None) if self.training else None
        return DF.batch_norm(
            x,
            self.weight,
            self.bias,
            self.running_mean,
            self.running_var,
            self.training,
            self.momentum,
            self.eps,
            self.fuse_relu,
            residual,
            pre_stats,
        )


class GroupedConv2dNHWC(nn.Module):
    """Grouped / depthwise conv (mobilenet, resnext). The hand HIP conv
    kernels are dense-gather only, so groups>1 routes through torch's conv
    on channels_last memory (MIOpen on ROCm) — the documented long-tail
    substrate path; groups==1 uses Conv2dNHWC instead."""

    def __init__(self, in_channels, out_channels, kernel_size, stride=1,
                 padding=0, groups=1, bias=False):
        super().__init__()
        assert in_channels % groups == 0 and out_channels % groups == 0
        self.stride = stride
        self.padding = padding
        self.groups = groups
        w = torch.empty(out_channels, in_channels // groups, kernel_size,
                        kernel_size)
        nn.init.kaiming_normal_(w, mode="fan_in", nonlinearity="relu")
        self.weight = nn.Parameter(w)
        self.bias = nn.Parameter(torch.zeros(out_channels)) if bias else None

    def forward(self, x):
        import torch.nn.functional as F
        return F.conv2d(x, self.weight.to(x.dtype),
                        self.bias.to(x.dtype) if self.bias is not None else None,
                        self.stride, self.padding, 1, self.groups)


class ReLU(nn.Module):
    def forward(self, x):
        return DF.relu(x)


class AddRelu(nn.Module):
    def forward(self, a, b):
        return DF.add_relu(a, b)


class MaxPool2dNHWC(nn.Module):
    def __init__(self, kernel_size, stride, padding=0):
        super().__init__()
        self.kernel_size = kernel_size
        self.stride = stride
        self.padding = padding

    def forward(self, x):
        return DF.max_pool2d(x, self.kernel_size, self.stride, self.padding)


class GlobalAvgPool(nn.Module):
    def forward(self, x):
        return DF.global_avg_pool(x)


class LRN(nn.Module):
    """Cross-channel local response norm (reference mx.sym.LRN,
    src/operator/nn/lrn.cc): y = x / (knorm + alpha/nsize * sum_n x^2)^beta.
    torch's local_response_norm computes exactly this convention — the
    substrate path for this long-tail op (alexnet/googlenet era only)."""

    def __init__(self, nsize=5, alpha=1e-4, beta=0.75, knorm=2.0):
        super().__init__()
        self.nsize = nsize
        self.alpha = alpha
        self.beta = beta
        self.knorm = knorm

    def forward(self, x):
        # HIP window-5 kernel on GPU 16-bit; torch op otherwise. (A
        # cumsum-over-channels torch reformulation measured SLOWER than
        # even pad+avg_pool3d — 18.8k vs 28.1k img/s alexnet bs2048.)
        return DF.local_response_norm(x, self.nsize, alpha=self.alpha,
                                      beta=self.beta, knorm=self.knorm)


class LinearBF16(nn.Module):
    def __init__(self, in_features, out_features, bias=True):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        w = torch.empty(out_features, in_features)
        nn.init.normal_(w, 0, 0.01)
        self.weight = nn.Parameter(w)
        self.bias = nn.Parameter(torch.zeros(out_features)) if bias else None

    def forward(self, x):
        return DF.linear(x, self.weight, self.bias)


class Dropout(nn.Module):
    """Train-time mask+scale dropout on the HIP counter-RNG kernel
    (reference nn/dropout.cu semantics; torch fallback on CPU)."""

    def __init__(self, p: float = 0.5):
        super().__init__()
        self.p = p

    def forward(self, x):
        return DF.dropout(x, self.p, self.training)

    def extra_repr(self):
        return f"p={self.p}"


class LayerNorm(nn.Module):
    """Last-dim LayerNorm with fp32 gamma/beta (reference nn/layer_norm.cu)."""

    def __init__(self, dim: int, eps: float = 1e-5):
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(dim, dtype=torch.float32))
        self.bias = nn.Parameter(torch.zeros(dim, dtype=torch.float32))

    def _apply(self, fn, recurse=True):  # keep affine params fp32 under .to(bf16)
        dev = fn(torch.empty(0))
        self.weight.data = self.weight.data.to(device=dev.device)
        self.bias.data = self.bias.data.to(device=dev.device)
        return self

    def forward(self, x):
        return DF.layer_norm(x, self.weight, self.bias, self.eps)


class SyncBatchNorm2dNHWC(BatchNorm2dNHWC):
    """Cross-rank BN (reference contrib/sync_batch_norm.cu): batch statistics
    are all-reduced over the data-parallel group each step, so every rank
    normalizes with the GLOBAL batch mean/var. Use for small per-rank batches
    where local statistics are too noisy."""

    def __init__(self, num_features, eps=1e-5, momentum=0.9, process_group=None):
        super().__init__(num_features, eps=eps, momentum=momentum, fuse_relu=False)
        self.process_group = process_group

    def forward(self, x, residual=None):
        assert residual is None, "SyncBatchNorm: residual fusion not supported"
        return DF.sync_batch_norm(x, self.weight, self.bias, self.running_mean,
                                  self.running_var, self.training, self.momentum,
                                  self.eps, self.process_group)


class Embedding(nn.Module):
    """Embedding lookup over a 2-D table (reference mx.sym.Embedding /
    src/operator/tensor/indexing_op.cu). Forward gathers rows with the HIP
    take kernel; backward is a scatter-add. For the row-sparse gradient
    workflow (kvstore row_sparse push of only touched rows), use
    DF.embedding_row_sparse_grad with the saved indices."""

    def __init__(self, input_dim: int, output_dim: int, dtype=None):
        super().__init__()
        self.input_dim = input_dim
        self.output_dim = output_dim
        w = torch.empty(input_dim, output_dim)
        nn.init.uniform_(w, -0.07, 0.07)  # mxnet default embedding init scale
        self.weight = nn.Parameter(w)

    def forward(self, idx):
        return DF.take(self.weight, idx)

    def extra_repr(self):
        return f"{self.input_dim}, {self.output_dim}"


class Deconv2dNHWC(nn.Module):
    """Transposed convolution (reference mx.sym.Deconvolution)."""

    def __init__(self, in_channels, out_channels, kernel_size, stride=1,
                 padding=0):
        super().__init__()
        self.stride = stride
        self.padding = padding
        w = torch.empty(in_channels, out_channels, kernel_size, kernel_size)
        nn.init.kaiming_normal_(w, mode="fan_in", nonlinearity="relu")
        self.weight = nn.Parameter(w)

    def forward(self, x):
        return DF.deconv2d(x, self.weight, self.stride, self.padding)
