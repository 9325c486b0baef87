"""Autograd ops: CDNA4 HIP kernels on GPU, torch reference on CPU.

Layout contract (MI355X-first): 4D activations are NHWC in memory (torch
"channels_last" on logical NCHW tensors); conv weights logical (K,C,R,S) with
channels_last memory = (K,R,S,C) — the implicit-GEMM B operand layout.
Compute dtype bf16 (fp32 accumulate inside kernels); CPU path is fp32 torch.

Reference kernel parity:
  conv fwd/dgrad/wgrad   <- src/operator/nn/convolution.cu (cuDNN/im2col)
  batch_norm fwd/bwd     <- src/operator/nn/batch_norm.cu:208-360
  pooling                <- src/operator/nn/pool.cuh
  relu / add(+relu)      <- mshadow_op elementwise kernels
  softmax+CE (fused)     <- src/operator/nn/softmax-inl.h:166-260, softmax_output
  linear                 <- fully_connected-inl.h (cuBLAS) -> dtmx MFMA GEMM
  sgd/momentum updates   <- optimizer_op-inl.h:86,305,430 -> fused multi-tensor
"""
from __future__ import annotations

from typing import Optional, Tuple

import os

import torch
import torch.nn.functional as F

from .hip import require_ext


_FALLBACK = set(os.environ.get("DTMX_FALLBACK", "").split(",")) - {""}


def _use_hip(*tensors, op=None) -> bool:
    if op is not None and op in _FALLBACK:  # debug: force torch fallback
        return False
    ts = [t for t in tensors if isinstance(t, torch.Tensor)]
    if not any(t.is_cuda for t in ts):
        return False
    # The MFMA/LDS kernels implement the training dtypes (bf16/fp16) only;
    # fp32 work routes to torch's rocBLAS/MIOpen-free ops instead of
    # erroring (reference supports fp32 paths, e.g. FullyConnected fp32).
    return all(
        t.dtype in (torch.bfloat16, torch.float16)
        for t in ts
        if t.is_floating_point()
    )


# --------------------------------------------------------------------- conv

class _ConvNHWC(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, stride, padding, want_stats):
        ext = require_ext()
        ctx.save_for_backward(x, w)
        ctx.stride = stride
        ctx.padding = padding
        if want_stats:
            y, ps, pss = ext.conv_fwd_stats(x, w, stride, padding)
            ctx.mark_non_differentiable(ps, pss)
            return y, ps, pss
        y = ext.conv_fwd(x, w, stride, padding)
        z = y.new_zeros(0)
        ctx.mark_non_differentiable(z)
        return y, z, z

    @staticmethod
    def backward(ctx, dy, _dps=None, _dpss=None):
        ext = require_ext()
        x, w = ctx.saved_tensors
        dy = dy.contiguous(memory_format=torch.channels_last)
        dx = dw = None
        if ctx.needs_input_grad[0]:
            dx = ext.conv_dgrad(dy, w, ctx.stride, ctx.padding, x.shape[2], x.shape[3])
        if ctx.needs_input_grad[1]:
            dw = ext.conv_wgrad(x, dy, w.shape[2], w.shape[3], ctx.stride, ctx.padding)
        return dx, dw, None, None, None


def conv2d(x, w, stride: int = 1, padding: int = 0, want_stats: bool = False):
    """want_stats: also return the fused per-block BN statistic slabs
    (psum, psumsq) computed in the conv epilogue — the following BatchNorm
    then skips its own whole-tensor stats pass."""
    if _use_hip(x, op="conv"):
        C = x.shape[1]
        if C % 8 != 0:
            # zero-pad channels to 8 (the 3-channel stem) so the implicit-GEMM
            # gather path applies — cheaper than materializing im2col
            # (~0.5 GB/step on resnet-50 bs128). F.pad is differentiable, so
            # dw slices back through autograd.
            padc = 8 - (C % 8)
            x = F.pad(x, (0, 0, 0, 0, 0, padc)).contiguous(
                memory_format=torch.channels_last
            )
            w = F.pad(w, (0, 0, 0, 0, 0, padc)).contiguous(
                memory_format=torch.channels_last
            )
        y, ps, pss = _ConvNHWC.apply(x, w, stride, padding, want_stats)
        if want_stats:
            return y, ps, pss
        return y
    return F.conv2d(x, w, None, stride, padding)


# --------------------------------------------------------------- batch norm

class _BatchNormNHWC(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var, training, momentum,
                eps, fuse_relu, residual, pre_psum=None, pre_psumsq=None):
        ext = require_ext()
        if training:
            y, save_mean, save_invstd = ext.bn_fwd_train(
                x, gamma, beta, running_mean, running_var, momentum, eps,
                fuse_relu, residual, pre_psum, pre_psumsq
            )
            ctx.save_for_backward(x, gamma, save_mean, save_invstd, y)
            ctx.fuse_relu = fuse_relu
            ctx.has_residual = residual is not None
        else:
            y = ext.bn_fwd_infer(x, gamma, beta, running_mean, running_var, eps,
                                 fuse_relu, residual)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext()
        x, gamma, save_mean, save_invstd, y = ctx.saved_tensors
        dy = dy.contiguous(memory_format=torch.channels_last)
        out = ext.bn_bwd(x, dy, gamma, save_mean, save_invstd,
                         ctx.fuse_relu, y, ctx.has_residual)
        dres = out[3] if ctx.has_residual else None
        return (out[0], out[1], out[2], None, None, None, None, None, None, dres,
                None, None)


def batch_norm(x, gamma, beta, running_mean, running_var, training: bool,
               momentum: float = 0.9, eps: float = 1e-5, fuse_relu: bool = False,
               residual=None, pre_stats=None):
    """NHWC batch norm. `momentum` follows mxnet semantics:
    moving = moving*momentum + batch*(1-momentum) (reference batch_norm-inl.h),
    i.e. torch's momentum is (1 - mxnet momentum).

    `residual`: fused y = [relu](bn(x) + residual) — the resnet block tail in
    one pass (bn_apply's extra read beats a separate add_relu round trip)."""
    if _use_hip(x, op="bn"):
        ps, pss = pre_stats if pre_stats is not None else (None, None)
        return _BatchNormNHWC.apply(x, gamma, beta, running_mean, running_var,
                                    training, momentum, eps, fuse_relu, residual,
                                    ps, pss)
    y = F.batch_norm(x, running_mean, running_var, gamma, beta, training,
                     1.0 - momentum, eps)
    if residual is not None:
        y = y + residual
    return F.relu(y) if fuse_relu else y


# ------------------------------------------------------------------ pooling

class _MaxPoolNHWC(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, kernel, stride, padding):
        ext = require_ext()
        y, idx = ext.maxpool_fwd(x, kernel, stride, padding)
        ctx.save_for_backward(idx)
        ctx.in_shape = x.shape
        ctx.params = (kernel, stride, padding)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext()
        (idx,) = ctx.saved_tensors
        k, s, p = ctx.params
        dy = dy.contiguous(memory_format=torch.channels_last)
        return (
            ext.maxpool_bwd(dy, idx, ctx.in_shape[2], ctx.in_shape[3], k, s, p),
            None, None, None,
        )


def max_pool2d(x, kernel: int, stride: int, padding: int = 0):
    if _use_hip(x, op="pool"):
        return _MaxPoolNHWC.apply(x, kernel, stride, padding)
    return F.max_pool2d(x, kernel, stride, padding)


class _GlobalAvgPoolNHWC(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = require_ext()
        ctx.in_shape = x.shape
        return ext.global_avgpool_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext()
        n, c, h, w = ctx.in_shape
        return ext.global_avgpool_bwd(dy.contiguous(), h, w)


def global_avg_pool(x):
    """(N,C,H,W) -> (N,C)"""
    if _use_hip(x, op="gap"):
        return _GlobalAvgPoolNHWC.apply(x)
    return x.mean(dim=(2, 3))


# -------------------------------------------------------------- elementwise

class _ReLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = require_ext()
        y = ext.relu_fwd(x)
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext()
        (y,) = ctx.saved_tensors
        return ext.relu_bwd(dy.contiguous(memory_format=_mf(dy)), y)


def relu(x):
    if _use_hip(x, op="relu"):
        return _ReLU.apply(x)
    return F.relu(x)


class _AddRelu(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b):
        ext = require_ext()
        y = ext.add_relu_fwd(a, b)
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext()
        (y,) = ctx.saved_tensors
        g = ext.relu_bwd(dy.contiguous(memory_format=_mf(dy)), y)
        return g, g


def add_relu(a, b):
    """Fused residual add + ReLU (resnet hot path: one HBM round trip)."""
    if _use_hip(a, op="relu"):
        return _AddRelu.apply(a, b)
    return F.relu(a + b)


def _mf(t):
    return (
        torch.channels_last
        if t.dim() == 4 and t.is_contiguous(memory_format=torch.channels_last)
        else torch.contiguous_format
    )


class _LRN(torch.autograd.Function):
    """Cross-channel LRN, window 5 (reference src/operator/nn/lrn.cc;
    alexnet/googlenet-era). HIP kernel streams the NHWC channel window
    (pool_elem.hip lrn_fwd/bwd); torch's pad+avg_pool3d lowering measured
    3.3x of the alexnet step."""

    @staticmethod
    def forward(ctx, x, alpha, beta, knorm):
        ext = require_ext()
        xc = x.contiguous(memory_format=torch.channels_last)
        ctx.save_for_backward(xc)
        ctx.hyper = (alpha, beta, knorm)
        return ext.lrn_fwd(xc, alpha, beta, knorm)

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext()
        (xc,) = ctx.saved_tensors
        alpha, beta, knorm = ctx.hyper
        return ext.lrn_bwd(xc, dy, alpha, beta, knorm), None, None, None


def local_response_norm(x, nsize=5, alpha=1e-4, beta=0.75, knorm=2.0):
    if (nsize == 5 and x.dim() == 4 and x.shape[1] % 8 == 0
            and _use_hip(x, op="lrn")):
        return _LRN.apply(x, alpha, beta, knorm)
    return F.local_response_norm(x, nsize, alpha=alpha, beta=beta, k=knorm)


# ------------------------------------------------------------------- linear

class _Linear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, bias):
        ext = require_ext()
        ctx.save_for_backward(x, w)
        ctx.has_bias = bias is not None
        return ext.linear_fwd(x, w, bias)

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext()
        x, w = ctx.saved_tensors
        dy = dy.contiguous()
        dx = ext.linear_dgrad(dy, w) if ctx.needs_input_grad[0] else None
        dw = ext.linear_wgrad(dy, x) if ctx.needs_input_grad[1] else None
        db = dy.float().sum(0).to(dy.dtype) if ctx.has_bias else None
        return dx, dw, db


def linear(x, w, bias=None):
    """x:(M,K) w:(N,K) -> (M,N)   (FullyConnected: X @ W^T + b)"""
    if _use_hip(x, op="linear"):
        return _Linear.apply(x, w, bias)
    return F.linear(x, w, bias)


# -------------------------------------------------------- fused softmax + CE

class _SoftmaxCE(torch.autograd.Function):
    """SoftmaxOutput semantics: forward returns sum-CE loss; backward emits
    d_logits = (softmax(logits) - onehot(label)) — NOT scaled by 1/batch
    (the optimizer's rescale_grad handles averaging; reference
    softmax_output-inl.h)."""

    @staticmethod
    def forward(ctx, logits, label):
        ext = require_ext()
        loss, probs = ext.softmax_ce_fwd(logits, label)
        ctx.save_for_backward(probs, label)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        ext = require_ext()
        probs, label = ctx.saved_tensors
        return ext.softmax_ce_bwd(probs, label, dloss), None


def softmax_cross_entropy_sum(logits, label):
    if _use_hip(logits, op="softmax"):
        return _SoftmaxCE.apply(logits, label.to(torch.int32))
    return F.cross_entropy(logits.float(), label.long(), reduction="sum")


# ------------------------------------------------------- dropout / layernorm

class _Dropout(torch.autograd.Function):
    """Mask-gen + scale dropout (reference src/operator/nn/dropout.cu) on the
    stateless splitmix64 counter RNG — no cuRAND state arrays."""

    @staticmethod
    def forward(ctx, x, p, seed):
        ext = require_ext()
        y, mask = ext.dropout_fwd(x, p, seed)
        ctx.save_for_backward(mask)
        ctx.p = p
        return y.view_as(x)

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext()
        (mask,) = ctx.saved_tensors
        return ext.dropout_bwd(dy, mask, ctx.p).view_as(dy), None, None


def dropout(x, p: float, training: bool, seed: int | None = None):
    if not training or p <= 0.0:
        return x
    if _use_hip(x) and os.environ.get("DTMX_TORCH_DROPOUT") != "1":
        if seed is None:
            seed = int(torch.randint(0, 2 ** 62, (1,)).item())
        return _Dropout.apply(x, float(p), seed)
    return F.dropout(x, p, training)


class _LayerNorm(torch.autograd.Function):
    """Row LayerNorm (reference src/operator/nn/layer_norm.cu): one block per
    row, fp32 accumulation; dgamma/dbeta via fp32 atomics."""

    @staticmethod
    def forward(ctx, x, gamma, beta, eps):
        ext = require_ext()
        y, mean, rstd = ext.layer_norm_fwd(x, gamma, beta, eps)
        ctx.save_for_backward(x, gamma, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext()
        x, gamma, mean, rstd = ctx.saved_tensors
        dx, dgamma, dbeta = ext.layer_norm_bwd(x, dy, gamma, mean, rstd)
        return dx, dgamma.to(gamma.dtype), dbeta.to(gamma.dtype), None


def layer_norm(x, gamma, beta, eps: float = 1e-5):
    """Normalize over the last dim of a 2D input."""
    if _use_hip(x) and x.dim() == 2:
        return _LayerNorm.apply(x, gamma, beta, eps)
    return F.layer_norm(x.float(), (x.shape[-1],), gamma.float(), beta.float(),
                        eps).to(x.dtype)


# ------------------------------------------------------- sync batch norm

class _SyncBatchNormNHWC(torch.autograd.Function):
    """Cross-rank BN (reference contrib/sync_batch_norm.cu): per-rank channel
    sums -> all-reduce -> normalize with GLOBAL mean/var. dgamma/dbeta are the
    LOCAL sums — the kvstore/bucketer gradient all-reduce makes them global
    (same division of labor as torch.nn.SyncBatchNorm under DDP)."""

    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var, momentum, eps,
                group):
        import torch.distributed as dist
        world = dist.get_world_size(group)
        rows = x.shape[0] * x.shape[2] * x.shape[3]
        if x.is_cuda:
            ext = require_ext()
            s, ss = ext.bn_local_sums(x)
        else:
            xf = x.float()
            s = xf.sum(dim=(0, 2, 3))
            ss = (xf * xf).sum(dim=(0, 2, 3))
        buf = torch.stack([s, ss])
        dist.all_reduce(buf, group=group)
        count = rows * world
        if x.is_cuda:
            y, save_mean, save_invstd = ext.bn_fwd_presummed(
                x, gamma, beta, running_mean, running_var, momentum, eps,
                False, None, buf[0].contiguous(), buf[1].contiguous(), count)
        else:
            mean = buf[0] / count
            var = (buf[1] / count - mean * mean).clamp_min(0)
            invstd = (var + eps).rsqrt()
            save_mean, save_invstd = mean, invstd
            xf = x.float()
            y = ((xf - mean.view(1, -1, 1, 1)) * invstd.view(1, -1, 1, 1)
                 * gamma.float().view(1, -1, 1, 1)
                 + beta.float().view(1, -1, 1, 1)).to(x.dtype)
            unbiased = var * count / (count - 1) if count > 1 else var
            running_mean.mul_(momentum).add_(mean * (1 - momentum))
            running_var.mul_(momentum).add_(unbiased * (1 - momentum))
        ctx.save_for_backward(x, gamma, save_mean, save_invstd, y)
        ctx.count = count
        ctx.group = group
        return y

    @staticmethod
    def backward(ctx, dy):
        import torch.distributed as dist
        x, gamma, save_mean, save_invstd, y = ctx.saved_tensors
        if x.is_cuda:
            ext = require_ext()
            dy = dy.contiguous(memory_format=torch.channels_last)
            tdb, tdg = ext.bn_bwd_sums(x, dy, y, save_mean, save_invstd, False)
        else:
            dyf = dy.float()
            xhat = ((x.float() - save_mean.view(1, -1, 1, 1))
                    * save_invstd.view(1, -1, 1, 1))
            tdb = dyf.sum(dim=(0, 2, 3))
            tdg = (dyf * xhat).sum(dim=(0, 2, 3))
        dgamma = tdg.to(gamma.dtype)
        dbeta = tdb.to(gamma.dtype)
        buf = torch.stack([tdb, tdg])
        dist.all_reduce(buf, group=ctx.group)
        if x.is_cuda:
            dx = ext.bn_bwd_dx_presummed(
                x, dy, y, save_mean, save_invstd, gamma, buf[0].contiguous(),
                buf[1].contiguous(), ctx.count, False, False)[0]
        else:
            xhat = ((x.float() - save_mean.view(1, -1, 1, 1))
                    * save_invstd.view(1, -1, 1, 1))
            g = gamma.float().view(1, -1, 1, 1)
            inv = 1.0 / ctx.count
            dx = (g * save_invstd.view(1, -1, 1, 1)
                  * (dy.float() - buf[0].view(1, -1, 1, 1) * inv
                     - xhat * buf[1].view(1, -1, 1, 1) * inv)).to(x.dtype)
        return dx, dgamma, dbeta, None, None, None, None, None


def sync_batch_norm(x, gamma, beta, running_mean, running_var, training: bool,
                    momentum: float = 0.9, eps: float = 1e-5, group=None):
    """Cross-rank NHWC batch norm; falls back to local BN when not distributed
    or in eval mode (running stats are already synchronized by construction)."""
    import torch.distributed as dist
    if (not training or not dist.is_available() or not dist.is_initialized()
            or dist.get_world_size(group) == 1):
        return batch_norm(x, gamma, beta, running_mean, running_var, training,
                          momentum, eps)
    return _SyncBatchNormNHWC.apply(x, gamma, beta, running_mean, running_var,
                                    momentum, eps, group)


# --------------------------------------------------- embedding / take / sparse

class RowSparse:
    """A row-sparse 2-D gradient: only `rows` carry values (reference
    NDArray kRowSparseStorage, include/mxnet/ndarray.h; the wire unit of
    kvstore_dist.h PushRowSparse/PullRowSparse_). `rows` is int64 sorted
    unique, `values` is [len(rows), D]."""

    __slots__ = ("rows", "values", "shape")

    def __init__(self, rows: torch.Tensor, values: torch.Tensor, shape):
        self.rows = rows
        self.values = values
        self.shape = tuple(shape)

    def to_dense(self) -> torch.Tensor:
        out = torch.zeros(self.shape, dtype=self.values.dtype,
                          device=self.values.device)
        out[self.rows] = self.values
        return out

    @staticmethod
    def from_dense(t: torch.Tensor) -> "RowSparse":
        nz = (t != 0).any(dim=1).nonzero(as_tuple=True)[0]
        return RowSparse(nz, t[nz].clone(), t.shape)


class _Take(torch.autograd.Function):
    """Embedding/take (reference src/operator/tensor/indexing_op.cu Take /
    AddTakeGrad): out[i] = table[idx[i]]; backward scatter-adds into an
    fp32 accumulator (HIP atomics on GPU)."""

    @staticmethod
    def forward(ctx, table, idx):
        ctx.save_for_backward(idx)
        ctx.V = table.shape[0]
        ctx.w_dtype = table.dtype
        if _use_hip(table, op="take") and table.shape[1] % 8 == 0:
            ext = require_ext()
            return ext.take_fwd(table, idx.reshape(-1).long()).view(
                *idx.shape, table.shape[1])
        return table.index_select(0, idx.reshape(-1).long().clamp_(
            0, table.shape[0] - 1)).view(*idx.shape, table.shape[1])

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        D = dy.shape[-1]
        dy2 = dy.reshape(-1, D)
        flat = idx.reshape(-1).long()
        if dy.is_cuda and dy.dtype in (torch.bfloat16, torch.float16):
            ext = require_ext()
            dtab = ext.take_bwd(dy2.contiguous(), flat, ctx.V)
        else:
            dtab = torch.zeros(ctx.V, D, dtype=torch.float32, device=dy.device)
            dtab.index_add_(0, flat, dy2.float())
        return dtab.to(ctx.w_dtype), None


def take(table: torch.Tensor, idx: torch.Tensor) -> torch.Tensor:
    """out[...] = table[idx[...]] over a 2-D table (Embedding forward)."""
    return _Take.apply(table, idx)


def embedding_row_sparse_grad(dy: torch.Tensor, idx: torch.Tensor,
                              vocab: int) -> RowSparse:
    """Compute the ROW-SPARSE gradient of an embedding lookup directly —
    only touched rows are materialized (reference kvstore row-sparse push:
    kvstore_dist.h:452-481 moves exactly these rows). `dy` is [..., D]."""
    D = dy.shape[-1]
    dy2 = dy.reshape(-1, D)
    flat = idx.reshape(-1).long()
    rows, inverse = torch.unique(flat, sorted=True, return_inverse=True)
    if dy.is_cuda and dy.dtype in (torch.bfloat16, torch.float16):
        ext = require_ext()
        vals = ext.take_bwd(dy2.contiguous(), inverse, rows.numel())
    else:
        vals = torch.zeros(rows.numel(), D, dtype=torch.float32,
                           device=dy.device)
        vals.index_add_(0, inverse, dy2.float())
    return RowSparse(rows, vals.to(dy.dtype), (vocab, D))


# ------------------------------------------------------------ deconvolution

class _DeconvNHWC(torch.autograd.Function):
    """Transposed convolution (reference src/operator/nn/deconvolution.cu) —
    no new kernels: deconv forward IS conv dgrad, its data-grad IS conv
    forward, and its weight-grad is conv wgrad with the operands swapped.
    Weight layout (Cin, Cout, R, S) channels_last, mxnet convention."""

    @staticmethod
    def forward(ctx, x, w, stride, padding):
        ext = require_ext()
        ctx.save_for_backward(x, w)
        ctx.stride, ctx.padding = stride, padding
        R = w.shape[2]
        H = (x.shape[2] - 1) * stride - 2 * padding + R
        W = (x.shape[3] - 1) * stride - 2 * padding + R
        return ext.conv_dgrad(x, w, stride, padding, H, W, None)

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext()
        x, w = ctx.saved_tensors
        dy = dy.contiguous(memory_format=torch.channels_last)
        dx = (ext.conv_fwd(dy, w, ctx.stride, ctx.padding)
              if ctx.needs_input_grad[0] else None)
        dw = None
        if ctx.needs_input_grad[1]:
            R = w.shape[2]
            dw = ext.conv_wgrad(dy, x, R, R, ctx.stride, ctx.padding)
        return dx, dw, None, None


def deconv2d(x, w, stride: int = 1, padding: int = 0):
    """Transposed conv, NHWC HIP path on GPU; torch fallback elsewhere.
    `w` is (Cin, Cout, R, S)."""
    if _use_hip(x, op="deconv") and w.shape[1] % 8 == 0 and w.shape[0] % 8 == 0:
        return _DeconvNHWC.apply(x, w, stride, padding)
    return F.conv_transpose2d(x, w, None, stride, padding)
