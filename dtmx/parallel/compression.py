"""2-bit gradient compression with error-feedback residual (reference
src/kvstore/gradient_compression-inl.h:40-135, gradient_compression.cu).

Scheme (same as the reference):
  - per element: quantize to {-threshold, 0, +threshold} by comparing
    (grad + residual) against +-threshold; residual accumulates the
    quantization error ("error feedback").
  - wire format packs 16 two-bit codes per uint32.

In the collective engine the compressed payload is what the all-reduce/
all-gather moves; `compress_decompress` (quantize -> dequantize locally,
then all-reduce the sparse-valued result) preserves the reference's
*numerics* for sync DP, since the server dequantized before merging
(kvstore_dist_server.h:606 DataHandleCompressed).

A HIP kernel (dtmx/csrc/compress.hip) implements quantize/dequantize on GPU;
this module holds the semantic reference (torch ops, CPU and GPU capable)
used for testing and as fallback.
"""
from __future__ import annotations

import torch


class TwoBitCompression:
    def __init__(self, threshold: float = 0.5):
        if threshold <= 0:
            raise ValueError("threshold must be positive")
        self.threshold = float(threshold)
        self._residual = {}

    def compress_decompress(self, grad: torch.Tensor, key=None) -> torch.Tensor:
        """Quantize+dequantize with persistent per-tensor residual.

        `key` should be the kvstore key: the reference keeps residual_[key]
        per key (kvstore_dist.h:778). Falling back to tensor metadata would
        alias the residuals of same-shaped parameters, corrupting error
        feedback — callers that push multiple tensors MUST pass the key."""
        if key is None:
            key = (grad.shape, grad.device, grad.dtype)
        res = self._residual.get(key)
        if res is None or res.shape != grad.shape:
            res = torch.zeros_like(grad, dtype=torch.float32)
            self._residual[key] = res
        if grad.is_cuda and grad.dtype == torch.bfloat16:
            from ..ops.hip import require_ext

            ext = require_ext()
            packed = ext.quantize_2bit(grad.contiguous().view(-1), res.view(-1),
                                       self.threshold)
            return ext.dequantize_2bit(packed, grad.numel(), self.threshold).view(
                grad.shape
            )
        g = grad.float() + res
        q = torch.where(
            g >= self.threshold,
            torch.full_like(g, self.threshold),
            torch.where(g <= -self.threshold, torch.full_like(g, -self.threshold),
                        torch.zeros_like(g)),
        )
        res.copy_(g - q)
        return q.to(grad.dtype)

    # -- wire format (used by tests and the bandwidth tool) -----------------
    def quantize(self, grad: torch.Tensor, residual: torch.Tensor) -> torch.Tensor:
        """Returns packed uint32 codes (16 x 2-bit per word): 01 = +thr,
        10 = -thr, 00 = zero (matches quantize_2bit struct layout)."""
        g = grad.float() + residual
        pos = g >= self.threshold
        neg = g <= -self.threshold
        q = torch.where(pos, torch.full_like(g, self.threshold),
                        torch.where(neg, torch.full_like(g, -self.threshold),
                                    torch.zeros_like(g)))
        residual.copy_(g - q)
        codes = torch.zeros_like(g, dtype=torch.int64)
        codes[pos] = 1
        codes[neg] = 2
        n = codes.numel()
        pad = (-n) % 16
        if pad:
            codes = torch.cat([codes.reshape(-1), codes.new_zeros(pad)])
        codes = codes.reshape(-1, 16)
        shifts = torch.arange(0, 32, 2, device=codes.device, dtype=torch.int64)
        packed = (codes << shifts).sum(dim=1)
        return packed.to(torch.int64)  # int64 holding uint32 range

    def dequantize(self, packed: torch.Tensor, numel: int,
                   dtype=torch.float32, device=None) -> torch.Tensor:
        shifts = torch.arange(0, 32, 2, device=packed.device, dtype=torch.int64)
        codes = (packed.reshape(-1, 1) >> shifts) & 0x3
        codes = codes.reshape(-1)[:numel]
        out = torch.zeros(numel, dtype=torch.float32, device=packed.device)
        out[codes == 1] = self.threshold
        out[codes == 2] = -self.threshold
        if device is not None:
            out = out.to(device)
        return out.to(dtype)
