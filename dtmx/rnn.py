"""RNN layer surface (reference python/mxnet/rnn/ + src/operator/rnn.cu /
cudnn_rnn-inl.h). Sequence models are OUTSIDE the reference's elastic-DP
hot path (its only sequence scale-out is RNN + cuDNN RNN),
so dtmx keeps the API surface on the torch-ROCm substrate rather than
hand-writing CDNA4 RNN kernels: the recurrences are GEMV/GEMM chains the
substrate already runs on rocBLAS. What this module guarantees is the
reference's layer semantics and state-shape conventions so RNN examples
port with import changes."""
from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn as nn


class RNNLayer(nn.Module):
    """FusedRNN-style multi-layer RNN (reference mx.rnn.FusedRNNCell):
    mode in {'rnn_tanh', 'rnn_relu', 'lstm', 'gru'}; input (T, N, C) or
    (N, T, C) with layout='NTC'."""

    def __init__(self, hidden_size: int, num_layers: int = 1,
                 mode: str = "lstm", bidirectional: bool = False,
                 dropout: float = 0.0, input_size: Optional[int] = None,
                 layout: str = "TNC"):
        super().__init__()
        assert input_size is not None, "input_size required (no deferred init)"
        self.mode = mode
        self.layout = layout
        kw = dict(input_size=input_size, hidden_size=hidden_size,
                  num_layers=num_layers, bidirectional=bidirectional,
                  dropout=dropout, batch_first=(layout == "NTC"))
        if mode == "lstm":
            self.rnn = nn.LSTM(**kw)
        elif mode == "gru":
            self.rnn = nn.GRU(**kw)
        elif mode in ("rnn_tanh", "rnn_relu"):
            self.rnn = nn.RNN(nonlinearity=mode.split("_")[1], **kw)
        else:
            raise ValueError(f"unknown RNN mode {mode}")

    def forward(self, x, states=None):
        out, st = self.rnn(x, states)
        return out, st

    def begin_state(self, batch_size: int, ctx=None) -> Tuple[torch.Tensor, ...]:
        dev = ctx.torch_device() if ctx is not None else "cpu"
        d = self.rnn.num_layers * (2 if self.rnn.bidirectional else 1)
        h = torch.zeros(d, batch_size, self.rnn.hidden_size, device=dev)
        if self.mode == "lstm":
            return (h, torch.zeros_like(h))
        return (h,)


class LSTMCell(nn.LSTMCell):
    """reference mx.rnn.LSTMCell analog."""


class GRUCell(nn.GRUCell):
    """reference mx.rnn.GRUCell analog."""
