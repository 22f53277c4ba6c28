"""Fused LSTM (csrc/lstm_seq.hip) — drop-in for ``nn.LSTM(batch_first=True)``.

Why: MIOpen's RNN path runs a batch-4 T=80 sequence in ~9 ms AND
segfaults under hipGraph capture, which left the Shakespeare benchmark
task at reference speed (PERF.md).  Here the input projection for ALL
timesteps is one hipBLASLt GEMM, the T-step recurrence is ONE kernel
launch per layer (forward and BPTT backward), and the weight gradients
are two GEMMs — ~25 graph-capturable nodes per batch.

Parameter names match ``nn.LSTM`` (``weight_ih_l{k}`` …) so checkpoints
interchange; on CPU (or hidden sizes other than 256) an unregistered
``nn.LSTM`` sharing the SAME Parameter objects runs the reference path.
"""

from __future__ import annotations

import torch
from torch import nn

from . import HAS_EXT, _C


def _pack_fwd(w_hh):
    """[4H, H] -> float4-packed [H/4, 4H, 4]: element (kk, j, d) =
    W_hh[j, 4kk+d], so the forward kernel's per-step weight read is one
    b128 per 4 hidden units (4x fewer loads in the latency-bound loop)."""
    H4, H = w_hh.shape
    return w_hh.t().reshape(H // 4, 4, H4).permute(0, 2, 1).contiguous()


def _pack_bwd(w_hh):
    """[4H, H] -> float4-packed [4H/4, H, 4]: element (jg, h, d) =
    W_hh[4jg+d, h] (the backward dot's b128 form)."""
    H4, H = w_hh.shape
    return w_hh.reshape(H4 // 4, 4, H).permute(0, 2, 1).contiguous()


class _LSTMSeq(torch.autograd.Function):
    @staticmethod
    def forward(ctx, xp, w_hh):
        h_seq, gates, c_seq = _C.lstm_seq_fwd(
            xp.contiguous(), _pack_fwd(w_hh))
        ctx.save_for_backward(gates, c_seq, w_hh, h_seq)
        return h_seq

    @staticmethod
    def backward(ctx, dh):
        gates, c_seq, w_hh, h_seq = ctx.saved_tensors
        dg = _C.lstm_seq_bwd(gates, c_seq, _pack_bwd(w_hh),
                             dh.contiguous())
        B, T, H = h_seq.shape
        h_prev = torch.cat([h_seq.new_zeros(B, 1, H), h_seq[:, :-1]], dim=1)
        # dW_hh = sum_t dgates_t^T h_{t-1} : one GEMM over the stacked steps
        dw_hh = dg.reshape(-1, 4 * H).t().mm(h_prev.reshape(-1, H))
        return dg, dw_hh


class FusedLSTM(nn.Module):
    """Unidirectional batch-first LSTM; inter-layer dropout unsupported
    (the benchmark models use dropout=0)."""

    def __init__(self, input_size, hidden_size, num_layers=1,
                 batch_first=True):
        super().__init__()
        assert batch_first
        self.input_size = input_size
        self.hidden_size = hidden_size
        self.num_layers = num_layers
        fallback = nn.LSTM(input_size, hidden_size, num_layers=num_layers,
                           batch_first=True)
        # register the fallback's Parameters under nn.LSTM's names; keep
        # the fallback itself unregistered so state_dict stays clean
        for name, p in list(fallback.named_parameters()):
            self.register_parameter(name, p)
        object.__setattr__(self, "_fallback", fallback)

    def _use_fused(self, x):
        return (x.is_cuda and HAS_EXT and self.hidden_size == 256
                and x.dtype == torch.float32)

    def forward(self, x, hx=None):
        if not self._use_fused(x):
            self._fallback.flatten_parameters = lambda: None  # shared params
            return self._fallback(x, hx)
        h = x
        for k in range(self.num_layers):
            w_ih = getattr(self, f"weight_ih_l{k}")
            w_hh = getattr(self, f"weight_hh_l{k}")
            b = getattr(self, f"bias_ih_l{k}") + getattr(self, f"bias_hh_l{k}")
            xp = h.matmul(w_ih.t()) + b
            h = _LSTMSeq.apply(xp, w_hh)
        return h, None


class _GRUSeq(torch.autograd.Function):
    """Fused GRU sequence (csrc/gru_seq.hip) — reference cell semantics
    (experiments/nlg_gru/model.py GRU2)."""

    @staticmethod
    def forward(ctx, gi, w_hh, b_hh):
        # float4-packed layouts (see _pack_fwd/_pack_bwd; 3H gate blocks)
        H3, H = w_hh.shape
        w_p = w_hh.t().reshape(H // 4, 4, H3).permute(0, 2, 1).contiguous()
        h_seq, gates, ghn = _C.gru_seq_fwd(
            gi.contiguous(), w_p, b_hh.contiguous())
        ctx.save_for_backward(gates, ghn, h_seq, w_hh)
        return h_seq

    @staticmethod
    def backward(ctx, dh):
        gates, ghn, h_seq, w_hh = ctx.saved_tensors
        H3, H = w_hh.shape
        w_pb = w_hh.reshape(H3 // 4, 4, H).permute(0, 2, 1).contiguous()
        dgi, dgh = _C.gru_seq_bwd(gates, ghn, h_seq, w_pb, dh.contiguous())
        B, T, H = h_seq.shape
        h_prev = torch.cat([h_seq.new_zeros(B, 1, H), h_seq[:, :-1]], dim=1)
        dw_hh = dgh.reshape(-1, 3 * H).t().mm(h_prev.reshape(-1, H))
        db_hh = dgh.sum(dim=(0, 1))
        return dgi, dw_hh, db_hh


def fused_gru_available(hidden_size, x):
    return HAS_EXT and x.is_cuda and hidden_size == 512 \
        and x.dtype == torch.float32


def fused_gru_seq(gi, w_hh_weight, w_hh_bias):
    return _GRUSeq.apply(gi, w_hh_weight, w_hh_bias)
