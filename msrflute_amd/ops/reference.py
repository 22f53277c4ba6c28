"""Pure-PyTorch reference implementations of the flat-arena ops.

These define the semantics that the CDNA4 HIP kernels in
``msrflute_amd/csrc`` must match (numerics tests compare kernel output
against these on fp32 random tensors).  They are also the CPU execution
path — the framework runs end-to-end on CPU (gloo) with exactly these.

Each op maps to a hot loop of the reference implementation, inventoried in
SURVEY.md §2.4 (K1-K14):

* K1+K2  pseudo_grad:   g = (w_server − w_trained) · weight
* K3     axpy:          y += α·x         (per-client weighted accumulate)
* K4     scale:         x ·= α           (normalize by Σweight)
* K5     sum_sumsq:     Σg, Σg²          (gradient sufficient stats)
* K6     clip_by_norm:  torch.nn.utils.clip_grad_norm_ semantics on a flat buffer
* K7/K8  add_gaussian_noise (DP)
* K10    fused optimizers (torch.optim semantics on flat buffers)
"""

from __future__ import annotations

import math
from typing import Optional

import torch


def pseudo_grad(out: torch.Tensor, w_server: torch.Tensor,
                w_trained: torch.Tensor, weight: float):
    """out = (w_server - w_trained) * weight  (reference: core/client.py:380-383
    + fedavg.py:80-86 fused)."""
    torch.sub(w_server, w_trained, out=out)
    if weight != 1.0:
        out.mul_(weight)
    return out


def axpy(y: torch.Tensor, x: torch.Tensor, alpha: float = 1.0):
    """y += alpha*x (reference: core/strategies/utils.py:21-33)."""
    y.add_(x, alpha=alpha)
    return y


def scale(x: torch.Tensor, alpha: float):
    x.mul_(alpha)
    return x


def sum_sumsq(x: torch.Tensor) -> torch.Tensor:
    """Single-pass Σx, Σx² -> tensor([sum, sumsq]) on x's device
    (reference: core/trainer.py:271-292, which round-trips through numpy
    per tensor per batch — here one fused pass, never leaving the device)."""
    return torch.stack([x.sum(), x.dot(x)])


def l2_norm(x: torch.Tensor) -> torch.Tensor:
    return x.norm()


def clip_by_norm(x: torch.Tensor, max_norm: float, eps: float = 1e-6) -> torch.Tensor:
    """torch.nn.utils.clip_grad_norm_ semantics on a flat buffer: scale by
    max_norm/(norm+eps) when norm > max_norm.  Returns the pre-clip norm
    (0-dim tensor, stays on device)."""
    norm = x.norm()
    coef = max_norm / (norm + eps)
    # clamp(coef, max=1) keeps the op free of host synchronization
    x.mul_(torch.clamp(coef, max=1.0))
    return norm


def add_gaussian_noise(x: torch.Tensor, sigma: float, generator=None):
    """x += sigma * N(0, 1) (reference: extensions/privacy/__init__.py:70-74)."""
    noise = torch.empty_like(x).normal_(0.0, 1.0, generator=generator)
    x.add_(noise, alpha=sigma)
    return x


# ---------------------------------------------------------------------------
# Fused optimizer steps on flat buffers — semantics match torch.optim.
# ---------------------------------------------------------------------------

def sgd_step(param: torch.Tensor, grad: torch.Tensor,
             momentum_buf: Optional[torch.Tensor], *, lr: float,
             momentum: float = 0.0, dampening: float = 0.0,
             weight_decay: float = 0.0, nesterov: bool = False,
             first_step: bool = False):
    """torch.optim.SGD single step on flat buffers."""
    d_p = grad
    if weight_decay != 0.0:
        d_p = d_p.add(param, alpha=weight_decay)
    if momentum != 0.0:
        assert momentum_buf is not None
        if first_step:
            momentum_buf.copy_(d_p)
        else:
            momentum_buf.mul_(momentum).add_(d_p, alpha=1.0 - dampening)
        d_p = d_p.add(momentum_buf, alpha=momentum) if nesterov else momentum_buf
    param.add_(d_p, alpha=-lr)


def adam_step(param: torch.Tensor, grad: torch.Tensor, exp_avg: torch.Tensor,
              exp_avg_sq: torch.Tensor, max_exp_avg_sq: Optional[torch.Tensor],
              *, step: int, lr: float, beta1: float = 0.9, beta2: float = 0.999,
              eps: float = 1e-8, weight_decay: float = 0.0,
              amsgrad: bool = False, adamw: bool = False):
    """torch.optim.Adam / AdamW single step on flat buffers.  ``step`` is the
    1-based step count."""
    if adamw and weight_decay != 0.0:
        param.mul_(1.0 - lr * weight_decay)
        g = grad
    elif weight_decay != 0.0:
        g = grad.add(param, alpha=weight_decay)
    else:
        g = grad
    exp_avg.mul_(beta1).add_(g, alpha=1.0 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g, g, value=1.0 - beta2)
    bc1 = 1.0 - beta1 ** step
    bc2 = 1.0 - beta2 ** step
    if amsgrad:
        torch.maximum(max_exp_avg_sq, exp_avg_sq, out=max_exp_avg_sq)
        denom = (max_exp_avg_sq / bc2).sqrt_().add_(eps)
    else:
        denom = (exp_avg_sq / bc2).sqrt_().add_(eps)
    param.addcdiv_(exp_avg, denom, value=-lr / bc1)


def adamax_step(param: torch.Tensor, grad: torch.Tensor, exp_avg: torch.Tensor,
                exp_inf: torch.Tensor, *, step: int, lr: float,
                beta1: float = 0.9, beta2: float = 0.999, eps: float = 1e-8,
                weight_decay: float = 0.0):
    """torch.optim.Adamax single step on flat buffers."""
    g = grad.add(param, alpha=weight_decay) if weight_decay != 0.0 else grad
    exp_avg.mul_(beta1).add_(g, alpha=1.0 - beta1)
    torch.maximum(exp_inf.mul_(beta2), g.abs().add_(eps), out=exp_inf)
    bc1 = 1.0 - beta1 ** step
    param.addcdiv_(exp_avg, exp_inf, value=-lr / bc1)


def segmented_sqnorm(x: torch.Tensor, seg_offsets: torch.Tensor) -> torch.Tensor:
    """Per-segment Σx² for S segments given offsets tensor of shape [S+1]
    (int64).  Needed by LAMB/LARS per-layer trust ratios on a flat arena
    (SURVEY.md §7.4 item 4)."""
    out = torch.empty(seg_offsets.numel() - 1, dtype=x.dtype, device=x.device)
    offs = seg_offsets.tolist()
    for i in range(len(offs) - 1):
        seg = x[offs[i]:offs[i + 1]]
        out[i] = seg.dot(seg)
    return out


def quantize_dequantize(x: torch.Tensor, n_bins: int, threshold_quantile: float):
    """Gradient binning + sparsification, matching the reference's
    quantize-then-dequantize-in-place semantics
    (extensions/quantization/quant.py:42-100): bins = linspace(min, max,
    n_bins); value -> bins[bucketize(value - binwidth/2)]; zero where
    |value| <= quantile(|value|, q).  In-place on x."""
    if x.numel() == 0:
        return x
    min_g, max_g = x.min(), x.max()
    thresh = torch.quantile(x.abs().float(), threshold_quantile)
    bins = torch.linspace(float(min_g), float(max_g), n_bins,
                          dtype=x.dtype, device=x.device)
    width = bins[1] - bins[0] if n_bins > 1 else torch.zeros_like(min_g)
    idx = torch.bucketize(x - 0.5 * width, bins, right=False)
    binned = bins[idx.clamp_(0, n_bins - 1)]
    x.copy_(torch.where(x.abs() > thresh, binned, torch.zeros_like(x)))
    return x


def gru_gates(g_i: torch.Tensor, g_h: torch.Tensor, h: torch.Tensor):
    """Fused GRU gate math given precomputed input/hidden projections
    (reference cell: experiments/nlg_gru/model.py:20-28).
    g_i, g_h: [B, 3H]; h: [B, H] -> new hidden [B, H]."""
    i_r, i_i, i_n = g_i.chunk(3, dim=-1)
    h_r, h_i, h_n = g_h.chunk(3, dim=-1)
    reset = torch.sigmoid(i_r + h_r)
    update = torch.sigmoid(i_i + h_i)
    new = torch.tanh(i_n + reset * h_n)
    return new + update * (h - new)
