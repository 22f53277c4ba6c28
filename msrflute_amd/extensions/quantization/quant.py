"""Gradient update quantization (reference: extensions/quantization/quant.py:9-100).

Binning semantics preserved: per-layer (or global) min/max + |·|-quantile
threshold, 2^bits linspace bins, values snapped to bin labels, components
under the threshold zeroed.  The reference quantizes-then-dequantizes in
place (the wire stays fp32); we keep that contract, fused into one device
op per segment (kernel K9).
"""

from __future__ import annotations

import logging
from typing import Optional

import torch

from ... import ops
from ...utils import print_rank


def quant_arena(arena, quant_bits: int = 8,
                quant_threshold: Optional[float] = None,
                global_stats: bool = False):
    """Quantize the flat gradient arena in place; no-op when
    quant_threshold is None (reference: quant.py:29-31)."""
    if quant_threshold is None:
        return
    print_rank(f"Performing Gradient Quantization with Prob. Threshold: "
               f"{quant_threshold}", loglevel=logging.INFO)
    n_bins = 2 ** quant_bits
    if global_stats:
        ops.quantize_dequantize(arena.grad, n_bins, quant_threshold)
    else:
        for i in range(len(arena.numels)):
            seg = arena.grad_segment(i)
            if seg.numel():
                ops.quantize_dequantize(seg, n_bins, quant_threshold)


def quant_model(model: torch.nn.Module, quant_bits: int = 8,
                quant_threshold: Optional[float] = None,
                global_stats: bool = False):
    """Per-tensor API parity with the reference (quant.py:9-50) for models
    without an arena."""
    if quant_threshold is None:
        return
    n_bins = 2 ** quant_bits
    if global_stats:
        flat = torch.cat([p.grad.data.flatten() for p in model.parameters()])
        ops.quantize_dequantize(flat, n_bins, quant_threshold)
        off = 0
        for p in model.parameters():
            p.grad.data.copy_(flat[off:off + p.numel()].view(p.shape))
            off += p.numel()
    else:
        for p in model.parameters():
            flat = p.grad.data.flatten().contiguous()
            ops.quantize_dequantize(flat, n_bins, quant_threshold)
            p.grad.data.copy_(flat.view(p.shape))
