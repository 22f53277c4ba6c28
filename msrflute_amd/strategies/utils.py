"""Strategy helpers (reference: core/strategies/utils.py:11-33)."""

from __future__ import annotations

import math

from .. import ops


def filter_weight(weight: float) -> float:
    """Sanitize an aggregation weight (reference: strategies/utils.py:11-19)."""
    if math.isnan(weight) or not math.isfinite(weight):
        weight = 0.0
    elif weight > 100:
        weight = 100.0
    return weight


def accumulate_flat_grad(worker_trainer, flat_grad, alpha: float = 1.0):
    """Accumulate a client's (pre-weighted) flat pseudo-gradient into the
    server grad arena — one fused axpy (reference per-tensor loop:
    strategies/utils.py:21-33)."""
    ops.axpy(worker_trainer.arena.grad, flat_grad, alpha)
