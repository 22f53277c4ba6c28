"""Small shared helpers (device placement, json status log, math utils).

Reference equivalents: utils/utils.py:497-560 (flatten/unflatten are NOT
reproduced — the arena design in msrflute_amd.comm.arena makes flat views
first-class instead of round-tripping through numpy), 543-544 (to_device),
546-560 (update_json_log), 598-678 (personalization math).
"""

from __future__ import annotations

import json
import os
from typing import Dict

import torch


def to_device(x):
    """Move a tensor/module to the bound GPU when one exists."""
    return x.cuda() if torch.cuda.is_available() else x


def update_json_log(log_path: str, status_info: Dict):
    """Merge ``status_info`` into the JSON file at ``log_path``
    (reference: utils/utils.py:546-560)."""
    elems = {}
    if os.path.exists(log_path):
        with open(log_path, "r") as fp:
            elems = json.load(fp)
    elems.update(status_info)
    with open(log_path, "w") as fp:
        json.dump(elems, fp)
    return elems


class AverageMeter:
    """Tracks a ratio numerator/denominator average."""

    def __init__(self, name=""):
        self.name = name
        self.num = 0.0
        self.den = 0.0

    def add(self, num, den):
        self.num += num
        self.den += den

    @property
    def value(self):
        return self.num / self.den if self.den else 0.0


def softmax_weights(losses, beta: float):
    """DGA softmax weighting ``exp(-beta * loss)`` (reference:
    core/strategies/dga.py:111-129)."""
    t = torch.as_tensor(losses, dtype=torch.float64)
    return torch.exp(-beta * t)


def alpha_update(local_model, global_model, alpha: float, lr: float) -> float:
    """Personalization convex-interpolation weight update (reference:
    utils/utils.py:598-616): gradient of the interpolated loss wrt alpha is
    sum over params of grad · (local - global); one SGD step on alpha,
    clipped to [0, 1].
    """
    grad_alpha = 0.0
    for lp, gp in zip(local_model.parameters(), global_model.parameters()):
        if lp.grad is None:
            continue
        diff = (lp.data - gp.data).flatten()
        grad = (alpha * lp.grad.data.flatten()
                + (1 - alpha) * gp.grad.data.flatten()) if gp.grad is not None \
            else alpha * lp.grad.data.flatten()
        grad_alpha += torch.dot(grad, diff).item()
    grad_alpha += 0.02 * alpha
    alpha = alpha - lr * grad_alpha
    return float(min(max(alpha, 0.0), 1.0))
