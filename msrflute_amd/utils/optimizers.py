"""Optimizer factory + layer-wise adaptive optimizers (LARS, LAMB).

Same optimizer menu as the reference factory (utils/utils.py:27-64 and
utils/optimizers/{lars,lamb,adamW}.py), written fresh: LARS (You et al.,
arXiv:1708.03888) and LAMB (You et al., arXiv:1904.00962) from their
published update rules; ``adamW`` maps to ``torch.optim.AdamW`` (modern
PyTorch ships it natively).

On a GPU these run on the flat parameter arena through the fused HIP
optimizer kernels in ``msrflute_amd.ops`` (see ServerOptimizer in
msrflute_amd/core/trainer.py); these torch classes remain the semantic
reference and the CPU path.
"""

from __future__ import annotations

import copy

import torch
from torch.optim import Optimizer


class LarsSGD(Optimizer):
    """SGD with layer-wise adaptive rate scaling (LARS, arXiv:1708.03888).

    local_lr = trust_coef * ||w|| / (||g|| + wd*||w|| + eps); momentum on
    the scaled update.  Reference equivalent: utils/optimizers/lars.py.
    """

    def __init__(self, params, lr=1e-2, momentum=0.9, weight_decay=0.0,
                 trust_coef=0.001, eps=1e-8):
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay,
                        trust_coef=trust_coef, eps=eps)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is None:
                    continue
                g = p.grad
                w_norm = p.norm()
                g_norm = g.norm()
                denom = g_norm + group["weight_decay"] * w_norm + group["eps"]
                local_lr = torch.where(
                    (w_norm > 0) & (g_norm > 0),
                    group["trust_coef"] * w_norm / denom,
                    torch.ones_like(w_norm))
                d_p = g + group["weight_decay"] * p
                state = self.state[p]
                if "momentum_buffer" not in state:
                    buf = state["momentum_buffer"] = torch.zeros_like(p)
                else:
                    buf = state["momentum_buffer"]
                buf.mul_(group["momentum"]).add_(d_p, alpha=float(local_lr * group["lr"]))
                p.add_(buf, alpha=-1.0)
        return loss


class LAMB(Optimizer):
    """LAMB (arXiv:1904.00962): Adam step scaled per layer by a trust ratio.

    Reference equivalent: utils/optimizers/lamb.py:33-134.
    """

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-6,
                 weight_decay=0.0):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                grad = p.grad
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p)
                    state["exp_avg_sq"] = torch.zeros_like(p)
                state["step"] += 1
                exp_avg, exp_avg_sq = state["exp_avg"], state["exp_avg_sq"]
                exp_avg.mul_(beta1).add_(grad, alpha=1 - beta1)
                exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
                bc1 = 1 - beta1 ** state["step"]
                bc2 = 1 - beta2 ** state["step"]
                update = (exp_avg / bc1) / ((exp_avg_sq / bc2).sqrt() + group["eps"])
                if group["weight_decay"] != 0:
                    update = update + group["weight_decay"] * p
                w_norm = p.norm()
                u_norm = update.norm()
                trust = torch.where((w_norm > 0) & (u_norm > 0),
                                    w_norm / u_norm, torch.ones_like(w_norm))
                p.add_(update, alpha=float(-group["lr"] * trust))
        return loss


def make_optimizer(optimizer_config, model) -> Optimizer:
    """Build an optimizer from an ``optimizer_config`` dict.

    Same type menu as reference utils/utils.py:27-64; extra keys in the
    config are passed through to the optimizer constructor.
    """
    cfg = copy.deepcopy(dict(optimizer_config))
    opt_type = cfg.pop("type")
    params = model.parameters() if hasattr(model, "parameters") else model
    if opt_type == "sgd":
        return torch.optim.SGD(params, **cfg)
    if opt_type == "adam":
        return torch.optim.Adam(params, **cfg)
    if opt_type == "adamax":
        cfg.pop("amsgrad", None)
        return torch.optim.Adamax(params, **cfg)
    if opt_type in ("lars", "LarsSGD"):
        cfg.pop("amsgrad", None)
        return LarsSGD(params, **cfg)
    if opt_type == "lamb":
        cfg.pop("amsgrad", None)
        return LAMB(params, **cfg)
    if opt_type == "adamW":
        cfg.pop("amsgrad", None)
        return torch.optim.AdamW(params, **cfg)
    raise ValueError(f"{opt_type} optimizer not supported")
