"""Arena-fused optimizers.

``torch.optim``-compatible optimizers whose ``step()`` is ONE fused kernel
over the flat parameter/gradient arena (reference equivalent: per-tensor
``torch.optim`` loops, utils/utils.py:27-64 + utils/optimizers/*).  State
(momentum, Adam moments) also lives in flat buffers; ``state_dict()`` emits
the torch per-parameter format (views split at arena segment boundaries) so
checkpoints stay interchangeable with the torch optimizers.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
from torch.optim import Optimizer

from . import adam_step, adamax_step, segmented_sqnorm, sgd_step
from .arena import ParameterArena


class _ArenaOptimizerBase(Optimizer):
    def __init__(self, arena: ParameterArena, defaults: dict):
        self.arena = arena
        params = list(arena.model.parameters())
        super().__init__(params, defaults)

    def zero_grad(self, set_to_none: bool = False):
        # set_to_none would unbind the arena grad views; always zero in place.
        self.arena.zero_grad()

    @property
    def _hp(self):
        return self.param_groups[0]

    def _split_flat(self, flat: torch.Tensor):
        """Per-param views of a flat state buffer, in parameter order."""
        a = self.arena
        return [flat[a.offsets[i]:a.offsets[i] + a.numels[i]].view(a.shapes[i])
                for i in range(len(a.numels))]


class ArenaSGD(_ArenaOptimizerBase):
    """torch.optim.SGD semantics, one kernel per step."""

    def __init__(self, arena: ParameterArena, lr: float, momentum: float = 0.0,
                 dampening: float = 0.0, weight_decay: float = 0.0,
                 nesterov: bool = False):
        super().__init__(arena, dict(lr=lr, momentum=momentum,
                                     dampening=dampening,
                                     weight_decay=weight_decay,
                                     nesterov=nesterov))
        self.momentum_buf: Optional[torch.Tensor] = None
        self._stepped = False

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        hp = self._hp
        if hp["momentum"] != 0.0 and self.momentum_buf is None:
            self.momentum_buf = self.arena.new_buffer()
        sgd_step(self.arena.data, self.arena.grad, self.momentum_buf,
                 lr=hp["lr"], momentum=hp["momentum"],
                 dampening=hp["dampening"], weight_decay=hp["weight_decay"],
                 nesterov=hp["nesterov"], first_step=not self._stepped)
        self._stepped = True
        return loss

    def reset_state(self):
        """Fresh-optimizer semantics without reallocating (per-client reuse)."""
        self._stepped = False

    def state_dict(self):
        state = {}
        if self.momentum_buf is not None and self._stepped:
            for i, v in enumerate(self._split_flat(self.momentum_buf)):
                state[i] = {"momentum_buffer": v.clone()}
        return {"state": state,
                "param_groups": [dict(self._hp, params=list(range(len(self.arena.numels))))]}

    def load_state_dict(self, sd):
        groups = sd.get("param_groups")
        if groups:
            g = dict(groups[0])
            g.pop("params", None)
            self._hp.update(g)
        state = sd.get("state", {})
        if state:
            if self.momentum_buf is None:
                self.momentum_buf = self.arena.new_buffer()
            views = self._split_flat(self.momentum_buf)
            for i, s in state.items():
                buf = s.get("momentum_buffer")
                if buf is not None:
                    views[int(i)].copy_(buf.to(views[int(i)].device))
            self._stepped = True


class ArenaAdam(_ArenaOptimizerBase):
    """torch.optim.Adam / AdamW / Adamax semantics, one kernel per step."""

    def __init__(self, arena: ParameterArena, lr: float = 1e-3,
                 betas=(0.9, 0.999), eps: float = 1e-8,
                 weight_decay: float = 0.0, amsgrad: bool = False,
                 adamw: bool = False, adamax: bool = False):
        super().__init__(arena, dict(lr=lr, betas=tuple(betas), eps=eps,
                                     weight_decay=weight_decay,
                                     amsgrad=amsgrad))
        self.adamw = adamw
        self.adamax = adamax
        self.exp_avg = arena.new_buffer()
        self.exp_avg_sq = arena.new_buffer()  # exp_inf for adamax
        self.max_exp_avg_sq = arena.new_buffer() if amsgrad else None
        self.step_count = 0

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        hp = self._hp
        beta1, beta2 = hp["betas"]
        self.step_count += 1
        if self.adamax:
            adamax_step(self.arena.data, self.arena.grad, self.exp_avg,
                        self.exp_avg_sq, step=self.step_count, lr=hp["lr"],
                        beta1=beta1, beta2=beta2, eps=hp["eps"],
                        weight_decay=hp["weight_decay"])
        else:
            adam_step(self.arena.data, self.arena.grad, self.exp_avg,
                      self.exp_avg_sq, self.max_exp_avg_sq,
                      step=self.step_count, lr=hp["lr"], beta1=beta1,
                      beta2=beta2, eps=hp["eps"],
                      weight_decay=hp["weight_decay"], amsgrad=hp["amsgrad"],
                      adamw=self.adamw)
        return loss

    def reset_state(self):
        self.exp_avg.zero_()
        self.exp_avg_sq.zero_()
        if self.max_exp_avg_sq is not None:
            self.max_exp_avg_sq.zero_()
        self.step_count = 0

    def state_dict(self):
        state = {}
        if self.step_count > 0:
            m_views = self._split_flat(self.exp_avg)
            v_views = self._split_flat(self.exp_avg_sq)
            x_views = (self._split_flat(self.max_exp_avg_sq)
                       if self.max_exp_avg_sq is not None else None)
            for i in range(len(self.arena.numels)):
                key = "exp_inf" if self.adamax else "exp_avg_sq"
                s = {"step": torch.tensor(float(self.step_count)),
                     "exp_avg": m_views[i].clone(), key: v_views[i].clone()}
                if x_views is not None:
                    s["max_exp_avg_sq"] = x_views[i].clone()
                state[i] = s
        return {"state": state,
                "param_groups": [dict(self._hp, params=list(range(len(self.arena.numels))))]}

    def load_state_dict(self, sd):
        groups = sd.get("param_groups")
        if groups:
            g = dict(groups[0])
            g.pop("params", None)
            for k, v in g.items():
                if k in self._hp:
                    self._hp[k] = tuple(v) if k == "betas" else v
        state = sd.get("state", {})
        if state:
            m_views = self._split_flat(self.exp_avg)
            v_views = self._split_flat(self.exp_avg_sq)
            x_views = (self._split_flat(self.max_exp_avg_sq)
                       if self.max_exp_avg_sq is not None else None)
            for i, s in state.items():
                i = int(i)
                m_views[i].copy_(s["exp_avg"].to(m_views[i].device))
                vv = s.get("exp_avg_sq", s.get("exp_inf"))
                v_views[i].copy_(vv.to(v_views[i].device))
                if x_views is not None and "max_exp_avg_sq" in s:
                    x_views[i].copy_(s["max_exp_avg_sq"].to(x_views[i].device))
                st = s.get("step", 0)
                self.step_count = int(st.item() if torch.is_tensor(st) else st)


def make_arena_optimizer(optimizer_config: dict, arena: ParameterArena):
    """Fused-optimizer factory; returns None when the type has no fused
    implementation yet (caller falls back to the torch factory)."""
    cfg = dict(optimizer_config)
    opt_type = cfg.pop("type")
    cfg.pop("amsgrad_keep", None)
    if opt_type == "sgd":
        allowed = {k: v for k, v in cfg.items()
                   if k in ("lr", "momentum", "dampening", "weight_decay", "nesterov")}
        return ArenaSGD(arena, **allowed)
    if opt_type in ("adam", "adamW", "adamax"):
        allowed = {k: v for k, v in cfg.items()
                   if k in ("lr", "betas", "eps", "weight_decay", "amsgrad")}
        return ArenaAdam(arena, adamw=(opt_type == "adamW"),
                         adamax=(opt_type == "adamax"), **allowed)
    return None
