"""BaseModel extension contract (reference: core/model.py:14-51).

Task plugins subclass ``BaseModel`` and implement ``loss(batch)`` and
``inference(batch)``; ``inference`` returns a dict with at least
``output``, ``acc`` and ``batch_size``, plus optional custom metrics as
``{'value': v, 'higher_is_better': bool}`` entries.
"""

from __future__ import annotations

from abc import ABC, abstractmethod

import torch


class BaseModel(ABC, torch.nn.Module):
    """Wrapper contract for task models."""

    @abstractmethod
    def __init__(self, **kwargs):
        super().__init__()

    @abstractmethod
    def loss(self, input):
        """Forward step returning the scalar training loss."""

    @abstractmethod
    def inference(self, input):
        """Forward step returning an eval-metrics dict:
        ``{'output': ..., 'acc': ..., 'batch_size': ..., <custom>: {'value': v,
        'higher_is_better': bool}}``."""

    def set_eval(self):
        self.eval()

    def set_train(self):
        self.train()
