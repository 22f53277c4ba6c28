"""Aggregation strategies (reference: core/strategies/__init__.py:9-23)."""

from .base import BaseStrategy
from .dga import DGA
from .fedavg import FedAvg
from .fedlabels import FedLabels


def select_strategy(strategy: str):
    """FedProx reuses FedAvg aggregation (weights ∝ client sample count);
    its proximal term lives in the Trainer (run_train_epoch_fedprox)."""
    s = strategy.lower()
    if s == "dga":
        return DGA
    if s in ("fedavg", "fedprox"):
        return FedAvg
    if s == "fedlabels":
        return FedLabels
    raise ValueError(f"cannot use strategy {strategy}")


__all__ = ["BaseStrategy", "DGA", "FedAvg", "FedLabels", "select_strategy"]
