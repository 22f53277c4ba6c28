"""BaseDataLoader extension contract (reference: core/dataloader.py:10-12)."""

from __future__ import annotations

from abc import ABC

from torch.utils.data import DataLoader as PyTorchDataLoader


class BaseDataLoader(ABC, PyTorchDataLoader):
    """Wrapper contract for task dataloaders."""

    def create_loader(self):
        return self
