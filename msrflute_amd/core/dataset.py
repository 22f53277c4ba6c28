"""BaseDataset extension contract (reference: core/dataset.py:10-27).

The universal FLUTE data blob convention (reference doc/sphinx/scenarios.rst
and testing/create_data.py:46-51) is::

    {"users": [...], "num_samples": [...],
     "user_data": {user: ...}, "user_data_label": {user: ...}}

Datasets expose ``user_list``, ``user_data``, ``user_data_label`` and
``num_samples`` attributes after ``load_data``.
"""

from __future__ import annotations

from abc import ABC, abstractmethod

from torch.utils.data import Dataset as PyTorchDataset


class BaseDataset(ABC, PyTorchDataset):
    """Wrapper contract for task datasets."""

    @abstractmethod
    def __init__(self, **kwargs):
        super().__init__()

    @abstractmethod
    def __getitem__(self, idx, **kwargs):
        """Fetch one sample."""

    @abstractmethod
    def __len__(self):
        """Dataset size."""

    @abstractmethod
    def load_data(self, **kwargs):
        """Read/instantiate the underlying data blob."""
