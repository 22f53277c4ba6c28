"""semisupervision dataloader: dict batches over the labeled view (the
unlabeled phases build their own torch DataLoaders inside the FedLabels
trainer).

Reference: experiments/semisupervision/dataloaders/dataloader.py.
"""

import os
from importlib.machinery import SourceFileLoader

import torch

from msrflute_amd.core.dataloader import BaseDataLoader

_Dataset = SourceFileLoader(
    "semisupervision_dataset",
    os.path.join(os.path.dirname(__file__), "dataset.py")).load_module().Dataset


class DataLoader(BaseDataLoader):
    def __init__(self, data, user_idx=0, mode="train", args=None, **kwargs):
        args = args or {}
        self.mode = mode
        self.batch_size = int(args.get("batch_size", 64))
        self.dataset = _Dataset(
            data, test_only=(mode != "train"),
            user_idx=user_idx if mode == "train" else -1, args=args)

    def create_loader(self):
        return self

    def __len__(self):
        n = len(self.dataset)
        return max(1, (n + self.batch_size - 1) // self.batch_size) if n else 0

    def __iter__(self):
        n = len(self.dataset)
        if n == 0:
            return
        order = torch.randperm(n) if self.mode == "train" else torch.arange(n)
        for s in range(0, n, self.batch_size):
            idx = order[s:s + self.batch_size]
            yield {"x": self.dataset.x[idx], "y": self.dataset.y[idx]}
