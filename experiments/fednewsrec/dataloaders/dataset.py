"""fednewsrec dataset: per-user click histories + candidate slates.

Blob convention (MIND-style, pretokenized):
``user_data[user] = {"history": [n, H, T], "candidates": [n, K, T],
"labels": [n] (train: index of the clicked candidate) or [n, K]
(eval: binary relevance)}``.
Reference: experiments/fednewsrec/dataloaders/dataset.py.
"""

import numpy as np
import torch

from msrflute_amd.core.dataset import BaseDataset
from msrflute_amd.models.generic_data import load_blob


class Dataset(BaseDataset):
    def __init__(self, data, test_only=False, user_idx=-1, args=None, **kwargs):
        self.test_only = test_only
        self.load_data(data, user_idx)

    def load_data(self, data=None, user_idx=-1):
        blob = load_blob(data)
        self.user_list = list(blob["users"])
        self.num_samples = list(blob["num_samples"])
        self.user_data = blob["user_data"]
        self.user_data_label = blob.get("user_data_label")
        users = (self.user_list if self.test_only or user_idx == -1
                 else [self.user_list[user_idx]])
        hist, cand, lab = [], [], []
        for u in users:
            ud = self.user_data[u]
            hist.append(np.asarray(ud["history"]))
            cand.append(np.asarray(ud["candidates"]))
            lab.append(np.asarray(ud["labels"]))
        self.history = torch.as_tensor(np.concatenate(hist), dtype=torch.int64)
        self.candidates = torch.as_tensor(np.concatenate(cand),
                                          dtype=torch.int64)
        self.labels = torch.as_tensor(np.concatenate(lab))

    def __len__(self):
        return len(self.labels)

    def __getitem__(self, idx):
        return (self.history[idx], self.candidates[idx]), self.labels[idx]
