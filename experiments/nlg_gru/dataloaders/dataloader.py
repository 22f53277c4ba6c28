"""nlg_gru dataloader: duration-packed dynamic batches padded with -1
(the model masks ids < 0).  Reference: experiments/nlg_gru/dataloaders/
dataloader.py:25-65."""

import os
from importlib.machinery import SourceFileLoader

import numpy as np
import torch

from msrflute_amd.core.dataloader import BaseDataLoader
from msrflute_amd.utils.data_utils import BatchSampler, DynamicBatchSampler

_Dataset = SourceFileLoader(
    "nlg_gru_dataset",
    os.path.join(os.path.dirname(__file__), "dataset.py")).load_module().Dataset

PAD_ID = -1


class _IndexSampler:
    """Minimal shuffled index sampler exposing ``.dataset`` for
    DynamicBatchSampler."""

    def __init__(self, dataset, shuffle=True):
        self.dataset = dataset
        self.shuffle = shuffle

    def __iter__(self):
        order = torch.randperm(len(self.dataset)).tolist() if self.shuffle \
            else range(len(self.dataset))
        return iter(order)

    def __len__(self):
        return len(self.dataset)


class DataLoader(BaseDataLoader):
    def __init__(self, data, user_idx=0, mode="train", args=None, **kwargs):
        args = args or {}
        self.mode = mode
        self.batch_size = int(args.get("batch_size", 64))
        self.dataset = _Dataset(
            data, test_only=(mode != "train"),
            user_idx=user_idx if mode == "train" else -1, args=args)
        if mode == "train":
            self.batch_sampler = DynamicBatchSampler(
                _IndexSampler(self.dataset, shuffle=True),
                frames_threshold=int(args.get("max_num_words", 25)) *
                self.batch_size,
                max_batch_size=self.batch_size,
                unsorted_batch=bool(args.get("unsorted_batch", False)),
                fps=1)
        else:
            self.batch_sampler = BatchSampler(
                self.dataset, batch_size=self.batch_size, randomize=False)

    def create_loader(self):
        return self

    def __len__(self):
        return max(1, len(self.batch_sampler.batches))

    def __iter__(self):
        for batch_idx in self.batch_sampler:
            seqs = [self.dataset[i][0] for i in batch_idx]
            if not seqs:
                continue
            T = max(len(s) for s in seqs)
            x = np.full((len(seqs), T), PAD_ID, dtype=np.int64)
            for r, s in enumerate(seqs):
                x[r, : len(s)] = s
            yield {"x": torch.from_numpy(x),
                   "total_frames": sum(len(s) for s in seqs)}
