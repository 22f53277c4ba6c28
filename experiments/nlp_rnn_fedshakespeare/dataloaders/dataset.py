"""Fed-Shakespeare blob dataset: int char-id sequences (len 80), label =
next-char sequence (reference dataloaders/dataset.py)."""

import torch

from msrflute_amd.models.generic_data import ArrayDataset


class Dataset(ArrayDataset):
    def __init__(self, data, test_only=False, user_idx=-1, args=None, **kwargs):
        super().__init__(data, test_only=test_only, user_idx=user_idx,
                         args=args, x_dtype=torch.int64, **kwargs)
