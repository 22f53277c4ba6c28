"""classif_cnn dataloader over the shared array machinery.

Reference: experiments/classif_cnn/dataloaders/dataloader.py.
"""

import os
from importlib.machinery import SourceFileLoader

from msrflute_amd.models.generic_data import ArrayDataLoader

_Dataset = SourceFileLoader(
    "classif_cnn_dataset",
    os.path.join(os.path.dirname(__file__), "dataset.py")).load_module().Dataset


class DataLoader(ArrayDataLoader):
    def __init__(self, data, user_idx=0, mode="train", args=None, **kwargs):
        super().__init__(data, user_idx=user_idx, mode=mode, args=args,
                         dataset_cls=_Dataset, **kwargs)
