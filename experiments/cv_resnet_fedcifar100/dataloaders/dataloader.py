"""cv_resnet_fedcifar100 dataloader over the shared array machinery.

Reference: experiments/cv_resnet_fedcifar100/dataloaders/dataloader.py.
"""

import os
from importlib.machinery import SourceFileLoader

from msrflute_amd.models.generic_data import ArrayDataLoader

_Dataset = SourceFileLoader(
    "cv_resnet_fedcifar100_dataset",
    os.path.join(os.path.dirname(__file__), "dataset.py")).load_module().Dataset


class DataLoader(ArrayDataLoader):
    def __init__(self, data, user_idx=0, mode="train", args=None, **kwargs):
        super().__init__(data, user_idx=user_idx, mode=mode, args=args,
                         dataset_cls=_Dataset, **kwargs)
