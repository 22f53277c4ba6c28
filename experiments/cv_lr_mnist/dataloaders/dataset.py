"""MNIST blob dataset (flat 784 features).

Reference: experiments/cv_lr_mnist/dataloaders/dataset.py (flat 784-feature MNIST rows, 1000 users).
"""

from msrflute_amd.models.generic_data import ArrayDataset


class Dataset(ArrayDataset):
    def __init__(self, data, test_only=False, user_idx=-1, args=None, **kwargs):
        super().__init__(data, test_only=test_only, user_idx=user_idx,
                         args=args, x_shape=(784,), **kwargs)
