"""classif_cnn blob dataset: 3x32x32 CIFAR-shaped images.

Reference: experiments/classif_cnn/dataloaders/dataset.py (CIFAR10 HDF5 blob; testing/create_data.py:142-152).
"""

from msrflute_amd.models.generic_data import ArrayDataset


class Dataset(ArrayDataset):
    def __init__(self, data, test_only=False, user_idx=-1, args=None, **kwargs):
        super().__init__(data, test_only=test_only, user_idx=user_idx,
                         args=args, x_shape=(3, 32, 32), **kwargs)
