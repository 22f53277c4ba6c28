"""FEMNIST blob dataset (28×28 images).

Reference: experiments/cv_cnn_femnist/dataloaders/dataset.py (FedEMNIST HDF5, 3400 users).
"""

from msrflute_amd.models.generic_data import ArrayDataset


class Dataset(ArrayDataset):
    def __init__(self, data, test_only=False, user_idx=-1, args=None, **kwargs):
        super().__init__(data, test_only=test_only, user_idx=user_idx,
                         args=args, x_shape=(28, 28), **kwargs)
