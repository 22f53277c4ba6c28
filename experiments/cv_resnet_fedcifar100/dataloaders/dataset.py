"""Fed-CIFAR100 blob dataset: 3x24x24 float images (FedML crop size;
reference experiments/cv_resnet_fedcifar100/dataloaders/dataset.py)."""

from msrflute_amd.models.generic_data import ArrayDataset


class Dataset(ArrayDataset):
    def __init__(self, data, test_only=False, user_idx=-1, args=None, **kwargs):
        super().__init__(data, test_only=test_only, user_idx=user_idx,
                         args=args, x_shape=(3, 24, 24), **kwargs)
