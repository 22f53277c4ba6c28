"""ecg_cnn blob dataset: 187-sample heartbeat traces, 5 classes.

Reference: experiments/ecg_cnn/dataloaders/dataset.py (heartbeat CSV -> HDF5 rows).
"""

from msrflute_amd.models.generic_data import ArrayDataset


class Dataset(ArrayDataset):
    def __init__(self, data, test_only=False, user_idx=-1, args=None, **kwargs):
        super().__init__(data, test_only=test_only, user_idx=user_idx,
                         args=args, **kwargs)
