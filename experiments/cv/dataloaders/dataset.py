"""cv dataset: federated blob OR flat {x, y} arrays partitioned on load.

A flat blob (keys ``x``/``y``) is split into ``args['total_num_clients']``
clients by the Dirichlet (``args['alpha']``) or fixed-label partitioner
(reference: experiments/cv/data.py) with seed ``args['partition_seed']``.
Optional per-client rotation augmentation (``args['want_transform']``)
rotates each client's images by a client-specific multiple of 90 degrees.
"""

import os
from importlib.machinery import SourceFileLoader

import numpy as np
import torch

from msrflute_amd.models.generic_data import ArrayDataset, load_blob

_data = SourceFileLoader(
    "cv_partition",
    os.path.join(os.path.dirname(__file__), "..", "data.py")).load_module()


class Dataset(ArrayDataset):
    def __init__(self, data, test_only=False, user_idx=-1, args=None, **kwargs):
        args = args or {}
        blob = load_blob(data)
        if "users" not in blob:  # flat arrays -> partition now
            blob = _data.partition_blob(
                blob["x"], blob["y"],
                n_clients=int(args.get("total_num_clients", 100)),
                alpha=float(args.get("alpha", 1.0)),
                seed=int(args.get("partition_seed", 2020)),
                mode=args.get("partition_mode", "dirichlet"),
                labels_per_client=int(args.get("labels_per_client", 2)))
        self._want_transform = bool(args.get("want_transform", False))
        self._user_idx = user_idx
        super().__init__(blob, test_only=test_only, user_idx=user_idx,
                         args=args, x_shape=tuple(args.get(
                             "x_shape", (3, 32, 32))))
        if self._want_transform and user_idx >= 0 and not test_only:
            k = _data.rotation_for_client(user_idx)
            if k:
                self.x = torch.rot90(self.x, k, dims=(-2, -1)).contiguous()
