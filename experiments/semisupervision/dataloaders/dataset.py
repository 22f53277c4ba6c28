"""semisupervision dataset: labeled / unlabeled / unlabeled+randaug views.

Reference: experiments/semisupervision/dataloaders/dataset.py + the
three-dataset client convention (reference client.py:92-94; our
Client.get_train_dataset caches user_idx -2 = unlabeled pool and -3 =
unlabeled with RandAugment).  Each user's samples are split: the first
``args['num_labeled_per_user']`` are labeled, the rest unlabeled.

``__getitem__`` returns ``(x, y)`` tuples (the FedLabels trainer feeds
these through plain torch DataLoaders).
"""

import os
from importlib.machinery import SourceFileLoader

import numpy as np
import torch

from msrflute_amd.core.dataset import BaseDataset
from msrflute_amd.models.generic_data import load_blob

_ra = SourceFileLoader(
    "semisup_randaugment",
    os.path.join(os.path.dirname(__file__), "..", "randaugment.py")
).load_module()


class Dataset(BaseDataset):
    def __init__(self, data, test_only=False, user_idx=-1, args=None, **kwargs):
        args = args or {}
        self.args = args
        self.test_only = test_only
        self.n_labeled = int(args.get("num_labeled_per_user", 10))
        self.randaug = None
        blob = load_blob(data)
        self.load_data(blob, user_idx)

    def load_data(self, blob, user_idx):
        self.user_list = list(blob["users"])
        all_samples = list(blob["num_samples"])
        src_data = blob["user_data"]
        src_label = blob.get("user_data_label") or {}
        variant = ("unlab_rand" if user_idx == -3
                   else "unlab" if user_idx == -2 else "labeled")

        # per-user views with the labeled/unlabeled split applied
        self.user_data, self.user_data_label, self.num_samples = {}, {}, []
        for u, ns in zip(self.user_list, all_samples):
            ud = src_data[u]
            randaug_flag = False
            if isinstance(ud, dict):
                x = np.asarray(ud["x"])
                randaug_flag = bool(ud.get("randaug", False))
            else:
                x = np.asarray(ud)
            y = np.asarray(src_label[u]) if u in src_label else \
                np.zeros(len(x), dtype=np.int64)
            k = min(self.n_labeled, len(x))
            if variant == "labeled":
                if isinstance(ud, dict) and ud.get("split_done"):
                    xs, ys = x, y
                else:
                    xs, ys = x[:k], y[:k]
            else:
                xs, ys = x[k:], y[k:]
            self.user_data[u] = {"x": xs, "split_done": True,
                                 "randaug": variant == "unlab_rand"
                                 or randaug_flag}
            self.user_data_label[u] = ys
            self.num_samples.append(len(xs))

        if variant == "unlab_rand" or any(
                d.get("randaug") for d in self.user_data.values()):
            self.randaug = _ra.RandAugment(
                n=int(self.args.get("randaug_n", 2)),
                magnitude=float(self.args.get("randaug_mag", 0.3)))

        users = (self.user_list if self.test_only or user_idx < 0
                 else [self.user_list[user_idx]])
        xs = [torch.as_tensor(self.user_data[u]["x"], dtype=torch.float32)
              for u in users]
        ys = [torch.as_tensor(self.user_data_label[u], dtype=torch.int64)
              for u in users]
        self.x = torch.cat(xs) if xs else torch.empty(0)
        self.y = torch.cat(ys) if ys else torch.empty(0, dtype=torch.int64)

    def __len__(self):
        return len(self.x)

    def __getitem__(self, idx):
        x = self.x[idx]
        if self.randaug is not None:
            x = self.randaug(x)
        return x, self.y[idx]
