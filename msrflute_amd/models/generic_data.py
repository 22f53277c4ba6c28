"""Shared dataset/dataloader machinery for array-style tasks.

Implements the universal FLUTE data-blob contract
(``{users, num_samples, user_data, user_data_label}`` — reference
doc/sphinx/scenarios.rst, testing/create_data.py:46-51) once, so each task
plugin only declares its sample shape.  Accepts a path to a JSON/NPZ blob
or an in-memory dict (the per-client slice handed out by
``Client.get_data``).
"""

from __future__ import annotations

import json
import os
from typing import Optional

import numpy as np
import torch

from ..core.dataloader import BaseDataLoader
from ..core.dataset import BaseDataset


def _normalize_blob(blob):
    """Normalize LEAF/FedML variants of the blob convention to FLUTE's
    ``{users, num_samples, user_data, user_data_label}``:

    * LEAF/FedML JSON stores ``user_data[user] = {"x": ..., "y": ...}``
      (reference experiments/cv_lr_mnist/dataloaders/preprocessing.py:42-68
      splits x/y the same way);
    * ``num_samples`` may be missing (derived from the x lists).
    """
    ud = blob.get("user_data", {})
    first = next(iter(ud.values()), None)
    if isinstance(first, dict) and "y" in first and "x" in first \
            and "user_data_label" not in blob:
        blob = dict(blob)
        blob["user_data"] = {u: v["x"] for u, v in ud.items()}
        blob["user_data_label"] = {u: v["y"] for u, v in ud.items()}
        ud = blob["user_data"]
    if "num_samples" not in blob and "users" in blob:
        blob = dict(blob)
        blob["num_samples"] = [len(ud[u]) for u in blob["users"]]
    return blob


def load_hdf5_blob(path):
    """Read the reference's HDF5 federated layout (groups ``users``,
    ``num_samples``, ``user_data/<user>``[, ``user_data_label/<user>``] —
    reference testing/create_data.py:95-140, experiments/*/preprocess.py).
    Requires h5py at runtime; this offline image ships without it, so the
    call fails with an actionable message instead of an ImportError."""
    try:
        import h5py
    except ImportError as e:
        raise RuntimeError(
            "reading .hdf5 federated datasets requires h5py (pip install "
            "h5py on a networked machine); offline runs can convert with "
            "tools/create_data.py to .pt/.npz/.json instead") from e
    with h5py.File(path, "r") as f:
        users = [u.decode() if isinstance(u, bytes) else str(u)
                 for u in f["users"][()]]
        blob = {"users": users,
                "num_samples": list(f["num_samples"][()]),
                "user_data": {}}
        for u in users:
            g = f["user_data"][u]
            # either a dataset per user or an {x, y} group
            if hasattr(g, "keys") and "x" in g:
                blob["user_data"][u] = np.asarray(g["x"])
                if "y" in g:
                    blob.setdefault("user_data_label", {})[u] = \
                        np.asarray(g["y"])
            else:
                blob["user_data"][u] = np.asarray(g)
        if "user_data_label" in f:
            blob["user_data_label"] = {
                u: np.asarray(f["user_data_label"][u]) for u in users}
    return _normalize_blob(blob)


def load_blob(data):
    """Load a data blob from dict / .json / .npz / torch .pt / .hdf5 file
    (LEAF/FedML per-user {x, y} layouts are normalized — _normalize_blob)."""
    if isinstance(data, dict):
        return _normalize_blob(data)
    if data is None:
        raise ValueError("no data provided")
    path = str(data)
    if path.endswith(".npz"):
        z = np.load(path, allow_pickle=True)
        return _normalize_blob(
            {k: z[k].item() if z[k].dtype == object and z[k].shape == ()
             else z[k] for k in z.files})
    if path.endswith(".pt"):
        return _normalize_blob(torch.load(path, weights_only=False))
    if path.endswith(".hdf5") or path.endswith(".h5"):
        return load_hdf5_blob(path)
    with open(path, "r") as f:
        return _normalize_blob(json.load(f))


class ArrayDataset(BaseDataset):
    """Array-feature dataset over the FLUTE blob convention.

    ``user_idx == -1`` (or ``test_only``): concatenates every user's rows
    (server-side eval usage); ``user_idx >= 0``: that single user's shard.
    """

    def __init__(self, data, test_only=False, user_idx=-1, args=None,
                 x_dtype=torch.float32, y_dtype=torch.int64,
                 x_shape: Optional[tuple] = None, **kwargs):
        self.test_only = test_only
        self.x_dtype = x_dtype
        self.y_dtype = y_dtype
        self.x_shape = x_shape
        self.args = args or {}
        self.load_data(data=data, user_idx=user_idx)

    def load_data(self, data=None, user_idx=-1):
        blob = load_blob(data)
        self.user_list = list(blob["users"])
        self.num_samples = list(blob["num_samples"])
        self.user_data = blob["user_data"]
        self.user_data_label = blob.get("user_data_label", None)

        if self.test_only or user_idx == -1:
            users = self.user_list
        else:
            users = [self.user_list[user_idx]]

        xs, ys = [], []
        for u in users:
            ud = self.user_data[u]
            x = ud["x"] if isinstance(ud, dict) else ud
            xs.append(torch.as_tensor(np.asarray(x), dtype=self.x_dtype))
            if self.user_data_label is not None:
                y = self.user_data_label[u]
                ys.append(torch.as_tensor(np.asarray(y), dtype=self.y_dtype))
        self.x = torch.cat(xs) if xs else torch.empty(0)
        if self.x_shape is not None and self.x.numel():
            self.x = self.x.reshape(-1, *self.x_shape)
        self.y = torch.cat(ys) if ys else None

    def __len__(self):
        return len(self.x)

    def __getitem__(self, idx):
        if self.y is not None:
            return self.x[idx], self.y[idx]
        return self.x[idx]


class ArrayDataLoader(BaseDataLoader):
    """Batching dataloader producing ``{'x': tensor, 'y': tensor}`` dicts.

    Slices the dataset's packed tensors directly (no per-sample Python
    collate) — batches are views, so the hot path does zero copies until
    ``to_device`` in the model.
    """

    def __init__(self, data, user_idx=0, mode="train", args=None,
                 dataset_cls=ArrayDataset, **kwargs):
        args = args or {}
        self.mode = mode
        self.args = args
        batch_size = args.get("batch_size", 40)
        self.dataset = dataset_cls(
            data, test_only=(mode != "train"),
            user_idx=user_idx if mode == "train" else -1, args=args)
        self.batch_size = max(1, int(batch_size))
        self.shuffle = (mode == "train")
        # intentionally NOT calling PyTorchDataLoader.__init__: this loader
        # iterates tensor slices itself (faster for small client shards)

    def create_loader(self):
        return self

    def to_device(self):
        """Move the whole shard to the GPU once; batches become device
        slices (kills the per-batch H2D copy of the reference path)."""
        if torch.cuda.is_available() and not self.dataset.x.is_cuda:
            self.dataset.x = self.dataset.x.cuda(non_blocking=True)
            if self.dataset.y is not None:
                self.dataset.y = self.dataset.y.cuda(non_blocking=True)
        return self

    def __len__(self):
        n = len(self.dataset)
        return max(1, (n + self.batch_size - 1) // self.batch_size) if n else 0

    def __iter__(self):
        n = len(self.dataset)
        if n == 0:
            return
        # shuffle order is drawn on CPU from the torch global RNG (keeps the
        # per-client seed discipline device-independent), then moved once
        order = torch.randperm(n) if self.shuffle else torch.arange(n)
        if self.dataset.x.is_cuda:
            order = order.cuda(non_blocking=True)
        for s in range(0, n, self.batch_size):
            idx = order[s:s + self.batch_size]
            batch = {"x": self.dataset.x[idx]}
            if self.dataset.y is not None:
                batch["y"] = self.dataset.y[idx]
            yield batch


class _PackedShardDataset:
    """View-dataset over one packed device tensor (a user's slice)."""

    __slots__ = ("x", "y")

    def __init__(self, x, y):
        self.x = x
        self.y = y

    def __len__(self):
        return len(self.x)


class PackedShardLoader(ArrayDataLoader):
    """ArrayDataLoader over pre-packed device views (no per-client dataset
    construction, no host->device traffic)."""

    def __init__(self, x, y, batch_size):
        self.mode = "train"
        self.args = {}
        self.dataset = _PackedShardDataset(x, y)
        self.batch_size = max(1, int(batch_size))
        self.shuffle = True

    def to_device(self):
        return self


class DeviceShardStore:
    """Entire federated train set packed into device HBM once
    (SURVEY.md §7.1 divergence 3): per-user slices are views into two
    contiguous tensors, staged through ONE pinned-host copy.

    288 GB per MI355X holds every reference dataset whole; ``budget_bytes``
    guards pathological cases (falls back to per-client loading)."""

    def __init__(self, dataset, x_shape, device="cuda",
                 budget_bytes=128 << 30):
        import numpy as np
        users = list(dataset.user_list)
        xs, ys, offsets = [], [], [0]
        for u in users:
            ud = dataset.user_data[u]
            x = np.asarray(ud["x"] if isinstance(ud, dict) else ud,
                           dtype=np.float32)
            y = np.asarray(dataset.user_data_label[u])
            xs.append(x.reshape(len(x), -1))
            ys.append(y)
            offsets.append(offsets[-1] + len(x))
        flat_x = np.concatenate(xs)
        flat_y = np.concatenate(ys)
        nbytes = flat_x.nbytes + flat_y.nbytes
        if nbytes > budget_bytes:
            raise MemoryError(f"shard store would need {nbytes >> 30} GiB")
        hx = torch.from_numpy(flat_x)
        hy = torch.from_numpy(flat_y).to(torch.int64)
        if torch.device(device).type == "cuda":
            hx, hy = hx.pin_memory(), hy.pin_memory()
        self.x = hx.to(device, non_blocking=True)
        self.y = hy.to(device, non_blocking=True)
        if x_shape:
            self.x = self.x.view(-1, *x_shape)
        self.offsets = offsets
        self.user_pos = {u: i for i, u in enumerate(users)}

    def loader_for(self, user, batch_size):
        i = self.user_pos.get(user)
        if i is None:
            return None
        lo, hi = self.offsets[i], self.offsets[i + 1]
        return PackedShardLoader(self.x[lo:hi], self.y[lo:hi], batch_size)
