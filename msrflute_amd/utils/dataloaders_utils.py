"""Dataloader factory functions (reference: utils/dataloaders_utils.py:25-116)."""

from __future__ import annotations

import os

from ..models import get_exp_dataloader, get_exp_dataset
from .logging import print_rank


def make_train_dataloader(data_config, data_path, clientx, task=None,
                          data_strct=None, replay_server=False):
    """Dataloader for client-side (or server-replay) training.
    Reference: utils/dataloaders_utils.py:25-55.
    """
    mode = "train"
    if clientx is None:
        if not data_config.get("train_data_server"):
            print_rank("No server training set is defined")
            return None
        data = os.path.join(data_path, data_config["train_data_server"])
        mode, clientx = "val", 0
    else:
        data = data_config.get("list_of_train_data")
        if data is not None and data_path:
            data = os.path.join(data_path, data)

    DataLoader = get_exp_dataloader(task)
    return DataLoader(data=data_strct if data_strct is not None else data,
                      user_idx=clientx, mode=mode, args=data_config)


def make_val_dataloader(data_config, data_path, task=None, data_strct=None):
    DataLoader = get_exp_dataloader(task)
    val_file = (os.path.join(data_path, data_config["val_data"])
                if data_config.get("val_data") and data_path is not None else data_config.get("val_data"))
    return DataLoader(data=data_strct if data_strct is not None else val_file,
                      user_idx=0, mode="val", args=data_config)


def make_test_dataloader(data_config, data_path, task=None, data_strct=None):
    DataLoader = get_exp_dataloader(task)
    test_file = (os.path.join(data_path, data_config["test_data"])
                 if data_config.get("test_data") and data_path is not None else data_config.get("test_data"))
    return DataLoader(data=data_strct if data_strct is not None else test_file,
                      user_idx=0, mode="test", args=data_config)


def get_data_config(config, mode):
    """Pick the per-mode data config, merging in any semisupervision keys
    (reference: utils/dataloaders_utils.py:100-116)."""
    if mode == "val":
        data_config = config["server_config"]["data_config"]["val"]
    elif mode == "test":
        data_config = config["server_config"]["data_config"]["test"]
    else:
        data_config = config["client_config"]["data_config"]["train"]
    semisup = config["client_config"].get("semisupervision")
    if semisup is None:
        return data_config
    merged = dict(data_config)
    merged.update(dict(semisup))
    return merged


def get_dataset(data_path, config, task, mode, test_only=False, user_idx=-1,
                data_strct=None):
    """Instantiate the task's train/val/test Dataset
    (reference: utils/dataloaders_utils.py:85-98)."""
    data_config = get_data_config(config, mode)
    Dataset = get_exp_dataset(task)
    key = "val_data" if mode == "val" else "test_data" if mode == "test" else "list_of_train_data"
    data_file = data_config.get(key)
    data_pointer = os.path.join(data_path, data_file) if data_file is not None else data_file
    return Dataset(data_pointer if data_strct is None else data_strct,
                   test_only=test_only, user_idx=user_idx, args=data_config)
