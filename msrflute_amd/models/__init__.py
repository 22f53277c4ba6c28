"""Task-plugin loading.

Tasks follow the FLUTE folder convention (reference
experiments/__init__.py:8-43, doc/sphinx/scenarios.rst): a task directory
holds ``model.py`` (BaseModel subclass named by ``model_type``),
``dataloaders/dataset.py`` (class ``Dataset``), ``dataloaders/dataloader.py``
(class ``DataLoader``) and ``config.yaml``.

Resolution order for a task name / model_folder:
1. an on-disk path (absolute, or relative to cwd) — external user tasks;
2. the built-in ``experiments/<task>/`` tree shipped at the repo root.
"""

from __future__ import annotations

import os
from importlib.machinery import SourceFileLoader

import torch

from ..utils import print_rank, to_device

_REPO_ROOT = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def resolve_task_path(*parts: str) -> str:
    """Find a plugin file relative to cwd or the repo root."""
    rel = os.path.join(*parts)
    for base in (os.getcwd(), _REPO_ROOT):
        cand = os.path.join(base, rel)
        if os.path.exists(cand):
            return cand
    return rel  # let the caller fail with a clear path


_MODULE_CACHE = {}


def _load_module(path: str, name: str):
    key = (os.path.abspath(path), name)
    mod = _MODULE_CACHE.get(key)
    if mod is None:
        mod = SourceFileLoader(name, path).load_module()
        _MODULE_CACHE[key] = mod
    return mod


def make_model(model_config, dataloader_type=None, input_dim=-1, output_dim=-1):
    """Instantiate the task model named ``model_type`` from ``model_folder``
    and apply the configured weight init (reference: experiments/__init__.py:8-43).
    """
    model_class = model_config["model_type"]
    folder = str(model_config["model_folder"])
    path = resolve_task_path(folder)
    if not os.path.exists(path):
        raise ValueError(
            f"{model_class} model not found at {folder}; set model_folder in the yaml")
    module = _load_module(path, model_class)
    model = getattr(module, model_class)(model_config)

    weight_init = model_config.get("weight_init", "default")
    if weight_init == "default":
        pass
    elif weight_init == "xavier_normal":
        for p in model.parameters():
            if p.dim() > 1:
                torch.nn.init.xavier_normal_(p.data)
            elif p.dim() == 1:
                p.data.zero_()
        for m in model.modules():
            if isinstance(m, (torch.nn.Embedding, torch.nn.LayerNorm, torch.nn.BatchNorm2d)):
                m.reset_parameters()
    else:
        raise ValueError(f"weight_init {weight_init} not supported")

    model = to_device(model)
    print_rank(f"model ready: {model.__class__.__name__} "
               f"({sum(p.numel() for p in model.parameters())} params)")
    return model


def get_exp_dataloader(task: str):
    """Load the task's DataLoader class (reference: utils/dataloaders_utils.py:9-23)."""
    path = resolve_task_path("experiments", task, "dataloaders", "dataloader.py")
    return _load_module(path, "DataLoader").DataLoader


def get_exp_dataset(task: str):
    """Load the task's Dataset class (reference: utils/dataloaders_utils.py:90-92)."""
    path = resolve_task_path("experiments", task, "dataloaders", "dataset.py")
    return _load_module(path, "Dataset").Dataset
