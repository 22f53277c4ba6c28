"""Evaluation rounds (reference: core/evaluation.py).

The reference dispatches TESTVAL commands to the worker pool and merges
sample-weighted metric dicts as results stream back (evaluation.py:146-183).
Here every rank evaluates its deterministic share of the eval-client groups
on the replicated global model and one ``all_gather_object`` merges the
per-group (metrics, count) pairs — every rank ends with identical metrics,
so best-model tracking and LR decisions stay in lockstep.
"""

from __future__ import annotations

import logging
import os
from typing import Dict, List, Optional

import numpy as np
import torch

from ..utils import print_rank
from ..utils.dataloaders_utils import make_test_dataloader, make_val_dataloader
from .client import Client
from .trainer import run_validation_generic


class Evaluation:

    def __init__(self, config, model_path, runtime, idx_val_clients,
                 idx_test_clients, val_dataset=None, test_dataset=None):
        self.config = config
        self.model_path = model_path
        self.runtime = runtime
        self.server_type = config["server_config"]["type"]
        self.idx_val_clients = idx_val_clients
        self.idx_test_clients = idx_test_clients
        self.val_dataset = val_dataset
        self.test_dataset = test_dataset
        self.send_dicts = config["server_config"].get("send_dicts", False)
        self.worker_trainer = None
        self.metrics: Dict = {}
        self.losses: List = []

    # ------------------------------------------------------------------
    def run(self, eval_list, req, metric_logger=None):
        """Run the modes in ``eval_list``; update best-metric tracking in
        ``req`` (reference: evaluation.py:35-111)."""
        self.worker_trainer = req["worker_trainer"]
        metric_logger = metric_logger or (lambda k, v: None)

        semisup = "tmp_unsup" in req
        save_model = False
        for mode in eval_list:
            if self.config["server_config"].get("wantRL", False) and mode == "val":
                continue
            self.metrics = self.run_distributed_inference(mode)
            req = self.initialize_req(req) if len(req) == 1 else req

            if semisup:
                for tag, sd in (("Unsup", req["tmp_unsup"]), ("Sup", req["tmp_sup"])):
                    cur = {k: v.clone() for k, v in
                           self.worker_trainer.model.state_dict().items()}
                    self.worker_trainer.model.load_state_dict(sd)
                    m = self.run_distributed_inference(mode)
                    self.worker_trainer.model.load_state_dict(cur)
                    for key, value in m.items():
                        metric_logger(str(tag + mode + " " + key).capitalize(),
                                      value["value"])

            for key, value in self.metrics.items():
                metric_logger(str(mode + " " + key).capitalize(), value["value"])
                print_rank(f"LOG: {mode}_{key}={value['value']}")

            for key, value in self.metrics.items():
                attr = f"best_{mode}_{key}"
                if attr not in req:
                    req[attr] = -1.0 if value["higher_is_better"] else float("inf")
                if value["higher_is_better"]:
                    if value["value"] > req[attr]:
                        req[attr] = value["value"]
                        save_model = True
                else:
                    if value["value"] < req[attr]:
                        req[attr] = value["value"]
                        save_model = True
                if save_model and mode == "val":
                    if self.runtime is None or self.runtime.rank == 0:
                        self.worker_trainer.save(
                            model_path=self.model_path,
                            token=f"best_{mode}_{key}",
                            config=self.config["server_config"])
                    save_model = False
        return req

    def initialize_req(self, req):
        """Reference: evaluation.py:113-126."""
        for mode in ["test", "val"]:
            for key in self.metrics.keys():
                attr = f"best_{mode}_{key}"
                req[attr] = -1.0 if self.metrics[key]["higher_is_better"] else float("inf")
        return req

    # ------------------------------------------------------------------
    def run_distributed_inference(self, mode):
        """Evaluate the current global model over the mode's eval clients,
        partitioned across ranks (reference: evaluation.py:128-183)."""
        clients = self.idx_val_clients if mode == "val" else self.idx_test_clients
        dataset = self.val_dataset if mode == "val" else self.test_dataset
        data_config = self.config["server_config"]["data_config"][mode]
        task = self.config["server_config"].get("task", self.config.get("task"))

        rt = self.runtime
        my_groups = rt.my_share(clients, [sum(c.num_samples) for c in clients]) \
            if rt is not None else clients

        local_results = []
        model = self.worker_trainer.model
        if not hasattr(self, "_dl_cache"):
            self._dl_cache = {}
        for gi, group in enumerate(my_groups):
            data_strct = Client.get_data(group.client_id, dataset)[0]
            key = (mode, gi)
            dataloader = self._dl_cache.get(key)
            if dataloader is None:
                if mode == "val":
                    dataloader = make_val_dataloader(
                        data_config, data_path=None, task=task,
                        data_strct=data_strct)
                else:
                    dataloader = make_test_dataloader(
                        data_config, data_path=None, task=task,
                        data_strct=data_strct)
                if hasattr(dataloader, "to_device"):
                    dataloader.to_device()
                self._dl_cache[key] = dataloader  # eval sets are static
            want_logits = data_config.get("wantLogits", False)
            output, metrics = run_validation_generic(model, dataloader)
            count = sum(data_strct["num_samples"])
            local_results.append((
                {k: {"value": float(v["value"]),
                     "higher_is_better": bool(v["higher_is_better"])}
                 for k, v in metrics.items()},
                count))

        all_results = (sum(rt.all_gather_object(local_results), [])
                       if rt is not None else local_results)

        total = 0
        val_metrics: Dict = {}
        for metrics, count in all_results:
            if not val_metrics:
                val_metrics = {k: {"value": 0.0,
                                   "higher_is_better": v["higher_is_better"]}
                               for k, v in metrics.items()}
            for k in val_metrics:
                val_metrics[k]["value"] += metrics[k]["value"] * count
            total += count
        for k in val_metrics:
            val_metrics[k]["value"] /= max(total, 1)

        self.losses = [val_metrics.get("loss", {}).get("value", None),
                       val_metrics.get("acc", {}).get("value", None)]
        return val_metrics

    def make_val_fn(self):
        """(val_loss, val_acc) callback for ModelUpdater.run_lr_scheduler."""
        def val_fn():
            m = self.run_distributed_inference("val")
            return (m.get("loss", {}).get("value"), m.get("acc", {}).get("value"))
        return val_fn


class _EvalGroup:
    """An eval client group with precomputed sample counts (so partitioning
    can balance by size, reference: evaluation.py:193-211)."""

    def __init__(self, idxs, dataset):
        self.client_id = idxs
        self.num_samples = [dataset.num_samples[i] for i in idxs]


def make_eval_clients(dataset, config, n_groups: Optional[int] = None):
    """Split eval users into ≈equal-sample groups (reference:
    evaluation.py:185-216).  Group count defaults to the world size so
    every rank gets about one group."""
    from ..comm.runtime import get_runtime
    total = sum(dataset.num_samples)
    if n_groups is None:
        rt = get_runtime()
        n_groups = max(rt.size, 1)
    if config["server_config"]["type"] == "personalization":
        return [_EvalGroup([i], dataset) for i in range(len(dataset.user_list))]
    delta = total / n_groups + 1
    groups = []
    current, current_total = [], 0
    for i in range(len(dataset.user_list)):
        current.append(i)
        current_total += dataset.num_samples[i]
        if current_total > delta:
            groups.append(_EvalGroup(current, dataset))
            current, current_total = [], 0
    if current:
        groups.append(_EvalGroup(current, dataset))
    return groups
