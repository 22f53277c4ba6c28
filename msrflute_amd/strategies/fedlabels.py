"""FedLabels semi-supervision strategy (reference: core/strategies/fedlabels.py:20-235).

Client payload = the supervised model's state dict + the unsupervised
model's state dict.  Server averages the supervised dicts uniformly and the
unsupervised dicts weighted by sample count, then blends ½/½
(reference: fedlabels.py:141-149, 170-216).

Symmetric-runtime form: each rank accumulates its local clients' dicts as
two FLAT vectors (Σ sup_i and Σ w_i·unsup_i) plus scalar (count, Σw); one
all-reduce of the two flats + scalars reproduces the reference's global
ratios exactly.
"""

from __future__ import annotations

import logging

import torch

from ..ops.arena import flatten_state_dict, unflatten_into_state_dict
from ..utils import print_rank
from .base import BaseStrategy


class FedLabels(BaseStrategy):

    def __init__(self, mode, config, model_path=None, runtime=None):
        super().__init__(mode=mode, config=config, model_path=model_path,
                         runtime=runtime)
        self.model_config = config["model_config"]
        self.client_config = config["client_config"]
        self.server_config = config["server_config"]
        self.dp_config = config.get("dp_config", None)
        self.tmp_sup = None
        self.tmp_unsup = None
        if mode == "client":
            self.stats_on_smooth_grad = self.client_config.get("stats_on_smooth_grad", False)
        else:
            self.skip_model_update = False
            self._sup_acc = None
            self._unsup_acc = None
            self._count = 0
            self.client_weights = []

    # -- client side -------------------------------------------------------
    def generate_client_payload(self, trainer):
        if self.mode != "client":
            raise RuntimeError("this method can only be invoked by the client")
        unsup_dict = trainer.algo_computation
        if self.stats_on_smooth_grad:
            trainer.reset_gradient_power()
            trainer.estimate_sufficient_stats()
        weight = 1 if trainer.num_samples == 0 else trainer.num_samples
        return {
            "weight": float(weight),
            "sup_flat": flatten_state_dict(trainer.model.state_dict()),
            "unsup_flat": flatten_state_dict(unsup_dict),
        }

    # -- server side -------------------------------------------------------
    def process_individual_payload(self, worker_trainer, payload):
        if self.mode != "server":
            raise RuntimeError("this method can only be invoked by the server")
        if payload["weight"] == 0.0:
            return False
        if self._sup_acc is None:
            self._sup_acc = torch.zeros_like(payload["sup_flat"])
            self._unsup_acc = torch.zeros_like(payload["unsup_flat"])
        self._sup_acc += payload["sup_flat"]
        self._unsup_acc += payload["unsup_flat"] * payload["weight"]
        self._count += 1
        self.client_weights.append(payload["weight"])
        return True

    def combine_payloads(self, worker_trainer, curr_iter, num_clients_curr_iter,
                         total_clients, client_stats, logger=None):
        if self.mode != "server":
            raise RuntimeError("this method can only be invoked by the server")

        sd = worker_trainer.model.state_dict()
        if self._sup_acc is None:
            self._sup_acc = torch.zeros_like(flatten_state_dict(sd))
            self._unsup_acc = self._sup_acc.clone()

        rt = self.runtime
        scal = torch.tensor([float(self._count), float(sum(self.client_weights))],
                            dtype=torch.float64)
        if rt is not None and rt.size > 1:
            dev = worker_trainer.arena.device if rt.backend == "nccl" else "cpu"
            sup = self._sup_acc.to(dev)
            unsup = self._unsup_acc.to(dev)
            rt.all_reduce_(sup)
            rt.all_reduce_(unsup)
            scal = scal.to(dev if rt.backend == "nccl" else "cpu")
            rt.all_reduce_(scal)
            self._sup_acc = sup
            self._unsup_acc = unsup
        count, weight_sum = float(scal[0].item()), float(scal[1].item())
        print_rank(f"Sum of weights: {weight_sum}", loglevel=logging.DEBUG)

        if count > 0:
            sup_mean = self._sup_acc / count
            unsup_mean = self._unsup_acc / weight_sum
            blended = 0.5 * sup_mean + 0.5 * unsup_mean
            dev = next(iter(sd.values())).device if sd else "cpu"
            unflatten_into_state_dict(blended.to(dev), sd)
            worker_trainer.model.load_state_dict(sd)
            # expose the two averaged models for semisup eval
            sup_sd = {k: v.clone() for k, v in sd.items()}
            unflatten_into_state_dict(sup_mean.to(dev), sup_sd)
            self.tmp_sup = sup_sd
            unsup_sd = {k: v.clone() for k, v in sd.items()}
            unflatten_into_state_dict(unsup_mean.to(dev), unsup_sd)
            self.tmp_unsup = unsup_sd

        self._sup_acc = None
        self._unsup_acc = None
        self._count = 0
        self.client_weights = []

        if self.skip_model_update:
            print_rank("Skipping model update")
            return

        worker_trainer.update_model()
        losses = worker_trainer.run_lr_scheduler(force_run_val=False)
        return losses
