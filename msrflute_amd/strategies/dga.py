"""Dynamic Gradient Aggregation (arXiv:2106.07578).

Reference: core/strategies/dga.py:27-406.  Client side computes a
softmax-of-train-loss aggregation weight, optionally applies local DP noise
and update quantization; server side supports simulated gradient staleness
and RL-based weight re-estimation.

Symmetric-runtime notes:
* staleness (reference dga.py:260-277) is handled rank-locally: each rank
  stacks its clients' payloads, flips a per-(round,client) seeded coin,
  holds the stale ones for the next round and contributes only the rest to
  the all-reduce — commutative, so identical to the reference semantics;
* the RL path (reference dga.py:286-406 — broken as shipped, SURVEY.md
  §7.5; reimplemented to the paper's intent in extensions/rl) runs
  replicated: every rank evaluates the same RL network on the same gathered
  stats, re-accumulates its local stack with the RL weights and joins a
  second all-reduce.
"""

from __future__ import annotations

import logging
import math

import torch

from .. import ops
from ..extensions import privacy
from ..extensions.quantization import quant_arena
from ..utils import print_rank
from .fedavg import FedAvg
from .utils import accumulate_flat_grad, filter_weight

MIN_WEIGHT = 1e-7


class DGA(FedAvg):

    def __init__(self, mode, config, model_path=None, runtime=None):
        super().__init__(mode=mode, config=config, model_path=model_path,
                         runtime=runtime)
        if mode == "client":
            self.quant_threshold = self.client_config.get("quant_thresh", None)
            self.quant_bits = self.client_config.get("quant_bits", 10)
        else:
            self.want_rl = self.server_config.get("wantRL", False)
            self.stale_prob = self.server_config.get("stale_prob", 0.0)
            if self.want_rl or self.stale_prob > 0.0 or self.dump_norm_stats:
                # these need per-client payloads kept around
                self.aggregate_fast = False
            if self.want_rl:
                from ..extensions.rl import RL
                self.rl = RL(config=self.server_config)
            self.client_parameters_stack_stale = []
            self.stale_weights = []
            self.weight_sum_stale = 0.0
            self._round_client_ids = []  # parallel to client_parameters_stack

    # -- client side -------------------------------------------------------
    def generate_client_payload(self, trainer):
        """Softmax weight + local DP + weighting + quantization
        (reference: dga.py:88-155)."""
        if self.mode != "client":
            raise RuntimeError("this method can only be invoked by the client")

        weight = 1.0
        add_weight_noise = False
        if self.stats_on_smooth_grad:
            trainer.reset_gradient_power()
            trainer.estimate_sufficient_stats()

        if self.server_config.get("aggregate_median") == "softmax":
            add_weight_noise = True
            wtl = self.server_config.get("weight_train_loss", "train_loss")
            if wtl == "train_loss":
                training_weight = trainer.train_loss / max(trainer.num_samples, 1)
            elif wtl == "mag_var_loss":
                training_weight = trainer.sufficient_stats["var"]
            elif wtl == "mag_mean_loss":
                training_weight = trainer.sufficient_stats["mean"]
            else:
                training_weight = trainer.sufficient_stats["mag"]
            try:
                weight = math.exp(-self.server_config["softmax_beta"] * training_weight)
            except OverflowError:
                weight = MIN_WEIGHT
            weight = filter_weight(weight)

        if weight > 0.0 and self.dp_config is not None \
                and self.dp_config.get("enable_local_dp", False):
            weight = privacy.apply_local_dp(trainer, weight, self.dp_config,
                                            add_weight_noise)

        if not add_weight_noise:
            assert self.server_config.get("aggregate_median") == "mean"
            assert weight == 1.0

        arena = trainer.arena
        if weight != 1.0:
            ops.scale(arena.grad, weight)
        freeze = self.model_config.get("freeze_layer", None)
        if freeze:
            off, n = arena.segment_of(freeze)
            arena.grad[off:off + n].zero_()

        quant_arena(arena, quant_threshold=self.quant_threshold,
                    quant_bits=self.quant_bits, global_stats=False)

        return {"weight": weight, "grad": arena.grad}

    # -- server side -------------------------------------------------------
    def process_individual_payload(self, worker_trainer, payload,
                                   client_id=None):
        ok = super().process_individual_payload(worker_trainer, payload)
        if ok and not self.aggregate_fast:
            self._round_client_ids.append(client_id)
        return ok

    def combine_payloads(self, worker_trainer, curr_iter, num_clients_curr_iter,
                         total_clients, client_stats, logger=None):
        if self.mode != "server":
            raise RuntimeError("this method can only be invoked by the server")
        log = logger or (lambda k, v: None)

        rl_model = None
        if self.want_rl:
            rl_model = self._run_rl_inference(worker_trainer, client_stats)

        weight_sum = self._aggregate_gradients_dga(worker_trainer, curr_iter,
                                                   num_clients_curr_iter, log)
        print_rank(f"Sum of weights: {weight_sum}", loglevel=logging.DEBUG)
        if weight_sum > 0:
            ops.scale(worker_trainer.arena.grad, 1.0 / weight_sum)

        if self.dump_norm_stats:
            self._dump_cosines(worker_trainer)

        # Global DP + accounting (reference: dga.py:222-226)
        privacy.apply_global_dp(self.config, worker_trainer,
                                num_clients_curr_iter=num_clients_curr_iter,
                                curr_iter=curr_iter, metric_logger=log)
        eps = privacy.update_privacy_accountant(
            self.config, total_clients, curr_iter=curr_iter,
            num_clients_curr_iter=num_clients_curr_iter)
        if eps:
            print_rank(f"DP result: {eps}")

        if self.skip_model_update:
            print_rank("Skipping model update")
            return

        worker_trainer.update_model()
        # the RL reward is the val-metric delta vs the RL-weighted model, so
        # RL mode forces a val pass (documented intent of arXiv:2106.07578;
        # the reference's RL path is broken as shipped — SURVEY.md §7.5)
        losses = worker_trainer.run_lr_scheduler(force_run_val=self.want_rl)

        if self.want_rl:
            losses = self._run_rl_training(worker_trainer, curr_iter, rl_model,
                                           client_stats, log, losses)
        return losses

    def _aggregate_gradients_dga(self, worker_trainer, curr_iter,
                                 num_clients_curr_iter, log):
        """Stack sum with staleness simulation, then the round collectives
        (reference: dga.py:243-284)."""
        local_weight_sum = 0.0
        if not self.aggregate_fast:
            n_stale = len(self.client_parameters_stack_stale)
            log("Stale Gradients Ratio", n_stale / max(num_clients_curr_iter, 1))
            # release the previous round's held gradients
            for flat, w in zip(self.client_parameters_stack_stale, self.stale_weights):
                accumulate_flat_grad(worker_trainer, flat)
                local_weight_sum += w
            self.client_parameters_stack_stale = []
            self.stale_weights = []

            rng = (self.runtime.round_rng(curr_iter, salt=7)
                   if self.runtime is not None else None)
            for flat, w, cid in zip(self.client_parameters_stack,
                                    self.client_weights,
                                    self._round_client_ids):
                hold = False
                if self.stale_prob > 0.0:
                    r = rng.random() if rng is not None else 0.5
                    hold = not (r > self.stale_prob)
                if hold:
                    self.client_parameters_stack_stale.append(flat)
                    self.stale_weights.append(w)
                else:
                    accumulate_flat_grad(worker_trainer, flat)
                    local_weight_sum += w
        else:
            local_weight_sum = float(sum(self.client_weights))

        rt = self.runtime
        if self._pending_reduce is not None:
            # overlapped round reduce already in flight (fast mode)
            weight_sum = rt.finish_grad_reduce(self._pending_reduce)
            self._pending_reduce = None
        elif rt is not None and rt._active:
            rt.all_reduce_(worker_trainer.arena.grad)
            t = torch.tensor([local_weight_sum], dtype=torch.float64,
                             device=worker_trainer.arena.device
                             if rt.backend == "nccl" else "cpu")
            rt.all_reduce_(t)
            weight_sum = float(t.item())
        else:
            weight_sum = local_weight_sum

        self._last_stack = self.client_parameters_stack
        self._last_weights = list(self.client_weights)
        self.client_parameters_stack = []
        self.client_weights = []
        self._round_client_ids = []
        return weight_sum

    # -- RL reweighting (reference: dga.py:286-406, reimplemented) ---------
    def _run_rl_inference(self, worker_trainer, client_stats):
        import numpy as np
        client_weights, mag, mean, var = client_stats
        state = np.concatenate((client_weights, mag, mean, var), axis=0)
        rl_weights = self.rl.forward(state)
        rl_weights = np.exp(np.asarray(rl_weights, dtype=np.float64))
        rl_weights[~np.isfinite(rl_weights)] = 0.0

        original = worker_trainer.arena.clone_data()
        original_grad = worker_trainer.arena.grad.clone()

        # Re-accumulate local stack with RL weights: grads are stored
        # pre-weighted by w_i, so scale by rl_w_i / w_i.
        worker_trainer.arena.grad.zero_()
        local_w = 0.0
        # map local clients to their global position: client ids were
        # gathered in sorted order, so look up by id
        for flat, w, cid in zip(self.client_parameters_stack, self.client_weights,
                                self._round_client_ids):
            pos = self._client_pos.get(cid, None) if hasattr(self, "_client_pos") else None
            rl_w = float(rl_weights[pos]) if pos is not None else float(w)
            if w != 0:
                accumulate_flat_grad(worker_trainer, flat, alpha=rl_w / w)
            local_w += rl_w
        rt = self.runtime
        if rt is not None and rt.size > 1:
            rt.all_reduce_(worker_trainer.arena.grad)
            t = torch.tensor([local_w], dtype=torch.float64,
                             device=worker_trainer.arena.device
                             if rt.backend == "nccl" else "cpu")
            rt.all_reduce_(t)
            local_w = float(t.item())
        if local_w > 0:
            ops.scale(worker_trainer.arena.grad, 1.0 / local_w)
        worker_trainer.update_model()
        rl_losses = worker_trainer.run_lr_scheduler(force_run_val=True)

        rl_model = worker_trainer.arena.clone_data()
        worker_trainer.arena.copy_data_(original)
        worker_trainer.arena.grad.copy_(original_grad)

        self.rl.set_weights(np.log(np.maximum(rl_weights, 1e-30)))
        self.rl.set_losses(rl_losses)
        return rl_model

    def _run_rl_training(self, worker_trainer, curr_iter, rl_model,
                         client_stats, log, losses):
        import numpy as np
        client_weights, mag, mean, var = client_stats
        val_loss, val_acc = losses if losses is not None else (None, None)
        rl_val_loss, rl_val_acc = self.rl.rl_losses
        if val_acc is None:
            return losses
        if abs(val_acc - rl_val_acc) < 0.001:
            reward = 0.1
            if self.server_config.get("marginal_update_RL", False):
                worker_trainer.arena.copy_data_(rl_model)
                losses = self.rl.rl_losses
        elif (val_acc - rl_val_acc) > 0:
            reward = 1.0
            worker_trainer.arena.copy_data_(rl_model)
            losses = self.rl.rl_losses
        else:
            reward = -1.0
        batch = (np.concatenate((client_weights, mag, mean, var), axis=0),
                 self.rl.rl_weights, [reward])
        self.rl.train(batch)
        if self.runtime is None or self.runtime.rank == 0:
            self.rl.save(curr_iter)
        log("RL Running Loss", float(self.rl.running_loss))
        log("RL Rewards", reward)
        return losses
