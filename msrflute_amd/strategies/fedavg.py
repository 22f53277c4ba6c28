"""Federated Averaging (reference: core/strategies/fedavg.py:20-194).

Aggregation here is the symmetric round-collective form:
clients' weighted pseudo-gradients accumulate rank-locally into the server
grad arena (one fused axpy per client), then ONE all-reduce of the arena +
one scalar all-reduce of Σweight replace the reference's per-client
point-to-point uploads.  Math is identical to the reference's
sum-then-normalize (fedavg.py:140-147).
"""

from __future__ import annotations

import json
import logging
import os

import torch

from .. import ops
from ..utils import print_rank
from .base import BaseStrategy
from .utils import accumulate_flat_grad


class FedAvg(BaseStrategy):

    def __init__(self, mode, config, model_path=None, runtime=None):
        super().__init__(mode=mode, config=config, model_path=model_path,
                         runtime=runtime)
        self.model_config = config["model_config"]
        self.client_config = config["client_config"]
        self.server_config = config["server_config"]
        self.dp_config = config.get("dp_config", None)

        if mode == "client":
            self.stats_on_smooth_grad = self.client_config.get("stats_on_smooth_grad", False)
        else:
            self.dump_norm_stats = self.config.get("dump_norm_stats", False)
            self.aggregate_fast = self.server_config.get("fast_aggregation", True)
            self.skip_model_update = False
            # rank-local accumulators
            self.client_parameters_stack = []  # stacked mode: clones of flat grads
            self.client_weights = []
            self._pending_reduce = None  # in-flight overlapped round reduce

    # -- client side -------------------------------------------------------
    def generate_client_payload(self, trainer):
        """weight = num_samples; grad ← weight·pseudo_grad with optional
        layer freeze (reference: fedavg.py:72-91)."""
        if self.mode != "client":
            raise RuntimeError("this method can only be invoked by the client")
        if self.stats_on_smooth_grad:
            trainer.reset_gradient_power()
            trainer.estimate_sufficient_stats()
        weight = float(trainer.num_samples)
        arena = trainer.arena
        if weight != 1.0:
            ops.scale(arena.grad, weight)
        freeze = self.model_config.get("freeze_layer", None)
        if freeze:
            off, n = arena.segment_of(freeze)
            arena.grad[off:off + n].zero_()
        return {"weight": weight, "grad": arena.grad}

    # -- server side -------------------------------------------------------
    def process_individual_payload(self, worker_trainer, payload):
        if self.mode != "server":
            raise RuntimeError("this method can only be invoked by the server")
        if payload["weight"] == 0.0:
            return False
        self.client_weights.append(payload["weight"])
        if payload.get("pooled"):
            # gradient already accumulated stream-locally by ClientPool
            return True
        if self.aggregate_fast:
            accumulate_flat_grad(worker_trainer, payload["grad"])
        else:
            self.client_parameters_stack.append(payload["grad"].clone())
        return True

    def combine_payloads(self, worker_trainer, curr_iter, num_clients_curr_iter,
                         total_clients, client_stats, logger=None):
        if self.mode != "server":
            raise RuntimeError("this method can only be invoked by the server")
        weight_sum = self._aggregate_gradients(worker_trainer,
                                               num_clients_curr_iter,
                                               self.client_weights,
                                               metric_logger=logger)
        print_rank(f"Sum of weights: {weight_sum}", loglevel=logging.DEBUG)
        if weight_sum > 0:
            ops.scale(worker_trainer.arena.grad, 1.0 / weight_sum)

        if self.dump_norm_stats:
            self._dump_cosines(worker_trainer)

        if self.skip_model_update:
            print_rank("Skipping model update")
            return

        worker_trainer.update_model()
        losses = worker_trainer.run_lr_scheduler(force_run_val=False)
        return losses

    def begin_aggregation(self, worker_trainer):
        """Start the round's grad + Σweight all-reduce on the comm stream
        as soon as local clients are folded (fast-aggregation mode), so it
        overlaps the host-side round bookkeeping; ``combine_payloads``
        joins it (comm/compute overlap — runtime.begin_grad_reduce)."""
        if (self.mode != "server" or not self.aggregate_fast
                or self.runtime is None or self._pending_reduce is not None):
            return
        local_weight_sum = float(sum(self.client_weights))
        if self.client_config.get("quant_wire", False):
            # 8-bit codes + per-segment scales on the wire (opt-in;
            # runtime.begin_grad_reduce_quant)
            arena = worker_trainer.arena
            if getattr(self, "_seg_expand", None) is None:
                lengths = torch.tensor(arena.numels, dtype=torch.int64)
                self._seg_expand = torch.repeat_interleave(
                    torch.arange(len(arena.numels)), lengths
                ).to(arena.device)
            self._pending_reduce = self.runtime.begin_grad_reduce_quant(
                arena.grad, local_weight_sum, self._seg_expand)
            return
        self._pending_reduce = self.runtime.begin_grad_reduce(
            worker_trainer.arena.grad, local_weight_sum)

    def _aggregate_gradients(self, worker_trainer, num_clients_curr_iter,
                             client_weights, metric_logger=None):
        """Local stack sum (if stacked), then the round-level collectives:
        all_reduce(grad arena) + all_reduce(Σweight) — joined from the
        overlapped handle when ``begin_aggregation`` already started it."""
        if self._pending_reduce is not None:
            weight_sum = self.runtime.finish_grad_reduce(self._pending_reduce)
            self._pending_reduce = None
            self.client_weights = []
            self._last_stack = self.client_parameters_stack
            self.client_parameters_stack = []
            return weight_sum
        if not self.aggregate_fast:
            for flat in self.client_parameters_stack:
                accumulate_flat_grad(worker_trainer, flat)
        local_weight_sum = float(sum(client_weights))

        rt = self.runtime
        if rt is not None and rt._active:
            rt.all_reduce_(worker_trainer.arena.grad)
            t = torch.tensor([local_weight_sum], dtype=torch.float64,
                             device=worker_trainer.arena.device
                             if rt.backend == "nccl" else "cpu")
            rt.all_reduce_(t)
            weight_sum = float(t.item())
        else:
            weight_sum = local_weight_sum

        self.client_weights = []
        # keep the local stack alive for diagnostics within this round
        self._last_stack = self.client_parameters_stack
        self.client_parameters_stack = []
        return weight_sum

    def _dump_cosines(self, worker_trainer):
        """Per-client grad vs aggregate cosine diagnostics
        (reference: fedavg.py:149-152, utils/utils.py:585-595); computed on
        each rank for its own clients, gathered to rank 0."""
        agg = worker_trainer.arena.grad
        agg_norm = float(agg.norm())
        cosines = []
        for flat in getattr(self, "_last_stack", []):
            denom = float(flat.norm()) * agg_norm
            cosines.append(float(torch.dot(flat, agg)) / denom if denom > 0 else 0.0)
        rt = self.runtime
        all_cos = sum(rt.all_gather_object(cosines), []) if rt is not None else cosines
        if (rt is None or rt.rank == 0) and self.model_path:
            with open(os.path.join(self.model_path, "cosines.txt"), "a",
                      encoding="utf-8") as f:
                f.write(f"{json.dumps(all_cos)}\n")
