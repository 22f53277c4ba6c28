"""OptimizationServer: the FL round loop (reference: core/server.py:47-597).

Every rank executes this loop on an identical replica of the server state;
round decisions (client sampling, partitioning, stale coin-flips, DP noise
seeds) derive deterministically from the shared round seed, client results
merge through ONE gradient all-reduce + one metadata all-gather per round,
and the server optimizer step is recomputed identically everywhere — so
the reference's model broadcast, command protocol and worker pool
 disappear.  Rank 0 alone touches disk and logs.
"""

from __future__ import annotations

import copy
import math
import json
import logging
import os
import shutil
import time
from collections import defaultdict

import numpy as np
import torch

from ..comm.runtime import FedRuntime
from ..strategies import select_strategy
from ..utils import log_metric, print_rank, update_json_log
from .client import Client, ClientExecutor
from .evaluation import Evaluation
from .trainer import (ModelUpdater, Trainer, flush_saves, get_lr,
                      set_component_wise_lr)


class OptimizationServer:
    def __init__(self, num_clients, model, optimizer, ss_scheduler, data_path,
                 model_path, server_train_dataloader, config, idx_val_clients,
                 idx_test_clients, runtime: FedRuntime, arena,
                 val_dataset=None, test_dataset=None, task=None):
        self.client_idx_list = list(range(num_clients))
        self.config = config
        self.runtime = runtime
        server_config = config["server_config"]
        decoder_config = config.get("decoder_config", None)

        self.max_iteration = server_config["max_iteration"]
        self.do_clustering = server_config.get("clustering", False)
        self.send_dicts = server_config.get("send_dicts", False)

        ncpi = server_config["num_clients_per_iteration"]
        self.num_clients_per_iteration = [int(x) for x in ncpi.split(",")] \
            if isinstance(ncpi, str) else [ncpi]

        self.val_freq = server_config["val_freq"]
        self.req_freq = server_config["rec_freq"]

        self.metrics = {}
        self.model_backup_freq = server_config.get("model_backup_freq", 100)
        self.worker_trainer_config = server_config.get("trainer_config", {})

        self.aggregate_median = server_config.get("aggregate_median")
        self.initial_lr_client = server_config.get("initial_lr_client", -1.0)
        self.lr_decay_factor = server_config.get("lr_decay_factor", 1.0)
        self.model_type = config["model_config"]["model_type"]
        self.quant_thresh = config["client_config"].get("quant_thresh", None)
        self.quant_bits = config["client_config"].get("quant_bits", 10)
        self.data_path = data_path
        self.task = task

        max_grad_norm = None
        if "train" in server_config["data_config"]:
            max_grad_norm = server_config["data_config"]["train"].get("max_grad_norm", None)

        self.evaluation = Evaluation(config, model_path, runtime,
                                     idx_val_clients, idx_test_clients,
                                     val_dataset=val_dataset,
                                     test_dataset=test_dataset)

        self.worker_trainer = ModelUpdater(
            model=model, optimizer=optimizer, ss_scheduler=ss_scheduler,
            train_dataloader=server_train_dataloader, val_dataloader=None,
            max_grad_norm=max_grad_norm,
            anneal_config=server_config["annealing_config"],
            model_type=self.model_type, decoder_config=decoder_config,
            arena=arena, val_fn=self.evaluation.make_val_fn())
        self.metrics["worker_trainer"] = self.worker_trainer
        # the RL path may force a val pass (run_lr_scheduler -> val_fn)
        # before any Evaluation.run has populated it
        self.evaluation.worker_trainer = self.worker_trainer

        # server-side replay trainer (reference: server.py:130-151)
        self.server_replay_iterations = None
        self.server_trainer = None
        if server_train_dataloader is not None:
            assert "server_replay_config" in server_config, "server_replay_config is not set"
            assert "optimizer_config" in server_config["server_replay_config"], \
                "server-side replay training optimizer is not set"
            self.server_optimizer_config = server_config["server_replay_config"]["optimizer_config"]
            self.server_trainer_config = server_config["server_replay_config"].get("trainer_config", {})
            self.server_replay_iterations = server_config["server_replay_config"]["server_iterations"]
            self.server_trainer = Trainer(
                model=model, optimizer=None, ss_scheduler=ss_scheduler,
                train_dataloader=server_train_dataloader,
                server_replay_config=server_config["server_replay_config"],
                max_grad_norm=server_config["server_replay_config"].get(
                    "max_grad_norm",
                    server_config["data_config"].get("train", {}).get("max_grad_norm", None)),
                anneal_config=server_config["server_replay_config"].get("annealing_config", None),
                ignore_subtask=server_config["server_replay_config"].get("ignore_subtask", False),
                arena=arena)

        self.skip_model_update = False
        self.train_loss = 0.0
        self.model_path = model_path
        self.best_model_criterion = server_config["best_model_criterion"]
        self.fall_back_to_best_model = server_config["fall_back_to_best_model"]
        self.last_model_path = os.path.join(model_path, "latest_model.tar")
        self.best_model_path = os.path.join(
            model_path, f"best_val_{self.best_model_criterion}_model.tar")
        self.log_path = os.path.join(model_path, "status_log.json")
        self.cur_iter_no = 0
        self.lr_weight = 1.0
        self.losses = []
        self.no_label_updates = 0

        if server_config.get("resume_from_checkpoint", False):
            self.load_saved_status()

        self.decoder_config = decoder_config
        self.spm_model = server_config["data_config"]["test"].get("spm_model", None)
        self.do_profiling = server_config.get("do_profiling", False)

        StrategyClass = select_strategy(config["strategy"])
        self.strategy = StrategyClass("server", config, model_path,
                                      runtime=runtime)

        # per-rank persistent client workspace; with a GPU and an
        # order-free strategy, a pool of stream-parallel executors trains
        # several clients concurrently (ClientPool)
        n_par = int(config["client_config"].get("parallel_clients", 8))
        # DGA in fast-aggregation mode is order-free like FedAvg (weighted
        # sum; softmax weights are per-client local), so it runs on the
        # stream pool too — the BASELINE config-5 shape (DGA + local DP +
        # 8-bit quant) trains with the fused epoch + pooled streams.
        # Stacked modes (staleness simulation, RL reweighting) need the
        # per-client stack and keep the single executor.
        dga_poolable = (config["strategy"] == "DGA"
                        and server_config.get("fast_aggregation", True)
                        and not server_config.get("wantRL", False)
                        and float(server_config.get("stale_prob", 0.0) or 0.0)
                        == 0.0)
        if (torch.cuda.is_available() and n_par > 1
                and (config["strategy"] in ("FedAvg", "FedProx")
                     or dga_poolable)
                and server_config.get("type") != "personalization"
                and server_config.get("fast_aggregation", True)
                and not config.get("dump_norm_stats", False)
                and not config.get("privacy_metrics_config", {}).get(
                    "apply_metrics", False)):
            from .client import ClientPool
            self.executor = ClientPool(config, self.task, data_path,
                                       server_arena=arena,
                                       model_path=model_path,
                                       n_parallel=n_par)
        else:
            self.executor = ClientExecutor(config, self.task, data_path,
                                           server_arena=arena,
                                           model_path=model_path)

        # per-client sample counts for size-aware partitioning
        from . import client as client_mod
        ds = client_mod.train_dataset
        self.client_num_samples_all = list(ds.num_samples) if ds is not None else None

    # ------------------------------------------------------------------
    def load_saved_status(self):
        """Resume from latest checkpoint + status log (reference: server.py:183-204)."""
        flush_saves()
        if os.path.exists(self.last_model_path):
            print_rank(f"Resuming from checkpoint model {self.last_model_path}")
            self.worker_trainer.load(self.last_model_path,
                                     update_lr_scheduler=True,
                                     update_ss_scheduler=True)
            if self.server_trainer is not None:
                self.server_trainer.model = self.worker_trainer.model
        if os.path.exists(self.log_path):
            with open(self.log_path, "r") as fp:
                elems = json.load(fp)
            self.cur_iter_no = elems.get("i", 0)
            self.metrics["best_val_loss"] = elems.get("best_val_loss", float("inf"))
            self.metrics["best_val_acc"] = elems.get("best_val_acc", 0)
            self.metrics["best_test_loss"] = elems.get("best_test_loss", float("inf"))
            self.metrics["best_test_acc"] = elems.get("best_test_acc", 0)
            self.lr_weight = elems.get("weight", 1.0)
            self.no_label_updates = elems.get("num_label_updates", 0)
            print_rank(f"Resuming from status_log: cur_iter: {self.cur_iter_no}")

    def run(self):
        try:
            self._run_impl()
        finally:
            flush_saves()  # all queued checkpoint writes hit disk

    def _run_impl(self):
        print_rank("server started")
        self.train()
        print_rank("server terminated")

    # ------------------------------------------------------------------
    def train(self):
        """Main loop (reference: server.py:215-528)."""
        self.train_setup()
        for i in range(self.cur_iter_no, self.max_iteration):
            self.run_one_round(i)

    def train_setup(self):
        """Initial eval + initial checkpoint saves (reference: server.py:218-258)."""
        rt = self.runtime
        self.run_stats = {
            "secsPerClientRound": [], "secsPerClient": [],
            "secsPerClientTraining": [], "secsPerClientSetup": [],
            "secsPerClientFull": [], "secsPerRoundHousekeeping": [],
            "secsPerRoundTotal": [], "communicationCosts": [],
        }
        eval_list = []
        if self.cur_iter_no == 0:
            if self.config["server_config"]["initial_rec"]:
                eval_list.append("test")
            if self.config["server_config"]["initial_val"]:
                eval_list.append("val")
            print_rank(f"Running {eval_list} at itr={self.cur_iter_no}")
            self.metrics = self.evaluation.run(eval_list, self.metrics,
                                               metric_logger=log_metric)
        if rt.rank == 0:
            for token in ["best_val_loss", "best_val_acc", "best_test_acc", "latest"]:
                self.worker_trainer.save(model_path=self.model_path,
                                         token=token,
                                         config=self.config["server_config"])
        self.worker_trainer.model.train()

    def run_one_round(self, i, housekeeping=True):
        """One FL round: sample → local client training → all-reduce
        aggregation → replicated server update (+ eval/checkpoint cadence
        when ``housekeeping``).  Reference round body: server.py:259-525."""
        rt = self.runtime
        is_chief = rt.rank == 0
        eval_list = []
        profiler = None
        if self.do_profiling:
            # reference wraps rounds in cProfile (server.py:327-331); on a
            # GPU additionally emit a roctx range so rocprofv3 runtime
            # traces show round boundaries
            import cProfile
            profiler = cProfile.Profile()
            profiler.enable()
            if torch.cuda.is_available():
                torch.cuda.nvtx.range_push(f"fl_round_{i}")
        begin = time.time()
        metrics_payload = {}

        def log_m(k, v):
            metrics_payload[k] = v

        print_rank(f"==== iteration {i}")
        log_m("Current iteration", i)

        initial_lr = self.initial_lr_client * self.lr_weight
        log_m("Client learning rate", initial_lr)

        # clear the server grad arena before accumulation
        self.worker_trainer.arena.zero_grad()
        self.train_loss = []

        # number of clients this round (int or random range)
        if len(self.num_clients_per_iteration) > 1:
            num_clients_curr_iter = rt.round_rng(i, salt=2).randint(
                self.num_clients_per_iteration[0],
                self.num_clients_per_iteration[1])
        else:
            num_clients_curr_iter = self.num_clients_per_iteration[0]
        log_m("Clients for round", num_clients_curr_iter)

        # quantization-threshold annealing (reference: server.py:295-298)
        if self.quant_thresh is not None:
            cc = self.config["client_config"]
            cc["quant_thresh"] = cc.get("quant_thresh", self.quant_thresh) * \
                cc.get("quant_anneal", 1.0)
            self.quant_thresh = cc["quant_thresh"]
            self.executor.client_strategy.quant_threshold = self.quant_thresh
            log_m("Quantization Thresh.", cc["quant_thresh"])

        # deterministic sampling — identical on every rank
        sampled_idx_clients = rt.sample_clients(self.client_idx_list,
                                                num_clients_curr_iter, i)
        weights = ([self.client_num_samples_all[c] for c in sampled_idx_clients]
                   if self.client_num_samples_all is not None else None)
        parts = rt.partition(sampled_idx_clients, weights)
        my_clients = parts[rt.rank]

        clients_begin = time.time()
        apply_privacy_metrics = bool(
            self.config.get("privacy_metrics_config", None)
            and self.config["privacy_metrics_config"]["apply_metrics"])
        adaptive_leakage = apply_privacy_metrics and \
            self.config["privacy_metrics_config"].get("adaptive_leakage_threshold", None)
        privacy_metrics_stats = defaultdict(list)

        for key in ["secsPerClient", "secsPerClientFull",
                    "secsPerClientTraining", "secsPerClientSetup",
                    "communicationCosts"]:
            self.run_stats[key].append([])

        # ---- local client training ----------------------------------
        # whole-round fused path: ONE _C.cnn_round call per executor
        # trains every local client (copy-in, epoch, pseudo-grad,
        # accumulate all inside the extension)
        fused_outputs = None
        # The fused driver trains plain SGD with no proximal term, so it
        # is only equivalent to the eager path under FedAvg (FedProx adds
        # (mu/2)||w-w_g||^2 every local step — reference trainer.py:463).
        if (hasattr(self.executor, "run_fused_round_batch")
                and self.config["strategy"] == "FedAvg"
                and self.config["client_config"].get("use_fused_round",
                                                     True)
                and not apply_privacy_metrics):
            seeds = [rt.round_rng(i, salt=100 + c).getrandbits(62)
                     for c in my_clients]
            fused_outputs = self.executor.run_fused_round_batch(
                my_clients, initial_lr, i, seeds)

        local_outputs = []
        for client_idx, pre_output in (
                fused_outputs if fused_outputs is not None
                else ((c, None) for c in my_clients)):
            if pre_output is not None:
                output = pre_output
            else:
                client = Client([client_idx], self.config, True)
                output = self.executor.process_round(
                    client, initial_lr, i,
                    round_seed=rt.round_rng(
                        i, salt=100 + client_idx).getrandbits(62))
            payload = output["pl"]
            if output.get("wt", None) == 0.0 and payload is not None:
                payload["weight"] = 0.0
            if isinstance(self.strategy, _needs_cid_cls()):
                ok = self.strategy.process_individual_payload(
                    self.worker_trainer, payload, client_id=client_idx)
            else:
                ok = self.strategy.process_individual_payload(
                    self.worker_trainer, payload)
            meta = {k: v for k, v in output.items() if k != "pl"}
            meta["accepted"] = bool(ok)
            meta["wt"] = payload["weight"] if payload is not None else 0.0
            local_outputs.append((client_idx, meta))
            self.run_stats["secsPerClient"][-1].append(time.time() - clients_begin)

        # join client streams + fold pool accumulators into the server
        # grad arena (no-op for the single-executor path)
        if hasattr(self.executor, "flush"):
            self.executor.flush(self.worker_trainer.arena.grad)

        # start the round's grad + Σweight all-reduce NOW on the comm
        # stream: it overlaps the lazy-stats finalize and metadata
        # all_gather below, and combine_payloads joins it (north-star
        # comm/compute overlap)
        if hasattr(self.strategy, "begin_aggregation"):
            self.strategy.begin_aggregation(self.worker_trainer)

        # ---- batched finalize of deferred client stats ---------------
        # lazy-stats clients carried device tensors; ONE host transfer
        # materializes every client's loss/Σg/Σg² for this rank
        lazy = [(idx, meta) for idx, (_, meta) in enumerate(local_outputs)
                if "_lazy" in meta]
        if lazy:
            flat = torch.stack(
                [torch.stack([meta["_lazy"][0].reshape(()),
                              meta["_lazy"][1][0], meta["_lazy"][1][1]])
                 for _, meta in lazy]).cpu().tolist()
            for (idx, meta), (tl, s, q) in zip(lazy, flat):
                n = max(meta["_lazy"][2], 1)
                mean = s / n
                meta["tl"] = tl
                meta["ng"] = mean
                meta["mg"] = math.sqrt(max(q / n, 0.0))
                meta["vg"] = max(q / n - mean ** 2, 0.0)
                meta["rg"] = math.sqrt(max(q, 0.0))
                del meta["_lazy"]

        # ---- metadata exchange (one all_gather per round) ------------
        # numeric metas travel as ONE fused tensor all_gather (no
        # pickle on the fabric); privacy-metric dicts fall back to the
        # object path
        if rt.size > 1 and not apply_privacy_metrics:
            cols = ["tl", "mg", "ng", "vg", "rg", "ns", "wt", "ts"]
            rows = [[float(cid), 1.0 if meta["accepted"] else 0.0]
                    + [float(meta.get(c, 0.0)) for c in cols]
                    for cid, meta in local_outputs]
            local_t = torch.tensor(rows, dtype=torch.float64).reshape(
                len(rows), 2 + len(cols))
            per_rank = rt.all_gather_rows(local_t,
                                          [len(p) for p in parts])
            gathered = []
            for r, block in enumerate(per_rank):
                if r == rt.rank:
                    gathered.extend(local_outputs)  # keep local cs dicts
                    continue
                for row in block.tolist():
                    meta = {"accepted": bool(row[1]),
                            **{c: row[2 + j] for j, c in enumerate(cols)},
                            "ns": int(row[7]),
                            "cs": {"full cost": 0.0, "training": 0.0,
                                   "setup": 0.0}}
                    gathered.append((int(row[0]), meta))
        else:
            gathered = sum(rt.all_gather_object(local_outputs), [])
        order = {c: k for k, c in enumerate(sampled_idx_clients)}
        gathered.sort(key=lambda t: order.get(t[0], 1 << 30))

        client_losses, client_mag_grads = [], []
        client_mean_grads, client_var_grads, client_norm_grads = [], [], []
        client_weights_all = []
        client_pos = {}
        for pos, (cid, meta) in enumerate(gathered):
            if not meta["accepted"]:
                num_clients_curr_iter -= 1
                continue
            client_pos[cid] = len(client_losses)
            self.train_loss.append(meta["tl"])
            client_losses.append(meta["tl"])
            client_mag_grads.append(meta["mg"])
            client_mean_grads.append(meta["ng"])
            client_var_grads.append(meta["vg"])
            client_norm_grads.append(meta["rg"])
            client_weights_all.append(meta["wt"])
            cs = meta["cs"]
            self.run_stats["secsPerClientFull"][-1].append(cs["full cost"])
            self.run_stats["secsPerClientTraining"][-1].append(cs["training"])
            self.run_stats["secsPerClientSetup"][-1].append(cs["setup"])
            self.run_stats["communicationCosts"][-1].append(time.time() - meta["ts"])
            if apply_privacy_metrics and "ps" in meta:
                for metric, value in meta["ps"].items():
                    privacy_metrics_stats[metric].append(value)
        self.strategy._client_pos = client_pos

        client_mag_grads = np.array(client_mag_grads)
        client_mean_grads = np.array(client_mean_grads)
        client_var_grads = np.array(client_var_grads)
        client_norm_grads = np.array(client_norm_grads)
        client_stats = (np.array(client_weights_all), client_mag_grads,
                        client_mean_grads, client_var_grads)

        dump_norm_stats = self.config.get("dump_norm_stats", False)
        if dump_norm_stats and is_chief:
            with open(os.path.join(self.model_path, "norm_stats.txt"), "a",
                      encoding="utf-8") as f:
                f.write(f"{json.dumps(list(client_norm_grads))}\n")

        if apply_privacy_metrics:
            for metric, values in privacy_metrics_stats.items():
                if metric == "Dropped clients":
                    log_m(metric, sum(values))
                else:
                    log_m(metric, max(values))
            if isinstance(adaptive_leakage, float):
                values = sorted(privacy_metrics_stats["Practical epsilon (Max leakage)"])
                if values:
                    new_threshold = values[int(adaptive_leakage * len(values))]
                    print_rank(f"Updating leakage threshold to {new_threshold}")
                    self.config["privacy_metrics_config"]["max_allowed_leakage"] = new_threshold

        end = time.time()
        self.run_stats["secsPerClientRound"].append(end - begin)
        begin = end
        log_m("Training loss", sum(self.train_loss))

        # ---- combine: all-reduce + replicated server update ----------
        cs_for_strategy = (client_mag_grads, client_mean_grads, client_var_grads)
        if hasattr(self.strategy, "want_rl") and self.strategy.want_rl:
            cs_for_strategy = client_stats
        self.losses = self.strategy.combine_payloads(
            worker_trainer=self.worker_trainer, curr_iter=i,
            num_clients_curr_iter=max(num_clients_curr_iter, 1),
            total_clients=len(self.client_idx_list),
            client_stats=cs_for_strategy, logger=log_m)

        # ---- server replay training (reference: server.py:430-442) ----
        if self.server_trainer is not None:
            print_rank("Running replay iterations on server")
            torch.manual_seed(rt.round_rng(i, salt=3).getrandbits(62))
            if "updatable_names" in self.server_trainer_config:
                set_component_wise_lr(self.worker_trainer.model,
                                      self.server_optimizer_config,
                                      self.server_trainer_config["updatable_names"])
            self.server_trainer.prepare_iteration(self.worker_trainer.model)
            self.server_trainer.train_desired_samples(self.server_replay_iterations)
            self.worker_trainer.model.load_state_dict(
                self.server_trainer.model.state_dict())

        self.worker_trainer.run_ss_scheduler()

        # ---- evaluation cadence --------------------------------------
        if housekeeping and ((i + 1) % self.val_freq) == 0:
            eval_list.append("val")
        if housekeeping and ((i + 1) % self.req_freq) == 0:
            eval_list.append("test")

        ran_val = "val" in eval_list
        if len(eval_list) > 0:
            print_rank(f"Running {eval_list} at itr={i+1}")
            self.metrics["worker_trainer"] = self.worker_trainer
            if hasattr(self.strategy, "tmp_unsup") and self.strategy.tmp_unsup is not None:
                self.metrics["tmp_sup"] = self.strategy.tmp_sup
                self.metrics["tmp_unsup"] = self.strategy.tmp_unsup
            self.metrics = self.evaluation.run(eval_list, self.metrics,
                                               metric_logger=log_m)
            self.losses = self.evaluation.losses
            eval_list = []

        # client-LR decay on val plateau.  NOTE: the reference checks
        # `'val' in eval_list` AFTER clearing the list (server.py:462-469)
        # making the decay dead code; we implement the documented intent.
        if ran_val and self.losses and self.losses[0] is not None:
            log_m("LR for agg. opt.", get_lr(self.worker_trainer.optimizer))
            if not (self.losses[0] < self.metrics.get("best_val_loss", float("inf"))):
                self.lr_weight *= self.lr_decay_factor
                print_rank(f"LOG: Client weight of learning rate {self.lr_weight}..")

        # ---- checkpoint / backup / fallback --------------------------
        if is_chief and housekeeping:
            self.backup_models(i)
        if self.fall_back_to_best_model and housekeeping:
            rt.barrier()
            self.fall_back_to_prev_best_status()

        if housekeeping and len(self.metrics) > 1 and is_chief:
            update_json_log(self.log_path, {
                "i": i + 1,
                "best_val_loss": float(self.metrics.get("best_val_loss", float("inf"))),
                "best_val_acc": float(self.metrics.get("best_val_acc", 0)),
                "best_test_loss": float(self.metrics.get("best_test_loss", float("inf"))),
                "best_test_acc": float(self.metrics.get("best_test_acc", 0)),
                "weight": float(self.lr_weight),
                "num_label_updates": int(self.no_label_updates),
            })

        end = time.time()
        self.run_stats["secsPerRoundHousekeeping"].append(end - begin)
        self.run_stats["secsPerRoundTotal"].append(
            self.run_stats["secsPerClientRound"][-1]
            + self.run_stats["secsPerRoundHousekeeping"][-1])
        log_m("secsPerRoundTotal", self.run_stats["secsPerRoundTotal"][-1])

        if self.do_profiling:
            for metric in ["secsPerClient", "secsPerClientTraining",
                           "secsPerClientFull", "secsPerClientSetup",
                           "communicationCosts"]:
                vals = self.run_stats[metric][-1]
                if vals:
                    log_m(f"{metric}Mean", float(np.mean(vals)))
                    log_m(f"{metric}Median", float(np.median(vals)))
                    log_m(f"{metric}Max", float(max(vals)))

        if is_chief:
            for k, v in metrics_payload.items():
                log_metric(k, v, step=i)

        if profiler is not None:
            if torch.cuda.is_available():
                torch.cuda.nvtx.range_pop()
            profiler.disable()
            import io
            import pstats
            buf = io.StringIO()
            pstats.Stats(profiler, stream=buf).sort_stats(
                "cumulative").print_stats(20)
            print_rank(f"round {i} profile:\n{buf.getvalue()}",
                       loglevel=logging.DEBUG)

    # ------------------------------------------------------------------
    def backup_models(self, i):
        """Reference: server.py:530-559."""
        if (i % self.model_backup_freq) == 0:
            flush_saves()  # epoch snapshots copy the best_* files
        self.worker_trainer.save(model_path=self.model_path, token="latest",
                                 config=self.config["server_config"])
        if (i % self.model_backup_freq) == 0:
            self.worker_trainer.save(model_path=self.model_path,
                                     token=f"epoch{i}",
                                     config=self.config["server_config"])
            for bodyname in ["best_val_acc", "best_val_loss", "best_test_acc"]:
                src = os.path.join(self.model_path, f"{bodyname}_model.tar")
                if os.path.exists(src):
                    dst = os.path.join(self.model_path,
                                       f"epoch{i}_{bodyname}_model.tar")
                    shutil.copyfile(src, dst)

    def fall_back_to_prev_best_status(self):
        """Reference: server.py:561-578."""
        flush_saves()
        if not self.fall_back_to_best_model:
            return
        if os.path.exists(self.best_model_path):
            print_rank(f"falling back to model {self.best_model_path}")
            tmp_lr = get_lr(self.worker_trainer.optimizer)
            self.worker_trainer.load(self.best_model_path,
                                     update_lr_scheduler=False,
                                     update_ss_scheduler=False)
            for g in self.worker_trainer.optimizer.param_groups:
                g["lr"] = tmp_lr
            if self.server_trainer is not None:
                self.server_trainer.model = self.worker_trainer.model


def _needs_cid_cls():
    from ..strategies.dga import DGA
    return DGA


def select_server(server_type):
    """Reference: server.py:581-597 (PersonalizationServer differs only in
    eval-client construction, handled by Evaluation/make_eval_clients)."""
    return OptimizationServer
