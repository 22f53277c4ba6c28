"""Client logic (reference: core/client.py).

Two pieces:

* ``Client`` — the lightweight per-task handle (same constructor contract
  as the reference: client ids + config + send_gradients) with the shared
  train-dataset cache and the data-slicing helper;
* ``ClientExecutor`` — NEW: a per-rank persistent workspace holding ONE
  resident model replica + flat arena + fused optimizer that every local
  client reuses.  The reference pays per-client model construction,
  parameter copy via Python loops and a fresh optimizer object per client
  (client.py:280-344); here per-client setup is one flat arena copy (K13)
  plus an optimizer-state reset.
"""

from __future__ import annotations

import logging
import os
import time
from typing import Dict, Optional

import torch

from .. import ops
from ..models import make_model
from ..ops.arena import ParameterArena
from ..ops.fused_optim import make_arena_optimizer
from ..strategies import select_strategy
from ..utils import (ScheduledSamplingScheduler, alpha_update, make_optimizer,
                     print_rank, to_device)
from ..utils.dataloaders_utils import get_dataset, make_train_dataloader
from .trainer import Trainer, set_component_wise_lr

# Worker-wide dataset cache (reference: client.py:45-47, 76-99)
train_dataset = None
trainset_unlab = None
trainset_unlab_rand = None


class Client:
    """Per-client handle (reference: core/client.py:49-124)."""

    def __init__(self, client_id, config, send_gradients):
        self.client_id = client_id
        self.config = config
        self.send_gradients = send_gradients

    def get_client_data(self, dataset=None):
        client_data = self.get_data(self.client_id, dataset)
        return self.client_id, client_data, self.config, self.send_gradients

    @staticmethod
    def get_train_dataset(data_path, client_train_config, task):
        """Load + cache the full train dataset once per worker
        (reference: client.py:76-99)."""
        global train_dataset, trainset_unlab, trainset_unlab_rand
        train_dataset = get_dataset(data_path, client_train_config, task, mode="train")
        if task == "semisupervision":
            trainset_unlab = get_dataset(data_path, client_train_config, task,
                                         mode="train", user_idx=-2)
            trainset_unlab_rand = get_dataset(data_path, client_train_config,
                                              task, mode="train", user_idx=-3)
        else:
            trainset_unlab = None
            trainset_unlab_rand = None
        return len(train_dataset.user_list)

    @staticmethod
    def get_data(clients, dataset):
        """Slice the cached dataset into one-or-more users' data blobs
        (reference: client.py:101-124)."""
        if dataset is None:
            datasets = ([train_dataset, trainset_unlab, trainset_unlab_rand]
                        if trainset_unlab is not None else [train_dataset])
        else:
            datasets = [dataset]
        data_with_labels = getattr(datasets[0], "user_data_label", None) is not None
        strcts = []
        for ds in datasets:
            s = {"users": [], "num_samples": [], "user_data": {}}
            if data_with_labels:
                s["user_data_label"] = {}
            for client in clients:
                user = ds.user_list[client]
                s["users"].append(user)
                s["num_samples"].append(ds.num_samples[client])
                s["user_data"][user] = ds.user_data[user]
                if data_with_labels:
                    s["user_data_label"][user] = ds.user_data_label[user]
            strcts.append(s)
        return strcts


class ClientExecutor:
    """Per-rank persistent client-training workspace."""

    def __init__(self, config, task, data_path, server_arena: ParameterArena,
                 model_path: Optional[str] = None):
        self.config = config
        self.task = task
        self.data_path = data_path
        self.server_arena = server_arena
        self.model_path = model_path or config.get("model_path")
        self.model_config = config["model_config"]
        self.client_config = config["client_config"]
        self.server_config = config["server_config"]

        # resident client model replica + arena
        self.model = make_model(self.model_config)
        self.arena = ParameterArena(self.model, bind_grads=True)
        # reusable fused optimizer (reset per client)
        opt_cfg = dict(self.client_config["optimizer_config"])
        opt_cfg.setdefault("lr", self.server_config.get("initial_lr_client", 1.0))
        self._fused_opt = make_arena_optimizer(opt_cfg, self.arena)
        self._torch_opt_cfg = opt_cfg

        strategy_cls = select_strategy(config["strategy"])
        self.client_strategy = strategy_cls("client", config, self.model_path)
        self.send_dicts = self.server_config.get("send_dicts", False)
        self.perf_acc = {}  # cumulative phase timings (bench diagnostics)

        # Device-resident client shard cache (SURVEY.md §7.1 divergence 3):
        # a client's packed (x, y) tensors live in HBM after first use, so
        # repeat sampling of a client costs zero host->device traffic and
        # zero Dataset re-construction.  288 GB/GPU holds every reference
        # dataset's full shard set; budget-capped for safety.
        self._shard_cache: Dict[str, object] = {}
        self._shard_cache_bytes = 0
        self._shard_cache_budget = int(self.client_config.get(
            "shard_cache_gb", 64)) << 30
        self._shard_store = None       # whole-dataset device pack
        self._shard_store_tried = False

        # hipGraph fast path for the per-batch client step (ops/graphs.py);
        # enabled on GPU for plain-SGD clients unless disabled by config
        # Fully-fused hand-written kernel path for the flagship CNN
        # (ops/fused_cnn.py): engaged when the arena layout matches and the
        # client optimizer is plain SGD (momentum/wd 0).
        self.fused_cnn = None
        opt_c = dict(self.client_config["optimizer_config"])
        if (torch.cuda.is_available() and ops.HAS_EXT
                and self.client_config.get("use_fused_cnn", True)
                and opt_c.get("type", "sgd") == "sgd"
                and float(opt_c.get("momentum", 0.0)) == 0.0
                and float(opt_c.get("weight_decay", 0.0)) == 0.0):
            from ..ops.fused_cnn import FusedCNNEpoch, matches_cnn_femnist
            C = matches_cnn_femnist(self.arena)
            bs_cfg = int(self.client_config["data_config"]["train"]
                         .get("batch_size", 20))
            if C is not None and bs_cfg <= 32:
                drops = [m.p for m in self.model.modules()
                         if isinstance(m, torch.nn.Dropout)]
                self.fused_cnn = FusedCNNEpoch(
                    self.arena, C,
                    bs=int(self.client_config["data_config"]["train"]
                           .get("batch_size", 20)),
                    p1=drops[0] if len(drops) > 0 else 0.25,
                    p2=drops[1] if len(drops) > 1 else 0.5,
                    max_grad_norm=self.client_config["data_config"]["train"]
                    .get("max_grad_norm"),
                    use_bf16=self.client_config.get(
                        "mixed_precision", "") == "bf16")

        # MIOpen RNN kernels segfault under hipGraph capture (hipblaslt
        # assert -> SIGSEGV, observed with the fedshakespeare LSTM), so
        # models containing RNN modules always run the eager path.
        has_rnn = any(isinstance(m, torch.nn.RNNBase)
                      for m in self.model.modules())
        self.graph_cache = None
        if (torch.cuda.is_available() and not has_rnn
                and self.fused_cnn is None  # fused path supersedes graphs
                and self.client_config.get("use_hip_graphs", True)):
            from ..ops.graphs import GraphCache
            cache = GraphCache(
                self.model, self.arena,
                dict(self.client_config["optimizer_config"]),
                self.client_config["data_config"]["train"].get("max_grad_norm"))
            if cache.supports():
                self.graph_cache = cache

    # ------------------------------------------------------------------
    def _get_shard_store(self, data_config):
        """Lazily pack the whole train set into device HBM (one pinned H2D)
        when the task uses the generic array machinery; None otherwise."""
        if self._shard_store_tried:
            return self._shard_store
        self._shard_store_tried = True
        import msrflute_amd.core.client as client_mod
        from ..models import get_exp_dataloader
        from ..models.generic_data import ArrayDataLoader, DeviceShardStore
        ds = client_mod.train_dataset
        try:
            dl_cls = get_exp_dataloader(self.task) if self.task else None
        except Exception:
            dl_cls = None
        eligible = (
            ds is not None
            and getattr(ds, "user_data_label", None) is not None
            and dl_cls is not None and issubclass(dl_cls, ArrayDataLoader)
            and not getattr(ds, "_want_transform", False)
            and self.client_config.get("cache_client_shards", True))
        if eligible:
            try:
                x_shape = tuple(getattr(ds, "x_shape", None) or ())
                self._shard_store = DeviceShardStore(
                    ds, x_shape, device=self.arena.device,
                    budget_bytes=int(self.client_config.get(
                        "shard_cache_gb", 128)) << 30)
                print_rank(
                    f"packed {len(ds.user_list)} client shards into device "
                    f"memory ({self._shard_store.x.numel() * 4 >> 20} MiB)")
            except MemoryError as e:
                print_rank(f"shard store disabled: {e}")
        return self._shard_store

    def _make_optimizer(self, initial_lr):
        if self._fused_opt is not None:
            self._fused_opt.reset_state()
            if initial_lr > 0:
                self._fused_opt.param_groups[0]["lr"] = initial_lr
            return self._fused_opt
        cfg = dict(self._torch_opt_cfg)
        if initial_lr > 0:
            cfg["lr"] = initial_lr
        return make_optimizer(cfg, self.model)

    def process_round(self, client: Client, initial_lr: float, iteration: int,
                      round_seed: int = 0) -> Dict:
        """Run one client's local training (reference: client.py:226-511)."""
        config = self.config
        client_config = self.client_config
        data_config = client_config["data_config"]["train"]
        privacy_metrics_config = config.get("privacy_metrics_config", None)
        begin = time.time()
        client_stats = {}

        client_id, data_strcts, _, send_gradients = client.get_client_data()
        data_strct = data_strcts[0]
        user = data_strct["users"][0]

        # Per-client seed discipline: a client's
        # local-SGD trajectory (shuffle order, dropout) is a function of
        # (round, client) only — independent of which rank runs it — so the
        # aggregated round result is world-size-invariant up to fp
        # reduction order.
        torch.manual_seed(round_seed & 0x7FFFFFFFFFFF)

        t_dl = time.time()
        train_dataloader = self._shard_cache.get(user)
        if train_dataloader is None and torch.cuda.is_available():
            store = self._get_shard_store(data_config)
            if store is not None:
                train_dataloader = store.loader_for(
                    user, data_config.get("batch_size", 40))
                if train_dataloader is not None:
                    self._shard_cache[user] = train_dataloader
        if train_dataloader is None:
            train_dataloader = make_train_dataloader(
                data_config, self.data_path, task=self.task, clientx=0,
                data_strct=data_strct)
            if hasattr(train_dataloader, "to_device"):
                train_dataloader.to_device()
            if (torch.cuda.is_available()
                    and hasattr(train_dataloader, "dataset")
                    and getattr(train_dataloader.dataset, "x", None) is not None
                    and torch.is_tensor(train_dataloader.dataset.x)
                    and train_dataloader.dataset.x.is_cuda
                    and self._shard_cache_bytes < self._shard_cache_budget):
                ds = train_dataloader.dataset
                nbytes = ds.x.numel() * ds.x.element_size()
                if ds.y is not None and torch.is_tensor(ds.y):
                    nbytes += ds.y.numel() * ds.y.element_size()
                self._shard_cache[user] = train_dataloader
                self._shard_cache_bytes += nbytes
        client_stats["dataloader"] = time.time() - t_dl

        # one flat copy-in instead of the reference's per-tensor clone loop
        # (client.py:294-301, K13)
        self.arena.copy_data_(self.server_arena.data)
        self.arena.zero_grad()

        trainer_config = client_config.get("trainer_config", {})
        if "updatable_names" in trainer_config:
            set_component_wise_lr(self.model, client_config["optimizer_config"],
                                  trainer_config["updatable_names"])

        optimizer = self._make_optimizer(initial_lr)

        ss_scheduler = None
        if client_config.get("ss_config") is not None:
            ss_scheduler = ScheduledSamplingScheduler(model=self.model,
                                                      **client_config["ss_config"])

        trainer = Trainer(
            model=self.model,
            optimizer=optimizer,
            ss_scheduler=ss_scheduler,
            train_dataloader=train_dataloader,
            # graph path engaged inside run_train_epoch when applicable

            server_replay_config=client_config,
            max_grad_norm=data_config.get("max_grad_norm", None),
            anneal_config=client_config.get("annealing_config", None),
            num_skips_threshold=client_config.get("num_skips_threshold", -1),
            ignore_subtask=client_config["ignore_subtask"],
            arena=self.arena,
        )
        trainer.graph_cache = self.graph_cache
        trainer.fused_cnn = self.fused_cnn
        trainer.round_seed = round_seed & 0x7FFFFFFFFFFF

        desired_max_samples = data_config.get("desired_max_samples", None)
        apply_privacy_metrics = bool(privacy_metrics_config
                                     and privacy_metrics_config["apply_metrics"])
        # FedAvg weights are num_samples (host-known), so the per-client
        # loss/stats host syncs can be deferred and batched round-level
        trainer.lazy_stats = (
            config["strategy"] == "FedAvg" and not apply_privacy_metrics
            and self.arena.device.type == "cuda"
            and not getattr(self.client_strategy, "stats_on_smooth_grad", False)
            and self.server_config.get("type") != "personalization")

        client_stats["setup"] = time.time() - begin
        begin_training = time.time()

        self.model.train()

        algo_payload = None
        strategy_algo = config["strategy"]
        if strategy_algo == "FedLabels":
            datasets = [get_dataset(self.data_path, config, self.task,
                                    mode="train", test_only=False,
                                    data_strct=data_strcts[i], user_idx=0)
                        for i in range(3)]
            algo_payload = {"strategy": "FedLabels", "data": datasets,
                            "iter": iteration,
                            "config": client_config.get("semisupervision")}
        elif strategy_algo == "FedProx":
            algo_payload = {"strategy": "FedProx",
                            "mu": client_config.get("mu", 0.001)}

        train_loss, num_samples, algo_computation = trainer.train_desired_samples(
            desired_max_samples=desired_max_samples,
            apply_privacy_metrics=apply_privacy_metrics,
            algo_payload=algo_payload)

        trainer.train_loss = train_loss
        trainer.num_samples = num_samples
        trainer.algo_computation = algo_computation

        # pseudo-gradient g = w_server − w_trained (K1; reference client.py:380-383)
        if not self.send_dicts:
            ops.pseudo_grad(self.arena.grad, self.server_arena.data,
                            self.arena.data, 1.0)

        payload = self.client_strategy.generate_client_payload(trainer) \
            if send_gradients else None

        if self.server_config.get("type") == "personalization":
            self._personalization_round(client, data_strct, data_config,
                                        client_config, trainer, initial_lr,
                                        user, desired_max_samples)

        client_stats["training"] = time.time() - begin_training
        client_stats["full cost"] = time.time() - begin
        for k, v in client_stats.items():
            self.perf_acc[k] = self.perf_acc.get(k, 0.0) + v
        self.perf_acc["clients"] = self.perf_acc.get("clients", 0) + 1

        if train_loss is None and trainer.loss_dev is not None:
            # deferred host sync: the server finalizes these in one batch
            client_output = {
                "cs": client_stats, "ns": num_samples, "pl": payload,
                "_lazy": (trainer.loss_dev, trainer.stats_dev,
                          trainer.counter),
            }
        else:
            client_output = {
                "cs": client_stats,
                "tl": train_loss,
                "mg": trainer.sufficient_stats["mag"],
                "vg": trainer.sufficient_stats["var"],
                "ng": trainer.sufficient_stats["mean"],
                "rg": trainer.sufficient_stats["norm"],
                "ns": num_samples,
                "pl": payload,
            }

        if apply_privacy_metrics:
            self._apply_privacy_metrics(client_output, trainer,
                                        privacy_metrics_config)

        client_output["ts"] = time.time()
        return client_output

    # ------------------------------------------------------------------
    def _personalization_round(self, client, data_strct, data_config,
                               client_config, trainer, initial_lr, user,
                               desired_max_samples):
        """Per-user local model + convex-interpolation alpha update
        (reference: client.py:387-443)."""
        alpha = client_config.get("convex_model_interp", 0.75)
        local_model = make_model(self.config["model_config"])
        train_dataloader = make_train_dataloader(
            data_config, self.data_path, task=self.task, clientx=0,
            data_strct=data_strct)
        local_optimizer = make_optimizer(dict(client_config["optimizer_config"]),
                                         local_model)
        local_trainer = Trainer(
            model=local_model, optimizer=local_optimizer, ss_scheduler=None,
            train_dataloader=train_dataloader, server_replay_config=client_config,
            max_grad_norm=data_config.get("max_grad_norm", None),
            anneal_config=client_config.get("annealing_config", None),
            ignore_subtask=client_config["ignore_subtask"])

        local_model_name = os.path.join(self.model_path, f"{user}_model.tar")
        local_alpha_name = os.path.join(self.model_path, f"{user}_alpha")
        if os.path.exists(local_model_name):
            local_trainer.load(local_model_name, update_lr_scheduler=False,
                               update_ss_scheduler=False)
        if os.path.exists(local_alpha_name):
            alpha = torch.load(local_alpha_name, weights_only=False)

        original_local_model = local_trainer.get_model()
        local_model.train()
        train_loss, num_samples, _ = local_trainer.train_desired_samples(
            desired_max_samples=desired_max_samples, apply_privacy_metrics=False)
        print_rank(f"user {user}: LOCAL training loss={train_loss}",
                   loglevel=logging.DEBUG)
        local_trainer.save(model_path=self.model_path, config=self.config,
                           token=user)
        for p, orig in zip(local_trainer.model.parameters(),
                           original_local_model.parameters()):
            p.grad = to_device(orig.data) - p.data
        alpha = alpha_update(local_trainer.model, trainer.model, alpha, initial_lr)
        torch.save(alpha, local_alpha_name)

    # ------------------------------------------------------------------
    def _apply_privacy_metrics(self, client_output, trainer,
                               privacy_metrics_config):
        """Token-extraction + leakage attacks; may zero the client's weight
        (reference: client.py:466-508)."""
        from ..extensions.privacy import metrics as privacy_metrics

        privacy_stats = {"Dropped clients": 0}
        batches = trainer.cached_batches
        trainer.cached_batches = []
        gradients = self.arena.grad  # already flat

        if privacy_metrics_config.get("apply_indices_extraction", False):
            allowed_word_rank = privacy_metrics_config.get("allowed_word_rank", 9000)
            embed_dim = self.model_config["embed_dim"]
            vocab_size = self.model_config["vocab_size"]
            overlap, indices = privacy_metrics.extract_indices_from_embeddings(
                gradients, batches, embed_dim, vocab_size)
            max_overlap = privacy_metrics_config.get("max_allowed_overlap", None)
            if max_overlap is not None and overlap > max_overlap:
                print_rank(f"Removing client: extracted {overlap*100}% words, "
                           f"max allowed {max_overlap*100}%")
                client_output["wt"] = 0.0
                privacy_stats["Dropped clients"] = 1
            privacy_stats["Extracted indices percentage"] = overlap
            privacy_stats[f"Words percentage above {allowed_word_rank} word rank"] = \
                float((indices > allowed_word_rank).mean()) if len(indices) > 0 else 0

        if privacy_metrics_config.get("apply_leakage_metric", False):
            import numpy as np
            orig_params = {n: self.server_arena.data[
                self.server_arena.offsets[i]:self.server_arena.offsets[i]
                + self.server_arena.numels[i]].view(self.server_arena.shapes[i])
                for i, n in enumerate(self.server_arena.names)}
            # practical_epsilon_leakage needs a full state dict
            sd = {k: v.clone() for k, v in self.model.state_dict().items()}
            for n, p in orig_params.items():
                if n in sd:
                    sd[n] = p.clone()
            max_ratio = float(np.exp(privacy_metrics_config["max_leakage"]))
            leakage = privacy_metrics.practical_epsilon_leakage(
                sd, self.model, batches,
                privacy_metrics_config.get("is_leakage_weighted", False),
                max_ratio,
                privacy_metrics_config.get("attacker_optimizer_config"))
            max_leakage = privacy_metrics_config.get("max_allowed_leakage", None)
            if max_leakage is not None and leakage > max_leakage:
                print_rank(f"Removing client: leakage {leakage} > {max_leakage}")
                client_output["wt"] = 0.0
                privacy_stats["Dropped clients"] = 1
            privacy_stats["Practical epsilon (Max leakage)"] = leakage

        client_output["ps"] = privacy_stats

def convex_inference(model_global, model_personal, alpha):
    """Personalized logit interpolation accuracy
    (reference: utils/utils.py:598-603)."""
    import numpy as np
    targets = torch.tensor(model_global["labels"])
    probs = alpha * model_personal["probabilities"] + \
        (1 - alpha) * model_global["probabilities"]
    preds = torch.argmax(torch.tensor(np.asarray(probs)), dim=1)
    return torch.mean((preds == targets).float()).item()


class ClientPool:
    """P client executors on P HIP streams: clients within a round train
    CONCURRENTLY on the GPU (the round is a weighted sum — order-free).

    Each executor owns a model replica, arena, graph cache and a round
    accumulator; its stream serializes its own clients.  Weighted
    pseudo-gradients accumulate stream-locally; ``flush()`` joins the
    streams and adds the P accumulators into the server grad arena.  This
    is the MI355X-native replacement for the reference's process-per-GPU
    worker pool: concurrency lives on streams inside one
    rank instead of extra processes.
    """

    def __init__(self, config, task, data_path, server_arena, model_path=None,
                 n_parallel=4):
        self.server_arena = server_arena
        self.executors = [ClientExecutor(config, task, data_path,
                                         server_arena, model_path)
                          for _ in range(max(1, int(n_parallel)))]
        self.streams = [torch.cuda.Stream() for _ in self.executors]
        self.round_accums = [server_arena.new_buffer() for _ in self.executors]
        self._rr = 0
        self._streams_dirty = False

    # properties the server reads off the executor
    @property
    def client_strategy(self):
        return self.executors[0].client_strategy

    @property
    def perf_acc(self):
        merged = {}
        for ex in self.executors:
            for k, v in ex.perf_acc.items():
                merged[k] = merged.get(k, 0) + v
        return merged

    def process_round(self, client, initial_lr, iteration, round_seed=0):
        k = self._rr % len(self.executors)
        self._rr += 1
        ex, st = self.executors[k], self.streams[k]
        # share the packed shard store across replicas (one copy in HBM)
        prim = self.executors[0]
        if ex is not prim and prim._shard_store_tried and not ex._shard_store_tried:
            ex._shard_store = prim._shard_store
            ex._shard_store_tried = True
        st.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(st):
            out = ex.process_round(client, initial_lr, iteration, round_seed)
            payload = out.get("pl")
            if payload is not None and payload.get("grad") is not None \
                    and payload["weight"] != 0.0:
                ops.axpy(self.round_accums[k], payload["grad"], 1.0)
                payload["grad"] = None
                payload["pooled"] = True
        self._streams_dirty = True
        return out

    def flush(self, target_grad):
        """Join client streams and fold the P accumulators into the server
        grad arena.  Must run before the round's lazy-stats finalize."""
        if not self._streams_dirty:
            return
        cur = torch.cuda.current_stream()
        for st in self.streams:
            cur.wait_stream(st)
        for acc in self.round_accums:
            ops.axpy(target_grad, acc, 1.0)
            acc.zero_()
        self._streams_dirty = False


def _fused_round_eligible(ex, data_cfg):
    """The fused whole-round driver trains full shards with plain SGD and
    aggregates every layer's gradient; fall back to the per-client path
    whenever the eager path would behave differently: a frozen layer
    (its pseudo-grad segment must be zeroed, fedavg.py:58) or a
    num_skips_threshold (clients can be zero-weighted mid-epoch).
    desired_max_samples is checked per client against shard sizes by the
    callers (the eager path truncates each client's epoch there)."""
    if ex.model_config.get("freeze_layer", None):
        return False
    if int(ex.client_config.get("num_skips_threshold", -1)) >= 0:
        return False
    return True


def _fused_round_impl(ex, client_ids, initial_lr, seeds):
    """One _C.cnn_round call training all of ``ex``'s clients for this
    round (copy-in, fused epoch, weighted pseudo-grad, accumulate into
    ``ex.round_accum_target``).  Returns outputs or None if ineligible."""
    if ex.fused_cnn is None:
        return None
    data_cfg = ex.client_config["data_config"]["train"]
    if not _fused_round_eligible(ex, data_cfg):
        return None
    store = ex._get_shard_store(data_cfg)
    if store is None or store.x[0].numel() != 784:
        return None
    import msrflute_amd.core.client as cm
    ds = cm.train_dataset
    if ds is None:
        return None
    fc = ex.fused_cnn
    dms = data_cfg.get("desired_max_samples", None)
    counts, row_bases, order_offs, orders, w_list = [], [], [], [], []
    off = 0
    for cid, seed in zip(client_ids, seeds):
        user = ds.user_list[cid]
        i = store.user_pos.get(user)
        if i is None:
            return None
        lo, hi = store.offsets[i], store.offsets[i + 1]
        n = hi - lo
        if n == 0 or (dms is not None and n > dms):
            return None
        torch.manual_seed(seed & 0x7FFFFFFFFFFF)
        orders.append(torch.randperm(n))
        counts.append(n)
        row_bases.append(lo)
        order_offs.append(off)
        off += n
        w_list.append(float(n))  # FedAvg weight = num_samples
    K = len(client_ids)
    orders_cat = torch.cat(orders) if orders else torch.empty(0, dtype=torch.int64)
    if getattr(ex, "_order_pin", None) is None or ex._order_pin.numel() < off:
        ex._order_pin = torch.empty(max(off, 1024),
                                    dtype=torch.int64).pin_memory()
        ex._order_pin_ev = None
    if ex._order_pin_ev is not None:
        # previous round's non_blocking H2D of this buffer may still be
        # queued; overwriting the pinned staging before it lands would
        # corrupt that round's shuffle orders
        ex._order_pin_ev.synchronize()
    ex._order_pin[:off].copy_(orders_cat)
    orders_dev = ex._order_pin[:off].to(ex.arena.device, non_blocking=True)
    ex._order_pin_ev = torch.cuda.Event()
    ex._order_pin_ev.record()
    if getattr(ex, "_round_stats", None) is None or \
            ex._round_stats.numel() < 2 * K:
        ex._round_stats = torch.zeros(2 * max(K, 8), device=ex.arena.device)
        ex._round_loss = torch.zeros(max(K, 8), device=ex.arena.device)
    ex._round_stats[: 2 * K].zero_()
    ex._round_loss[:K].zero_()
    fc.lr_t.fill_(float(initial_lr))
    ops_mod = ops
    _ = ops_mod  # (ops imported at module top)
    from msrflute_amd import _C
    _C.cnn_round(
        store.x.reshape(len(store.y), -1), store.y, orders_dev,
        torch.tensor(row_bases, dtype=torch.int64),
        torch.tensor(order_offs, dtype=torch.int64),
        torch.tensor(counts, dtype=torch.int64),
        torch.tensor(w_list, dtype=torch.float32),
        torch.tensor([s & 0x7FFFFFFFFFFF for s in seeds], dtype=torch.int64),
        fc.bs, fc.C, ex.server_arena.data, ex.arena.data, ex.arena.grad,
        ex.round_accum_target, fc.work_f, fc.work_i, fc.work_b, fc.work_d,
        fc.lr_t, fc.max_norm, fc.p1, fc.p2,
        ex._round_stats, ex._round_loss, fc.use_bf16)
    outputs = []
    now = time.time()
    for k, cid in enumerate(client_ids):
        n_batches = (counts[k] + fc.bs - 1) // fc.bs
        outputs.append((cid, {
            "cs": {"setup": 0.0, "training": 0.0, "full cost": 0.0,
                   "dataloader": 0.0},
            "ns": counts[k],
            "pl": {"weight": w_list[k], "grad": None, "pooled": True},
            "_lazy": (ex._round_loss[k].reshape(()),
                      ex._round_stats[2 * k: 2 * k + 2],
                      n_batches * ex.arena.total),
            "ts": now,
        }))
        ex.perf_acc["clients"] = ex.perf_acc.get("clients", 0) + 1
    return outputs


def _try_mega_lstm_round(self, prim, data_cfg, client_ids, initial_lr,
                         seeds):
    """Cross-client MEGA round for the Shakespeare char-LSTM
    (ops/mega_shakespeare.py): all K clients' epochs in one graph-captured
    launch set (batched recurrence kernels + bmm projections over
    K-stacked weights).  Returns outputs or None if not this task /
    ineligible."""
    if self._mega_lstm is False:
        return None
    if not prim.client_config.get("use_mega_round", True):
        return None
    if prim.client_config.get("mixed_precision"):
        return None  # the LSTM path is fp32
    if prim.arena is None or prim.arena.device.type != "cuda":
        return None
    opt_cfg = prim.client_config.get("optimizer_config", {})
    if (opt_cfg.get("type", "sgd") != "sgd"
            or opt_cfg.get("momentum", 0) or opt_cfg.get("weight_decay", 0)
            or opt_cfg.get("nesterov", False)):
        return None
    if self._mega_lstm is None:
        from ..ops import HAS_EXT
        from ..ops.mega_shakespeare import (ShakespeareMegaRound,
                                            matches_char_lstm)
        if not HAS_EXT or matches_char_lstm(prim.arena) is None:
            self._mega_lstm = False  # not this model: stop probing
            return None
        self._mega_lstm = ShakespeareMegaRound(
            prim.arena, data_cfg.get("batch_size", 4),
            data_cfg.get("max_grad_norm"))
    if not self._mega_lstm.supports(len(client_ids)):
        return None
    store = prim._get_shard_store(data_cfg)
    if store is None or store.x.dim() != 2:
        return None
    import msrflute_amd.core.client as cm
    ds = cm.train_dataset
    if ds is None:
        return None
    dms = data_cfg.get("desired_max_samples", None)
    for cid in client_ids:
        i = store.user_pos.get(ds.user_list[cid])
        if i is None:
            return None
        n = store.offsets[i + 1] - store.offsets[i]
        if n == 0 or (dms is not None and n > dms):
            return None
    out = self._mega_lstm.run(store, ds, client_ids, seeds, initial_lr,
                              self.server_arena, self.round_accums[0])
    if out is not None:
        self._streams_dirty = True
        prim.perf_acc["clients"] = (prim.perf_acc.get("clients", 0)
                                    + len(client_ids))
    return out


def _try_mega_resnet_round(self, prim, data_cfg, client_ids, initial_lr,
                           seeds):
    """Cross-client MEGA round for the fed-CIFAR100 ResNet-18
    (ops/mega_resnet.py): all K clients' epochs as one graph-captured
    launch set of grouped convs + GroupNorm over K-stacked weights.

    OPT-IN (use_mega_round_resnet): measured SLOWER than the per-client
    epoch-graph path on ROCm 7.2 — MIOpen decomposes groups=K convs into
    K per-group kernels (no batching win) and grouped bwd-weight falls
    to multi-ms CK batched-GEMM fallbacks (PERF.md, profiles/).  The
    formulation itself is exact (tests/test_mega_cpu.py, f64) and flips
    on wholesale when a ROCm release makes grouped conv competitive."""
    if self._mega_resnet is False:
        return None
    if not prim.client_config.get("use_mega_round_resnet", False):
        return None
    if not prim.client_config.get("use_mega_round", True):
        return None
    if prim.client_config.get("mixed_precision"):
        return None
    if prim.arena is None or prim.arena.device.type != "cuda":
        return None
    opt_cfg = prim.client_config.get("optimizer_config", {})
    if (opt_cfg.get("type", "sgd") != "sgd"
            or opt_cfg.get("momentum", 0) or opt_cfg.get("weight_decay", 0)
            or opt_cfg.get("nesterov", False)):
        return None
    if self._mega_resnet is None:
        from ..ops import HAS_EXT
        from ..ops.mega_resnet import ResNetMegaRound, matches_resnet18
        cpg = int(prim.model_config.get("group_norm", 0) or 0)
        if (not HAS_EXT or cpg <= 0
                or matches_resnet18(prim.arena) is None):
            self._mega_resnet = False  # not this model: stop probing
            return None
        self._mega_resnet = ResNetMegaRound(
            prim.arena, data_cfg.get("batch_size", 20),
            data_cfg.get("max_grad_norm"), cpg)
    if not self._mega_resnet.supports(len(client_ids)):
        return None
    store = prim._get_shard_store(data_cfg)
    if store is None or store.x.dim() != 4:
        return None
    import msrflute_amd.core.client as cm
    ds = cm.train_dataset
    if ds is None:
        return None
    dms = data_cfg.get("desired_max_samples", None)
    for cid in client_ids:
        i = store.user_pos.get(ds.user_list[cid])
        if i is None:
            return None
        n = store.offsets[i + 1] - store.offsets[i]
        if n == 0 or (dms is not None and n > dms):
            return None
    out = self._mega_resnet.run(store, ds, client_ids, seeds, initial_lr,
                                self.server_arena, self.round_accums[0])
    if out is not None:
        self._streams_dirty = True
        prim.perf_acc["clients"] = (prim.perf_acc.get("clients", 0)
                                    + len(client_ids))
    return out


def _pool_run_fused_round_batch(self, client_ids, initial_lr, iteration,
                                seeds):
    """ClientPool: one _C.cnn_round per executor (on its stream) covering
    its chunk of the round's clients.  Returns outputs or None if the
    fused path is ineligible (caller falls back to the per-client path).
    Eligibility is validated UP FRONT so no partial state is mutated on
    the fallback path."""
    prim = self.executors[0]
    data_cfg = prim.client_config["data_config"]["train"]
    if not _fused_round_eligible(prim, data_cfg):
        return None
    if getattr(self, "_mega_lstm", "missing") == "missing":
        self._mega_lstm = None
    if getattr(self, "_mega_resnet", "missing") == "missing":
        self._mega_resnet = None
    out = _try_mega_lstm_round(self, prim, data_cfg, client_ids,
                               initial_lr, seeds)
    if out is None:
        out = _try_mega_resnet_round(self, prim, data_cfg, client_ids,
                                     initial_lr, seeds)
    if out is not None:
        return out
    if prim.fused_cnn is None:
        return None
    store = prim._get_shard_store(data_cfg)
    if store is None or store.x[0].numel() != 784:
        return None
    import msrflute_amd.core.client as cm
    ds = cm.train_dataset
    if ds is None:
        return None
    dms = data_cfg.get("desired_max_samples", None)
    for cid in client_ids:
        i = store.user_pos.get(ds.user_list[cid])
        if i is None:
            return None
        n = store.offsets[i + 1] - store.offsets[i]
        if n == 0 or (dms is not None and n > dms):
            return None

    # cross-client MEGA round: one launch set per batch-step covering all
    # K clients (fp32 path; bf16 keeps the per-executor fused rounds)
    fc = prim.fused_cnn
    if prim.client_config.get("use_mega_round", True):
        if getattr(self, "_mega", None) is None:
            from ..ops.fused_cnn import MegaRound
            self._mega = MegaRound(prim.arena, fc.C, fc.bs, fc.p1, fc.p2,
                                   prim.client_config["data_config"]["train"]
                                   .get("max_grad_norm"),
                                   use_bf16=fc.use_bf16)
        if self._mega.supports(len(client_ids)):
            out = self._mega.run(store, ds, client_ids, seeds, initial_lr,
                                 self.server_arena, self.round_accums[0])
            if out is not None:
                self._streams_dirty = True
                prim.perf_acc["clients"] = (prim.perf_acc.get("clients", 0)
                                            + len(client_ids))
                return out
    for ex in self.executors[1:]:
        if not ex._shard_store_tried:
            ex._shard_store = store
            ex._shard_store_tried = True

    P = len(self.executors)
    chunks = [client_ids[k::P] for k in range(P)]
    seed_chunks = [seeds[k::P] for k in range(P)]
    outputs = []
    for k, (chunk, schunk) in enumerate(zip(chunks, seed_chunks)):
        if not chunk:
            continue
        ex, st = self.executors[k], self.streams[k]
        ex.round_accum_target = self.round_accums[k]
        st.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(st):
            out = _fused_round_impl(ex, chunk, initial_lr, schunk)
        assert out is not None  # pre-validated above
        outputs.extend(out)
        self._streams_dirty = True
    return outputs


ClientPool.run_fused_round_batch = _pool_run_fused_round_batch
