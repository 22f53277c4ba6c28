"""Training engine: client-side Trainer + server-side ModelUpdater.

Reference: core/trainer.py (TrainerBase 30-79, ModelUpdater 82-197,
Trainer 200-687, run_validation_generic 690-723, save_model 753-775).

MI355X-first differences (behavior-preserving):

* the model's params/grads are bound to a flat ``ParameterArena``; gradient
  clipping, gradient sufficient statistics and (when the optimizer type
  allows) the optimizer step are single fused kernels over the arena
  instead of per-tensor loops;
* gradient statistics (reference trainer.py:271-292 round-trips every
  tensor through ``.cpu().numpy()`` per batch) accumulate in a device
  tensor and hit the host exactly once per epoch;
* the FedProx proximal term (reference trainer.py:463-466, which buggily
  re-adds partial sums inside the parameter loop — SURVEY.md §7.5) is
  applied exactly: ``grad += mu*(w - w_global)`` as two fused axpys and
  ``loss += (mu/2)*||w - w_global||^2`` for reporting.

Known deliberate deviation: the reference's ``var`` sufficient stat is
identically zero by construction (``sq_sum/n - (sqrt(sq_sum/n))**2``,
trainer.py:300-301); we compute the actual variance ``sq_sum/n - mean**2``.
"""

from __future__ import annotations

import copy
import logging
import math
import os
from typing import Optional

import torch
import yaml

from .. import ops
from ..ops.arena import ParameterArena
from ..utils import (make_lr_scheduler, make_optimizer, print_rank, to_device)
from .metrics import Metrics


def get_lr(optimizer) -> float:
    return optimizer.param_groups[0]["lr"]


class TrainerBase:
    """Common trainer interface (reference: core/trainer.py:30-79)."""

    def __init__(self, model, train_dataloader, optimizer, max_grad_norm=None,
                 ignore_subtask=True, model_type="LanguageModel",
                 decoder_config=None, arena: Optional[ParameterArena] = None):
        self.model = model
        self.train_dataloader = train_dataloader
        self.optimizer = optimizer
        self.max_grad_norm = max_grad_norm
        self.model_type = model_type
        self.decoder_config = decoder_config
        self.arena = arena
        self.step = 0
        self.ignore_subtask = ignore_subtask

    def epoch_boundary(self):
        return self.step % len(self.train_dataloader.create_loader()) == 0 and self.step != 0

    # -- gradient plumbing -------------------------------------------------
    def zero_grad(self):
        if self.arena is not None:
            self.arena.zero_grad()
        else:
            for p in self.model.parameters():
                if p.grad is not None:
                    p.grad.detach_()
                    p.grad.zero_()

    def clip_gradients(self):
        if self.max_grad_norm is None:
            return None
        if self.arena is not None:
            return ops.clip_by_norm(self.arena.grad, float(self.max_grad_norm))
        return torch.nn.utils.clip_grad_norm_(self.model.parameters(),
                                              self.max_grad_norm)

    def train_desired_samples(self, desired_max_samples, apply_privacy_metrics=False):
        pass

    def save(self):
        pass

    def load(self):
        pass


class ModelUpdater(TrainerBase):
    """Server-side trainer without data: applies the aggregated
    pseudo-gradient (reference: core/trainer.py:82-197)."""

    def __init__(self, model, optimizer, ss_scheduler, train_dataloader,
                 val_dataloader, max_grad_norm, anneal_config,
                 model_type="LanguageModel", decoder_config=None,
                 arena: Optional[ParameterArena] = None, val_fn=None):
        super().__init__(model=model, train_dataloader=train_dataloader,
                         optimizer=optimizer, max_grad_norm=max_grad_norm,
                         model_type=model_type, decoder_config=decoder_config,
                         arena=arena)
        self.val_dataloader = val_dataloader
        self.annealing_type = anneal_config["type"] if anneal_config is not None else None
        self.lr_scheduler = make_lr_scheduler(anneal_config, self.optimizer) \
            if anneal_config is not None else None
        self.ss_scheduler = ss_scheduler
        # callback returning (val_loss, val_acc); wired to the distributed
        # evaluation by the server (replaces reference's local val loader).
        self.val_fn = val_fn

    def update_model(self):
        """Clip aggregated gradient then step the server optimizer
        (reference: core/trainer.py:127-137)."""
        self.clip_gradients()
        self.optimizer.step()
        self.optimizer.zero_grad() if self.arena is None else self.arena.zero_grad()

    def run_lr_scheduler(self, force_run_val=False):
        val_loss = val_acc = None
        if (force_run_val or self.annealing_type == "val_loss") and self.val_fn is not None:
            val_loss, val_acc = self.val_fn()
        if self.lr_scheduler is not None:
            if self.annealing_type == "val_loss":
                self.lr_scheduler.step(val_loss)
            else:
                self.lr_scheduler.step()
        return (val_loss, val_acc)

    def run_ss_scheduler(self):
        if self.ss_scheduler is not None:
            self.ss_scheduler.step()

    def save(self, model_path, token=None, config=None):
        save_model(model_path=model_path, config=config, model=self.model,
                   optimizer=self.optimizer, lr_scheduler=self.lr_scheduler,
                   ss_scheduler=self.ss_scheduler, token=token)

    def load(self, save_path, update_lr_scheduler, update_ss_scheduler):
        _load_checkpoint(self, save_path, update_lr_scheduler, update_ss_scheduler)


class Trainer(TrainerBase):
    """Client-side local-SGD engine (reference: core/trainer.py:200-687)."""

    def __init__(self, model, ss_scheduler, train_dataloader,
                 server_replay_config=None, optimizer=None, max_grad_norm=None,
                 anneal_config=None, num_skips_threshold=-1,
                 ignore_subtask=True, arena: Optional[ParameterArena] = None):
        super().__init__(model=model, train_dataloader=train_dataloader,
                         optimizer=optimizer, max_grad_norm=max_grad_norm,
                         ignore_subtask=ignore_subtask, arena=arena)
        # optional hipGraph / fused-kernel fast paths, injected by
        # ClientExecutor
        self.graph_cache = None
        self.fused_cnn = None
        self.round_seed = 0
        self.server_replay_config = server_replay_config
        self.anneal_config = anneal_config
        self.lr_scheduler = None
        if self.optimizer is None and server_replay_config is not None \
                and "optimizer_config" in server_replay_config:
            self.optimizer = make_optimizer(server_replay_config["optimizer_config"], model)
        if self.optimizer is not None and self.anneal_config is not None:
            self.lr_scheduler = make_lr_scheduler(self.anneal_config, self.optimizer)
        self.cached_batches = []
        self.ss_scheduler = ss_scheduler
        # lazy-stats mode (set by ClientExecutor): keep per-epoch loss and
        # Σg/Σg² on device; the server syncs all clients' stats at once
        self.lazy_stats = False
        self.loss_dev = None
        self.stats_dev = None
        self.reset_gradient_power()

    # -- gradient sufficient statistics (K5) ------------------------------
    def _device(self):
        if self.arena is not None:
            return self.arena.device
        p = next(self.model.parameters(), None)
        return p.device if p is not None else torch.device("cpu")

    def reset_gradient_power(self):
        dev = self._device()
        self._stats_acc = torch.zeros(2, dtype=torch.float32, device=dev)
        self.counter = 0
        self.sum_grad = 0.0
        self.sum_grad2 = 0.0
        self.sufficient_stats = {}

    def accumulate_gradient_power(self):
        """Accumulate Σg, Σg² for the current gradient — one fused pass on
        the arena, no host transfer (reference: trainer.py:271-292)."""
        if self.arena is not None:
            self._stats_acc += ops.sum_sumsq(self.arena.grad)
            self.counter += self.arena.total
        else:
            for p in self.model.parameters():
                if p.grad is None:
                    continue
                g = p.grad.detach().reshape(-1)
                self._stats_acc[0] += g.sum()
                self._stats_acc[1] += g.dot(g)
                self.counter += g.numel()

    def estimate_sufficient_stats(self):
        """Accumulate current grad stats + finalize (reference API,
        trainer.py:294-312: called per batch there; here the per-batch flow
        calls accumulate_gradient_power and the epoch end finalizes once)."""
        self.accumulate_gradient_power()
        return self._finalize_sufficient_stats()

    def finalize_stats_from(self, s, q):
        """Fill the stats dict from already-synced host scalars (lazy path:
        the server syncs every client's accumulator in ONE transfer)."""
        self.sum_grad, self.sum_grad2 = float(s), float(q)
        n = max(self.counter, 1)
        mean_grad = self.sum_grad / n
        mag_grad = math.sqrt(max(self.sum_grad2 / n, 0.0))
        var_grad = max(self.sum_grad2 / n - mean_grad ** 2, 0.0)
        self.sufficient_stats = {
            "n": n, "sum": self.sum_grad, "sq_sum": self.sum_grad2,
            "var": var_grad, "mean": mean_grad, "mag": mag_grad,
            "norm": math.sqrt(max(self.sum_grad2, 0.0)),
        }
        return self.sufficient_stats

    def _finalize_sufficient_stats(self):
        """One host sync: turn the device accumulator into the stats dict."""
        acc = self._stats_acc.tolist()
        self.sum_grad, self.sum_grad2 = acc[0], acc[1]
        n = max(self.counter, 1)
        mean_grad = self.sum_grad / n
        mag_grad = math.sqrt(max(self.sum_grad2 / n, 0.0))
        var_grad = max(self.sum_grad2 / n - mean_grad ** 2, 0.0)
        norm_grad = math.sqrt(max(self.sum_grad2, 0.0))
        self.sufficient_stats = {
            "n": n, "sum": self.sum_grad, "sq_sum": self.sum_grad2,
            "var": var_grad, "mean": mean_grad, "mag": mag_grad,
            "norm": norm_grad,
        }
        return self.sufficient_stats

    # -- training loops ----------------------------------------------------
    def train_desired_samples(self, desired_max_samples=None,
                              apply_privacy_metrics=False, algo_payload=None):
        """One local-training pass; returns (train_loss, num_samples,
        algo_computation) (reference: trainer.py:314-339)."""
        algo_computation = None
        if algo_payload is None:
            num_samples, train_loss = self.run_train_epoch(
                desired_max_samples, apply_privacy_metrics)
        elif algo_payload["strategy"] == "FedProx":
            num_samples, train_loss = self.run_train_epoch_fedprox(
                desired_max_samples, apply_privacy_metrics, algo_payload)
        elif algo_payload["strategy"] == "FedLabels":
            num_samples, train_loss, algo_computation = self.run_train_epoch_sup(
                desired_max_samples, apply_privacy_metrics, algo_payload)
        else:
            raise ValueError(f"unknown algo payload {algo_payload['strategy']}")
        return train_loss, num_samples, algo_computation

    def _batch_loss(self, batch, apply_privacy_metrics):
        if self.ignore_subtask and hasattr(self.model, "single_task_loss"):
            return self.model.single_task_loss(batch)
        if apply_privacy_metrics:
            if "x" in batch:
                self.cached_batches.append(to_device(batch["x"]))
            elif "input_ids" in batch:
                self.cached_batches.append(to_device(batch["input_ids"]))
        return self.model.loss(batch)

    @staticmethod
    def _batch_samples(batch):
        if "attention_mask" in batch:
            return int(torch.sum(batch["attention_mask"].detach().cpu() == 1).item())
        if "total_frames" in batch:
            return int(batch["total_frames"])
        x = batch["x"]
        if isinstance(x, (tuple, list)):  # multi-input tasks (fednewsrec)
            x = x[0]
        return len(x)

    def _train_step(self, loss):
        loss.backward()
        self.clip_gradients()
        self.accumulate_gradient_power()
        if self.optimizer is not None:
            self.optimizer.step()

    def run_train_epoch(self, desired_max_samples=None,
                        apply_privacy_metrics=False, prox=None):
        """Reference: trainer.py:341-414 (and 416-501 when ``prox`` is set:
        prox = (mu, w_global_flat))."""
        fast_ok = prox is None and not apply_privacy_metrics
        if fast_ok and self.fused_cnn is not None:
            ds = getattr(self.train_dataloader, "dataset", None)
            if (ds is not None and torch.is_tensor(getattr(ds, "x", None))
                    and ds.x.is_cuda and torch.is_tensor(getattr(ds, "y", None))
                    and ds.y.dim() == 1 and ds.x[0].numel() == 784
                    and getattr(self.train_dataloader, "shuffle", False)
                    and (desired_max_samples is None
                         or desired_max_samples >= len(ds.x))):
                return self._run_epoch_fused_cnn(ds)
        if (self.graph_cache is not None and fast_ok
                and self.graph_cache.supports()):
            return self._run_train_epoch_graphed(desired_max_samples)
        num_samples = 0
        self.reset_gradient_power()
        self.zero_grad()
        # loss accumulates in a device scalar: ONE host sync per epoch
        # instead of the reference's per-batch loss.item()
        dev = self._device()
        loss_acc = torch.zeros((), device=dev)

        train_loader = self.train_dataloader.create_loader()
        for batch in train_loader:
            if desired_max_samples is not None and num_samples >= desired_max_samples:
                break
            # NOTE: not optimizer.zero_grad() — torch's set_to_none default
            # would unbind the arena grad views.
            self.zero_grad()
            loss = self._batch_loss(batch, apply_privacy_metrics)
            if prox is not None:
                mu, w_global = prox
                loss.backward()
                if self.arena is not None:
                    # exact prox gradient: g += mu*(w - w_global)
                    ops.axpy(self.arena.grad, self.arena.data, mu)
                    ops.axpy(self.arena.grad, w_global, -mu)
                    loss_acc += loss.detach() + 0.5 * mu * \
                        (self.arena.data - w_global).square().sum()
                else:
                    off = 0
                    reg = torch.zeros((), device=dev)
                    for p in self.model.parameters():
                        wg = w_global[off:off + p.numel()].view(p.shape)
                        p.grad.add_(p.data - wg, alpha=mu)
                        reg += 0.5 * mu * (p.data - wg).square().sum()
                        off += p.numel()
                    loss_acc += loss.detach() + reg
                self.clip_gradients()
                self.accumulate_gradient_power()
                if self.optimizer is not None:
                    self.optimizer.step()
            else:
                loss_acc += loss.detach()
                self._train_step(loss)
            num_samples += self._batch_samples(batch)
            self.step += 1

        if self.lr_scheduler is not None:
            self.lr_scheduler.step()
        if self.lazy_stats and self.arena is not None \
                and self.arena.device.type == "cuda":
            self.loss_dev = loss_acc
            self.stats_dev = self._stats_acc.clone()
            return num_samples, None
        self._finalize_sufficient_stats()
        return num_samples, float(loss_acc)

    def _run_train_epoch_graphed(self, desired_max_samples=None):
        """hipGraph fast path: each full-shape batch is one graph replay
        (capture: zero-grad → fwd → bwd → fused clip+stats → fused SGD).
        Ragged tail batches run the identical ops eagerly against the same
        graph-owned accumulators/momentum so semantics match the eager
        epoch exactly (ops/graphs.py)."""
        cache = self.graph_cache
        num_samples = 0
        self.reset_gradient_power()
        g = None
        n_graph_batches = 0

        cache.set_lr(get_lr(self.optimizer) if self.optimizer is not None
                     else float(cache.lr_t[0]))

        # whole-epoch fast path: device-resident uniform shard ⇒ the entire
        # local epoch is ONE graph replay (ops/graphs.py GraphedClientEpoch).
        # Attempted FIRST — it needs no probe batch, and the probe costs
        # real host time on fast tasks (an extra loader iteration per
        # client on a ~5 ms round)
        ds = getattr(self.train_dataloader, "dataset", None)
        bs = getattr(self.train_dataloader, "batch_size", 0)
        if (ds is not None and bs
                and torch.is_tensor(getattr(ds, "x", None)) and ds.x.is_cuda
                and torch.is_tensor(getattr(ds, "y", None))
                and getattr(self.train_dataloader, "shuffle", False)
                and (desired_max_samples is None
                     or desired_max_samples >= len(ds.x))):
            from ..ops.graphs import epoch_graph_for
            eg = epoch_graph_for(cache, ds.x, ds.y, bs)
            if eg is not None:
                return self._run_epoch_one_graph(eg, ds, bs, cache)

        # probe one batch: per-batch graphs need dict batches of plain
        # (x, y) tensors (nlg/mlm/newsrec batch shapes run the generic
        # eager path).  The probe must NOT consume the per-client-seeded
        # host RNG (a shuffled draw would shift the epoch's randperm,
        # diverging this path from the eager/fused/mega paths' first-draw
        # order): peek with shuffle off when the loader supports it, else
        # save/restore the RNG state
        dl = self.train_dataloader
        if hasattr(dl, "shuffle"):
            sh = dl.shuffle
            dl.shuffle = False
            probe = next(iter(dl.create_loader()), None)
            dl.shuffle = sh
        else:
            rng_state = torch.get_rng_state()
            probe = next(iter(dl.create_loader()), None)
            torch.set_rng_state(rng_state)
        if not (isinstance(probe, dict) and torch.is_tensor(probe.get("x"))
                and torch.is_tensor(probe.get("y"))):
            self.graph_cache = None
            return self.run_train_epoch(desired_max_samples)

        train_loader = self.train_dataloader.create_loader()
        for batch in train_loader:
            if desired_max_samples is not None and num_samples >= desired_max_samples:
                break
            x, y = batch["x"], batch["y"]
            if g is None:
                g = cache.get(x, y)
                if g is None:  # capture failed — this model runs eager
                    self.graph_cache = None
                    return self.run_train_epoch(desired_max_samples)
                g.reset_client()
            if tuple(x.shape) == tuple(g.static_x.shape):
                g.run_batch(x, y)
            else:
                # ragged tail: same ops, eager, same accumulators
                self.arena.grad.zero_()
                loss = self.model.loss({"x": x, "y": y})
                loss.backward()
                ops.clip_stats_accumulate(
                    self.arena.grad,
                    float(self.max_grad_norm) if self.max_grad_norm else -1.0,
                    g.stats_acc)
                ops.sgd_step_devlr(self.arena.data, self.arena.grad,
                                   g.momentum_buf, cache.lr_t,
                                   momentum=cache.momentum, dampening=0.0,
                                   weight_decay=cache.weight_decay,
                                   nesterov=cache.nesterov, first_step=False)
                g.loss_acc += loss.detach()
            n_graph_batches += 1
            num_samples += self._batch_samples(batch)
            self.step += 1

        if self.lr_scheduler is not None:
            self.lr_scheduler.step()

        if g is not None:
            self._stats_acc += g.stats_acc
            self.counter += n_graph_batches * self.arena.total
            if self.lazy_stats:
                # defer the host sync: the server batches every local
                # client's (loss, Σg, Σg²) into ONE transfer per round
                self.loss_dev = g.loss_acc.clone()
                self.stats_dev = self._stats_acc.clone()
                return num_samples, None
            loss_total = float(g.loss_acc)
        else:
            loss_total = 0.0
        self._finalize_sufficient_stats()
        return num_samples, loss_total

    def _run_epoch_fused_cnn(self, ds):
        """One local epoch via the hand-written fused CNN kernels
        (ops/fused_cnn.py) — no autograd, no graphs."""
        self.reset_gradient_power()
        fc = self.fused_cnn
        fc.lr_t  # noqa: B018 — touch to assert constructed
        lr = get_lr(self.optimizer) if self.optimizer is not None else 0.0
        order = torch.randperm(len(ds.x))
        n, n_batches = fc.run_epoch(ds.x, ds.y, order, lr, self.round_seed)
        self.step += n_batches
        if self.lr_scheduler is not None:
            self.lr_scheduler.step()
        self._stats_acc += fc.stats_acc
        self.counter += n_batches * self.arena.total
        if self.lazy_stats:
            self.loss_dev = fc.loss_acc.reshape(()).clone()
            self.stats_dev = self._stats_acc.clone()
            return n, None
        loss_total = float(fc.loss_acc)
        self._finalize_sufficient_stats()
        return n, loss_total

    def _run_epoch_one_graph(self, eg, ds, bs, cache):
        """Run one client epoch as a single graph replay + eager ragged
        tail.  Shuffle order comes from the host torch RNG (per-client seed
        discipline preserved)."""
        n = len(ds.x)
        order = torch.randperm(n)
        eg.reset_client()
        eg.run_epoch(ds.x, ds.y, order)
        n_batches = eg.n_batches
        tail = order[n_batches * bs:]
        if len(tail):
            idx = tail.to(ds.x.device, non_blocking=True)
            x_t = ds.x.index_select(0, idx)
            y_t = ds.y.index_select(0, idx)
            # the ragged tail used to run EAGER (autograd) — ~3 ms of
            # host-blocking python per client that serialized the stream
            # pool.  A per-batch graph for the tail shape (captured once,
            # shared by every client with the same shard geometry) replays
            # in ~0.1 ms.  Momentum needs the epoch graph's buffer, so
            # momentum>0 keeps the eager path.
            g = cache.get(x_t, y_t) if cache.momentum == 0.0 else None
            if g is not None:
                g.reset_client()
                g.run_batch(x_t, y_t)
                eg.stats_acc += g.stats_acc
                eg.loss_acc += g.loss_acc
            else:
                self.arena.grad.zero_()
                loss = self.model.loss({"x": x_t, "y": y_t})
                loss.backward()
                ops.clip_stats_accumulate(
                    self.arena.grad,
                    float(self.max_grad_norm) if self.max_grad_norm else -1.0,
                    eg.stats_acc)
                ops.sgd_step_devlr(self.arena.data, self.arena.grad,
                                   eg.momentum_buf, cache.lr_t,
                                   momentum=cache.momentum, dampening=0.0,
                                   weight_decay=cache.weight_decay,
                                   nesterov=cache.nesterov, first_step=False)
                eg.loss_acc += loss.detach()
            n_batches += 1
        self.step += n_batches
        if self.lr_scheduler is not None:
            self.lr_scheduler.step()
        self._stats_acc += eg.stats_acc
        self.counter += n_batches * self.arena.total
        if self.lazy_stats:
            self.loss_dev = eg.loss_acc.clone()
            self.stats_dev = self._stats_acc.clone()
            return n, None
        loss_total = float(eg.loss_acc)
        self._finalize_sufficient_stats()
        return n, loss_total

    def run_train_epoch_fedprox(self, desired_max_samples=None,
                                apply_privacy_metrics=False, algo_payload=None):
        """FedProx local training (reference: trainer.py:416-501; the
        proximal term is applied exactly once per batch — see module
        docstring)."""
        mu = algo_payload["mu"]
        if self.arena is not None:
            w_global = self.arena.clone_data()
        else:
            w_global = torch.cat([p.detach().reshape(-1).clone()
                                  for p in self.model.parameters()])
        return self.run_train_epoch(desired_max_samples, apply_privacy_metrics,
                                    prox=(mu, w_global))

    def run_train_epoch_sup(self, desired_max_samples=None,
                            apply_privacy_metrics=False, algo_payload=None):
        """FedLabels semi-supervised local training
        (reference: trainer.py:503-619)."""
        from ..extensions.fedlabels_train import run_train_epoch_sup as _sup
        return _sup(self, desired_max_samples, apply_privacy_metrics, algo_payload)

    # -- misc --------------------------------------------------------------
    def get_model(self):
        return copy.deepcopy(self.model)

    def prepare_iteration(self, model=None):
        """Reference: trainer.py:624-638."""
        if model is not None:
            self.model.load_state_dict(model.state_dict())
            self.lr_scheduler = None
            if self.optimizer is None and self.server_replay_config is not None \
                    and "optimizer_config" in self.server_replay_config:
                self.optimizer = make_optimizer(
                    self.server_replay_config["optimizer_config"], self.model)
            if self.optimizer is not None and self.anneal_config is not None:
                self.lr_scheduler = make_lr_scheduler(self.anneal_config, self.optimizer)

    def reset_optimizer(self, optimizer_state_dict, annealing_config=None):
        assert self.optimizer is not None, "This trainer does not have an optimizer"
        self.optimizer.load_state_dict(optimizer_state_dict)
        self.lr_scheduler = None
        if annealing_config is not None:
            self.lr_scheduler = make_lr_scheduler(annealing_config, self.optimizer)

    def save(self, model_path, token=None, config=None):
        save_model(model_path=model_path, config=config, model=self.model,
                   optimizer=self.optimizer, lr_scheduler=self.lr_scheduler,
                   ss_scheduler=self.ss_scheduler, token=token)

    def load(self, save_path, update_lr_scheduler, update_ss_scheduler):
        _load_checkpoint(self, save_path, update_lr_scheduler, update_ss_scheduler)


def run_validation_generic(model, val_dataloader):
    """Run the generic metrics loop (reference: trainer.py:690-723)."""
    model.set_eval()
    val_loader = val_dataloader.create_loader()
    return Metrics().compute_metrics(dataloader=val_loader, model=model)


def set_component_wise_lr(model, optimizer_config, updatable_names):
    """Freeze layers by zero LR (reference: trainer.py:725-751)."""
    import re

    def matched(name):
        return any(re.match(u, name) is not None for u in updatable_names)

    parameters = []
    for name, params in model.named_parameters():
        if matched(name):
            parameters.append({"params": params, "lr": optimizer_config["lr"]})
        else:
            parameters.append({"params": params, "lr": 0.0})
    return parameters


_SAVE_POOL = None


def _save_pool():
    global _SAVE_POOL
    if _SAVE_POOL is None:
        from concurrent.futures import ThreadPoolExecutor
        _SAVE_POOL = ThreadPoolExecutor(max_workers=1,
                                        thread_name_prefix="ckpt")
    return _SAVE_POOL


def flush_saves():
    """Block until all queued checkpoint writes hit disk."""
    if _SAVE_POOL is not None:
        _SAVE_POOL.shutdown(wait=True)
        globals()["_SAVE_POOL"] = None


def _clone_state(d):
    if isinstance(d, dict):
        return {k: _clone_state(v) for k, v in d.items()}
    if isinstance(d, (list, tuple)):
        return type(d)(_clone_state(v) for v in d)
    if torch.is_tensor(d):
        return d.detach().to("cpu", copy=True)
    return d


def save_model(model_path, config, model, optimizer, lr_scheduler,
               ss_scheduler, token=None, async_ok=True):
    """Checkpoint in the reference's exact .tar dict layout
    (reference: trainer.py:753-775; format parity required by BASELINE).

    The state is snapshotted to host memory synchronously; the disk write
    runs on a background thread (the reference blocks the round loop on
    torch.save every round — ~40% of its housekeeping cost here).  Call
    ``flush_saves()`` before reading checkpoints back.
    """
    save_state = {
        "model_state_dict": _clone_state(model.state_dict()),
        "optimizer_state_dict": _clone_state(optimizer.state_dict())
        if optimizer is not None else None,
        "lr_scheduler_state_dict": lr_scheduler.state_dict()
        if lr_scheduler is not None else None,
    }
    if ss_scheduler is not None:
        save_state["ss_scheduler_state_dict"] = ss_scheduler.state_dict()
    name = f"{token}_model.tar" if token else "model.tar"
    save_path = os.path.join(model_path, name)

    def _write():
        for attempt in range(3):  # retry (reference: utils/utils.py:348-359)
            try:
                torch.save(save_state, save_path)
                break
            except Exception as e:
                print_rank(f"save attempt {attempt} failed: {e}",
                           loglevel=logging.WARNING)
    if async_ok:
        _save_pool().submit(_write)
    else:
        _write()
    if config is not None:
        cfg = config.to_dict() if hasattr(config, "to_dict") else dict(config)
        with open(os.path.join(model_path, "config.yaml"), "w") as f:
            yaml.safe_dump(cfg, f)


def _load_checkpoint(trainer, save_path, update_lr_scheduler, update_ss_scheduler):
    if not os.path.isfile(save_path):
        return
    print_rank(f"Loading checkpoint: {save_path}")
    checkpoint = torch.load(save_path, map_location="cpu", weights_only=False)
    trainer.model.load_state_dict(checkpoint["model_state_dict"])
    if trainer.arena is not None:
        # state_dict load wrote through the arena views; nothing to rebind,
        # but make sure grads are still views after any torn state.
        trainer.arena.zero_grad()
    if trainer.optimizer is not None and checkpoint.get("optimizer_state_dict"):
        trainer.optimizer.load_state_dict(checkpoint["optimizer_state_dict"])
    anl = checkpoint.get("lr_scheduler_state_dict")
    if anl and trainer.lr_scheduler is not None and update_lr_scheduler:
        trainer.lr_scheduler.load_state_dict(anl)
    sss = checkpoint.get("ss_scheduler_state_dict")
    if sss and trainer.ss_scheduler is not None and update_ss_scheduler:
        trainer.ss_scheduler.load_state_dict(sss)
