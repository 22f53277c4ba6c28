"""hipGraph capture of the per-batch client training step.

The FL client workload is thousands of tiny fixed-shape batches: eager
execution is launch-bound (measured ~4% GPU busy on CNN-FEMNIST — see
profiles/).  This captures the whole per-batch step

    zero grad arena → forward → backward → fused clip+stats → fused SGD

as ONE hip graph replayed per batch; the host copies the batch into static
buffers and replays.  The LR is read from a device scalar so a single
capture serves every round even as the server retunes the client LR.

Constraints (checked by ``supports``): CUDA + arena-bound model, plain SGD
client optimizer with dampening 0 (covers every reference benchmark task:
all use client SGD — BASELINE.md), fixed batch shape (ragged tail batches
fall back to the eager path).

Note on RNG: dropout inside the graph draws from the torch CUDA
generator's graph-captured Philox stream, which advances per replay — so
the GPU fast path does not reproduce the CPU path's per-client dropout
bit-pattern (world-size invariance of numerics remains exact on the CPU
path, which is the one under test).
"""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch

from .. import ops
from .arena import ParameterArena


class GraphedClientStep:
    """One captured (model, arena, batch-shape) training step."""

    def __init__(self, model, arena: ParameterArena, lr_t: torch.Tensor,
                 max_grad_norm: Optional[float], momentum: float,
                 weight_decay: float, nesterov: bool,
                 x_shape: Tuple[int, ...], y_shape: Tuple[int, ...],
                 y_dtype: torch.dtype, x_dtype: torch.dtype = torch.float32):
        self.model = model
        self.arena = arena
        self.lr_t = lr_t
        self.momentum = momentum
        dev = arena.device
        self.static_x = torch.zeros(x_shape, dtype=x_dtype, device=dev)
        self.static_y = torch.zeros(y_shape, dtype=y_dtype, device=dev)
        self.stats_acc = torch.zeros(2, device=dev)
        self.loss_acc = torch.zeros((), device=dev)
        self.momentum_buf = arena.new_buffer() if momentum != 0.0 else None
        mn = float(max_grad_norm) if max_grad_norm is not None else -1.0

        def step_body():
            self.arena.grad.zero_()
            loss = self.model.loss({"x": self.static_x, "y": self.static_y})
            loss.backward()
            ops.clip_stats_accumulate(self.arena.grad, mn, self.stats_acc)
            ops.sgd_step_devlr(self.arena.data, self.arena.grad,
                               self.momentum_buf, self.lr_t,
                               momentum=momentum, dampening=0.0,
                               weight_decay=weight_decay, nesterov=nesterov,
                               first_step=False)
            self.loss_acc += loss.detach()

        # warmup on a side stream (required before capture), then capture.
        # warmup EXECUTES the step and would corrupt the live client
        # weights — snapshot and restore around it.
        saved = arena.data.clone()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):
                step_body()
        torch.cuda.current_stream().wait_stream(s)

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            step_body()

        arena.data.copy_(saved)
        arena.grad.zero_()
        self.reset_client()
        torch.cuda.synchronize()

    def reset_client(self):
        """Per-client optimizer/stat reset (fresh-optimizer semantics:
        dampening==0 makes a zeroed momentum buffer equal to first_step)."""
        self.stats_acc.zero_()
        self.loss_acc.zero_()
        if self.momentum_buf is not None:
            self.momentum_buf.zero_()

    def run_batch(self, x: torch.Tensor, y: torch.Tensor):
        self.static_x.copy_(x, non_blocking=True)
        self.static_y.copy_(y, non_blocking=True)
        self.graph.replay()


class GraphCache:
    """Per-executor cache of captured steps keyed by batch shape."""

    def __init__(self, model, arena: ParameterArena, optimizer_config: dict,
                 max_grad_norm: Optional[float]):
        self.model = model
        self.arena = arena
        self.max_grad_norm = max_grad_norm
        cfg = dict(optimizer_config)
        self.opt_type = cfg.get("type", "sgd")
        self.momentum = float(cfg.get("momentum", 0.0))
        self.dampening = float(cfg.get("dampening", 0.0))
        self.weight_decay = float(cfg.get("weight_decay", 0.0))
        self.nesterov = bool(cfg.get("nesterov", False))
        self.lr_t = torch.zeros(1, device=arena.device)
        self._graphs: Dict[Tuple, GraphedClientStep] = {}

    def supports(self) -> bool:
        return (torch.cuda.is_available() and self.arena.device.type == "cuda"
                and ops.HAS_EXT and self.opt_type == "sgd"
                and self.dampening == 0.0)

    def set_lr(self, lr: float):
        self.lr_t.fill_(float(lr))

    def get(self, x: torch.Tensor, y: torch.Tensor):
        key = (tuple(x.shape), x.dtype, tuple(y.shape), y.dtype)
        if key in self._graphs:
            return self._graphs[key]
        try:
            g = GraphedClientStep(self.model, self.arena, self.lr_t,
                                  self.max_grad_norm, self.momentum,
                                  self.weight_decay, self.nesterov,
                                  tuple(x.shape), tuple(y.shape), y.dtype,
                                  x_dtype=x.dtype)
        except Exception as e:  # capture-unsafe model op — run eager
            import logging
            logging.getLogger().warning(
                f"hipGraph capture failed ({e}); falling back to eager")
            g = None
        self._graphs[key] = g
        return g


class GraphedClientEpoch:
    """Whole-epoch capture: every full-size batch of a client's local epoch
    replays as ONE hipGraph.

    The client's device-resident shard is copied (D2D) into a static
    buffer, a fresh shuffle order is staged through a pinned host buffer,
    and the graph gathers each batch with ``index_select`` before the
    captured fwd/bwd/clip+stats/SGD sequence — so per client the host does
    two tiny copies and one graph launch, regardless of batch count.
    Keyed by (n_samples, n_batches, batch shape); the ragged tail (n %
    batch_size) stays on the per-batch path.
    """

    def __init__(self, model, arena: ParameterArena, lr_t: torch.Tensor,
                 max_grad_norm, momentum: float, weight_decay: float,
                 nesterov: bool, shard_x_shape, y_dtype, n: int, bs: int):
        self.model = model
        self.arena = arena
        self.lr_t = lr_t
        self.momentum = momentum
        self.n = n
        self.bs = bs
        self.n_batches = n // bs
        dev = arena.device
        x_shape, x_dtype, y_shape = shard_x_shape
        self.static_x = torch.zeros((n, *x_shape), dtype=x_dtype, device=dev)
        self.static_y = torch.zeros((n, *y_shape), dtype=y_dtype, device=dev)
        self.static_idx = torch.zeros((n,), dtype=torch.int64, device=dev)
        self.host_idx = torch.zeros((n,), dtype=torch.int64).pin_memory()
        self._idx_ev = None
        self.stats_acc = torch.zeros(2, device=dev)
        self.loss_acc = torch.zeros((), device=dev)
        self.momentum_buf = arena.new_buffer() if momentum != 0.0 else None
        mn = float(max_grad_norm) if max_grad_norm is not None else -1.0

        def epoch_body():
            for b in range(self.n_batches):
                idx = self.static_idx[b * bs:(b + 1) * bs]
                x = self.static_x.index_select(0, idx)
                y = self.static_y.index_select(0, idx)
                self.arena.grad.zero_()
                loss = self.model.loss({"x": x, "y": y})
                loss.backward()
                ops.clip_stats_accumulate(self.arena.grad, mn, self.stats_acc)
                ops.sgd_step_devlr(self.arena.data, self.arena.grad,
                                   self.momentum_buf, self.lr_t,
                                   momentum=momentum, dampening=0.0,
                                   weight_decay=weight_decay,
                                   nesterov=nesterov, first_step=False)
                self.loss_acc += loss.detach()

        saved = arena.data.clone()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                epoch_body()
        torch.cuda.current_stream().wait_stream(s)

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            epoch_body()

        arena.data.copy_(saved)
        arena.grad.zero_()
        self.reset_client()
        torch.cuda.synchronize()

    def reset_client(self):
        self.stats_acc.zero_()
        self.loss_acc.zero_()
        if self.momentum_buf is not None:
            self.momentum_buf.zero_()

    def run_epoch(self, x_shard, y_shard, order_cpu):
        """order_cpu: CPU int64 tensor of length n (fresh shuffle)."""
        self.static_x.copy_(x_shard, non_blocking=True)
        self.static_y.copy_(y_shard, non_blocking=True)
        if self._idx_ev is not None:
            # the previous client's non_blocking H2D of host_idx may still
            # be queued (no host sync between clients in lazy-stats mode)
            self._idx_ev.synchronize()
        self.host_idx.copy_(order_cpu)
        self.static_idx.copy_(self.host_idx, non_blocking=True)
        self._idx_ev = torch.cuda.Event()
        self._idx_ev.record()
        self.graph.replay()


def epoch_graph_for(cache: "GraphCache", x_shard, y_shard, bs: int):
    """Fetch/capture the whole-epoch graph for this shard geometry."""
    n = x_shard.shape[0]
    if n < bs:  # single ragged batch — per-batch path handles it
        return None
    key = ("epoch", n, bs, tuple(x_shard.shape[1:]), x_shard.dtype,
           tuple(y_shard.shape[1:]), y_shard.dtype)
    if key in cache._graphs:  # a cached None remembers a failed capture
        return cache._graphs[key]
    try:
        g = GraphedClientEpoch(
            cache.model, cache.arena, cache.lr_t, cache.max_grad_norm,
            cache.momentum, cache.weight_decay, cache.nesterov,
            (tuple(x_shard.shape[1:]), x_shard.dtype,
             tuple(y_shard.shape[1:])), y_shard.dtype, n, bs)
    except Exception as e:  # capture-unsafe model op — run eager
        import logging
        logging.getLogger().warning(
            f"epoch-graph capture failed ({e}); falling back to eager")
        g = None
    cache._graphs[key] = g
    return g
