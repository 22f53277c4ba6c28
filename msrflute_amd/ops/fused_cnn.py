"""Fully-fused CNN-FEMNIST client epoch (csrc/fused_cnn.hip).

One host call trains a client's whole local epoch — per batch: gather,
conv1+relu, conv2+relu (LDS-tiled), pool+dropout, fc1(+relu+dropout),
fc2+softmax+CE, full backward into the flat grad arena, fused clip+stats,
SGD — with ~17 raw kernel launches per batch instead of a ~230-node
autograd graph (hipGraphLaunch costs ~4.6us/node on this stack; PERF.md).

Numerics match the eager path batch-for-batch given the same shuffle
order (dropout masks come from our Philox stream keyed by (seed, batch),
not torch's — same distribution, different bits).
"""

from __future__ import annotations

from typing import Optional

import torch

from . import HAS_EXT, _C
from .arena import ParameterArena

# expected parameter layout of the flagship CNN
# (experiments/cv_cnn_femnist/model.py)
_SHAPES = [
    ("net.conv1.weight", (32, 1, 3, 3)),
    ("net.conv1.bias", (32,)),
    ("net.conv2.weight", (64, 32, 3, 3)),
    ("net.conv2.bias", (64,)),
    ("net.fc1.weight", (128, 9216)),
    ("net.fc1.bias", (128,)),
    ("net.fc2.weight", None),  # (C, 128)
    ("net.fc2.bias", None),    # (C,)
]


def matches_cnn_femnist(arena: ParameterArena) -> Optional[int]:
    """Return num_classes C if the arena holds exactly the flagship CNN's
    parameter layout, else None."""
    if len(arena.names) != len(_SHAPES):
        return None
    for (name, shape), got_name, got_shape in zip(_SHAPES, arena.names,
                                                  arena.shapes):
        if got_name != name:
            return None
        if shape is not None and tuple(got_shape) != shape:
            return None
    c_shape = tuple(arena.shapes[6])
    if len(c_shape) != 2 or c_shape[1] != 128:
        return None
    if tuple(arena.shapes[7]) != (c_shape[0],):
        return None
    return c_shape[0]


class FusedCNNEpoch:
    """Per-executor reusable workspace + driver for the fused epoch."""

    def __init__(self, arena: ParameterArena, num_classes: int, bs: int,
                 p1: float = 0.25, p2: float = 0.5,
                 max_grad_norm: Optional[float] = None,
                 use_bf16: bool = False):
        assert HAS_EXT and arena.device.type == "cuda"
        self.arena = arena
        self.C = int(num_classes)
        self.bs = int(bs)
        self.p1, self.p2 = float(p1), float(p2)
        self.max_norm = float(max_grad_norm) if max_grad_norm else -1.0
        # mixed precision: bf16 MFMA GEMMs (conv2 fwd/bwd, fc1 fwd) with
        # fp32 accumulators/master weights; everything else stays fp32
        self.use_bf16 = bool(use_bf16)
        dev = arena.device
        B = self.bs
        n_float = B * (784 + 21632 + 36864 + 9216 + 128 + 128 + self.C
                       + 128 + 9216 + 36864 + 21632
                       + 18432) + 2 * 18432  # + split-K slab + W2 layouts
        self.work_f = torch.empty(n_float, dtype=torch.float32, device=dev)
        self.work_i = torch.empty(B, dtype=torch.int32, device=dev)
        self.work_b = torch.empty(B * (9216 * 2 + 128), dtype=torch.uint8,
                                  device=dev)
        self.work_d = torch.empty(2 * 2048 + 2, dtype=torch.float64,
                                  device=dev)
        self.lr_t = torch.zeros(1, dtype=torch.float32, device=dev)
        self.stats_acc = torch.zeros(2, dtype=torch.float32, device=dev)
        self.loss_acc = torch.zeros(1, dtype=torch.float32, device=dev)
        self._order_pin = None
        self._order_ev = None

    def run_epoch(self, shard_x: torch.Tensor, shard_y: torch.Tensor,
                  order_cpu: torch.Tensor, lr: float, seed: int):
        """Train one local epoch over the device-resident shard.

        Returns (n_samples, n_batches); loss/stats stay in
        ``self.loss_acc`` / ``self.stats_acc`` (device, no sync).
        """
        n = shard_y.numel()
        x = shard_x.reshape(n, -1)
        assert x.shape[1] == 784, "fused CNN epoch expects 784-feature rows"
        if self._order_pin is None or self._order_pin.numel() < n:
            self._order_pin = torch.empty(n, dtype=torch.int64).pin_memory()
            self._order_ev = None
        if self._order_ev is not None:
            # in lazy-stats mode there is no host sync between clients, so
            # the previous client's non_blocking H2D of this pinned buffer
            # may still be queued — wait for it before overwriting
            self._order_ev.synchronize()
        self._order_pin[:n].copy_(order_cpu)
        order_dev = self._order_pin[:n].to(shard_x.device, non_blocking=True)
        self._order_ev = torch.cuda.Event()
        self._order_ev.record()
        self.lr_t.fill_(float(lr))
        self.stats_acc.zero_()
        self.loss_acc.zero_()
        _C.cnn_epoch(x.contiguous(), shard_y.contiguous(), order_dev,
                     self.bs, self.C, self.arena.data, self.arena.grad,
                     self.work_f, self.work_i, self.work_b, self.work_d,
                     self.lr_t, self.max_norm, self.p1, self.p2,
                     self.stats_acc, self.loss_acc, int(seed),
                     self.use_bf16)
        return n, (n + self.bs - 1) // self.bs


class MegaRound:
    """Cross-client mega-batched round driver (csrc/fused_cnn_mega.hip):
    ONE _C.cnn_round_mega call trains ALL K sampled clients with one
    launch set per batch-step — grids scale by K (conv2 fwd: K*bs*9
    blocks fills the 256-CU chip) and the host enqueues ~22 kernels per
    step instead of ~22 per client per step.  fp32 path; per-client
    dropout Philox streams are bit-identical to the per-client fused
    epoch (client-local indices)."""

    def __init__(self, arena: ParameterArena, num_classes: int, bs: int,
                 p1: float, p2: float, max_grad_norm, k_cap: int = 32,
                 use_bf16: bool = False):
        assert HAS_EXT and arena.device.type == "cuda"
        self.arena = arena
        self.C = int(num_classes)
        self.bs = int(bs)
        self.p1, self.p2 = float(p1), float(p2)
        self.max_norm = float(max_grad_norm) if max_grad_norm else -1.0
        self.k_cap = int(k_cap)
        self.use_bf16 = bool(use_bf16)
        self._alloc_k = 0
        dev = arena.device
        self.lr_t = torch.zeros(1, dtype=torch.float32, device=dev)
        self._pins = None

    def _ensure(self, K: int):
        if K <= self._alloc_k:
            return
        k = min(self.k_cap, max(8, K))
        dev = self.arena.device
        P = self.arena.total
        G = k * self.bs
        per_row = (784 + 21632 + 36864 + 9216 + 128 + 128 + self.C
                   + 128 + 9216 + 36864 + 21632 + 18432)
        self.params_stack = torch.empty(k * P, dtype=torch.float32,
                                        device=dev)
        self.grads_stack = torch.empty(k * P, dtype=torch.float32,
                                       device=dev)
        self.work_f = torch.empty(G * per_row + k * 36864,
                                  dtype=torch.float32, device=dev)
        self.work_i = torch.empty(G, dtype=torch.int32, device=dev)
        self.work_b = torch.empty(G * (9216 * 2 + 128), dtype=torch.uint8,
                                  device=dev)
        self.work_d = torch.empty(2 * k, dtype=torch.float64, device=dev)
        self.loss_out = torch.zeros(k, dtype=torch.float32, device=dev)
        self.stats_out = torch.zeros(2 * k, dtype=torch.float32, device=dev)
        self._alloc_k = k

    def supports(self, K: int) -> bool:
        return K <= self.k_cap

    def run(self, store, ds, client_ids, seeds, initial_lr: float,
            server_arena: ParameterArena, round_accum: torch.Tensor):
        """Returns the per-client outputs list [(cid, meta), ...] or None
        if a client is missing from the device shard store."""
        import time as _time
        K = len(client_ids)
        if K == 0 or K > self.k_cap:
            return None
        counts, row_bases, order_offs, orders = [], [], [], []
        off = 0
        for cid, seed in zip(client_ids, seeds):
            user = ds.user_list[cid]
            i = store.user_pos.get(user)
            if i is None:
                return None
            lo, hi = store.offsets[i], store.offsets[i + 1]
            n = hi - lo
            if n == 0:
                return None
            torch.manual_seed(seed & 0x7FFFFFFFFFFF)
            orders.append(torch.randperm(n))
            counts.append(n)
            row_bases.append(lo)
            order_offs.append(off)
            off += n
        self._ensure(K)
        dev = self.arena.device
        orders_cat = torch.cat(orders)
        meta = torch.tensor(row_bases + order_offs + counts
                            + [s & 0x7FFFFFFFFFFF for s in seeds],
                            dtype=torch.int64)
        if self._pins is None or self._pins[0].numel() < off \
                or self._pins[1].numel() < meta.numel():
            self._pins = (torch.empty(max(off, 4096),
                                      dtype=torch.int64).pin_memory(),
                          torch.empty(max(meta.numel(), 256),
                                      dtype=torch.int64).pin_memory(),
                          [None])
        if self._pins[2][0] is not None:
            self._pins[2][0].synchronize()
        self._pins[0][:off].copy_(orders_cat)
        self._pins[1][:meta.numel()].copy_(meta)
        orders_dev = self._pins[0][:off].to(dev, non_blocking=True)
        meta_dev = self._pins[1][:meta.numel()].to(dev, non_blocking=True)
        ev = torch.cuda.Event()
        ev.record()
        self._pins[2][0] = ev
        row_bases_dev = meta_dev[:K]
        order_offs_dev = meta_dev[K:2 * K]
        counts_dev = meta_dev[2 * K:3 * K]
        seeds_dev = meta_dev[3 * K:4 * K]
        counts_host = torch.tensor(counts, dtype=torch.int64)
        weights_dev = counts_dev.to(torch.float32)  # FedAvg: num_samples
        self.lr_t.fill_(float(initial_lr))
        self.loss_out[:K].zero_()
        self.stats_out[:2 * K].zero_()
        _C.cnn_round_mega(
            store.x.reshape(len(store.y), -1), store.y, orders_dev,
            row_bases_dev, order_offs_dev, counts_dev, counts_host,
            weights_dev, seeds_dev, self.bs, self.C,
            server_arena.data, self.params_stack, self.grads_stack,
            round_accum, self.work_f, self.work_i, self.work_b, self.work_d,
            self.lr_t, self.max_norm, self.p1, self.p2,
            self.stats_out, self.loss_out, self.use_bf16)
        now = _time.time()
        outputs = []
        for k, cid in enumerate(client_ids):
            n_batches = (counts[k] + self.bs - 1) // self.bs
            outputs.append((cid, {
                "cs": {"setup": 0.0, "training": 0.0, "full cost": 0.0,
                       "dataloader": 0.0},
                "ns": counts[k],
                "pl": {"weight": float(counts[k]), "grad": None,
                       "pooled": True},
                "_lazy": (self.loss_out[k].reshape(()),
                          self.stats_out[2 * k: 2 * k + 2],
                          n_batches * self.arena.total),
                "ts": now,
            }))
        return outputs
