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
