"""Cross-client MEGA round for the fed-CIFAR100 ResNet-18 (BASELINE
benchmark task 3).

The per-client path runs ~22 small MIOpen convs per batch per client —
measured 8.1x slower than ONE grouped conv covering all K clients
(tools/diag_grouped_conv.py): per-op dispatch plus grids that underfill
the 256-CU chip.  Here all K sampled clients train together:

* one [K, P] parameter stack (single autograd leaf; per-client weights
  are strided views), exactly the Shakespeare mega pattern
  (ops/mega_shakespeare.py);
* batched compute: the super-batch is [bs, K*C, H, W] (client k owns
  channel block k) and every conv is ONE F.conv2d with groups=K over
  the K-stacked filters; normalization is GroupNorm (the benchmark
  config trains with group_norm=2), which is per-sample — so stacked
  channels keep EXACT per-client semantics with no cross-client or
  cross-batch stat pollution and no running buffers;
* per-client clip + sufficient stats + SGD and the weighted pseudo-grad
  accumulate reuse the generic K-stacked kernels (_C.mega_clip_sgd /
  _C.mega_pseudo_accum, csrc/fused_cnn_mega.hip);
* the WHOLE local epoch is captured as one hipGraph, captured on the
  first round's REAL gather indices (capture-safety lesson from the
  Shakespeare round: degenerate capture data can bake value-dependent
  kernel choices that fault on replay);
* ragged epochs are masked by labels: inactive rows get y = -100
  (CrossEntropyLoss's default ignore_index), and since conv/GroupNorm/
  pool are all per-sample, inactive rows contribute exactly zero
  gradients — finished clients' SGD no-ops.

Only conv algorithm selection (grouped vs per-client) and float
accumulation orders differ from the per-client path.
"""

from __future__ import annotations

import os
import time
from typing import Dict, Optional, Tuple

import torch
import torch.nn.functional as F

from . import HAS_EXT, _C
from .arena import ParameterArena

_PLANES = (64, 128, 256, 512)


def _expected_names():
    names = ["net.conv1.weight", "net.n1.weight", "net.n1.bias"]
    for li in range(1, 5):
        for bi in range(2):
            p = f"net.layer{li}.{bi}."
            names += [p + "conv1.weight", p + "n1.weight", p + "n1.bias",
                      p + "conv2.weight", p + "n2.weight", p + "n2.bias"]
            if li > 1 and bi == 0:
                names += [p + "down.0.weight", p + "down.1.weight",
                          p + "down.1.bias"]
    return names + ["net.fc.weight", "net.fc.bias"]


def matches_resnet18(arena: ParameterArena) -> Optional[int]:
    """Return num_classes when the arena is the fed-CIFAR100 ResNet-18
    (2-2-2-2 BasicBlocks, 7x7 stem), else None."""
    if arena.names != _expected_names():
        return None
    if tuple(arena.shapes[0]) != (64, 3, 7, 7):
        return None
    fc = arena.shapes[-2]
    if len(fc) != 2 or fc[1] != 512:
        return None
    return int(fc[0])


class ResNetMegaRound:
    def __init__(self, arena: ParameterArena, bs: int, max_grad_norm,
                 cpg: int, k_cap: int = 32):
        assert HAS_EXT and arena.device.type == "cuda" and cpg > 0
        self.arena = arena
        self.bs = int(bs)
        self.max_norm = float(max_grad_norm) if max_grad_norm else -1.0
        self.cpg = int(cpg)
        self.k_cap = int(k_cap)
        nc = matches_resnet18(arena)
        assert nc is not None
        self.NC = nc
        self.lr_t = torch.zeros(1, dtype=torch.float32, device=arena.device)
        self._graphs: Dict[Tuple, dict] = {}

    def supports(self, K: int) -> bool:
        return 0 < K <= self.k_cap

    # ------------------------------------------------------------------
    def _views(self, flat):
        a = self.arena
        return {n: flat[:, off:off + cnt].view(-1, *a.shapes[i])
                for i, (n, off, cnt) in enumerate(
                    zip(a.names, a.offsets, a.numels))}

    def _forward(self, v, x, K):
        """Grouped functional ResNet-18 over the K-stacked views;
        x [bs, K*3, H, W] -> logits [K, bs, NC]."""
        cpg = self.cpg

        def conv(h, name, stride, pad):
            w = v[name]
            return F.conv2d(h, w.reshape(-1, *w.shape[2:]), None, stride,
                            pad, 1, K)

        def gn(h, p):
            w, b = v[p + "weight"], v[p + "bias"]
            C = w.shape[1]
            return F.group_norm(h, K * max(1, C // cpg), w.reshape(-1),
                                b.reshape(-1), 1e-5)

        h = F.max_pool2d(
            F.relu(gn(conv(x, "net.conv1.weight", 2, 3), "net.n1.")), 3, 2, 1)
        for li in range(1, 5):
            for bi in range(2):
                p = f"net.layer{li}.{bi}."
                stride = 2 if (li > 1 and bi == 0) else 1
                out = F.relu(gn(conv(h, p + "conv1.weight", stride, 1),
                                p + "n1."))
                out = gn(conv(out, p + "conv2.weight", 1, 1), p + "n2.")
                if p + "down.0.weight" in v:
                    sc = gn(conv(h, p + "down.0.weight", stride, 0),
                            p + "down.1.")
                else:
                    sc = h
                h = F.relu(out + sc)
        bs = x.shape[0]
        z = F.adaptive_avg_pool2d(h, 1).reshape(bs, K, -1).permute(1, 0, 2)
        return torch.baddbmm(v["net.fc.bias"].unsqueeze(1), z,
                             v["net.fc.weight"].transpose(1, 2))

    def _step(self, flat, x, y, K, loss_dev):
        """x [bs, K*3, H, W]; y [K*bs] long with -100 on inactive rows
        (client-major).  Per-client loss = mean CE over its active rows
        (the per-client path's CrossEntropyLoss batch mean)."""
        bs = self.bs
        logits = self._forward(self._views(flat), x, K)
        ce = F.cross_entropy(logits.reshape(K * bs, self.NC), y,
                             reduction="none",
                             ignore_index=-100).view(K, bs)
        n_act = (y.view(K, bs) != -100).sum(dim=1).clamp_min(1)
        loss_k = ce.sum(dim=1) / n_act
        loss_dev += loss_k.detach()
        return loss_k.sum()

    # ------------------------------------------------------------------
    def _alloc(self, K, steps):
        dev = self.arena.device
        P = self.arena.total
        bs = self.bs
        R = K * bs
        store = self._store
        C, H, W = store.x.shape[1:]
        g = {
            "flat": torch.zeros(K, P, device=dev, requires_grad=True),
            "idx": torch.zeros(steps, R, dtype=torch.int64, device=dev),
            "ymask": torch.zeros(steps, R, dtype=torch.bool, device=dev),
            "loss_dev": torch.zeros(K, device=dev),
            "stats_out": torch.zeros(2 * K, device=dev),
            "acc2k": torch.zeros(2 * K, dtype=torch.float64, device=dev),
            "weights": torch.zeros(K, device=dev),
            "accum": torch.zeros(P, device=dev),
        }
        x_all, y_all = store.x, store.y
        server = self._server_data

        def epoch_body():
            flat = g["flat"]
            flat.data.copy_(server.unsqueeze(0).expand(K, P))
            g["loss_dev"].zero_()
            for t in range(steps):
                idx = g["idx"][t]
                # [K*bs, C, H, W] client-major -> [bs, K*C, H, W]
                x = x_all.index_select(0, idx).view(K, bs, C, H, W) \
                    .transpose(0, 1).reshape(bs, K * C, H, W)
                y = torch.where(g["ymask"][t],
                                y_all.index_select(0, idx).long(),
                                torch.full((1,), -100, dtype=torch.int64,
                                           device=dev))
                if flat.grad is not None:
                    flat.grad.zero_()
                loss = self._step(flat, x, y, K, g["loss_dev"])
                loss.backward()
                _C.mega_clip_sgd(flat.data.reshape(-1),
                                 flat.grad.reshape(-1), K, g["acc2k"],
                                 self.max_norm, self.lr_t, g["stats_out"])
            _C.mega_pseudo_accum(flat.grad.reshape(-1), server,
                                 flat.data.reshape(-1), g["weights"],
                                 g["accum"])

        g["body"] = epoch_body
        g["graph"] = None
        g["_pins"] = (x_all, y_all, server)
        return g

    def _capture(self, g):
        # exhaustive MIOpen find during warmup: grouped-conv algorithms
        # from immediate mode are fallbacks; the capture then bakes the
        # found kernels (find cost is paid once, outside the hot loop)
        torch.backends.cudnn.benchmark = True
        if os.environ.get("MEGA_RESNET_EAGER") == "1":
            return
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                g["body"]()
        torch.cuda.current_stream().wait_stream(s)
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            g["body"]()
        g["graph"] = graph
        torch.cuda.synchronize()

    # ------------------------------------------------------------------
    def run(self, store, ds, client_ids, seeds, initial_lr: float,
            server_arena: ParameterArena, round_accum: torch.Tensor):
        K = len(client_ids)
        if not self.supports(K):
            return None
        bs = self.bs
        self._store = store
        if store.x.dim() != 4:
            return None
        if getattr(self, "_server_data", None) is None:
            self._server_data = server_arena.data
        assert self._server_data is server_arena.data
        counts, row_lo = [], []
        for cid in client_ids:
            i = store.user_pos.get(ds.user_list[cid])
            if i is None:
                return None
            lo, hi = store.offsets[i], store.offsets[i + 1]
            if hi - lo == 0:
                return None
            counts.append(int(hi - lo))
            row_lo.append(int(lo))
        steps = max((c + bs - 1) // bs for c in counts)
        key = (K, steps, bs)
        if key not in self._graphs:
            self._graphs[key] = self._alloc(K, steps)
        g = self._graphs[key]

        R = K * bs
        idx = torch.zeros(steps, R, dtype=torch.int64)
        mask = torch.zeros(steps, R, dtype=torch.bool)
        for k, (cid, seed) in enumerate(zip(client_ids, seeds)):
            n = counts[k]
            torch.manual_seed(seed & 0x7FFFFFFFFFFF)
            order = torch.randperm(n) + row_lo[k]
            nb = (n + bs - 1) // bs
            for t in range(nb):
                chunk = order[t * bs:(t + 1) * bs]
                idx[t, k * bs:k * bs + len(chunk)] = chunk
                mask[t, k * bs:k * bs + len(chunk)] = True
        g["idx"].copy_(idx)
        g["ymask"].copy_(mask)
        g["weights"].copy_(torch.tensor([float(c) for c in counts]))
        self.lr_t.fill_(float(initial_lr))
        if g["graph"] is None and not g.get("_captured"):
            self._capture(g)
            g["_captured"] = True
        g["stats_out"].zero_()
        g["accum"].zero_()
        if g["graph"] is None:
            g["body"]()
        else:
            g["graph"].replay()
        round_accum += g["accum"]

        now = time.time()
        outputs = []
        for k, cid in enumerate(client_ids):
            nb = (counts[k] + bs - 1) // bs
            outputs.append((cid, {
                "cs": {"setup": 0.0, "training": 0.0, "full cost": 0.0,
                       "dataloader": 0.0},
                "ns": counts[k],
                "pl": {"weight": float(counts[k]), "grad": None,
                       "pooled": True},
                "_lazy": (g["loss_dev"][k].reshape(()),
                          g["stats_out"][2 * k:2 * k + 2],
                          nb * self.arena.total),
                "ts": now,
            }))
        return outputs
