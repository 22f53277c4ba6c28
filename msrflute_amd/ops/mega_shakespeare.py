"""Cross-client MEGA round for the Shakespeare char-LSTM (BASELINE
benchmark task 4).

The per-client path runs each client's epoch on its own stream;
concurrent recurrence kernels co-schedule only ~2.5x on this stack
(tools/diag_streams.py), which left Shakespeare at ~7 rounds/s.  Here
ALL K sampled clients train together with K-stacked per-client weights:

* one flat [K, P] parameter stack (leaf tensor); every per-client weight
  is a strided view into it, so autograd accumulates a [K, P] gradient
  stack and the per-client clip + sufficient-stats + SGD is the SAME
  gfx950 kernel suite as the CNN mega round (_C.mega_clip_sgd);
* batched compute: per-client embedding via one offset index_select
  (value-independent backward — capture-safe), input/output projections
  via torch.bmm over the K-stacked weights (hipBLASLt batched GEMMs),
  and the T-step recurrences via the cross-client kernels
  (_C.lstm_seq_fwd_b/_bwd_b — grid = all K*bs rows, each block indexing
  its client's float4-packed W_hh);
* the WHOLE local epoch (all batch-steps, forward + autograd backward +
  clip/SGD) is captured as ONE hipGraph keyed by (K, steps, bs, T):
  per round the host refreshes the static gather-index/label buffers
  and replays;
* ragged epochs are masked by labels: inactive rows get pad labels
  (ignore_index 0), so their CE contributions AND gradients are exactly
  zero — finished clients' SGD no-ops (the CNN mega round's trick).

Per-client numerics match the per-client FusedLSTM path: same recurrence
kernel math, same per-batch mean-CE loss, same clip-then-SGD semantics;
only GEMM scheduling (bmm vs mm) and float accumulation orders differ.
"""

from __future__ import annotations

import os
import time
from typing import Dict, Optional, Tuple

import torch
import torch.nn.functional as F

from . import HAS_EXT, _C
from .arena import ParameterArena

H = 256
# expected arena layout of the Shakespeare model
# (experiments/nlp_rnn_fedshakespeare/model.py)
_NAMES = ["net.embeddings.weight", "net.lstm.weight_ih_l0",
          "net.lstm.weight_hh_l0", "net.lstm.bias_ih_l0",
          "net.lstm.bias_hh_l0", "net.lstm.weight_ih_l1",
          "net.lstm.weight_hh_l1", "net.lstm.bias_ih_l1",
          "net.lstm.bias_hh_l1", "net.fc.weight", "net.fc.bias"]


def matches_char_lstm(arena: ParameterArena) -> Optional[Tuple[int, int]]:
    """Return (vocab, embed_dim) when the arena is the Shakespeare
    2-layer H=256 char LSTM, else None."""
    if arena.names != _NAMES:
        return None
    V, E = arena.shapes[0]
    if (tuple(arena.shapes[1]) != (4 * H, E)
            or tuple(arena.shapes[2]) != (4 * H, H)
            or tuple(arena.shapes[5]) != (4 * H, H)
            or tuple(arena.shapes[9]) != (V, H)):
        return None
    return int(V), int(E)


class _LSTMSeqB(torch.autograd.Function):
    """Cross-client batched LSTM layer: xp [R, T, 4H] with R = K*bs rows
    (row r -> client r // bs), w_hh_s [K, 4H, H] stacked weights."""

    @staticmethod
    def forward(ctx, xp, w_hh_s, rows_per_client):
        K = w_hh_s.shape[0]
        # per-client float4 packing (ops/lstm._pack_fwd, K-stacked)
        w_p = w_hh_s.transpose(1, 2).reshape(K, H // 4, 4, 4 * H) \
            .permute(0, 1, 3, 2).contiguous()
        h_seq, gates, c_seq = _C.lstm_seq_fwd_b(
            xp.contiguous(), w_p.reshape(-1), rows_per_client)
        ctx.save_for_backward(gates, c_seq, w_hh_s, h_seq)
        ctx.rows_per_client = rows_per_client
        return h_seq

    @staticmethod
    def backward(ctx, dh):
        gates, c_seq, w_hh_s, h_seq = ctx.saved_tensors
        rpc = ctx.rows_per_client
        K = w_hh_s.shape[0]
        w_pb = w_hh_s.reshape(K, H, 4, H).permute(0, 1, 3, 2).contiguous()
        dg = _C.lstm_seq_bwd_b(gates, c_seq, w_pb.reshape(-1),
                               dh.contiguous(), rpc)
        R, T, _ = h_seq.shape
        h_prev = torch.cat([h_seq.new_zeros(R, 1, H), h_seq[:, :-1]], dim=1)
        # dW_hh per client: one batched GEMM over the client's rows
        dgk = dg.view(K, rpc * T, 4 * H).transpose(1, 2)
        hpk = h_prev.view(K, rpc * T, H)
        dw_hh = torch.bmm(dgk, hpk)
        return dg, dw_hh, None


class ShakespeareMegaRound:
    def __init__(self, arena: ParameterArena, bs: int,
                 max_grad_norm, k_cap: int = 32):
        assert HAS_EXT and arena.device.type == "cuda"
        self.arena = arena
        self.bs = int(bs)
        self.max_norm = float(max_grad_norm) if max_grad_norm else -1.0
        self.k_cap = int(k_cap)
        vc = matches_char_lstm(arena)
        assert vc is not None
        self.V, self.E = vc
        self.lr_t = torch.zeros(1, dtype=torch.float32, device=arena.device)
        self._graphs: Dict[Tuple, dict] = {}

    def supports(self, K: int) -> bool:
        return 0 < K <= self.k_cap

    # ------------------------------------------------------------------
    def _views(self, flat):
        """Per-parameter [K, *shape] strided views into the [K, P] stack."""
        a = self.arena
        out = {}
        for i, n in enumerate(a.names):
            off, cnt = a.offsets[i], a.numels[i]
            out[n] = flat[:, off:off + cnt].view(-1, *a.shapes[i])
        return out

    def _step(self, flat, x, y, K, loss_dev):
        """One batched training step's forward + loss (autograd does the
        backward); x [R, T] long, y [R, T] long (pad 0 masks)."""
        bs, V, E = self.bs, self.V, self.E
        R, T = x.shape
        v = self._views(flat)
        # per-client embedding: offset rows into the stacked table.
        # index_select, NOT F.embedding: embedding_dense_backward's
        # implementation is index-value-dependent, which is unsafe under
        # hipGraph capture (captured on round-1 data, replayed on other
        # rounds'); index_select's backward is a plain index_add with
        # value-independent kernels
        offs = (torch.arange(K, device=x.device) * V).repeat_interleave(bs)
        e = v["net.embeddings.weight"].reshape(K * V, E).index_select(
            0, (x + offs[:, None]).view(-1))
        e = e.view(K, bs * T, E)
        b0 = (v["net.lstm.bias_ih_l0"] + v["net.lstm.bias_hh_l0"])
        xp0 = torch.baddbmm(b0.unsqueeze(1), e,
                            v["net.lstm.weight_ih_l0"].transpose(1, 2))
        h0 = _LSTMSeqB.apply(xp0.view(R, T, 4 * H),
                             v["net.lstm.weight_hh_l0"], bs)
        b1 = (v["net.lstm.bias_ih_l1"] + v["net.lstm.bias_hh_l1"])
        xp1 = torch.baddbmm(b1.unsqueeze(1), h0.view(K, bs * T, H),
                            v["net.lstm.weight_ih_l1"].transpose(1, 2))
        h1 = _LSTMSeqB.apply(xp1.view(R, T, 4 * H),
                             v["net.lstm.weight_hh_l1"], bs)
        logits = torch.baddbmm(v["net.fc.bias"].unsqueeze(1),
                               h1.view(K, bs * T, H),
                               v["net.fc.weight"].transpose(1, 2))
        ce = F.cross_entropy(logits.reshape(-1, V), y.reshape(-1),
                             ignore_index=0, reduction="none").view(K, -1)
        n_tok = (y.view(K, -1) != 0).sum(dim=1).clamp_min(1)
        loss_k = ce.sum(dim=1) / n_tok
        loss_dev += loss_k.detach()
        return loss_k.sum()

    def _build_graph(self, K, steps, T):
        dev = self.arena.device
        P = self.arena.total
        bs = self.bs
        R = K * bs
        g = {
            "flat": torch.zeros(K, P, device=dev, requires_grad=True),
            "idx": torch.zeros(steps, R, dtype=torch.int64, device=dev),
            "ymask": torch.zeros(steps, R, 1, dtype=torch.bool, device=dev),
            "loss_dev": torch.zeros(K, device=dev),
            "stats_out": torch.zeros(2 * K, device=dev),
            "acc2k": torch.zeros(2 * K, dtype=torch.float64, device=dev),
            "weights": torch.zeros(K, device=dev),
            "accum": torch.zeros(P, device=dev),
        }
        store = self._store
        x_all, y_all = store.x, store.y
        server = self._server_data
        V, E = self.V, self.E
        emb_off = self.arena.offsets[0]  # net.embeddings.weight

        def epoch_body():
            flat = g["flat"]
            flat.data.copy_(server.unsqueeze(0).expand(K, P))
            g["loss_dev"].zero_()
            for t in range(steps):
                idx = g["idx"][t]
                x = x_all.index_select(0, idx).long()
                y = torch.where(g["ymask"][t],
                                y_all.index_select(0, idx).long(),
                                torch.zeros(1, dtype=torch.int64,
                                            device=dev))
                if flat.grad is not None:
                    flat.grad.zero_()
                loss = self._step(flat, x, y, K, g["loss_dev"])
                loss.backward()
                # nn.Embedding(padding_idx=0) never accumulates into the
                # pad row; F.embedding has no per-client padding_idx, so
                # zero those rows before the clip norm sees them
                flat.grad[:, emb_off:emb_off + V * E] \
                    .view(K, V, E)[:, 0, :].zero_()
                _C.mega_clip_sgd(flat.data.reshape(-1),
                                 flat.grad.reshape(-1), K, g["acc2k"],
                                 self.max_norm, self.lr_t, g["stats_out"])
            _C.mega_pseudo_accum(flat.grad.reshape(-1), server,
                                 flat.data.reshape(-1), g["weights"],
                                 g["accum"])

        g["body"] = epoch_body
        g["graph"] = None
        # the compute reads these tensors by baked pointer: pin lifetime
        g["_pins"] = (x_all, y_all, server)
        return g

    def _capture(self, g):
        """Warm up + capture the epoch.  MUST run after the round's real
        idx/ymask/lr are staged: capturing on degenerate data risks
        value-dependent kernel selection that faults on replay."""
        if os.environ.get("MEGA_SHK_EAGER") == "1":
            return
        # warmup (establishes flat.grad + autograd buffers), then capture;
        # nothing here writes the server arena (flat is the working copy)
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
        if store.x.dim() != 2:
            return None
        if getattr(self, "_server_data", None) is None:
            self._server_data = server_arena.data
        # the captured graphs bake this pointer in
        assert self._server_data is server_arena.data
        T = store.x.shape[1]
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
        key = (K, steps, bs, T)
        if key not in self._graphs:
            self._graphs[key] = self._build_graph(K, steps, T)
        g = self._graphs[key]

        # host-side: per-client shuffle orders -> gather indices + masks
        R = K * bs
        idx = torch.zeros(steps, R, dtype=torch.int64)
        mask = torch.zeros(steps, R, 1, dtype=torch.bool)
        for k, (cid, seed) in enumerate(zip(client_ids, seeds)):
            n = counts[k]
            torch.manual_seed(seed & 0x7FFFFFFFFFFF)
            order = torch.randperm(n) + row_lo[k]
            nb = (n + bs - 1) // bs
            for t in range(nb):
                chunk = order[t * bs:(t + 1) * bs]
                idx[t, k * bs:k * bs + len(chunk)] = chunk
                mask[t, k * bs:k * bs + len(chunk), 0] = True
        g["idx"].copy_(idx)
        g["ymask"].copy_(mask)
        g["weights"].copy_(torch.tensor([float(c) for c in counts]))
        self.lr_t.fill_(float(initial_lr))
        if g["graph"] is None and not g.get("_captured"):
            # first use of this shape: capture on THIS round's real data
            self._capture(g)
            g["_captured"] = True
        # warmup/capture side effects land in these accumulators: zero
        # them after capture, before the replay that counts
        g["stats_out"].zero_()
        g["accum"].zero_()
        if g["graph"] is None:
            g["body"]()
        else:
            g["graph"].replay()
        # fold the epoch's accumulated weighted pseudo-gradients
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
