"""Step-granular divergence diagnosis: Shakespeare mega round vs a
hand-rolled eager replica of the per-client path (same fused recurrence
kernels, same clip-then-SGD).  Run configs with 1..N steps to see where
the mega trajectory departs: a per-step diff ~1e-6 that grows with step
count is GEMM-order noise amplified by the T=80 recurrence; a big
step-1 diff is a bug."""

import os
import sys

import torch
import torch.nn.functional as F

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from msrflute_amd.models import make_model
from msrflute_amd.ops.arena import ParameterArena
from msrflute_amd.ops.mega_shakespeare import ShakespeareMegaRound
from tools.create_data import make_char_lm_blob

DEV = "cuda"


class Store:
    pass


def eager_round(model, arena, server_data, x_all, y_all, counts, row_lo,
                seeds, bs, lr, max_norm):
    """Replicates client.py's per-client path: seed -> randperm -> per
    batch loss.backward + clip + SGD; returns (final stacks, losses)."""
    outs, losses = [], []
    for k, (n, lo, seed) in enumerate(zip(counts, row_lo, seeds)):
        arena.data.copy_(server_data)
        torch.manual_seed(seed & 0x7FFFFFFFFFFF)
        order = torch.randperm(n).cuda()
        loss_sum = 0.0
        for s in range(0, n, bs):
            idx = order[s:s + bs] + lo
            x = x_all.index_select(0, idx).long()
            y = y_all.index_select(0, idx).long()
            arena.grad.zero_()
            loss = model.loss({"x": x, "y": y})
            loss.backward()
            gn = arena.grad.double().norm()
            if max_norm and max_norm > 0:
                scale = torch.clamp(max_norm / (gn.float() + 1e-6), max=1.0)
                arena.grad.mul_(scale)
            arena.data.add_(arena.grad, alpha=-lr)
            loss_sum += loss.item()
        outs.append(arena.data.clone())
        losses.append(loss_sum)
    return outs, losses


def main():
    torch.manual_seed(5)
    model = make_model({"model_type": "RNN",
                        "model_folder":
                        "experiments/nlp_rnn_fedshakespeare/model.py",
                        "vocab_size": 90, "embed_dim": 8,
                        "hidden_dim": 256}).cuda()
    arena = ParameterArena(model, bind_grads=True)
    server_data = arena.data.clone()

    for n_samp, label in [(4, "1 step"), (8, "2 steps"), (24, "6 steps"),
                          (45, "12 steps ragged")]:
        blob = make_char_lm_blob(n_users=3, samples_per_user=n_samp, seed=3)
        xs = torch.cat([torch.tensor(blob["user_data"][u]["x"],
                                     dtype=torch.float32)
                        for u in blob["users"]]).cuda()
        ys = torch.cat([torch.tensor(blob["user_data_label"][u])
                        for u in blob["users"]]).long().cuda()
        store = Store()
        store.x, store.y = xs, ys
        store.offsets = [0, n_samp, 2 * n_samp, 3 * n_samp]
        store.user_pos = {u: i for i, u in enumerate(blob["users"])}

        class DS:
            user_list = blob["users"]

        bs, lr, max_norm = 4, 0.8, 10.0
        seeds = [11, 22, 33]
        counts = [n_samp] * 3
        row_lo = [0, n_samp, 2 * n_samp]

        ref_stacks, ref_losses = eager_round(
            model, arena, server_data, xs, ys, counts, row_lo, seeds, bs,
            lr, max_norm)

        class SA:
            data = server_data
        mega = ShakespeareMegaRound(arena, bs, max_norm, k_cap=8)
        accum = torch.zeros(arena.total, device=DEV)
        out = mega.run(store, DS, [0, 1, 2], seeds, lr, SA, accum)
        assert out is not None
        g = list(mega._graphs.values())[-1]
        torch.cuda.synchronize()
        mega_flat = g["flat"].data
        mega_losses = g["loss_dev"].tolist()
        print(f"--- {label} (n={n_samp}) ---")
        for k in range(3):
            wd = float((ref_stacks[k] - mega_flat[k]).norm()
                       / ref_stacks[k].norm())
            print(f"  client {k}: wdiff {wd:.3e}  "
                  f"loss ref {ref_losses[k]:.7f} mega {mega_losses[k]:.7f}")


if __name__ == "__main__":
    main()
