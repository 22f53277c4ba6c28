"""Fused CNN-FEMNIST kernels vs the torch-autograd path (GPU only).

With dropout disabled both paths are deterministic given the same shuffle
order, so the trained arena, per-batch losses and sufficient stats must
agree to fp32 tolerance.  Dropout-on gets a distribution-level smoke.
"""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _build(seed=3):
    from msrflute_amd.models import make_model
    from msrflute_amd.ops.arena import ParameterArena
    torch.manual_seed(seed)
    model = make_model({"model_type": "CNN",
                        "model_folder": "experiments/cv_cnn_femnist/model.py",
                        "num_classes": 62})
    arena = ParameterArena(model, bind_grads=True)
    return model, arena


def _eager_epoch(model, arena, xs, ys, order, bs, lr, max_norm):
    from msrflute_amd import ops
    for m in model.modules():
        if isinstance(m, torch.nn.Dropout):
            m.p = 0.0
    stats = torch.zeros(2, device="cuda")
    loss_acc = torch.zeros((), device="cuda")
    n = len(ys)
    for s in range(0, n, bs):
        idx = order[s:s + bs].cuda()
        arena.grad.zero_()
        loss = model.loss({"x": xs[idx], "y": ys[idx]})
        loss.backward()
        ops.clip_stats_accumulate(arena.grad, max_norm, stats)
        ops.sgd_step(arena.data, arena.grad, None, lr=lr)
        loss_acc += loss.detach()
    return stats, loss_acc


def test_fused_epoch_matches_autograd_no_dropout():
    from msrflute_amd.ops.fused_cnn import FusedCNNEpoch, matches_cnn_femnist

    n, bs, lr, mn = 20 * 4 + 13, 20, 0.1, 10.0  # ragged tail of 13
    torch.manual_seed(7)
    xs = torch.randn(n, 28, 28, device="cuda")
    ys = torch.randint(0, 62, (n,), device="cuda")
    order = torch.randperm(n)

    m1, a1 = _build()
    s1, l1 = _eager_epoch(m1, a1, xs, ys, order, bs, lr, mn)

    m2, a2 = _build()
    # a1 was trained by the eager epoch above; a fresh same-seed arena
    # must differ from it (i.e. training actually moved the weights)
    assert not torch.allclose(a1.data, a2.data)
    assert matches_cnn_femnist(a2) == 62
    fc = FusedCNNEpoch(a2, 62, bs=bs, p1=0.0, p2=0.0, max_grad_norm=mn)
    n_out, n_batches = fc.run_epoch(xs, ys, order, lr, seed=123)
    torch.cuda.synchronize()
    assert n_out == n and n_batches == 5

    diff = (a1.data - a2.data).abs().max().item()
    assert torch.allclose(a1.data, a2.data, rtol=1e-4, atol=1e-5), diff
    assert torch.allclose(l1.reshape(1), fc.loss_acc, rtol=1e-4,
                          atol=1e-5), (float(l1), float(fc.loss_acc))
    assert torch.allclose(s1, fc.stats_acc, rtol=1e-3, atol=1e-4), \
        (s1.tolist(), fc.stats_acc.tolist())


def test_fused_epoch_dropout_smoke():
    from msrflute_amd.ops.fused_cnn import FusedCNNEpoch

    n, bs = 100, 20
    xs = torch.randn(n, 28, 28, device="cuda")
    ys = torch.randint(0, 62, (n,), device="cuda")
    _, arena = _build(seed=9)
    before = arena.data.clone()
    fc = FusedCNNEpoch(arena, 62, bs=bs, p1=0.25, p2=0.5, max_grad_norm=10.0)
    fc.run_epoch(xs, ys, torch.randperm(n), 0.1, seed=5)
    torch.cuda.synchronize()
    assert torch.isfinite(fc.loss_acc).all()
    assert float(fc.loss_acc) > 0
    assert not torch.allclose(before, arena.data)  # it trained
    # determinism: same seed+order reproduces bitwise
    arena.data.copy_(before)
    fc.run_epoch(xs, ys, torch.arange(n), 0.1, seed=5)
    torch.cuda.synchronize()
    w1 = arena.data.clone()
    arena.data.copy_(before)
    fc.run_epoch(xs, ys, torch.arange(n), 0.1, seed=5)
    torch.cuda.synchronize()
    assert torch.equal(w1, arena.data)
