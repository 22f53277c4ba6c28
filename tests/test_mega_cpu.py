"""CPU-side checks of the autograd mega rounds' model math and arena
matchers (the GPU harness equivalence lives in test_mega_*_gpu.py)."""

import os

import pytest
import torch
import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _resnet_model(seed):
    from msrflute_amd.models import make_model
    with open(os.path.join(REPO, "configs/cv_resnet_fedcifar100.yaml")) as f:
        cfg = yaml.safe_load(f)
    torch.manual_seed(seed)
    return make_model(cfg["model_config"]).double()


def test_resnet_mega_forward_exact_in_f64():
    """The grouped-conv + stacked-GroupNorm functional forward must be
    SEMANTICALLY identical to K independent per-client forwards — in
    float64 the difference is exactly zero (f32 differences are conv
    algorithm scheduling only)."""
    from msrflute_amd.ops.arena import ParameterArena
    from msrflute_amd.ops.mega_resnet import (ResNetMegaRound,
                                              matches_resnet18)
    m1, m2 = _resnet_model(0), _resnet_model(7)
    a = ParameterArena(m1, bind_grads=True, dtype=torch.float64)
    a2 = ParameterArena(m2, bind_grads=True, dtype=torch.float64)
    assert matches_resnet18(a) == 100
    K, bs = 2, 3
    flat = torch.stack([a.data.clone(), a2.data.clone()])
    obj = ResNetMegaRound.__new__(ResNetMegaRound)
    obj.arena, obj.cpg, obj.NC, obj.bs = a, 2, 100, bs
    torch.manual_seed(3)
    x1 = torch.randn(bs, 3, 24, 24, dtype=torch.float64)
    x2 = torch.randn(bs, 3, 24, 24, dtype=torch.float64)
    logits = obj._forward(obj._views(flat), torch.cat([x1, x2], 1), K)
    with torch.no_grad():
        r1, r2 = m1.net(x1), m2.net(x2)
    assert torch.equal(logits[0], r1)
    assert torch.equal(logits[1], r2)


def test_resnet_mega_masked_rows_zero_grad():
    """Inactive (ragged) rows: y=-100 must produce exactly zero gradient
    contribution — conv/GroupNorm/pool are per-sample, so a masked row
    cannot leak into the client's weights."""
    import torch.nn.functional as F
    from msrflute_amd.ops.arena import ParameterArena
    from msrflute_amd.ops.mega_resnet import ResNetMegaRound
    m1 = _resnet_model(1)
    a = ParameterArena(m1, bind_grads=True, dtype=torch.float64)
    K, bs = 1, 3
    obj = ResNetMegaRound.__new__(ResNetMegaRound)
    obj.arena, obj.cpg, obj.NC, obj.bs = a, 2, 100, bs
    torch.manual_seed(5)
    x = torch.randn(bs, 3, 24, 24, dtype=torch.float64)
    y = torch.tensor([4, 7, -100])
    flat = a.data.clone().unsqueeze(0).requires_grad_(True)
    loss_dev = torch.zeros(K, dtype=torch.float64)
    loss = obj._step(flat, x, y, K, loss_dev)
    loss.backward()
    g_masked = flat.grad.clone()
    # same two active rows alone
    flat2 = a.data.clone().unsqueeze(0).requires_grad_(True)
    obj.bs = 2
    loss_dev2 = torch.zeros(K, dtype=torch.float64)
    loss2 = obj._step(flat2, x[:2], y[:2], K, loss_dev2)
    loss2.backward()
    assert torch.allclose(g_masked, flat2.grad, atol=1e-12)
    assert torch.allclose(loss_dev, loss_dev2, atol=1e-12)


def test_arena_matchers_reject_other_models():
    from msrflute_amd.models import make_model
    from msrflute_amd.ops.arena import ParameterArena
    from msrflute_amd.ops.mega_resnet import matches_resnet18
    from msrflute_amd.ops.mega_shakespeare import matches_char_lstm
    cnn = make_model({"model_type": "CNN",
                      "model_folder": "experiments/cv_cnn_femnist/model.py",
                      "num_classes": 62})
    a = ParameterArena(cnn, bind_grads=True)
    assert matches_resnet18(a) is None
    assert matches_char_lstm(a) is None
    rnn = make_model({"model_type": "RNN",
                      "model_folder":
                      "experiments/nlp_rnn_fedshakespeare/model.py",
                      "vocab_size": 90, "embed_dim": 8,
                      "hidden_dim": 256})
    ar = ParameterArena(rnn, bind_grads=True)
    assert matches_char_lstm(ar) == (90, 8)
    assert matches_resnet18(ar) is None
