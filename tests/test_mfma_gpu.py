"""Numerics of the hand-written MFMA implicit-GEMM kernels vs plain
PyTorch fp32 references (conv2 fwd/bwd, fc1 fwd/bwd of the flagship
CNN — csrc/fused_cnn.hip).  gfx950's f32-input MFMA is exact f32
(a k-ordered fmaf chain), so tolerances only cover summation-order
differences vs torch's reductions.
"""

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from msrflute_amd import _C


def _rand(*shape):
    return torch.randn(*shape, device="cuda", dtype=torch.float32)


@pytest.fixture(scope="module", params=[20, 13, 7])
def batch(request):
    return request.param


def test_conv2_fwd_mfma(batch):
    B = batch
    torch.manual_seed(0)
    a1 = _rand(B, 32, 26, 26).abs()  # post-relu activations
    w2, b2 = _rand(64, 32, 3, 3), _rand(64)
    ref = torch.nn.functional.relu(
        torch.nn.functional.conv2d(a1, w2, b2))
    out = _C.dbg_conv2_fwd_mfma(a1.reshape(-1).contiguous(),
                                w2.reshape(-1).contiguous(), b2, B)
    out = out.view(B, 64, 24, 24)
    assert torch.allclose(out, ref, rtol=1e-4, atol=5e-4), \
        (out - ref).abs().max().item()


def test_conv2_bwd_x_mfma(batch):
    B = batch
    torch.manual_seed(1)
    a1 = _rand(B, 32, 26, 26)  # pre-relu sign matters for the mask
    w2 = _rand(64, 32, 3, 3)
    dz2 = _rand(B, 64, 24, 24)
    # reference: autograd through conv2d + relu mask at a1
    x = a1.clamp(min=0).detach().requires_grad_(True)
    y = torch.nn.functional.conv2d(x, w2)
    y.backward(dz2)
    ref = x.grad * (a1 > 0)
    out = _C.dbg_conv2_bwd_x_mfma(dz2.reshape(-1).contiguous(),
                                  w2.reshape(-1).contiguous(),
                                  a1.reshape(-1).contiguous(), B)
    out = out.view(B, 32, 26, 26)
    assert torch.allclose(out, ref, rtol=1e-4, atol=5e-4), \
        (out - ref).abs().max().item()


def test_conv2_bwd_w_mfma(batch):
    B = batch
    torch.manual_seed(2)
    a1 = _rand(B, 32, 26, 26).abs()
    dz2 = _rand(B, 64, 24, 24)
    x = a1.detach().requires_grad_(False)
    w = torch.zeros(64, 32, 3, 3, device="cuda", requires_grad=True)
    y = torch.nn.functional.conv2d(x, w)
    y.backward(dz2)
    ref_w, ref_b = w.grad, dz2.sum(dim=(0, 2, 3))
    dw2, db2 = _C.dbg_conv2_bwd_w_mfma(dz2.reshape(-1).contiguous(),
                                       a1.reshape(-1).contiguous(), B)
    assert torch.allclose(dw2.view(64, 32, 3, 3), ref_w,
                          rtol=1e-3, atol=1e-3), \
        (dw2.view(64, 32, 3, 3) - ref_w).abs().max().item()
    assert torch.allclose(db2, ref_b, rtol=1e-3, atol=1e-3)


def test_fc1_fwd_mfma(batch):
    B = batch
    torch.manual_seed(3)
    a2 = _rand(B, 9216)
    w3, b3 = _rand(128, 9216) * 0.02, _rand(128)
    ref_z = a2 @ w3.t() + b3
    z3, a3, m3 = _C.dbg_fc1_fwd_mfma(a2.reshape(-1).contiguous(),
                                     w3.reshape(-1).contiguous(), b3,
                                     B, 0.0, 123, 0)
    assert torch.allclose(z3.view(B, 128), ref_z, rtol=1e-4, atol=5e-4), \
        (z3.view(B, 128) - ref_z).abs().max().item()
    # p2=0: a3 == relu(z3), mask all-keep
    assert torch.equal(a3.view(B, 128), z3.view(B, 128).clamp(min=0))
    assert int(m3.min()) == 1


def test_fc1_fwd_mfma_dropout_mask_distribution():
    B = 20
    torch.manual_seed(4)
    a2, w3, b3 = _rand(B, 9216), _rand(128, 9216) * 0.02, _rand(128)
    z3, a3, m3 = _C.dbg_fc1_fwd_mfma(a2.reshape(-1), w3.reshape(-1), b3,
                                     B, 0.5, 99, 1)
    keep = m3.float().mean().item()
    assert 0.35 < keep < 0.65
    kept = m3.view(B, 128).bool()
    assert torch.allclose(a3.view(B, 128)[kept],
                          z3.view(B, 128).clamp(min=0)[kept] / 0.5,
                          rtol=1e-6, atol=0)
    assert (a3.view(B, 128)[~kept] == 0).all()
    # same (seed, offset) -> bitwise same mask (per-client determinism)
    _, _, m3b = _C.dbg_fc1_fwd_mfma(a2.reshape(-1), w3.reshape(-1), b3,
                                    B, 0.5, 99, 1)
    assert torch.equal(m3, m3b)


def test_fc1_bwd_w_mfma(batch):
    B = batch
    torch.manual_seed(5)
    dz3, a2 = _rand(B, 128), _rand(B, 9216)
    ref_w = dz3.t() @ a2
    ref_b = dz3.sum(0)
    dw3, db3 = _C.dbg_fc1_bwd_w_mfma(dz3.reshape(-1).contiguous(),
                                     a2.reshape(-1).contiguous(), B)
    assert torch.allclose(dw3.view(128, 9216), ref_w, rtol=1e-4, atol=1e-4)
    assert torch.allclose(db3, ref_b, rtol=1e-4, atol=1e-4)


def test_fc1_bwd_x_mfma(batch):
    B = batch
    torch.manual_seed(6)
    dz3, w3 = _rand(B, 128), _rand(128, 9216)
    ref = dz3 @ w3
    da2 = _C.dbg_fc1_bwd_x_mfma(dz3.reshape(-1).contiguous(),
                                w3.reshape(-1).contiguous(), B)
    assert torch.allclose(da2.view(B, 9216), ref, rtol=1e-4, atol=1e-4)
