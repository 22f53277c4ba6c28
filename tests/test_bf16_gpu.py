"""bf16 MFMA path: fragment-layout probe, per-kernel numerics vs fp32
torch references (bf16 rounding tolerances), and an epoch-level
divergence bound of the bf16 mixed-precision client step vs the fp32
one (VERDICT round-1 item 2)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from msrflute_amd import _C


def _rand(*shape):
    return torch.randn(*shape, device="cuda", dtype=torch.float32)


def test_mfma_bf16_fragment_layout():
    """Pins A[i=l&15][k=(l>>4)*8+e] / B[k][j=l&15] / D[(l>>4)*4+r][l&15]
    for v_mfma_f32_16x16x32_bf16 with transpose-detecting (asymmetric)
    operands."""
    torch.manual_seed(0)
    A = (torch.randn(16, 32) * 2).to(torch.bfloat16).cuda()
    B = (torch.randn(32, 16) * 2).to(torch.bfloat16).cuda()
    D = _C.dbg_mfma_bf16_probe(A.contiguous(), B.contiguous())
    ref = A.float() @ B.float()
    assert torch.allclose(D, ref, rtol=1e-3, atol=1e-3), \
        (D - ref).abs().max().item()


def _bf16_ref_conv2(a1, w2, b2=None):
    a1b = a1.to(torch.bfloat16).float()
    w2b = w2.to(torch.bfloat16).float()
    out = torch.nn.functional.conv2d(a1b, w2b, b2)
    return out


def test_conv2_fwd_mfma_bf16():
    B = 13
    torch.manual_seed(1)
    a1 = _rand(B, 32, 26, 26).abs()
    w2, b2 = _rand(64, 32, 3, 3), _rand(64)
    ref = torch.nn.functional.relu(_bf16_ref_conv2(a1, w2, b2))
    out = _C.dbg_conv2_fwd_mfma_bf16(a1.reshape(-1).contiguous(),
                                     w2.reshape(-1).contiguous(), b2, B)
    out = out.view(B, 64, 24, 24)
    # same bf16-rounded inputs, different f32 accumulation order
    assert torch.allclose(out, ref, rtol=2e-2, atol=5e-2), \
        (out - ref).abs().max().item()


def test_conv2_bwd_x_mfma_bf16():
    B = 7
    torch.manual_seed(2)
    a1 = _rand(B, 32, 26, 26)
    w2 = _rand(64, 32, 3, 3)
    dz2 = _rand(B, 64, 24, 24)
    x = a1.clamp(min=0).detach().to(torch.bfloat16).float().requires_grad_(True)
    y = torch.nn.functional.conv2d(x, w2.to(torch.bfloat16).float())
    y.backward(dz2.to(torch.bfloat16).float())
    ref = x.grad * (a1 > 0)
    out = _C.dbg_conv2_bwd_x_mfma_bf16(dz2.reshape(-1).contiguous(),
                                       w2.reshape(-1).contiguous(),
                                       a1.reshape(-1).contiguous(), B)
    out = out.view(B, 32, 26, 26)
    assert torch.allclose(out, ref, rtol=2e-2, atol=5e-2), \
        (out - ref).abs().max().item()


def test_conv2_bwd_w_mfma_bf16():
    B = 20
    torch.manual_seed(3)
    a1 = _rand(B, 32, 26, 26).abs()
    dz2 = _rand(B, 64, 24, 24)
    w = torch.zeros(64, 32, 3, 3, device="cuda", requires_grad=True)
    y = torch.nn.functional.conv2d(a1.to(torch.bfloat16).float(), w)
    y.backward(dz2.to(torch.bfloat16).float())
    ref_w = w.grad
    dw2, db2 = _C.dbg_conv2_bwd_w_mfma_bf16(dz2.reshape(-1).contiguous(),
                                            a1.reshape(-1).contiguous(), B)
    # K = 11520 bf16 products in f32 accum: scale tolerance to magnitude
    scale = ref_w.abs().max()
    assert torch.allclose(dw2.view(64, 32, 3, 3), ref_w,
                          rtol=2e-2, atol=2e-2 * float(scale)), \
        (dw2.view(64, 32, 3, 3) - ref_w).abs().max().item()
    assert torch.allclose(db2, dz2.sum(dim=(0, 2, 3)), rtol=1e-3, atol=1e-2)


def test_fc1_fwd_mfma_bf16():
    B = 20
    torch.manual_seed(4)
    a2 = _rand(B, 9216)
    w3, b3 = _rand(128, 9216) * 0.02, _rand(128)
    ref_z = (a2.to(torch.bfloat16).float()
             @ w3.to(torch.bfloat16).float().t() + b3)
    z3, a3, m3 = _C.dbg_fc1_fwd_mfma_bf16(a2.reshape(-1).contiguous(),
                                          w3.reshape(-1).contiguous(), b3,
                                          B, 0.0, 123, 0)
    assert torch.allclose(z3.view(B, 128), ref_z, rtol=2e-2, atol=5e-2), \
        (z3.view(B, 128) - ref_z).abs().max().item()
    assert torch.equal(a3.view(B, 128), z3.view(B, 128).clamp(min=0))


def test_bf16_epoch_divergence_bounded():
    """Train one client epoch fp32 and bf16 from the same init: the bf16
    mixed-precision step must stay close to the fp32 trajectory (fp32
    master weights; bf16 only rounds GEMM inputs)."""
    import sys
    sys.path.insert(0, ".")
    from msrflute_amd.models import make_model
    from msrflute_amd.ops.arena import ParameterArena
    from msrflute_amd.ops.fused_cnn import FusedCNNEpoch, matches_cnn_femnist

    def build():
        torch.manual_seed(11)
        m = make_model({"model_type": "CNN",
                        "model_folder": "experiments/cv_cnn_femnist/model.py",
                        "num_classes": 62}).cuda()
        return ParameterArena(m, bind_grads=True)

    n, bs = 60, 20
    torch.manual_seed(12)
    xs = torch.randn(n, 28, 28, device="cuda")
    ys = torch.randint(0, 62, (n,), device="cuda")
    order = torch.randperm(n)

    outs = {}
    for mode in [False, True]:
        a = build()
        fc = FusedCNNEpoch(a, 62, bs=bs, p1=0.0, p2=0.0, max_grad_norm=10.0,
                           use_bf16=mode)
        fc.run_epoch(xs, ys, order, lr=0.05, seed=5)
        torch.cuda.synchronize()
        outs[mode] = (a.data.clone(), float(fc.loss_acc))

    w_fp32, loss_fp32 = outs[False]
    w_bf16, loss_bf16 = outs[True]
    # weights moved (training happened) and stayed close across precisions
    assert not torch.equal(w_fp32, w_bf16)  # genuinely different path
    rel = (w_fp32 - w_bf16).norm() / w_fp32.norm()
    assert float(rel) < 5e-3, float(rel)
    assert abs(loss_fp32 - loss_bf16) / abs(loss_fp32) < 2e-2, \
        (loss_fp32, loss_bf16)
