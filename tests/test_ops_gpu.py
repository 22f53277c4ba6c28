"""gfx950 HIP kernel numerics: every kernel vs. the fp32 torch reference.

Runs on the MI355X box (`pytest -m gpu`).  The ops dispatcher routes GPU
tensors to msrflute_amd._C, so these exercise the HIP kernels directly and
fail loudly if the extension is missing (by design).
"""

import math

import pytest
import torch

import msrflute_amd.ops as ops
from msrflute_amd.ops import reference as ref

pytestmark = pytest.mark.gpu

# odd size exercises the scalar tail path of the vectorized kernels
N = 1_000_003


def dev(seed=0, n=N):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(n, generator=g).cuda()


def test_extension_is_loaded():
    assert ops.HAS_EXT, "HIP extension must be present on a GPU box"


def test_pseudo_grad_kernel():
    a, b = dev(1), dev(2)
    out = torch.empty_like(a)
    ops.pseudo_grad(out, a, b, 2.5)
    expected = (a - b) * 2.5
    assert torch.allclose(out, expected, rtol=1e-6, atol=1e-6)


def test_axpy_scale_kernels():
    y, x = dev(3), dev(4)
    y0 = y.clone()
    ops.axpy(y, x, 1.7)
    assert torch.allclose(y, y0 + 1.7 * x, rtol=1e-6, atol=1e-6)
    ops.scale(y, 0.25)
    assert torch.allclose(y, (y0 + 1.7 * x) * 0.25, rtol=1e-6, atol=1e-6)


def test_sum_sumsq_kernel():
    x = dev(5)
    s = ops.sum_sumsq(x)
    assert torch.allclose(s[0], x.sum(), rtol=1e-4, atol=1e-2)
    assert torch.allclose(s[1], x.double().pow(2).sum().float(), rtol=1e-5)


@pytest.mark.parametrize("max_norm", [0.5, 1e9])
def test_clip_by_norm_kernel(max_norm):
    x = dev(6)
    expected = x.clone()
    expected_norm = ref.clip_by_norm(expected, max_norm)
    norm = ops.clip_by_norm(x, max_norm)
    assert torch.allclose(norm.cpu(), expected_norm.cpu(), rtol=1e-5)
    assert torch.allclose(x, expected, rtol=1e-5, atol=1e-6)


def test_gaussian_noise_kernel_stats_and_determinism():
    x = torch.zeros(N, device="cuda")
    ops.add_gaussian_noise(x, 3.0, seed=42)
    assert abs(x.mean().item()) < 0.02
    assert abs(x.std().item() - 3.0) < 0.02
    y = torch.zeros(N, device="cuda")
    ops.add_gaussian_noise(y, 3.0, seed=42)
    assert torch.equal(x, y), "same seed must give identical noise"
    z = torch.zeros(N, device="cuda")
    ops.add_gaussian_noise(z, 3.0, seed=43)
    assert not torch.equal(x, z)


@pytest.mark.parametrize("momentum,nesterov,wd", [
    (0.0, False, 0.0), (0.9, False, 0.0), (0.9, True, 1e-4)])
def test_sgd_kernel_matches_reference(momentum, nesterov, wd):
    p_ref = dev(8).clone()
    p = p_ref.clone()
    buf_ref = torch.zeros_like(p) if momentum else None
    buf = torch.zeros_like(p) if momentum else None
    for step in range(3):
        g = dev(10 + step)
        ref.sgd_step(p_ref, g.clone(), buf_ref, lr=0.1, momentum=momentum,
                     weight_decay=wd, nesterov=nesterov, first_step=(step == 0))
        ops.sgd_step(p, g.clone(), buf, lr=0.1, momentum=momentum,
                     weight_decay=wd, nesterov=nesterov, first_step=(step == 0))
        assert torch.allclose(p, p_ref, rtol=1e-5, atol=1e-6), step


@pytest.mark.parametrize("amsgrad,adamw", [(False, False), (True, False),
                                           (False, True)])
def test_adam_kernel_matches_reference(amsgrad, adamw):
    p_ref, p = dev(20).clone(), dev(20).clone()
    m_r, v_r = torch.zeros_like(p), torch.zeros_like(p)
    m, v = torch.zeros_like(p), torch.zeros_like(p)
    vm_r = torch.zeros_like(p) if amsgrad else None
    vm = torch.zeros_like(p) if amsgrad else None
    for step in range(1, 4):
        g = dev(30 + step)
        ref.adam_step(p_ref, g.clone(), m_r, v_r, vm_r, step=step, lr=1e-2,
                      weight_decay=1e-3, amsgrad=amsgrad, adamw=adamw)
        ops.adam_step(p, g.clone(), m, v, vm, step=step, lr=1e-2,
                      weight_decay=1e-3, amsgrad=amsgrad, adamw=adamw)
        assert torch.allclose(p, p_ref, rtol=1e-4, atol=1e-6), step


def test_adamax_kernel_matches_reference():
    p_ref, p = dev(40).clone(), dev(40).clone()
    m_r, u_r = torch.zeros_like(p), torch.zeros_like(p)
    m, u = torch.zeros_like(p), torch.zeros_like(p)
    for step in range(1, 4):
        g = dev(50 + step)
        ref.adamax_step(p_ref, g.clone(), m_r, u_r, step=step, lr=1e-2)
        ops.adamax_step(p, g.clone(), m, u, step=step, lr=1e-2)
        assert torch.allclose(p, p_ref, rtol=1e-4, atol=1e-6), step


def test_segmented_sqnorm_kernel():
    x = dev(60)
    offs = torch.tensor([0, 17, 100_000, N], dtype=torch.int64, device="cuda")
    out = ops.segmented_sqnorm(x, offs)
    expected = ref.segmented_sqnorm(x, offs.cpu())
    assert torch.allclose(out, expected, rtol=1e-5)


def test_quantize_kernel_matches_reference():
    x = dev(70, n=100_001)
    expected = ref.quantize_dequantize(x.clone(), 256, 0.5)
    got = ops.quantize_dequantize(x.clone(), 256, 0.5)
    width = float((x.max() - x.min()) / 255)
    # borderline elements may land one bin apart due to fp rounding
    close = torch.isclose(got, expected, atol=width + 1e-6)
    assert close.float().mean().item() > 0.9999
    frac_zero = (got == 0).float().mean().item()
    assert 0.45 < frac_zero < 0.55


def test_gru_gates_kernel_matches_reference():
    torch.manual_seed(80)
    B, H = 33, 129
    g_i = torch.randn(B, 3 * H, device="cuda")
    g_h = torch.randn(B, 3 * H, device="cuda")
    h = torch.randn(B, H, device="cuda")
    with torch.no_grad():
        got = ops.gru_gates(g_i, g_h, h)
    expected = ref.gru_gates(g_i.cpu(), g_h.cpu(), h.cpu())
    assert torch.allclose(got.cpu(), expected, rtol=1e-4, atol=1e-5)
