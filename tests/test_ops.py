"""Flat-arena op semantics: torch-reference implementations on CPU.

These define the contract the gfx950 HIP kernels must match; the GPU
mirror of every case lives in tests/test_ops_gpu.py.
"""

import math

import pytest
import torch

from msrflute_amd import ops
from msrflute_amd.ops import reference as ref


def randvec(n=10001, seed=0):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(n, generator=g)


def test_pseudo_grad():
    a, b = randvec(seed=1), randvec(seed=2)
    out = torch.empty_like(a)
    ops.pseudo_grad(out, a, b, 3.5)
    assert torch.allclose(out, (a - b) * 3.5)


def test_axpy_scale():
    y, x = randvec(seed=3), randvec(seed=4)
    y0 = y.clone()
    ops.axpy(y, x, 2.0)
    assert torch.allclose(y, y0 + 2.0 * x)
    ops.scale(y, 0.5)
    assert torch.allclose(y, (y0 + 2.0 * x) * 0.5)


def test_sum_sumsq():
    x = randvec(seed=5)
    s = ops.sum_sumsq(x)
    assert torch.allclose(s[0], x.sum(), atol=1e-4)
    assert torch.allclose(s[1], (x * x).sum(), rtol=1e-5)


def test_clip_by_norm_matches_torch_clip_grad():
    for max_norm in (0.5, 1e6):
        x = randvec(seed=6)
        p = torch.nn.Parameter(x.clone())
        p.grad = x.clone()
        expected_norm = torch.nn.utils.clip_grad_norm_([p], max_norm)
        got = x.clone()
        norm = ops.clip_by_norm(got, max_norm)
        assert torch.allclose(norm, expected_norm, rtol=1e-6)
        assert torch.allclose(got, p.grad, rtol=1e-6)


def test_gaussian_noise_statistics():
    x = torch.zeros(200_000)
    g = torch.Generator().manual_seed(7)
    ops.add_gaussian_noise(x, 2.0, seed=7, generator=g)
    assert abs(x.mean().item()) < 0.02
    assert abs(x.std().item() - 2.0) < 0.02


@pytest.mark.parametrize("momentum,nesterov,wd", [
    (0.0, False, 0.0), (0.9, False, 0.0), (0.9, True, 1e-4), (0.5, False, 1e-2)])
def test_sgd_step_matches_torch(momentum, nesterov, wd):
    n = 5000
    p_ref = torch.nn.Parameter(randvec(n, seed=8).clone())
    opt = torch.optim.SGD([p_ref], lr=0.1, momentum=momentum,
                          nesterov=nesterov, weight_decay=wd)
    p = p_ref.detach().clone()
    buf = torch.zeros(n) if momentum else None
    for step in range(4):
        g = randvec(n, seed=10 + step)
        p_ref.grad = g.clone()
        opt.step()
        ops.sgd_step(p, g.clone(), buf, lr=0.1, momentum=momentum,
                     weight_decay=wd, nesterov=nesterov, first_step=(step == 0))
        assert torch.allclose(p, p_ref.detach(), rtol=1e-5, atol=1e-6), step


@pytest.mark.parametrize("amsgrad,adamw,wd", [
    (False, False, 0.0), (True, False, 0.0), (False, True, 1e-2),
    (False, False, 1e-3)])
def test_adam_step_matches_torch(amsgrad, adamw, wd):
    n = 4000
    p_ref = torch.nn.Parameter(randvec(n, seed=20).clone())
    cls = torch.optim.AdamW if adamw else torch.optim.Adam
    opt = cls([p_ref], lr=1e-2, amsgrad=amsgrad, weight_decay=wd)
    p = p_ref.detach().clone()
    m, v = torch.zeros(n), torch.zeros(n)
    vmax = torch.zeros(n) if amsgrad else None
    for step in range(1, 5):
        g = randvec(n, seed=30 + step)
        p_ref.grad = g.clone()
        opt.step()
        ops.adam_step(p, g.clone(), m, v, vmax, step=step, lr=1e-2,
                      weight_decay=wd, amsgrad=amsgrad, adamw=adamw)
        assert torch.allclose(p, p_ref.detach(), rtol=1e-4, atol=1e-6), step


def test_adamax_step_matches_torch():
    n = 3000
    p_ref = torch.nn.Parameter(randvec(n, seed=40).clone())
    opt = torch.optim.Adamax([p_ref], lr=1e-2)
    p = p_ref.detach().clone()
    m, u = torch.zeros(n), torch.zeros(n)
    for step in range(1, 5):
        g = randvec(n, seed=50 + step)
        p_ref.grad = g.clone()
        opt.step()
        ops.adamax_step(p, g.clone(), m, u, step=step, lr=1e-2)
        assert torch.allclose(p, p_ref.detach(), rtol=1e-4, atol=1e-6), step


def test_segmented_sqnorm():
    x = randvec(seed=60)
    offs = torch.tensor([0, 100, 5000, 10001])
    out = ops.segmented_sqnorm(x, offs)
    for i, (a, b) in enumerate(zip(offs[:-1], offs[1:])):
        seg = x[a:b]
        assert torch.allclose(out[i], seg.dot(seg), rtol=1e-5)


def test_quantize_dequantize_semantics():
    """Matches the reference quant.py pipeline computed with torch ops."""
    torch.manual_seed(70)
    x = torch.randn(5000)
    orig = x.clone()
    n_bins = 2 ** 6
    q = 0.5
    got = ops.quantize_dequantize(x.clone(), n_bins, q)
    # reference formulation
    mn, mx = orig.min(), orig.max()
    thresh = torch.quantile(orig.abs(), q)
    bins = torch.linspace(mn, mx, n_bins)
    width = bins[1] - bins[0]
    idx = torch.bucketize(orig - 0.5 * width, bins, right=False).clamp(0, n_bins - 1)
    expected = torch.where(orig.abs() > thresh, bins[idx], torch.tensor(0.0))
    assert torch.allclose(got, expected, atol=float(width) + 1e-6)
    # sparsity: about half the entries zeroed
    frac_zero = (got == 0).float().mean().item()
    assert 0.4 < frac_zero < 0.6
