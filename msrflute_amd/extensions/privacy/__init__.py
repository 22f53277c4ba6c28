"""Differential-privacy extensions.

Reference: extensions/privacy/__init__.py.  The flat arena makes the
reference's unroll/update round-trips (privacy/__init__.py:105-126)
unnecessary — the gradient already IS a flat vector, and clip/normalize/
noise are fused device ops (kernels K7/K8).

Noise determinism: GPU noise is Philox keyed by an explicit seed so local
DP is reproducible per (round, client) and global DP is IDENTICAL across
the symmetric rank replicas (required for the replicated server update).
"""

from __future__ import annotations

import json
import logging
import math
from typing import Optional

import numpy as np
import torch

from ... import ops
from ...utils import print_rank, log_metric


def compute_LDP_noise_std(eps, max_sensitivity, delta):
    """Gaussian-mechanism sigma (reference: privacy/__init__.py:15-16)."""
    return np.sqrt(2 * np.log(1.25 / delta)) * max_sensitivity / eps


def _noise(flat: torch.Tensor, sigma: float, seed: Optional[int]):
    if flat.is_cuda:
        ops.add_gaussian_noise(flat, sigma, seed if seed is not None else 0)
    else:
        gen = None
        if seed is not None:
            gen = torch.Generator(device="cpu")
            gen.manual_seed(seed & 0x7FFFFFFFFFFFFFFF)
        ops.add_gaussian_noise(flat, sigma, 0, generator=gen)
    return flat


def add_gaussian_noise(grad: torch.Tensor, eps, max_grad, delta,
                       seed: Optional[int] = None):
    """noisy = grad + sigma*N(0,1) (reference: privacy/__init__.py:70-74);
    in-place on the flat gradient."""
    sigma = float(compute_LDP_noise_std(eps, max_grad, delta))
    return _noise(grad, sigma, seed), sigma


def apply_local_dp(trainer, weight, dp_config, add_weight_noise,
                   seed: Optional[int] = None):
    """Client-side DP on the flat pseudo-gradient
    (reference: privacy/__init__.py:154-201).

    eps < 0: L2-clip to max_grad, no noise.  Otherwise: normalize the
    gradient to norm == max_grad, add Gaussian noise with sensitivity
    sqrt(max_grad² + max_weight²), and noise the (scaled, clamped)
    aggregation weight with the same sigma.
    """
    flat = trainer.arena.grad
    grad_norm = float(flat.norm())

    if dp_config["eps"] < 0:
        if grad_norm > dp_config["max_grad"]:
            ops.scale(flat, dp_config["max_grad"] / grad_norm)
        return weight

    dp_eps = dp_config["eps"]
    delta = dp_config.get("delta", 1e-7)
    weight_ = weight
    weight = dp_config.get("weight_scaler", 1) * weight
    weight = min(dp_config["max_weight"], weight)
    if grad_norm > 0:
        ops.scale(flat, dp_config["max_grad"] / grad_norm)
    max_sensitivity = math.sqrt(
        dp_config["max_grad"] ** 2
        + (dp_config["max_weight"] ** 2 if add_weight_noise else 0.0))
    _, sigma = add_gaussian_noise(flat, dp_eps, max_sensitivity, delta, seed=seed)

    # the reference draws one joint Gaussian over [grad ‖ weight]; an
    # independent draw for the scalar weight is distributionally identical
    wseed = None if seed is None else seed ^ 0x9E3779B97F4A7C15
    rng = np.random.default_rng(wseed if wseed is not None else None)
    noisy_weight = weight + sigma * float(rng.standard_normal())
    weight = min(max(noisy_weight, dp_config["min_weight"]), dp_config["max_weight"])
    weight = weight / dp_config.get("weight_scaler", 1)
    if not add_weight_noise:
        weight = weight_
    return weight


def apply_global_dp(config, worker_trainer, num_clients_curr_iter,
                    curr_iter=0, select_grad=True, metric_logger=None,
                    seed: Optional[int] = None):
    """Server-side DP noise on the aggregated gradient
    (reference: privacy/__init__.py:128-151).  The seed must be identical
    on every rank (derived from the round number by the caller) so the
    replicated server updates stay in lockstep."""
    dp_config = config.get("dp_config", None)
    if dp_config is None or not dp_config.get("enable_global_dp", False):
        return
    assert dp_config["enable_local_dp"]
    flat = worker_trainer.arena.grad
    sigma = dp_config["global_sigma"]
    max_grad = dp_config["max_grad"]
    noise_scale = sigma * max_grad / max(num_clients_curr_iter, 1)
    pre_norm = float(flat.norm())
    if seed is None:
        seed = 0xD1F00000 + curr_iter
    _noise(flat, noise_scale, seed)
    print_rank(f"global DP: noise_scale={noise_scale} grad_norm={pre_norm}",
               loglevel=logging.DEBUG)
    (metric_logger or log_metric)("Gradient Norm", pre_norm)


def update_privacy_accountant(config, num_clients, curr_iter,
                              num_clients_curr_iter):
    """RDP accounting of the subsampled Gaussian mechanism
    (reference: privacy/__init__.py:204-260)."""
    dp_config = config.get("dp_config", None)
    if dp_config is None or not (dp_config.get("enable_global_dp", False)
                                 or dp_config.get("enable_local_dp", False)):
        return None
    if dp_config.get("eps", 0) < 0 and dp_config.get("global_sigma") is None:
        # eps < 0 = clip-only local DP (reference privacy/__init__.py:
        # 167-171): no noise is added, so there is nothing to account
        return None
    from . import analysis as privacy_analysis

    K = 1
    B = num_clients_curr_iter
    n = num_clients
    T_steps = curr_iter + 1
    _delta = dp_config.get("delta", min(1e-7, 1.0 / (n * math.log(n))))
    if dp_config.get("global_sigma", None) is None:
        max_sensitivity = np.sqrt(dp_config["max_grad"] ** 2
                                  + dp_config["max_weight"] ** 2)
        noise_scale = compute_LDP_noise_std(dp_config["eps"], max_sensitivity,
                                            _delta)
        global_sigma = noise_scale * np.sqrt(B) / max_sensitivity
    else:
        global_sigma = dp_config["global_sigma"]
        noise_scale = global_sigma * dp_config["max_grad"] / B

    try:
        mu = K * B / n * math.sqrt(T_steps * math.exp((1.0 / global_sigma) ** 2 - 1))
    except OverflowError:
        mu = -1

    orders = ([1.25, 1.5, 1.75, 2.0, 2.25, 2.5, 3.0, 3.5, 4.0, 4.5]
              + list(range(5, 64)) + [128, 256, 512])
    q = B / n
    rdp = privacy_analysis.compute_rdp(q, global_sigma, T_steps, orders)
    rdp_epsilon, opt_order = privacy_analysis.get_privacy_spent(orders, rdp, _delta)

    props = {
        "dp_global_K": K, "dp_global_B": B, "dp_global_n": n,
        "dp_global_T": T_steps, "dp_sigma": global_sigma, "dp_global_mu": mu,
        "dp_epsilon_rdp": rdp_epsilon, "dp_opt_order": opt_order,
        "dp_delta": _delta, "dp_noise_scale": noise_scale,
    }
    print_rank(f"DP accounting: {json.dumps(props)}")
    for k, v in props.items():
        log_metric(k, v)
    return rdp_epsilon


# ---------------------------------------------------------------------------
# Exotic mechanisms (reference: privacy/__init__.py:18-102) — standalone
# ---------------------------------------------------------------------------

def _beta2betainc_ratio(a, x):
    from scipy.special import betainc
    return 1 / betainc(a, a, x)


def _efficient_m(d, gamma, p):
    from scipy.special import betaln
    alpha = (d - 1) / 2
    tau = (1 + gamma) / 2
    log_m1 = alpha * np.log(1 - gamma ** 2) - (d - 2) * np.log(2) - np.log(d - 1)
    log_m2 = (np.log(p / (_beta2betainc_ratio(alpha, tau) - 1) - (1 - p))
              + np.log(_beta2betainc_ratio(alpha, tau)) - betaln(alpha, alpha))
    return np.exp(log_m1 + log_m2)


def privacy_parameters(eps0, eps, d):
    exp_eps0 = np.exp(eps0)
    exp_eps = np.exp(eps)
    p0 = 1 if np.isinf(exp_eps0) else exp_eps0 / (1 + exp_eps0)
    if np.isinf(exp_eps):
        gamma = math.sqrt(math.pi / (2 * (d - 1)))
    else:
        gamma = ((exp_eps - 1) / (exp_eps + 1)) * math.sqrt(math.pi / (2 * (d - 1)))
    return p0, gamma


def private_unit2(grad: torch.Tensor, gamma: float, prob: float):
    """PrivUnit₂ mechanism on a unit vector (reference: privacy/__init__.py:51-65)."""
    np.testing.assert_almost_equal(float(grad.norm()), 1, decimal=5)
    assert prob >= 0.5 and 0 <= gamma <= 1
    p = torch.rand(())
    while True:
        V = torch.normal(0, 1, grad.shape, device=grad.device)
        V = V / V.norm()
        dot = torch.dot(V, grad)
        if (dot >= gamma and p < prob) or (dot < gamma and p >= prob):
            break
    m = _efficient_m(grad.shape[0], gamma, prob)
    return V / m


def add_private_unit2_noise(eps, grad):
    eps0 = 0.01 * eps
    eps1 = 0.99 * eps
    samp_prob, gamma = privacy_parameters(eps0, eps1, grad.shape[0])
    return private_unit2(grad, gamma, samp_prob)


def scalar_DP(r, eps, k, r_max):
    """Randomized-rounding scalar mechanism (reference: privacy/__init__.py:82-98)."""
    r = min(r, r_max)
    val = k * r / r_max
    f_val, c_val = math.floor(val), math.ceil(val)
    J = f_val if torch.rand(()) < (c_val - val) else c_val
    exp_eps = np.exp(eps)
    rand_prob = exp_eps / (exp_eps + k)
    if torch.rand(()) >= rand_prob:
        while True:
            J_ = int(torch.randint(0, k + 1, ()).item())
            if J != J_:
                J = J_
                break
    a = ((exp_eps + k) / (exp_eps - 1)) * (r_max / k)
    b = (k * (k + 1)) / (2 * (exp_eps + k))
    return a * (J - b)


def laplace_noise(max_sens, eps, vocab_size):
    return np.random.laplace(0.0, max_sens / eps, vocab_size)


# legacy helpers kept for API parity (arena makes them trivial)
def unroll_network(named_params, select_grad=False):
    params_ids, flat_params = {}, []
    cur = 0
    for n, p in named_params:
        dat = p.grad if select_grad else p.data
        flat_params.append(dat.reshape(-1))
        params_ids[n] = (cur, cur + flat_params[-1].shape[0])
        cur = params_ids[n][1]
    return torch.cat(flat_params), params_ids


def update_network(named_params, params_ids, flat_params, apply_to_grad=False):
    for n, p in named_params:
        s, e = params_ids[n]
        if apply_to_grad:
            p.grad.copy_(flat_params[s:e].view(*p.grad.shape))
        else:
            p.data.copy_(flat_params[s:e].view(*p.data.shape))
