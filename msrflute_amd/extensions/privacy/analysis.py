"""Rényi DP accounting for the subsampled Gaussian mechanism.

Implements the standard RDP accountant (Mironov 2017, "Rényi Differential
Privacy"; Mironov/Talwar/Zhang 2019, "Rényi Differential Privacy of the
Sampled Gaussian Mechanism", arXiv:1908.10530) used by the reference
(extensions/privacy/analysis.py:245-308): ``compute_rdp`` evaluates the
per-order RDP of one sampled-Gaussian step × number of steps, and
``get_privacy_spent`` converts the RDP curve to an (ε, δ) pair via
ε = min_α [ rdp_α − log(δ)/(α−1) ].

Written from the published formulas: integer orders use the exact binomial
expansion in log space; fractional orders use the two-part integral split
at z₀ = σ²·log(1/q − 1) + 1/2 with erfc-based closed forms.
"""

from __future__ import annotations

import math
import sys

import numpy as np
from scipy import special


def _log_add(a, b):
    """log(exp(a) + exp(b)) stably."""
    if a == -math.inf:
        return b
    if b == -math.inf:
        return a
    m, n = max(a, b), min(a, b)
    return m + math.log1p(math.exp(n - m))


def _log_sub(a, b):
    """log(exp(a) - exp(b)) stably, a >= b."""
    if b == -math.inf:
        return a
    if a == b:
        return -math.inf
    return a + math.log1p(-math.exp(b - a))


def _log_erfc(x):
    return math.log(2) + special.log_ndtr(-x * 2 ** 0.5)


def _compute_log_a_int(q, sigma, alpha: int):
    """log A_α for integer α: A = Σ_k C(α,k) (1-q)^(α-k) q^k e^{k(k-1)/2σ²}."""
    log_a = -math.inf
    for k in range(alpha + 1):
        log_coef = (math.lgamma(alpha + 1) - math.lgamma(k + 1)
                    - math.lgamma(alpha - k + 1)
                    + k * math.log(q) + (alpha - k) * math.log(1 - q))
        log_a = _log_add(log_a, log_coef + (k * k - k) / (2 * sigma ** 2))
    return log_a


def _compute_log_a_frac(q, sigma, alpha):
    """log A_α for fractional α via the infinite binomial series split into
    the two integrals at z0 (arXiv:1908.10530, §3.3)."""
    log_a0, log_a1 = -math.inf, -math.inf
    i = 0
    z0 = sigma ** 2 * math.log(1 / q - 1) + 0.5
    while True:
        coef = special.binom(alpha, i)
        log_coef = math.log(abs(coef)) if coef != 0 else -math.inf
        j = alpha - i
        log_t0 = log_coef + i * math.log(q) + j * math.log(1 - q)
        log_t1 = log_coef + j * math.log(q) + i * math.log(1 - q)
        log_e0 = 0.5 * math.log(0.5) + _log_erfc((i - z0) / (math.sqrt(2) * sigma))
        log_e1 = 0.5 * math.log(0.5) + _log_erfc((z0 - j) / (math.sqrt(2) * sigma))
        log_s0 = log_t0 + (i * i - i) / (2 * sigma ** 2) + log_e0
        log_s1 = log_t1 + (j * j - j) / (2 * sigma ** 2) + log_e1
        if coef > 0:
            log_a0 = _log_add(log_a0, log_s0)
            log_a1 = _log_add(log_a1, log_s1)
        else:
            log_a0 = _log_sub(log_a0, log_s0)
            log_a1 = _log_sub(log_a1, log_s1)
        i += 1
        if max(log_s0, log_s1) < -30:
            break
        if i > 1000:
            break
    return _log_add(log_a0, log_a1)


def _compute_rdp_order(q, sigma, alpha):
    """RDP of one sampled-Gaussian step at order alpha."""
    if q == 0:
        return 0.0
    if q == 1.0:
        return alpha / (2 * sigma ** 2)
    if math.isinf(alpha):
        return math.inf
    if float(alpha).is_integer():
        log_a = _compute_log_a_int(q, sigma, int(alpha))
    else:
        log_a = _compute_log_a_frac(q, sigma, alpha)
    return log_a / (alpha - 1)


def compute_rdp(q, noise_multiplier, steps, orders):
    """RDP at each order for ``steps`` compositions of the sampled Gaussian
    with sampling rate q and noise multiplier sigma
    (reference: analysis.py:245-269)."""
    if np.isscalar(orders):
        rdp = _compute_rdp_order(q, noise_multiplier, orders)
    else:
        rdp = np.array([_compute_rdp_order(q, noise_multiplier, a)
                        for a in orders])
    return rdp * steps


def get_privacy_spent(orders, rdp, target_delta):
    """(ε, optimal_order) from the RDP curve
    (reference: analysis.py:272-308): ε = min over α of rdp − log δ/(α−1)."""
    orders_vec = np.atleast_1d(orders)
    rdp_vec = np.atleast_1d(rdp)
    if len(orders_vec) != len(rdp_vec):
        raise ValueError("orders and rdp must have the same length")
    eps = rdp_vec - math.log(target_delta) / (orders_vec - 1)
    eps = np.where(np.isfinite(eps), eps, np.inf)
    idx = int(np.nanargmin(eps))
    return float(eps[idx]), float(orders_vec[idx])
