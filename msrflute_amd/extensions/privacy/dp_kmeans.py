"""Differentially-private K-Means.

Reference: extensions/privacy/dp_kmeans.py (sphere-packing center init at
:23-47, per-iteration Gaussian mechanism on centers+counts at :49-73,
sklearn Lloyd-iteration monkey-patch at :141-188).  Standalone utility —
like the reference, it is not wired into the FL round loop.

MI355X-native divergence: instead of monkey-patching sklearn's private
``lloyd_iter_chunked_dense``, this implements the DP Lloyd iteration
directly (numpy/torch), which is equivalent and keeps sklearn internals
out of the trusted path.  Each iteration: clip sample weights, rescale
samples to ``max_cluster_l2``, assign, then noise the weighted center sums
and cluster counts with the Gaussian mechanism.
"""

from __future__ import annotations

import numpy as np


def _sample_ball(n_dim, r, rng):
    """Uniform sample inside the L2 ball of radius r."""
    v = rng.standard_normal(n_dim)
    v /= np.linalg.norm(v) + 1e-12
    return v * r * rng.random() ** (1.0 / n_dim)


def sphere_packing_initialization(n_clusters, n_dim, min_cluster_radius,
                                  max_space_size, max_failed_cases=100,
                                  rng=None, verbose=False):
    """Data-independent center init: rejection-sample centers at pairwise
    distance >= 2a inside the radius-(R-a) ball, halving a on repeated
    failure (reference: dp_kmeans.py:23-47)."""
    rng = rng or np.random.default_rng()
    a, R = min_cluster_radius, max_space_size
    centers = np.empty((n_clusters, n_dim))
    cid = fails = 0
    r = R - a
    while cid < n_clusters:
        v = _sample_ball(n_dim, r, rng)
        if cid > 0 and np.min(
                np.linalg.norm(centers[:cid] - v, axis=-1)) < 2 * a:
            fails += 1
            if fails >= max_failed_cases:
                fails, cid = 0, 0
                a /= 2
                r = R - a
                if verbose:
                    print(f"halving min_cluster_radius to {a}")
            continue
        centers[cid] = v
        cid += 1
    return centers, a


def _noise_centers(center_sums, counts, eps, delta, max_cluster_l2,
                   max_sample_weight, cluster_to_weight_ratio, rng):
    """Gaussian mechanism over the concatenated (weighted-sum, count)
    per-cluster vector (reference: dp_kmeans.py:49-73)."""
    scaler = 1.0
    if cluster_to_weight_ratio > 0:
        scaler = max_cluster_l2 / (max_sample_weight * cluster_to_weight_ratio)
    msw = max_sample_weight * scaler
    sens = np.sqrt(max_cluster_l2 ** 2 + msw ** 2)
    sigma = np.sqrt(2 * np.log(1.25 / delta)) * sens / eps
    noisy_sums = center_sums + rng.normal(scale=sigma, size=center_sums.shape)
    noisy_counts = np.maximum(
        1e-10, counts * scaler + rng.normal(scale=sigma, size=counts.shape)
    ) / scaler
    return noisy_sums / noisy_counts.reshape(-1, 1), noisy_counts, sigma


def dp_kmeans(X, n_clusters, eps, max_cluster_l2, sample_weight=None,
              max_sample_weight=1.0, cluster_to_weight_ratio=-1.0,
              delta=1e-7, max_iter=10, seed=0, verbose=False):
    """Run DP K-Means.  Returns (centers, labels, spent_sigma_per_iter).

    Privacy model matches the reference: inputs are rescaled into the
    ``max_cluster_l2`` ball, sample weights clipped to
    ``max_sample_weight``, and every Lloyd iteration releases noised
    sums/counts — so eps/delta are *per iteration* (compose across
    ``max_iter`` with the caller's accountant).
    """
    X = np.asarray(X, dtype=np.float64)
    n, d = X.shape
    rng = np.random.default_rng(seed)
    w = np.ones(n) if sample_weight is None else \
        np.minimum(np.asarray(sample_weight, dtype=np.float64),
                   max_sample_weight)

    # rescale each row into the max_cluster_l2 ball (reference clips by
    # scaling rows over the limit)
    norms = np.linalg.norm(X, axis=1, keepdims=True)
    scale = np.minimum(1.0, max_cluster_l2 / np.maximum(norms, 1e-12))
    Xs = X * scale

    centers, _ = sphere_packing_initialization(
        n_clusters, d, max_cluster_l2 / (2 * n_clusters ** (1.0 / d)),
        max_cluster_l2, rng=rng, verbose=verbose)

    sigmas = []
    labels = np.zeros(n, dtype=np.int64)
    for _ in range(max_iter):
        d2 = ((Xs[:, None, :] - centers[None, :, :]) ** 2).sum(-1)
        labels = np.argmin(d2, axis=1)
        sums = np.zeros((n_clusters, d))
        counts = np.full(n_clusters, 1e-10)
        np.add.at(sums, labels, Xs * w[:, None])
        np.add.at(counts, labels, w)
        centers, counts, sigma = _noise_centers(
            sums, counts, eps, delta, max_cluster_l2, max_sample_weight,
            cluster_to_weight_ratio, rng)
        sigmas.append(sigma)
    return centers, labels, sigmas
