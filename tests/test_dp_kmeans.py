"""DP-KMeans extension tests (reference: extensions/privacy/dp_kmeans.py)."""

import numpy as np

from msrflute_amd.extensions.privacy.dp_kmeans import (
    dp_kmeans, sphere_packing_initialization)


def test_sphere_packing_centers_are_separated():
    rng = np.random.default_rng(0)
    centers, a = sphere_packing_initialization(8, 4, 0.2, 1.0, rng=rng)
    assert centers.shape == (8, 4)
    d = np.linalg.norm(centers[:, None] - centers[None, :], axis=-1)
    d += np.eye(8) * 10
    assert d.min() >= 2 * a - 1e-9
    assert np.linalg.norm(centers, axis=1).max() <= 1.0 + 1e-9


def test_dp_kmeans_recovers_clusters_with_large_eps():
    rng = np.random.default_rng(1)
    c0 = rng.standard_normal((3, 5))
    X = np.concatenate([c + 0.05 * rng.standard_normal((200, 5)) for c in c0])
    centers, labels, sigmas = dp_kmeans(X, 3, eps=1e6, max_cluster_l2=10.0,
                                        max_iter=8, seed=2)
    assert centers.shape == (3, 5)
    assert len(sigmas) == 8
    # with eps huge, noise ~0: cluster assignment should be near-pure
    purity = 0
    for k in range(3):
        seg = labels[k * 200:(k + 1) * 200]
        purity += np.bincount(seg, minlength=3).max()
    assert purity / 600 > 0.95


def test_dp_kmeans_noise_scales_with_eps():
    X = np.random.default_rng(3).standard_normal((50, 4))
    _, _, s_hi = dp_kmeans(X, 2, eps=1.0, max_cluster_l2=1.0, max_iter=1)
    _, _, s_lo = dp_kmeans(X, 2, eps=10.0, max_cluster_l2=1.0, max_iter=1)
    assert s_hi[0] > s_lo[0] * 5
