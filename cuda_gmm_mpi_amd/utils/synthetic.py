"""Synthetic FCS-shaped data generators.

BASELINE.json prescribes synthetic flow-cytometry-shaped data with
random-init mixture parameters (no network access for real datasets). FCS
channels are non-negative, O(10^2..10^3) scaled, cluster-structured.
"""
from __future__ import annotations

import numpy as np


def make_blobs(num_events: int, num_dims: int, num_clusters: int,
               seed: int = 0, scale: float = 250.0,
               spread: float = 25.0) -> tuple[np.ndarray, np.ndarray]:
    """Gaussian blobs with random means/covariances, FCS-like scale.

    Returns (data [N, D] float32, labels [N] int32).
    """
    rng = np.random.default_rng(seed)
    weights = rng.dirichlet(np.full(num_clusters, 5.0))
    counts = rng.multinomial(num_events, weights)
    means = rng.uniform(0.2 * scale, 3.0 * scale, size=(num_clusters, num_dims))
    data = np.empty((num_events, num_dims), dtype=np.float32)
    labels = np.empty(num_events, dtype=np.int32)
    pos = 0
    for c in range(num_clusters):
        n_c = int(counts[c])
        if n_c == 0:
            continue
        # Random SPD covariance with controlled condition number
        a = rng.standard_normal((num_dims, num_dims))
        q, _ = np.linalg.qr(a)
        eig = rng.uniform(0.3, 1.7, size=num_dims) * spread**2
        cov = (q * eig) @ q.T
        x = rng.multivariate_normal(means[c], cov, size=n_c)
        data[pos:pos + n_c] = x.astype(np.float32)
        labels[pos:pos + n_c] = c
        pos += n_c
    perm = rng.permutation(num_events)
    return data[perm], labels[perm]


def make_bench_data(num_events: int, num_dims: int, num_clusters: int,
                    seed: int = 1234) -> np.ndarray:
    """Deterministic benchmark dataset of the shape BASELINE.json names."""
    data, _ = make_blobs(num_events, num_dims, num_clusters, seed=seed)
    return data


def make_supported_blobs(num_events: int, num_dims: int, num_clusters: int,
                         seed: int = 0, scale: float = 250.0,
                         spread: float = 25.0
                         ) -> tuple[np.ndarray, np.ndarray]:
    """Equal-weight, separation-checked blobs: every cluster carries
    ~N/K events and mean pairs are at least 8*spread apart, so an MDL
    sweep's empty-cluster elimination cannot mass-kill clusters and the
    merge path steps K down one at a time (BASELINE config 5 needs the
    sweep to LAND on the target K, reference save path gaussian.cu:839).
    """
    rng = np.random.default_rng(seed)
    k, d = num_clusters, num_dims
    means = np.empty((k, d))
    have = 0
    min_dist = 8.0 * spread
    while have < k:
        cand = rng.uniform(0.2 * scale, 3.0 * scale, size=(k, d))
        for row in cand:
            if have == k:
                break
            if have == 0 or np.min(
                np.linalg.norm(means[:have] - row, axis=1)) >= min_dist:
                means[have] = row
                have += 1
    base, rem = divmod(num_events, k)
    counts = np.full(k, base, dtype=np.int64)
    counts[:rem] += 1
    data = np.empty((num_events, d), dtype=np.float32)
    labels = np.empty(num_events, dtype=np.int32)
    pos = 0
    for c in range(k):
        n_c = int(counts[c])
        a = rng.standard_normal((d, d))
        q, _ = np.linalg.qr(a)
        eig = rng.uniform(0.3, 1.7, size=d) * spread**2
        cov = (q * eig) @ q.T
        x = rng.multivariate_normal(means[c], cov, size=n_c)
        data[pos:pos + n_c] = x.astype(np.float32)
        labels[pos:pos + n_c] = c
        pos += n_c
    perm = rng.permutation(num_events)
    data, labels = data[perm], labels[perm]
    # The reference seeds cluster c's mean from the event at row
    # c*(N-1)/(K-1) (gaussian.cu:108-123). Arrange the row order so each
    # seed position holds an event from a DISTINCT cluster — every blob
    # starts seeded, so EM at K0=K keeps all clusters supported and the
    # MDL sweep cannot mass-eliminate its way past the target.
    if k > 1:
        seed_rows = np.round(np.arange(k) * (num_events - 1.0)
                             / (k - 1.0)).astype(np.int64)
        used = set(seed_rows.tolist())
        for c, p in enumerate(seed_rows):
            if labels[p] == c:
                continue
            cand = np.nonzero(labels == c)[0]
            swap = next(int(s) for s in cand
                        if int(s) not in used or int(s) == int(p))
            data[[p, swap]] = data[[swap, p]]
            labels[[p, swap]] = labels[[swap, p]]
            used.add(swap)
    return data, labels
