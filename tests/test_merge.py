import math

import numpy as np
import pytest

from cuda_gmm_mpi_amd.models.merge import (
    HostClusters, add_clusters, cluster_distance, eliminate_empty_clusters,
    reduce_order,
)
from cuda_gmm_mpi_amd.ops.invert import invert_cpu


def make_clusters(rng, k, d):
    means = rng.standard_normal((k, d)).astype(np.float32) * 2
    rs = []
    for _ in range(k):
        a = rng.standard_normal((d, d))
        rs.append((a @ a.T + d * np.eye(d)).astype(np.float32))
    r = np.stack(rs)
    rinv = np.stack([np.linalg.inv(r[c]) for c in range(k)]).astype(np.float32)
    n = rng.uniform(10, 100, k).astype(np.float32)
    pi = (n / n.sum()).astype(np.float32)
    const = np.array(
        [-d / 2 * math.log(2 * math.pi)
         - 0.5 * np.linalg.slogdet(r[c].astype(np.float64))[1]
         for c in range(k)], dtype=np.float32)
    return HostClusters(
        N=n, pi=pi, constant=const,
        avgvar=np.full(k, 0.5, dtype=np.float32), means=means, R=r, Rinv=rinv,
    )


def test_add_clusters_moments(rng):
    """Merged N/pi additive; means = weighted avg; R = law of total
    covariance in the reference's exact form (gaussian.cu:1210-1241)."""
    hc = make_clusters(rng, 3, 4)
    m = add_clusters(hc, 0, 2, bug_compat=False)
    n1, n2 = hc.N[0], hc.N[2]
    wt1 = np.float32(n1 / (n1 + n2))
    wt2 = np.float32(1.0) - wt1
    assert m.N == pytest.approx(float(n1 + n2))
    assert m.pi == pytest.approx(float(hc.pi[0] + hc.pi[2]))
    np.testing.assert_allclose(
        m.means, wt1 * hc.means[0] + wt2 * hc.means[2], rtol=1e-6)
    d = hc.num_dimensions
    ref_r = np.zeros((d, d), dtype=np.float64)
    for i in range(d):
        for j in range(d):
            c1 = ((m.means[i] - hc.means[0, i]) * (m.means[j] - hc.means[0, j])
                  + hc.R[0, i, j]) * wt1
            c2 = ((m.means[i] - hc.means[2, i]) * (m.means[j] - hc.means[2, j])
                  + hc.R[2, i, j]) * wt2
            ref_r[i, j] = c1 + c2
    np.testing.assert_allclose(m.R, ref_r, rtol=1e-4, atol=1e-5)
    assert m.avgvar == pytest.approx(float(hc.avgvar[0]))


def test_bug_compat_constant_uses_log10(rng):
    """bug_compat (default) must reproduce the reference's log10
    determinant consumed as ln."""
    hc = make_clusters(rng, 2, 5)
    m_bug = add_clusters(hc, 0, 1, bug_compat=True)
    m_fix = add_clusters(hc, 0, 1, bug_compat=False)
    np.testing.assert_allclose(m_bug.R, m_fix.R)
    _, ld10 = invert_cpu(m_bug.R, base10=True)
    _, ldln = invert_cpu(m_fix.R, base10=False)
    d = hc.num_dimensions
    assert m_bug.constant == pytest.approx(
        -d * 0.5 * math.log(2 * math.pi) - 0.5 * ld10, rel=1e-5)
    assert m_fix.constant == pytest.approx(
        -d * 0.5 * math.log(2 * math.pi) - 0.5 * ldln, rel=1e-5)
    assert m_bug.constant != pytest.approx(m_fix.constant)


def test_cluster_distance_formula(rng):
    hc = make_clusters(rng, 2, 3)
    dist, merged = cluster_distance(hc, 0, 1)
    expect = (float(hc.N[0]) * float(hc.constant[0])
              + float(hc.N[1]) * float(hc.constant[1])
              - merged.N * merged.constant)
    assert dist == pytest.approx(expect)


def test_eliminate_empty_clusters(rng):
    hc = make_clusters(rng, 5, 3)
    hc.N[1] = 0.2
    hc.N[3] = 0.0
    kept = [hc.means[0].copy(), hc.means[2].copy(), hc.means[4].copy()]
    k = eliminate_empty_clusters(hc)
    assert k == 3
    for i, m in enumerate(kept):
        np.testing.assert_array_equal(hc.means[i], m)


def test_reduce_order_merges_closest(rng):
    hc = make_clusters(rng, 4, 3)
    # find expected min pair independently
    best = None
    for c1 in range(4):
        for c2 in range(c1 + 1, 4):
            dcopy = make_copy(hc)
            dist, _ = cluster_distance(dcopy, c1, c2)
            if best is None or dist < best[0]:
                best = (dist, c1, c2)
    k, c1, c2 = reduce_order(hc)
    assert k == 3
    assert (c1, c2) == (best[1], best[2])
    assert hc.num_clusters == 4  # storage unchanged; logical k shrinks


def make_copy(hc):
    return HostClusters(
        N=hc.N.copy(), pi=hc.pi.copy(), constant=hc.constant.copy(),
        avgvar=hc.avgvar.copy(), means=hc.means.copy(), R=hc.R.copy(),
        Rinv=hc.Rinv.copy(),
    )


def test_reduce_order_batched_matches_exhaustive(rng):
    from cuda_gmm_mpi_amd.models.merge import reduce_order_batched
    for trial in range(5):
        hc_a = make_clusters(rng, 8, 5)
        hc_b = make_copy(hc_a)
        k_a, c1_a, c2_a = reduce_order(hc_a)
        k_b, c1_b, c2_b = reduce_order_batched(hc_b)
        assert (k_a, c1_a, c2_a) == (k_b, c1_b, c2_b)
        np.testing.assert_allclose(hc_a.means[:k_a], hc_b.means[:k_b],
                                   rtol=1e-6)
        np.testing.assert_allclose(hc_a.R[:k_a], hc_b.R[:k_b], rtol=1e-6)
        np.testing.assert_array_equal(hc_a.N[:k_a], hc_b.N[:k_b])


def test_reduce_order_batched_no_bugcompat(rng):
    from cuda_gmm_mpi_amd.models.merge import reduce_order_batched
    hc_a = make_clusters(rng, 6, 4)
    hc_b = make_copy(hc_a)
    res_a = reduce_order(hc_a, bug_compat=False)
    res_b = reduce_order_batched(hc_b, bug_compat=False)
    assert res_a == res_b


def test_reduce_order_after_mass_elimination(rng):
    """5 clusters where 3 are empty: elimination leaves 2, which must then
    merge into 1 in the same reduce_order step (the path that can jump
    past target_num_clusters, gaussian.cu:866-907)."""
    hc = make_clusters(rng, 5, 3)
    hc.N[0] = 0.1
    hc.N[2] = 0.4
    hc.N[4] = 0.0
    n_live = hc.N[1] + hc.N[3]
    from cuda_gmm_mpi_amd.models.merge import reduce_order_batched
    for fn in (reduce_order, reduce_order_batched):
        k, c1, c2 = fn(make_copy(hc))
        assert k == 1 and (c1, c2) == (0, 1)
    k, _, _ = reduce_order(hc)
    assert hc.N[0] == pytest.approx(n_live, rel=1e-6)


def test_merge_identical_clusters_is_idempotent(rng):
    """Merging a cluster with an exact copy of itself must reproduce it:
    means and R unchanged (the cross terms vanish), N doubled."""
    hc = make_clusters(rng, 2, 3)
    for f in ("means", "R", "pi", "N", "avgvar", "constant"):
        getattr(hc, f)[1] = getattr(hc, f)[0]
    m = add_clusters(hc, 0, 1, bug_compat=False)
    assert m.N == pytest.approx(float(2 * hc.N[0]))
    np.testing.assert_allclose(m.means, hc.means[0], rtol=1e-6)
    np.testing.assert_allclose(m.R, hc.R[0], rtol=1e-5, atol=1e-6)


def test_cluster_distance_self_merge_is_zero(rng):
    """distance(a, a-copy) = N1·c1 + N2·c2 − (N1+N2)·c12 with c12 = c1
    when the merged cluster equals its parents — the scan can never
    prefer a worse pair over an exact duplicate."""
    from cuda_gmm_mpi_amd.models.merge import cluster_distance
    hc = make_clusters(rng, 3, 3)
    for f in ("means", "R", "pi", "N", "avgvar", "constant"):
        getattr(hc, f)[1] = getattr(hc, f)[0]
    dist, m = cluster_distance(hc, 0, 1, bug_compat=False)
    assert m.N == pytest.approx(float(2 * hc.N[0]))
    # c12 is recomputed from the merged R (fp32): allow fp noise only
    scale = abs(float(hc.N[0] * hc.constant[0])) + 1.0
    assert abs(dist) < 1e-4 * scale
