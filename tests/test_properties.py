"""Property-based tests (hypothesis) over the IO and numerics layers."""
import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from cuda_gmm_mpi_amd.ops import cpu_reference as cpu
from cuda_gmm_mpi_amd.utils import io as gio


@settings(max_examples=25, deadline=None)
@given(
    n=st.integers(1, 40), d=st.integers(1, 8),
    seed=st.integers(0, 2**31 - 1),
)
def test_bin_roundtrip_property(tmp_path_factory, n, d, seed):
    rng = np.random.default_rng(seed)
    data = (rng.standard_normal((n, d)) * 100).astype(np.float32)
    path = str(tmp_path_factory.mktemp("bin") / "x.bin")
    gio.write_bin(path, data)
    np.testing.assert_array_equal(gio.read_bin(path), data)


@settings(max_examples=25, deadline=None)
@given(k=st.integers(1, 12), n=st.integers(1, 200),
       scale=st.floats(0.1, 50), seed=st.integers(0, 2**31 - 1))
def test_posteriors_always_normalized(k, n, scale, seed):
    rng = np.random.default_rng(seed)
    logw = torch.from_numpy(
        (rng.standard_normal((k, n)) * scale).astype(np.float32))
    w, lik = cpu.estep_posteriors(logw.clone())
    assert torch.isfinite(w).all()
    np.testing.assert_allclose(w.sum(dim=0).numpy(), np.ones(n), rtol=1e-4)
    assert float(lik) == float(lik)  # finite, not NaN


@settings(max_examples=30, deadline=None)
@given(d=st.integers(1, 16), seed=st.integers(0, 2**31 - 1),
       ridge=st.floats(1e-2, 100.0))
def test_lu_invert_property(d, seed, ridge):
    """No-pivot LU inverse + logdet across conditioning: ridge down to
    1e-2 gives condition numbers up to ~1e5 (fp32 headroom shrinks with
    cond, so tolerances scale with the inverse's magnitude)."""
    rng = np.random.default_rng(seed)
    a = rng.standard_normal((d, d))
    m = (a @ a.T + ridge * d * np.eye(d)).astype(np.float32)
    inv, logdet = cpu.lu_invert_nopivot(torch.from_numpy(m)[None])
    ref = np.linalg.slogdet(m.astype(np.float64))[1]
    assert abs(float(logdet[0]) - ref) < max(1e-2, 2e-3 * abs(ref))
    scale = max(1.0, float(np.abs(inv[0].numpy()).max()))
    np.testing.assert_allclose(
        (inv[0].numpy() @ m), np.eye(d), atol=5e-4 * scale * d)


@settings(max_examples=15, deadline=None)
@given(st.lists(st.floats(-1e6, 1e6, allow_nan=False, width=32),
                min_size=2, max_size=20),
       st.integers(2, 6))
def test_csv_roundtrip_values(values, d):
    import tempfile, os
    rows = [values[i:i + d] for i in range(0, len(values) - d + 1, d)]
    if not rows:
        return
    with tempfile.TemporaryDirectory() as td:
        path = os.path.join(td, "x.csv")
        with open(path, "w") as f:
            f.write(",".join("h" * d) or "h")
            f.write("\n")
            for r in rows:
                f.write(",".join(f"{v!r}" for v in r) + "\n")
        data = gio.read_csv(path)
    assert data.shape == (len(rows), d)
    np.testing.assert_allclose(
        data, np.array(rows, dtype=np.float32), rtol=1e-6)


@given(n=st.integers(0, 10_000_000), world=st.integers(1, 64))
@settings(max_examples=200, deadline=None)
def test_shard_bounds_partition(n, world):
    """shard_bounds partitions [0, n) exactly: contiguous, ordered, no
    dropped or double-counted tail events (the reference's remainder bugs,
    SURVEY 2.6 #4). The split follows the reference scheme — every rank
    gets floor(n/world) events and the LAST rank absorbs the remainder
    (gaussian.cu:348-352) — which gather_memberships also assumes."""
    from cuda_gmm_mpi_amd.parallel.dist import shard_bounds
    prev_end = 0
    per = n // world
    for r in range(world):
        s, e = shard_bounds(n, world, r)
        assert s == prev_end and e >= s
        assert e - s == (per if r < world - 1 else n - per * (world - 1))
        prev_end = e
    assert prev_end == n


@given(
    n=st.integers(12, 300), d=st.integers(1, 8), k=st.integers(1, 12),
    tgt_mode=st.integers(0, 2), diag=st.booleans(), bug=st.booleans(),
    center=st.booleans(), seed=st.integers(0, 1000),
)
@settings(max_examples=25, deadline=None)
def test_sweep_config_fuzz(n, d, k, tgt_mode, diag, bug, center, seed):
    """Whole-sweep fuzz over the config space on tiny CPU problems: any
    valid (N, D, K, target, diag, bug_compat, center) combination must
    complete with a finite best model — no crashes in seeding, EM,
    elimination, merging, or bookkeeping."""
    from cuda_gmm_mpi_amd.engine import build_engine
    from cuda_gmm_mpi_amd.utils.config import GmmConfig
    from cuda_gmm_mpi_amd.utils.synthetic import make_blobs

    target = 0 if tgt_mode == 0 else (1 if tgt_mode == 1 else min(k, 3))
    data, _ = make_blobs(n, d, max(2, min(k, 4)), seed=seed)
    cfg = GmmConfig(num_clusters=k, target_num_clusters=target,
                    min_iters=1, max_iters=2, diag_only=diag,
                    bug_compat=bug, center_data=center)
    res = build_engine(data, cfg, device="cpu").sweep()
    assert 1 <= res.num_clusters <= k
    assert np.isfinite(res.min_rissanen)
    assert all(np.isfinite(v) for v in res.rissanen_by_k.values())
    st_best = res.state
    assert int(st_best.num_clusters) == res.num_clusters
    assert np.isfinite(st_best.means.numpy()).all()
