"""GPU end-to-end: full EM / MDL sweep on MI355X vs the CPU golden path."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from cuda_gmm_mpi_amd.engine import build_engine  # noqa: E402
from cuda_gmm_mpi_amd.utils.config import GmmConfig  # noqa: E402
from cuda_gmm_mpi_amd.utils.synthetic import make_blobs  # noqa: E402


def run(device, cfg, data, k=None):
    eng = build_engine(data, cfg, device=device)
    lik = eng.run_em(k or cfg.num_clusters)
    return eng, lik


def test_em_gpu_matches_cpu():
    data, _ = make_blobs(20000, 6, 4, seed=31)
    cfg = GmmConfig(num_clusters=4, target_num_clusters=4,
                    min_iters=15, max_iters=15)
    eng_c, lik_c = run("cpu", cfg, data)
    eng_g, lik_g = run("cuda", cfg, data)
    assert lik_g == pytest.approx(lik_c, rel=1e-4)
    np.testing.assert_allclose(eng_g.state.N.cpu().numpy(),
                               eng_c.state.N.numpy(), rtol=1e-3)
    np.testing.assert_allclose(eng_g.state.means.cpu().numpy(),
                               eng_c.state.means.numpy(), rtol=1e-2,
                               atol=1e-2)
    np.testing.assert_allclose(eng_g.state.R.cpu().numpy(),
                               eng_c.state.R.numpy(), rtol=5e-2, atol=5e-2)


def test_em_gpu_bf16_estep_close():
    data, _ = make_blobs(20000, 24, 4, seed=37)
    cfg32 = GmmConfig(num_clusters=4, target_num_clusters=4,
                      min_iters=10, max_iters=10)
    cfg16 = GmmConfig(num_clusters=4, target_num_clusters=4,
                      min_iters=10, max_iters=10, estep_dtype="bf16")
    _, lik32 = run("cuda", cfg32, data)
    _, lik16 = run("cuda", cfg16, data)
    assert lik16 == pytest.approx(lik32, rel=2e-2)


def test_sweep_gpu_matches_cpu():
    data, _ = make_blobs(8000, 3, 3, seed=41)
    cfg = GmmConfig(num_clusters=6, target_num_clusters=2,
                    min_iters=5, max_iters=5)
    eng_c = build_engine(data, cfg, device="cpu")
    res_c = eng_c.sweep()
    # strict trajectory comparison on the VALU E-step (same math as CPU);
    # the fused f32-MFMA path reorganizes the quadratic form (Cholesky
    # factors), which can legitimately flip near-tie merge decisions
    eng_g = build_engine(data, cfg, device="cuda")
    eng_g.use_fused_estep = False
    eng_g.mfac32 = None
    res_g = eng_g.sweep()
    assert res_g.num_clusters == res_c.num_clusters
    assert res_g.min_rissanen == pytest.approx(res_c.min_rissanen, rel=1e-3)
    np.testing.assert_allclose(res_g.state.means.cpu().numpy(),
                               res_c.state.means.numpy(), rtol=1e-2,
                               atol=1e-2)
    # fused f32-MFMA sweep: with the post-merge factor refresh it tracks
    # the CPU trajectory (near-tie merge flips aside)
    eng_f = build_engine(data, cfg, device="cuda")
    res_f = eng_f.sweep()
    assert res_f.num_clusters == res_c.num_clusters
    assert res_f.min_rissanen == pytest.approx(res_c.min_rissanen, rel=2e-3)


def test_gpu_em_deterministic():
    data, _ = make_blobs(30000, 24, 8, seed=43)
    cfg = GmmConfig(num_clusters=8, target_num_clusters=8,
                    min_iters=8, max_iters=8)
    _, lik1 = run("cuda", cfg, data)
    _, lik2 = run("cuda", cfg, data)
    assert lik1 == lik2  # bitwise: deterministic chunked reductions


def test_gpu_big_d_config4_shape():
    """Config 4 shape (scaled down): K=256, D=128."""
    rng = np.random.default_rng(47)
    data = rng.standard_normal((60000, 128)).astype(np.float32) * 10
    cfg = GmmConfig(num_clusters=256, target_num_clusters=256,
                    min_iters=2, max_iters=2)
    eng, lik = run("cuda", cfg, data)
    assert np.isfinite(lik)
    assert float(eng.posteriors(256).sum(dim=0).mean()) == pytest.approx(1.0, abs=1e-3)


def test_gpu_cli_end_to_end(tmp_path):
    from cuda_gmm_mpi_amd.cli import main
    from cuda_gmm_mpi_amd.utils import io as gio
    data, _ = make_blobs(5000, 4, 3, seed=53)
    binpath = str(tmp_path / "d.bin")
    gio.write_bin(binpath, data)
    out = str(tmp_path / "o")
    rc = main(["3", binpath, out, "3", "--min-iters", "5", "--max-iters", "5",
               "--device", "cuda"])
    assert rc == 0
    assert open(out + ".summary").read().count("Cluster #") == 3
    lines = open(out + ".results").read().splitlines()
    assert len(lines) == 5000


def test_gpu_cli_sweep_checkpoint_metrics(tmp_path):
    """Full user path on GPU: bin input, MDL sweep, checkpoint, metrics."""
    import json
    from cuda_gmm_mpi_amd.cli import main
    from cuda_gmm_mpi_amd.utils import io as gio
    data, _ = make_blobs(40000, 10, 6, seed=71)
    binpath = str(tmp_path / "d.bin")
    gio.write_bin(binpath, data)
    out = str(tmp_path / "o")
    rc = main(["8", binpath, out, "4", "--min-iters", "10", "--max-iters",
               "10", "--device", "cuda", "--estep-dtype", "bf16",
               "--mstep-precision", "bf16x3",
               "--checkpoint-dir", str(tmp_path / "ck"),
               "--metrics-out", str(tmp_path / "m.json")])
    assert rc == 0
    m = json.load(open(tmp_path / "m.json"))
    # empty-cluster elimination may jump past the target K, in which case
    # the first saved model is kept (reference semantics, gaussian.cu:839)
    kk = m["num_clusters"]
    assert kk == 4 or kk == 8
    assert open(out + ".summary").read().count("Cluster #") == kk
    lines = open(out + ".results").read().splitlines()
    assert len(lines) == 40000
    assert len(lines[0].split("\t")[1].split(",")) == kk
    import os
    assert os.path.exists(tmp_path / "ck" / "gmm_sweep.npz")


def test_em_gpu_fp32_fused_matches_valu():
    """The f32-MFMA fused E-step path vs the VALU path: same trajectory."""
    import os
    data, _ = make_blobs(20000, 24, 4, seed=77)
    cfg = GmmConfig(num_clusters=4, target_num_clusters=4,
                    min_iters=10, max_iters=10)  # fp32 everywhere
    eng_f = build_engine(data, cfg, device="cuda")
    assert eng_f.use_fused_estep and eng_f.mfac32 is not None
    lik_f = eng_f.run_em(4)
    # force the VALU path
    eng_v = build_engine(data, cfg, device="cuda")
    eng_v.use_fused_estep = False
    eng_v.mfac32 = None
    lik_v = eng_v.run_em(4)
    assert lik_f == pytest.approx(lik_v, rel=1e-4)
    np.testing.assert_allclose(eng_f.state.means.cpu().numpy(),
                               eng_v.state.means.cpu().numpy(),
                               rtol=1e-3, atol=1e-2)


def test_sweep_bf16_fused_post_merge_factors():
    """bf16 fused sweep must refresh factor tables after each merge: the
    post-merge E-step otherwise runs on stale/misaligned factors."""
    data, _ = make_blobs(12000, 8, 4, seed=83)
    cfg = GmmConfig(num_clusters=6, target_num_clusters=3,
                    min_iters=5, max_iters=5, estep_dtype="bf16",
                    mstep_precision="bf16x3")
    cfg_ref = GmmConfig(num_clusters=6, target_num_clusters=3,
                        min_iters=5, max_iters=5)
    eng_c = build_engine(data, cfg_ref, device="cpu")
    res_c = eng_c.sweep()
    eng_g = build_engine(data, cfg, device="cuda")
    assert eng_g.use_fused_estep
    res_g = eng_g.sweep()
    assert res_g.num_clusters == res_c.num_clusters
    # bf16-class tolerance on the score trajectory
    assert res_g.min_rissanen == pytest.approx(res_c.min_rissanen, rel=2e-2)


def test_gpu_soak_100_iterations_monotone():
    """Reference-length EM (100 iterations) at scale: likelihood monotone
    (exact fp32 path) and posteriors normalized throughout."""
    data, _ = make_blobs(500_000, 24, 64, seed=97)
    cfg = GmmConfig(num_clusters=64, target_num_clusters=64,
                    min_iters=1, max_iters=1)
    eng = build_engine(data, cfg, device="cuda")
    liks = [eng._reduce_likelihood(eng._estep(64))]
    for _ in range(100):
        eng.em_iteration(64)
        liks.append(float(eng._lik_dev.item()))
    liks = np.array(liks)
    tol = abs(liks[-1]) * 1e-6
    assert (np.diff(liks) > -tol).all(), \
        f"non-monotone at {np.argmin(np.diff(liks))}"
    s = eng.posteriors(64).sum(dim=0)
    assert float((s - 1).abs().max()) < 1e-3


def test_gpu_soak_bf16_close_to_fp32():
    """bf16 fast path tracks the fp32 path over a 50-iteration run."""
    data, _ = make_blobs(300_000, 24, 32, seed=101)
    liks = {}
    for name, ed, mp in (("fp32", "fp32", "fp32"),
                         ("bf16", "bf16", "bf16x3")):
        cfg = GmmConfig(num_clusters=32, target_num_clusters=32,
                        min_iters=50, max_iters=50, estep_dtype=ed,
                        mstep_precision=mp)
        eng = build_engine(data, cfg, device="cuda")
        liks[name] = eng.run_em(32)
    assert liks["bf16"] == pytest.approx(liks["fp32"], rel=5e-3)


def test_gpu_k1_and_d1():
    """Degenerate shapes on every GPU kernel path."""
    rng = np.random.default_rng(9)
    d1 = np.concatenate([rng.normal(0, 1, 2000),
                         rng.normal(8, 1, 2000)]).astype(np.float32)
    for ed in ("fp32", "bf16"):
        cfg = GmmConfig(num_clusters=2, target_num_clusters=2,
                        min_iters=10, max_iters=10, estep_dtype=ed)
        eng = build_engine(d1.reshape(-1, 1), cfg, device="cuda")
        lik = eng.run_em(2)
        assert np.isfinite(lik)
        mu = np.sort((eng.state.means.cpu() + eng.center.cpu()).numpy().ravel())
        np.testing.assert_allclose(mu, [0.0, 8.0], atol=0.5)
    data, _ = make_blobs(5000, 4, 2, seed=3)
    cfg = GmmConfig(num_clusters=1, target_num_clusters=1,
                    min_iters=4, max_iters=4, estep_dtype="bf16",
                    mstep_precision="bf16x3")
    eng = build_engine(data, cfg, device="cuda")
    res = eng.sweep()
    assert res.num_clusters == 1


def test_gpu_max_clusters_512():
    """MAX_CLUSTERS=512 (reference bound) runs through the large-K path —
    since the online-softmax redesign the fused kernel covers any K at
    D <= 31 (no K-sized logw LDS buffer)."""
    data, _ = make_blobs(60000, 16, 32, seed=111)
    cfg = GmmConfig(num_clusters=512, target_num_clusters=512,
                    min_iters=2, max_iters=2, estep_dtype="bf16",
                    mstep_precision="bf16x3")
    eng = build_engine(data, cfg, device="cuda")
    assert eng.use_fused_estep and not eng.use_big_estep
    lik = eng.run_em(512)
    assert np.isfinite(lik)
    s = eng.posteriors(512).sum(dim=0)
    assert float((s - 1).abs().max()) < 1e-2
    assert abs(float(eng.state.pi.sum()) - 1.0) < 1e-2


def test_gpu_diag_only_engine():
    """DIAG_ONLY mode end-to-end on GPU (diag kernels + diag finalize)."""
    data, _ = make_blobs(20000, 6, 3, seed=121)
    cfg = GmmConfig(num_clusters=3, target_num_clusters=3,
                    min_iters=8, max_iters=8, diag_only=True)
    eng_c = build_engine(data, cfg, device="cpu")
    lik_c = eng_c.run_em(3)
    eng_g = build_engine(data, cfg, device="cuda")
    # diag mode routes through the fused factor path: a diagonal Rinv
    # yields a diagonal Cholesky factor, so q = ||Uz+u0||^2 is exactly
    # the diagonal quadratic form (VERDICT r1 task 9)
    assert eng_g.use_fused_estep
    lik_g = eng_g.run_em(3)
    assert lik_g == pytest.approx(lik_c, rel=1e-4)
    r = eng_g.state.R.cpu().numpy()
    for c in range(3):
        off = r[c] - np.diag(np.diag(r[c]))
        assert np.abs(off).max() == 0.0


def test_gpu_em_likelihood_monotone_fp32():
    """EM ascent on the exact-fp32 GPU path (fused f32 MFMA E-step +
    fp32 moments): likelihood non-decreasing up to reduction noise."""
    data, _ = make_blobs(20000, 6, 4, seed=9)
    cfg = GmmConfig(num_clusters=4, target_num_clusters=4,
                    min_iters=1, max_iters=1,
                    estep_dtype="fp32", mstep_precision="fp32")
    eng = build_engine(data, cfg, device="cuda")
    eng.run_em(4)
    liks = [eng.likelihood]
    for _ in range(25):
        eng.em_iteration(4)
        liks.append(float(eng._lik_dev.item()))
    diffs = np.diff(np.array(liks))
    assert np.all(diffs >= -1e-6 * abs(liks[-1])), diffs


@pytest.mark.parametrize("d,ed", [(140, "fp32"), (145, "fp32"), (145, "bf16")])
def test_engine_past_factor_dcap_matches_cpu(d, ed):
    """D in (128, 142]: fp32 MFMA E-step + rocBLAS covariance GEMMs;
    D > 142: VALU E-step + CPU-LU constants (the factor-emission LDS
    working set crosses gfx950's 160 KB at D=143) + GEMM covariance.
    Both quadrants must still match the CPU golden path."""
    data, _ = make_blobs(3000, d, 3, seed=d)
    cfg = GmmConfig(num_clusters=3, target_num_clusters=3,
                    min_iters=4, max_iters=4, estep_dtype=ed,
                    mstep_precision="bf16x3" if ed == "bf16" else "fp32")
    eng_g, lik_g = run("cuda", cfg, data)
    cfg_c = GmmConfig(num_clusters=3, target_num_clusters=3,
                      min_iters=4, max_iters=4)
    eng_c, lik_c = run("cpu", cfg_c, data)
    assert np.isfinite(lik_g)
    assert lik_g == pytest.approx(lik_c, rel=5e-3 if ed == "bf16" else 1e-4)
    np.testing.assert_allclose(eng_g.state.N.cpu().numpy(),
                               eng_c.state.N.numpy(), rtol=2e-2)


def test_diag_sweep_with_merge_matches_cpu():
    """DIAG_ONLY through an MDL merge: the first post-merge E-step must
    use diag(inv(R_merged_full)) — the host merge's full inverse, carried
    by quirk #8 — not the full quadratic form of chol(R_merged). The GPU
    factor path and the CPU golden path must take the same trajectory."""
    data, _ = make_blobs(6000, 4, 3, seed=77)
    for ed in ("fp32", "bf16"):
        cfg = GmmConfig(num_clusters=5, target_num_clusters=2,
                        min_iters=6, max_iters=6, diag_only=True,
                        estep_dtype=ed)
        eng_g = build_engine(data, cfg, device="cuda")
        res_g = eng_g.sweep()
        cfg_c = GmmConfig(num_clusters=5, target_num_clusters=2,
                          min_iters=6, max_iters=6, diag_only=True)
        eng_c = build_engine(data, cfg_c, device="cpu")
        res_c = eng_c.sweep()
        assert res_g.num_clusters == res_c.num_clusters
        assert res_g.min_rissanen == pytest.approx(
            res_c.min_rissanen, rel=5e-3 if ed == "bf16" else 1e-4)
        assert res_g.rissanen_by_k.keys() == res_c.rissanen_by_k.keys()
        for kk in res_g.rissanen_by_k:
            assert res_g.rissanen_by_k[kk] == pytest.approx(
                res_c.rissanen_by_k[kk], rel=1e-2 if ed == "bf16" else 1e-3)
