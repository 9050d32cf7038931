"""GPU kernel numerics: every gfx950 kernel vs the plain torch fp32
reference of the same op (run on MI355X via gpurun)."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from cuda_gmm_mpi_amd.ops import cpu_reference as cpu  # noqa: E402
from cuda_gmm_mpi_amd.ops import functional as F  # noqa: E402


def random_model(rng, k, d, device):
    means = torch.from_numpy(
        rng.standard_normal((k, d)).astype(np.float32) * 2).to(device)
    rs = []
    for _ in range(k):
        a = rng.standard_normal((d, d))
        rs.append((a @ a.T + d * np.eye(d)).astype(np.float32))
    r = torch.from_numpy(np.stack(rs)).to(device)
    pi = torch.from_numpy(rng.dirichlet(np.ones(k)).astype(np.float32)).to(device)
    return means, r, pi


@pytest.fixture
def device():
    assert torch.cuda.is_available()
    return torch.device("cuda:0")


def test_extension_loaded_and_native(device):
    """The HIP extension must be the loaded in-tree .so, not a fallback."""
    from cuda_gmm_mpi_amd.ops.backend import hip_ext
    ext = hip_ext()
    assert "cuda_gmm_mpi_amd" in ext.__file__
    assert ext.__file__.endswith(".so")


@pytest.mark.parametrize("d", [2, 8, 16, 24, 32, 48, 128])
def test_estep_logw_matches_cpu(device, d):
    rng = np.random.default_rng(d)
    k, n = 5, 4096
    means, r, pi = random_model(rng, k, d, device)
    rinv, const = cpu.compute_constants(r.cpu())
    x = torch.from_numpy(rng.standard_normal((d, n)).astype(np.float32))
    ref = cpu.estep_logw(x, means.cpu(), rinv, const, pi.cpu())
    got = F.estep_logw(x.to(device), means, rinv.to(device), const.to(device),
                       pi)
    np.testing.assert_allclose(got.cpu().numpy(), ref.numpy(),
                               rtol=2e-3, atol=2e-3)


def test_estep_logw_bf16(device):
    rng = np.random.default_rng(0)
    k, d, n = 4, 24, 8192
    means, r, pi = random_model(rng, k, d, device)
    rinv, const = cpu.compute_constants(r.cpu())
    x = torch.from_numpy(rng.standard_normal((d, n)).astype(np.float32))
    ref = cpu.estep_logw(x, means.cpu(), rinv, const, pi.cpu())
    got = F.estep_logw(x.to(device).to(torch.bfloat16), means,
                       rinv.to(device), const.to(device), pi)
    # bf16 data reads, fp32 accumulate: ~3 decimal digits on inputs
    np.testing.assert_allclose(got.cpu().numpy(), ref.numpy(),
                               rtol=5e-2, atol=5e-1)


def test_estep_logw_diag_only(device):
    rng = np.random.default_rng(1)
    k, d, n = 3, 16, 2048
    means, r, pi = random_model(rng, k, d, device)
    rd = torch.diag_embed(torch.diagonal(r, dim1=1, dim2=2))
    rinv, const = cpu.compute_constants(rd.cpu(), diag_only=True)
    ref = cpu.estep_logw(
        torch.from_numpy(rng.standard_normal((d, n)).astype(np.float32)),
        means.cpu(), rinv, const, pi.cpu(), diag_only=True)
    # regenerate same x stream
    rng = np.random.default_rng(1)
    _ = rng.standard_normal((k, d))
    for _ in range(k):
        rng.standard_normal((d, d))
    rng.dirichlet(np.ones(k))
    x = rng.standard_normal((d, n)).astype(np.float32)
    got = F.estep_logw(torch.from_numpy(x).to(device), means,
                       rinv.to(device), const.to(device), pi, diag_only=True)
    np.testing.assert_allclose(got.cpu().numpy(), ref.numpy(),
                               rtol=2e-3, atol=2e-3)


@pytest.mark.parametrize("k", [1, 3, 64])
def test_estep_posteriors_matches_cpu(device, k):
    rng = np.random.default_rng(k)
    n = 10000
    logw = rng.standard_normal((k, n)).astype(np.float32) * 10
    ref_w, ref_lik = cpu.estep_posteriors(torch.from_numpy(logw.copy()))
    t = torch.from_numpy(logw.copy()).to(device)
    w, lik = F.estep_posteriors(t)
    np.testing.assert_allclose(w.cpu().numpy(), ref_w.numpy(),
                               rtol=1e-4, atol=1e-5)
    assert float(lik) == pytest.approx(float(ref_lik), rel=1e-4)
    np.testing.assert_allclose(w.sum(dim=0).cpu().numpy(), np.ones(n),
                               rtol=1e-4)


@pytest.mark.parametrize("d,n", [(2, 5000), (24, 10000), (64, 4000),
                                 (128, 2000)])
def test_mstep_covariance_matches_cpu(device, d, n):
    rng = np.random.default_rng(d + n)
    k = 7
    x = rng.standard_normal((d, n)).astype(np.float32)
    w = rng.uniform(0, 1, (k, n)).astype(np.float32)
    ref = cpu.mstep_sufficient_stats(torch.from_numpy(x),
                                     torch.from_numpy(w))[2]
    got = F.mstep_covariance_s(torch.from_numpy(x).to(device),
                               torch.from_numpy(w).to(device))
    np.testing.assert_allclose(got.cpu().numpy(), ref.numpy(),
                               rtol=2e-3, atol=2e-2)
    # exact symmetry by construction (packed storage)
    sym = got - got.transpose(1, 2)
    assert float(sym.abs().max()) == 0.0


def test_mstep_covariance_determinism(device):
    rng = np.random.default_rng(9)
    d, n, k = 24, 100000, 8
    x = torch.from_numpy(rng.standard_normal((d, n)).astype(np.float32)).to(device)
    w = torch.from_numpy(rng.uniform(0, 1, (k, n)).astype(np.float32)).to(device)
    a = F.mstep_covariance_s(x, w)
    b = F.mstep_covariance_s(x, w)
    assert torch.equal(a, b)


@pytest.mark.parametrize("d", [1, 2, 24, 64, 128])
def test_constants_matches_cpu(device, d):
    rng = np.random.default_rng(d)
    k = 6
    rs = []
    for _ in range(k):
        a = rng.standard_normal((d, d))
        rs.append((a @ a.T + d * np.eye(d)).astype(np.float32))
    r = torch.from_numpy(np.stack(rs))
    ref_rinv, ref_const = cpu.compute_constants(r)
    rinv, const = F.constants(r.to(device))
    np.testing.assert_allclose(const.cpu().numpy(), ref_const.numpy(),
                               rtol=1e-3, atol=1e-3)
    for c in range(k):
        np.testing.assert_allclose(
            rinv[c].cpu().numpy(), np.linalg.inv(r[c].numpy()),
            rtol=5e-2, atol=5e-3)


def test_constants_diag_only(device):
    rng = np.random.default_rng(5)
    k, d = 4, 24
    diag = rng.uniform(0.5, 3.0, (k, d)).astype(np.float32)
    r = torch.diag_embed(torch.from_numpy(diag))
    rinv, const = F.constants(r.to(device), diag_only=True)
    ref_rinv, ref_const = cpu.compute_constants(r, diag_only=True)
    np.testing.assert_allclose(rinv.cpu().numpy(), ref_rinv.numpy(),
                               rtol=1e-5)
    np.testing.assert_allclose(const.cpu().numpy(), ref_const.numpy(),
                               rtol=1e-4)


def test_missing_extension_raises_loudly(monkeypatch, device):
    """On GPU, a missing extension must fail, never silently fall back."""
    from cuda_gmm_mpi_amd.ops import backend
    monkeypatch.setattr(backend, "_ext", None)
    monkeypatch.setattr(backend, "_ext_error", ImportError("simulated"))
    x = torch.zeros(2, 16, device=device)
    with pytest.raises(RuntimeError, match="HIP extension"):
        F.estep_logw(x, torch.zeros(1, 2, device=device),
                     torch.zeros(1, 2, 2, device=device),
                     torch.zeros(1, device=device),
                     torch.ones(1, device=device))


def test_mfma_bf16_probe(device):
    """Empirically verify the assumed bf16 MFMA fragment layout (guide G9:
    asymmetric B catches operand/output transposes)."""
    from cuda_gmm_mpi_amd.ops.backend import hip_ext
    rng = np.random.default_rng(77)
    a = rng.standard_normal((16, 32)).astype(np.float32)
    b = (rng.standard_normal((32, 16)) + np.arange(16)[None, :]).astype(
        np.float32)  # asymmetric
    at = torch.from_numpy(a).to(device).to(torch.bfloat16)
    bt = torch.from_numpy(b).to(device).to(torch.bfloat16)
    c = torch.zeros(16, 16, dtype=torch.float32, device=device)
    hip_ext().mfma_probe(at.contiguous(), bt.contiguous(), c)
    ref = at.float() @ bt.float()
    np.testing.assert_allclose(c.cpu().numpy(), ref.cpu().numpy(),
                               rtol=2e-2, atol=2e-2)


@pytest.mark.parametrize("d,n", [(2, 3001), (15, 5000), (16, 5000),
                                 (24, 100000), (31, 4097)])
def test_mstep_moments_matches_cpu(device, d, n):
    from cuda_gmm_mpi_amd.ops import functional as F
    rng = np.random.default_rng(d * 1000 + 7)
    k = 6
    x = rng.standard_normal((d, n)).astype(np.float32)
    w = rng.uniform(0, 1, (k, n)).astype(np.float32)
    packed = F.mstep_moments(torch.from_numpy(x).to(device),
                             torch.from_numpy(w).to(device))
    n_c, mean_num, s = F.moments_views(packed, d)
    rn, rm, rs = cpu.mstep_sufficient_stats(torch.from_numpy(x),
                                            torch.from_numpy(w))
    np.testing.assert_allclose(n_c.cpu().numpy(), rn.numpy(), rtol=1e-4)
    np.testing.assert_allclose(mean_num.cpu().numpy(), rm.numpy(),
                               rtol=1e-3, atol=1e-2)
    np.testing.assert_allclose(s.cpu().numpy(), rs.numpy(),
                               rtol=2e-3, atol=5e-2)


def test_mstep_moments_determinism(device):
    from cuda_gmm_mpi_amd.ops import functional as F
    rng = np.random.default_rng(3)
    d, n, k = 24, 200000, 16
    x = torch.from_numpy(rng.standard_normal((d, n)).astype(np.float32)).to(device)
    w = torch.from_numpy(rng.uniform(0, 1, (k, n)).astype(np.float32)).to(device)
    assert torch.equal(F.mstep_moments(x, w), F.mstep_moments(x, w))


def test_estep_fused_matches_cpu(device):
    """Fused bf16 MFMA E-step vs plain fp32 torch reference."""
    from cuda_gmm_mpi_amd.ops import functional as F
    rng = np.random.default_rng(11)
    k, d, n = 8, 24, 20000
    means, r, pi = random_model(rng, k, d, device)
    rinv, const = cpu.compute_constants(r.cpu())
    x = rng.standard_normal((d, n)).astype(np.float32) * 2
    xt = torch.from_numpy(x).to(device)
    # emit mfac via the constants kernel
    mfac = torch.empty(k, 2, 32, 32, dtype=torch.bfloat16, device=device)
    rinv_g, const_g = F.constants(r, means, False, mfac)
    add = const_g + torch.log(pi)
    w_out = torch.empty(k, n, dtype=torch.float32, device=device)
    lse = torch.empty(n, dtype=torch.float32, device=device)
    _, lik = F.estep_fused(xt.to(torch.bfloat16), mfac, add, w_out, lse)
    w = torch.exp(w_out - lse.unsqueeze(0))  # w_out holds logw
    # reference: logw + posteriors in fp32
    ref_logw = cpu.estep_logw(torch.from_numpy(x), means.cpu(), rinv, const,
                              pi.cpu())
    ref_w, ref_lik = cpu.estep_posteriors(ref_logw)
    # bf16 data + bf16 hi/lo factors: posteriors within bf16-class tolerance
    np.testing.assert_allclose(w.cpu().numpy(), ref_w.numpy(),
                               rtol=5e-2, atol=2e-2)
    assert float(lik) == pytest.approx(float(ref_lik), rel=2e-3)
    np.testing.assert_allclose(w.sum(dim=0).cpu().numpy(), np.ones(n),
                               rtol=1e-3)


def test_estep_fused_tail_events(device):
    """n not a multiple of the 128-event block."""
    from cuda_gmm_mpi_amd.ops import functional as F
    rng = np.random.default_rng(13)
    k, d, n = 4, 8, 1000 + 37
    means, r, pi = random_model(rng, k, d, device)
    mfac = torch.empty(k, 2, 32, 32, dtype=torch.bfloat16, device=device)
    rinv_g, const_g = F.constants(r, means, False, mfac)
    add = const_g + torch.log(pi)
    x = rng.standard_normal((d, n)).astype(np.float32)
    w_out = torch.empty(k, n, dtype=torch.float32, device=device)
    lse = torch.empty(n, dtype=torch.float32, device=device)
    _, lik = F.estep_fused(
        torch.from_numpy(x).to(device).to(torch.bfloat16), mfac, add, w_out,
        lse)
    w = torch.exp(w_out - lse.unsqueeze(0))
    rinv, const = cpu.compute_constants(r.cpu())
    ref_logw = cpu.estep_logw(torch.from_numpy(x), means.cpu(), rinv, const,
                              pi.cpu())
    ref_w, ref_lik = cpu.estep_posteriors(ref_logw)
    np.testing.assert_allclose(w.cpu().numpy(), ref_w.numpy(),
                               rtol=5e-2, atol=2e-2)
    assert float(lik) == pytest.approx(float(ref_lik), rel=5e-3)


def test_constants_emits_valid_cholesky(device):
    """mfac hi+lo must reconstruct M with U^T U = Rinv and u0 = -U mu."""
    from cuda_gmm_mpi_amd.ops import functional as F
    rng = np.random.default_rng(19)
    k, d = 5, 24
    means, r, pi = random_model(rng, k, d, device)
    mfac = torch.empty(k, 2, 32, 32, dtype=torch.bfloat16, device=device)
    rinv, const = F.constants(r, means, False, mfac)
    m = mfac.float()
    mm = (m[:, 0] + m[:, 1]).cpu().numpy()  # [K, 32, 32] fragment k-order
    # identity k-map: column j of the stored fragment == M column j
    for c in range(k):
        u = mm[c][:d, :d]
        np.testing.assert_allclose(u.T @ u, rinv[c].cpu().numpy(),
                                   rtol=2e-2, atol=2e-2)
        u0 = mm[c][:d, d]
        np.testing.assert_allclose(
            u0, -(u @ means[c].cpu().numpy()), rtol=2e-2, atol=2e-2)
        assert np.abs(mm[c][:, d + 1:]).max() == 0.0
        assert np.abs(mm[c][d:, :]).max() == 0.0


def test_mfma_bf16_probe32(device):
    """32x32x16 bf16 fragment layout (used by the fused E-step)."""
    from cuda_gmm_mpi_amd.ops.backend import hip_ext
    rng = np.random.default_rng(78)
    a = rng.standard_normal((32, 16)).astype(np.float32)
    b = (rng.standard_normal((16, 32)) + np.arange(32)[None, :]).astype(
        np.float32)
    at = torch.from_numpy(a).to(device).to(torch.bfloat16)
    bt = torch.from_numpy(b).to(device).to(torch.bfloat16)
    c = torch.zeros(32, 32, dtype=torch.float32, device=device)
    hip_ext().mfma_probe32(at.contiguous(), bt.contiguous(), c)
    ref = at.float() @ bt.float()
    np.testing.assert_allclose(c.cpu().numpy(), ref.cpu().numpy(),
                               rtol=2e-2, atol=5e-2)


@pytest.mark.parametrize("d,n", [(2, 3001), (24, 100000), (31, 4097)])
def test_mstep_moments_b16_matches_cpu(device, d, n):
    """Split-precision bf16x3 moments: ~1e-5-class relative accuracy."""
    from cuda_gmm_mpi_amd.ops import functional as F
    rng = np.random.default_rng(d * 31 + 5)
    k = 6
    x = rng.standard_normal((d, n)).astype(np.float32) * 3
    w = rng.uniform(0, 1, (k, n)).astype(np.float32)
    packed = F.mstep_moments(torch.from_numpy(x).to(device),
                             torch.from_numpy(w).to(device),
                             precision="bf16x3")
    n_c, mean_num, s = F.moments_views(packed, d)
    rn, rm, rs = cpu.mstep_sufficient_stats(torch.from_numpy(x).double(),
                                            torch.from_numpy(w).double())
    np.testing.assert_allclose(n_c.cpu().numpy(), rn.numpy(), rtol=1e-4)
    np.testing.assert_allclose(mean_num.cpu().numpy(), rm.numpy(),
                               rtol=3e-4, atol=1e-2)
    scale = float(rs.abs().max())
    np.testing.assert_allclose(s.cpu().numpy(), rs.numpy(),
                               rtol=3e-4, atol=3e-4 * scale)


@pytest.mark.parametrize("d", [32, 33, 40, 64, 100, 128, 142])
def test_estep_logw_big_matches_cpu(device, d):
    """Big-D MFMA logw vs fp32 torch reference (bf16-class tolerance)."""
    from cuda_gmm_mpi_amd.ops import functional as F
    rng = np.random.default_rng(d)
    k, n = 6, 2000 + 57
    means, r, pi = random_model(rng, k, d, device)
    mfac = torch.empty(k, *F.mfac_shape(d), dtype=torch.bfloat16,
                       device=device)
    rinv_g, const_g = F.constants(r, means, False, mfac)
    add = const_g + torch.log(pi)
    x = rng.standard_normal((d, n)).astype(np.float32)
    out = torch.empty(k, n, dtype=torch.float32, device=device)
    F.estep_logw_big(torch.from_numpy(x).to(device).to(torch.bfloat16),
                     mfac, add, out)
    rinv, const = cpu.compute_constants(r.cpu())
    ref = cpu.estep_logw(torch.from_numpy(x), means.cpu(), rinv, const,
                         pi.cpu())
    # |logw| grows with D; tolerance scales accordingly
    np.testing.assert_allclose(out.cpu().numpy(), ref.numpy(),
                               rtol=5e-2, atol=2.0)


@pytest.mark.parametrize("d,n", [(32, 4000), (40, 5000), (64, 3001),
                                 (128, 2000), (143, 1500)])
def test_mstep_moments_big_matches_cpu(device, d, n):
    from cuda_gmm_mpi_amd.ops import functional as F
    rng = np.random.default_rng(d * 7 + 1)
    k = 5
    x = rng.standard_normal((d, n)).astype(np.float32) * 2
    w = rng.uniform(0, 1, (k, n)).astype(np.float32)
    packed = F.mstep_moments(torch.from_numpy(x).to(device),
                             torch.from_numpy(w).to(device),
                             precision="bf16x3")
    n_c, mean_num, s = F.moments_views(packed, d)
    rn, rm, rs = cpu.mstep_sufficient_stats(torch.from_numpy(x).double(),
                                            torch.from_numpy(w).double())
    np.testing.assert_allclose(n_c.cpu().numpy(), rn.numpy(), rtol=1e-4)
    np.testing.assert_allclose(mean_num.cpu().numpy(), rm.numpy(),
                               rtol=3e-4, atol=1e-2)
    scale = float(rs.abs().max())
    np.testing.assert_allclose(s.cpu().numpy(), rs.numpy(),
                               rtol=5e-4, atol=5e-4 * scale)


def test_engine_big_d_bf16_matches_fp32(device=None):
    """Config-4-shaped engine run (scaled down): big-D MFMA paths vs the
    fp32 VALU/compose paths."""
    from cuda_gmm_mpi_amd.engine import build_engine
    from cuda_gmm_mpi_amd.utils.config import GmmConfig
    from cuda_gmm_mpi_amd.utils.synthetic import make_blobs
    data, _ = make_blobs(30000, 64, 8, seed=61)
    res = {}
    for name, ed, mp in (("fp32", "fp32", "fp32"),
                         ("bf16", "bf16", "bf16x3")):
        cfg = GmmConfig(num_clusters=8, target_num_clusters=8,
                        min_iters=6, max_iters=6, estep_dtype=ed,
                        mstep_precision=mp)
        eng = build_engine(data, cfg, device="cuda")
        lik = eng.run_em(8)
        res[name] = (lik, eng.state.means.cpu().numpy().copy(),
                     eng.use_big_estep)
    # both precisions use the big-D MFMA path now (bf16 hi/lo split and
    # the exact-f32 MFMA variant respectively)
    assert res["bf16"][2] is True and res["fp32"][2] is True
    assert res["bf16"][0] == pytest.approx(res["fp32"][0], rel=2e-3)
    np.testing.assert_allclose(res["bf16"][1], res["fp32"][1],
                               rtol=5e-2, atol=5e-1)


def test_estep_fused_f32_matches_cpu(device):
    """Exact-f32 MFMA fused E-step vs the fp32 torch reference (tight)."""
    from cuda_gmm_mpi_amd.ops import functional as F
    rng = np.random.default_rng(23)
    k, d, n = 8, 24, 20000 + 57
    means, r, pi = random_model(rng, k, d, device)
    mfac = torch.empty(k, 2, 32, 32, dtype=torch.bfloat16, device=device)
    mfac32 = torch.empty(k, 32, 32, dtype=torch.float32, device=device)
    rinv_g, const_g = F.constants(r, means, False, mfac, mfac32)
    add = const_g + torch.log(pi)
    x = rng.standard_normal((d, n)).astype(np.float32) * 2
    w_out = torch.empty(k, n, dtype=torch.float32, device=device)
    lse = torch.empty(n, dtype=torch.float32, device=device)
    _, lik = F.estep_fused_f32(torch.from_numpy(x).to(device), mfac32, add,
                               w_out, lse)
    w = torch.exp(w_out - lse.unsqueeze(0))
    rinv, const = cpu.compute_constants(r.cpu())
    ref_logw = cpu.estep_logw(torch.from_numpy(x), means.cpu(), rinv, const,
                              pi.cpu())
    ref_w, ref_lik = cpu.estep_posteriors(ref_logw)
    # exact f32 MFMA: tolerance limited by the Cholesky-vs-Rinv algebra
    np.testing.assert_allclose(w.cpu().numpy(), ref_w.numpy(),
                               rtol=2e-3, atol=2e-4)
    assert float(lik) == pytest.approx(float(ref_lik), rel=1e-4)


def test_fuzz_kernels_vs_cpu(device):
    """Randomized shape fuzz: fused/big E-step + moments vs CPU reference."""
    from cuda_gmm_mpi_amd.ops import functional as F
    rng = np.random.default_rng(2024)
    for trial in range(12):
        d = int(rng.integers(1, 32)) if trial % 2 == 0 else int(
            rng.integers(32, 129))
        k = int(rng.integers(1, 20))
        n = int(rng.integers(50, 4000))
        x = (rng.standard_normal((d, n)) * rng.uniform(0.5, 5)).astype(
            np.float32)
        w = rng.uniform(0, 1, (k, n)).astype(np.float32)
        xt = torch.from_numpy(x).to(device)
        wt = torch.from_numpy(w).to(device)
        packed = F.mstep_moments(xt, wt, precision="bf16x3")
        n_c, mean_num, s = F.moments_views(packed, d)
        rn, rm, rs = cpu.mstep_sufficient_stats(
            torch.from_numpy(x).double(), torch.from_numpy(w).double())
        np.testing.assert_allclose(n_c.cpu().numpy(), rn.numpy(),
                                   rtol=1e-3, atol=1e-3)
        scale = float(rs.abs().max()) + 1e-6
        np.testing.assert_allclose(s.cpu().numpy(), rs.numpy(),
                                   rtol=1e-3, atol=1e-3 * scale)
        # exact fp32 moments too (D <= 31 fast path or compose path)
        packed32 = F.mstep_moments(xt, wt, precision="fp32")
        n32, m32, s32 = F.moments_views(packed32, d)
        np.testing.assert_allclose(s32.cpu().numpy(), rs.numpy(),
                                   rtol=1e-4, atol=1e-4 * scale)


def test_estep_big_small_d_large_k(device):
    """D <= 31 with K beyond the fused LDS gate routes through the big-D
    MFMA kernel (KCT=2 tier)."""
    from cuda_gmm_mpi_amd.ops import functional as F
    rng = np.random.default_rng(29)
    k, d, n = 7, 12, 3000
    means, r, pi = random_model(rng, k, d, device)
    mfac = torch.empty(k, *F.mfac_shape(d), dtype=torch.bfloat16,
                       device=device)
    rinv_g, const_g = F.constants(r, means, False, mfac)
    add = const_g + torch.log(pi)
    x = rng.standard_normal((d, n)).astype(np.float32)
    out = torch.empty(k, n, dtype=torch.float32, device=device)
    F.estep_logw_big(torch.from_numpy(x).to(device).to(torch.bfloat16),
                     mfac, add, out)
    rinv, const = cpu.compute_constants(r.cpu())
    ref = cpu.estep_logw(torch.from_numpy(x), means.cpu(), rinv, const,
                         pi.cpu())
    np.testing.assert_allclose(out.cpu().numpy(), ref.numpy(),
                               rtol=5e-2, atol=1.0)


def test_mstep_finalize_kernel_matches_cpu(device):
    """The one-kernel finalize vs the torch reference chain (exact rules:
    empty-cluster resets, G*avgvar, pi floor)."""
    from cuda_gmm_mpi_amd.ops import functional as F
    from cuda_gmm_mpi_amd.ops.backend import hip_ext
    rng = np.random.default_rng(33)
    k, d, n = 6, 24, 5000
    x = rng.standard_normal((d, n)).astype(np.float32)
    w = rng.uniform(0, 1, (k, n)).astype(np.float32)
    w[3] *= 0.0       # empty cluster (N=0 -> identity reset, pi floor)
    w[4] *= 1.5e-4    # 0.5 < N < 1 edge
    packed = F.mstep_moments(torch.from_numpy(x).to(device),
                             torch.from_numpy(w).to(device))
    avgvar = torch.full((k,), 0.37, device=device)
    n_o = torch.empty(k, device=device)
    mu_o = torch.empty(k, d, device=device)
    r_o = torch.empty(k, d, d, device=device)
    pi_o = torch.empty(k, device=device)
    hip_ext().mstep_finalize(packed, avgvar, 3, n_o, mu_o, r_o, pi_o, False)
    # reference chain
    n_c, mean_num, s = F.moments_views(packed, d)
    ref_mu = cpu.finalize_means(n_c, mean_num)
    ref_r = cpu.finalize_covariance(n_c, ref_mu, s, avgvar, world_size=3)
    ref_pi = cpu.compute_pi(n_c)
    np.testing.assert_allclose(n_o.cpu().numpy(), n_c.cpu().numpy())
    np.testing.assert_allclose(mu_o.cpu().numpy(), ref_mu.cpu().numpy(),
                               rtol=1e-5, atol=1e-6)
    np.testing.assert_allclose(r_o.cpu().numpy(), ref_r.cpu().numpy(),
                               rtol=1e-4, atol=1e-5)
    np.testing.assert_allclose(pi_o.cpu().numpy(), ref_pi.cpu().numpy(),
                               rtol=1e-6)
    assert pi_o.cpu().numpy()[3] == 1e-10
    np.testing.assert_allclose(r_o[3].cpu().numpy(), np.eye(d))


@pytest.mark.parametrize("k", [1, 3, 64, 104])
def test_estep_fused_odd_k(device, k):
    """Pass-2 half-split must cover odd K and K=1 (likelihood non-zero)."""
    from cuda_gmm_mpi_amd.ops import functional as F
    rng = np.random.default_rng(k)
    d, n = 8, 1000
    means, r, pi = random_model(rng, k, d, device)
    mfac = torch.empty(k, *F.mfac_shape(d), dtype=torch.bfloat16,
                       device=device)
    rinv_g, const_g = F.constants(r, means, False, mfac)
    add = const_g + torch.log(pi)
    x = rng.standard_normal((d, n)).astype(np.float32)
    w_out = torch.empty(k, n, dtype=torch.float32, device=device)
    lse = torch.empty(n, dtype=torch.float32, device=device)
    _, lik = F.estep_fused(
        torch.from_numpy(x).to(device).to(torch.bfloat16), mfac, add, w_out,
        lse)
    w = torch.exp(w_out - lse.unsqueeze(0))
    rinv, const = cpu.compute_constants(r.cpu())
    ref_logw = cpu.estep_logw(torch.from_numpy(x), means.cpu(), rinv, const,
                              pi.cpu())
    ref_w, ref_lik = cpu.estep_posteriors(ref_logw)
    assert float(lik) == pytest.approx(float(ref_lik), rel=1e-2)
    np.testing.assert_allclose(w.sum(dim=0).cpu().numpy(), np.ones(n),
                               rtol=1e-3)


@pytest.mark.parametrize("d", [40, 64, 128])
def test_estep_logw_big_f32_matches_cpu(device, d):
    """Exact-f32 big-D MFMA logw vs the fp32 torch reference (tight —
    f32 MFMA is bitwise an fmaf chain; only the Cholesky-vs-Rinv algebra
    differs). Fills the fp32 + D > 31 quadrant (VERDICT r1 missing #4)."""
    from cuda_gmm_mpi_amd.ops import functional as F
    rng = np.random.default_rng(d + 3)
    k, n = 6, 2000 + 57
    means, r, pi = random_model(rng, k, d, device)
    mfac = torch.empty(k, *F.mfac_shape(d), dtype=torch.bfloat16,
                       device=device)
    mfac32 = torch.empty(k, *F.mfac_shape(d)[1:], dtype=torch.float32,
                         device=device)
    rinv_g, const_g = F.constants(r, means, False, mfac, mfac32)
    add = const_g + torch.log(pi)
    x = rng.standard_normal((d, n)).astype(np.float32)
    out = torch.empty(k, n, dtype=torch.float32, device=device)
    F.estep_logw_big_f32(torch.from_numpy(x).to(device), mfac32, add, out)
    rinv, const = cpu.compute_constants(r.cpu())
    ref = cpu.estep_logw(torch.from_numpy(x), means.cpu(), rinv, const,
                         pi.cpu())
    scale = float(ref.abs().max())
    np.testing.assert_allclose(out.cpu().numpy(), ref.numpy(),
                               rtol=1e-3, atol=1e-5 * scale)


def test_engine_big_d_fp32_mfma_matches_valu(device=None):
    """Engine-level: the exact-f32 big-D MFMA E-step vs the VALU gen
    kernel on the same fp32 data — same trajectory."""
    from cuda_gmm_mpi_amd.engine import build_engine
    from cuda_gmm_mpi_amd.utils.config import GmmConfig
    from cuda_gmm_mpi_amd.utils.synthetic import make_blobs
    data, _ = make_blobs(20000, 48, 6, seed=67)
    cfg = GmmConfig(num_clusters=6, target_num_clusters=6,
                    min_iters=6, max_iters=6)  # fp32 everywhere
    eng_m = build_engine(data, cfg, device="cuda")
    assert eng_m.use_big_estep and eng_m.mfac32 is not None
    lik_m = eng_m.run_em(6)
    eng_v = build_engine(data, cfg, device="cuda")
    eng_v.use_big_estep = False
    eng_v.mfac32 = None
    lik_v = eng_v.run_em(6)
    assert lik_m == pytest.approx(lik_v, rel=1e-4)
    np.testing.assert_allclose(eng_m.state.means.cpu().numpy(),
                               eng_v.state.means.cpu().numpy(),
                               rtol=1e-3, atol=1e-2)


def test_engine_diag_only_fused_matches_valu(device=None):
    """DIAG_ONLY through the fused factor path vs the VALU diag kernel:
    identical math (diagonal Cholesky factor), same trajectory."""
    from cuda_gmm_mpi_amd.engine import build_engine
    from cuda_gmm_mpi_amd.utils.config import GmmConfig
    from cuda_gmm_mpi_amd.utils.synthetic import make_blobs
    data, _ = make_blobs(20000, 12, 4, seed=131)
    for ed in ("fp32", "bf16"):
        cfg = GmmConfig(num_clusters=4, target_num_clusters=4,
                        min_iters=6, max_iters=6, diag_only=True,
                        estep_dtype=ed)
        eng_f = build_engine(data, cfg, device="cuda")
        assert eng_f.use_fused_estep
        lik_f = eng_f.run_em(4)
        eng_v = build_engine(data, cfg, device="cuda")
        eng_v.use_fused_estep = False
        eng_v.use_big_estep = False
        eng_v.mfac32 = None
        lik_v = eng_v.run_em(4)
        rel = 1e-4 if ed == "fp32" else 2e-2
        assert lik_f == pytest.approx(lik_v, rel=rel), ed


def test_estep_fused_lds_matches_cpu(device):
    """v1 lw-in-LDS fused E-step (the small-K fast path: posteriors
    written directly) vs the fp32 torch reference."""
    from cuda_gmm_mpi_amd.ops import functional as F
    rng = np.random.default_rng(311)
    k, d, n = 8, 24, 20000 + 57
    means, r, pi = random_model(rng, k, d, device)
    mfac = torch.empty(k, 2, 32, 32, dtype=torch.bfloat16, device=device)
    mfac32 = torch.empty(k, 32, 32, dtype=torch.float32, device=device)
    rinv_g, const_g = F.constants(r, means, False, mfac, mfac32)
    add = const_g + torch.log(pi)
    x = rng.standard_normal((d, n)).astype(np.float32) * 2
    rinv, const = cpu.compute_constants(r.cpu())
    ref_logw = cpu.estep_logw(torch.from_numpy(x), means.cpu(), rinv, const,
                              pi.cpu())
    ref_w, ref_lik = cpu.estep_posteriors(ref_logw)
    w_out = torch.empty(k, n, dtype=torch.float32, device=device)
    w, lik = F.estep_fused_lds(
        torch.from_numpy(x).to(device).to(torch.bfloat16), mfac, add, w_out)
    np.testing.assert_allclose(w.cpu().numpy(), ref_w.numpy(),
                               rtol=5e-2, atol=2e-2)
    assert float(lik) == pytest.approx(float(ref_lik), rel=2e-3)
    w32, lik32 = F.estep_fused_f32_lds(torch.from_numpy(x).to(device),
                                       mfac32, add, w_out)
    np.testing.assert_allclose(w32.cpu().numpy(), ref_w.numpy(),
                               rtol=2e-3, atol=2e-4)
    assert float(lik32) == pytest.approx(float(ref_lik), rel=1e-4)


def test_engine_lds_vs_lse_estep_equivalent(device=None):
    """The small-K lds fast path and the any-K online-softmax path give
    the same trajectory (engine-level)."""
    from cuda_gmm_mpi_amd.engine import build_engine
    from cuda_gmm_mpi_amd.utils.config import GmmConfig
    from cuda_gmm_mpi_amd.utils.synthetic import make_blobs
    data, _ = make_blobs(30000, 24, 8, seed=211)
    cfg = GmmConfig(num_clusters=8, target_num_clusters=8,
                    min_iters=8, max_iters=8, estep_dtype="bf16",
                    mstep_precision="bf16x3")
    eng_a = build_engine(data, cfg, device="cuda")
    assert eng_a.use_lds_estep
    lik_a = eng_a.run_em(8)
    eng_b = build_engine(data, cfg, device="cuda")
    eng_b.use_lds_estep = False  # force the online-softmax variant
    lik_b = eng_b.run_em(8)
    assert lik_a == pytest.approx(lik_b, rel=1e-3)
    np.testing.assert_allclose(eng_a.state.means.cpu().numpy(),
                               eng_b.state.means.cpu().numpy(),
                               rtol=1e-2, atol=1e-1)
