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
