"""Reference (CPU, torch fp32) implementations of every EM op.

These are the golden-path semantics the HIP kernels are tested against and
the compute path for CPU-only runs (BASELINE config 1). Shapes follow the
MI355X-native layout:

 - data is dimension-major: ``x`` is [D, N] (gaussian.cu:212-218 transpose,
   kept resident in this layout from the start)
 - memberships / log-weights are cluster-major: ``w`` is [K, N]
   (gaussian.h:75, memberships[c*num_events+e])
 - sufficient statistics: N [K], mean numerators [K, D], second moments
   S [K, D, D] with S_c = sum_e w_ce * x_e x_e^T.

The covariance is finalized as R_c = (S_c - N_c mu_c mu_c^T + reg) / N_c,
algebraically identical to the reference's per-event centered sum
(gaussian_kernel.cu:644-646) because mu_c is the exact weighted mean.
"""
from __future__ import annotations

import math

import torch


LOG_2PI = math.log(2.0 * math.pi)


def estep_logw(x: torch.Tensor, means: torch.Tensor, rinv: torch.Tensor,
               constant: torch.Tensor, pi: torch.Tensor,
               diag_only: bool = False) -> torch.Tensor:
    """Log weighted likelihoods (estep1, gaussian_kernel.cu:383-444).

    x [D, N]; means [K, D]; rinv [K, D, D]; constant [K]; pi [K].
    Returns logw [K, N] = -0.5 * (x-mu)^T Rinv (x-mu) + constant + ln(pi).
    """
    d, n = x.shape
    k = means.shape[0]
    logw = torch.empty((k, n), dtype=torch.float32, device=x.device)
    for c in range(k):
        xc = x - means[c].unsqueeze(1)  # [D, N]
        if diag_only:
            q = (xc * xc * torch.diagonal(rinv[c]).unsqueeze(1)).sum(dim=0)
        else:
            q = (xc * (rinv[c] @ xc)).sum(dim=0)
        logw[c] = -0.5 * q + constant[c] + torch.log(pi[c])
    return logw


def estep_posteriors(logw: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """Posteriors + total log-likelihood (estep2, gaussian_kernel.cu:446-512).

    Returns (w [K, N] with columns summing to 1, likelihood scalar tensor).
    Log-sum-exp with per-event max for stability, like the reference.
    """
    m = logw.max(dim=0).values  # [N]
    denom = m + torch.log(torch.exp(logw - m.unsqueeze(0)).sum(dim=0))
    w = torch.exp(logw - denom.unsqueeze(0))
    return w, denom.sum()


def mstep_sufficient_stats(
    x: torch.Tensor, w: torch.Tensor
) -> tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Per-shard sufficient statistics (mstep_N / mstep_means /
    mstep_covariance1 numerators, gaussian_kernel.cu:522-677).

    Returns (N [K], mean numerators [K, D], S [K, D, D]).
    S is the *uncentered* weighted second moment; centering happens in
    finalize (translation by the global mean is handled by the engine).
    """
    n_c = w.sum(dim=1)                                 # [K]
    mean_num = w @ x.T                                  # [K, D]
    # S_c = (x * w_c) @ x^T, batched over clusters
    k, n = w.shape
    d = x.shape[0]
    s = torch.empty((k, d, d), dtype=x.dtype, device=x.device)
    for c in range(k):
        xw = x * w[c].unsqueeze(0)
        s[c] = xw @ x.T
    return n_c, mean_num, s


def finalize_means(n_c: torch.Tensor, mean_num: torch.Tensor) -> torch.Tensor:
    """means = mean_num / N where N > 0.5 else 0 (gaussian.cu:610-622)."""
    safe = n_c > 0.5
    means = torch.where(
        safe.unsqueeze(1), mean_num / n_c.clamp(min=1e-30).unsqueeze(1),
        torch.zeros_like(mean_num),
    )
    return means


def finalize_covariance(
    n_c: torch.Tensor, means: torch.Tensor, s: torch.Tensor,
    avgvar: torch.Tensor, world_size: int = 1, diag_only: bool = False,
) -> torch.Tensor:
    """Covariance finalize, reproducing the reference exactly.

    Per reference semantics with G total GPUs:
     - each GPU's kernel writes its partial centered sum if global N >= 1.0
       else 0 (gaussian_kernel.cu:658-667), then adds avgvar to the diagonal
       (gaussian_kernel.cu:673-675);
     - the global reduction sums G partials => diagonal receives G*avgvar;
     - the host divides by global N when N > 0.5, else resets to identity
       (gaussian.cu:663-679).

    Here: cov_num = S - N mu mu^T when N >= 1.0 else 0; diagonal += G*avgvar;
    R = cov_num / N when N > 0.5 else I.
    """
    k, d, _ = s.shape
    mu_outer = means.unsqueeze(2) * means.unsqueeze(1)      # [K, D, D]
    cov_num = s - n_c.view(k, 1, 1) * mu_outer
    cov_num = torch.where(
        (n_c >= 1.0).view(k, 1, 1), cov_num, torch.zeros_like(cov_num)
    )
    if diag_only:
        cov_num = torch.diag_embed(torch.diagonal(cov_num, dim1=1, dim2=2))
    eye = torch.eye(d, dtype=s.dtype, device=s.device)
    cov_num = cov_num + (world_size * avgvar).view(k, 1, 1) * eye
    r = torch.where(
        (n_c > 0.5).view(k, 1, 1),
        cov_num / n_c.clamp(min=1e-30).view(k, 1, 1),
        eye.expand(k, d, d),
    )
    return r.contiguous()


def lu_invert_nopivot(a: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """Batched no-pivot LU inversion + ln|det| on fp32 tensors.

    Same elimination as the device `invert` (gaussian_kernel.cu:107-169):
    Doolittle LU without pivoting, ln determinant from |diag|, triangular
    inversion, composition. a: [K, D, D]. Returns (inv [K, D, D], logdet [K]).
    """
    k, d, _ = a.shape
    data = a.clone()
    if d == 1:
        logdet = torch.log(data[:, 0, 0])
        return (1.0 / data).reshape(k, 1, 1), logdet
    data[:, 0, 1:] /= data[:, 0, 0].unsqueeze(1)
    for i in range(1, d):
        data[:, i:, i] -= torch.einsum("kjp,kp->kj", data[:, i:, :i], data[:, :i, i])
        if i == d - 1:
            continue
        data[:, i, i + 1:] = (
            data[:, i, i + 1:]
            - torch.einsum("kp,kpj->kj", data[:, i, :i], data[:, :i, i + 1:])
        ) / data[:, i, i].unsqueeze(1)
    diag = torch.diagonal(data, dim1=1, dim2=2).abs()
    logdet = torch.log(diag).sum(dim=1)
    # invert L (Crout storage: L holds the diagonal, U is unit-diagonal)
    for i in range(d):
        for j in range(i, d):
            if i == j:
                x = torch.ones(k, dtype=a.dtype, device=a.device)
            else:
                x = -torch.einsum("kp,kp->k", data[:, j, i:j], data[:, i:j, i])
            data[:, j, i] = x / data[:, j, j]
    # invert U (unit diagonal)
    for i in range(d):
        for j in range(i + 1, d):
            ks = torch.arange(i, j, device=a.device)
            lhs = data[:, ks, j]
            rhs = torch.where(
                (ks == i).unsqueeze(0), torch.ones_like(lhs), data[:, i, ks]
            )
            data[:, i, j] = -(lhs * rhs).sum(dim=1)
    # final composition out[j,i] = sum_k Uinv[j,k>=] * Linv[k,i]
    out = torch.empty_like(data)
    for i in range(d):
        for j in range(d):
            k0 = max(i, j)
            ks = torch.arange(k0, d, device=a.device)
            lhs = torch.where(
                (ks == j).unsqueeze(0),
                torch.ones(1, len(ks), dtype=a.dtype, device=a.device),
                data[:, j, ks],
            )
            out[:, j, i] = (lhs * data[:, ks, i]).sum(dim=1)
    return out, logdet


def lu_logdet_nopivot(a: torch.Tensor) -> torch.Tensor:
    """Batched ln|det| via the same no-pivot elimination as
    lu_invert_nopivot, without the inversion phases. a: [K, D, D]."""
    k, d, _ = a.shape
    data = a.clone()
    if d == 1:
        return torch.log(data[:, 0, 0].abs())
    data[:, 0, 1:] /= data[:, 0, 0].unsqueeze(1)
    for i in range(1, d):
        data[:, i:, i] -= torch.einsum("kjp,kp->kj", data[:, i:, :i], data[:, :i, i])
        if i == d - 1:
            continue
        data[:, i, i + 1:] = (
            data[:, i, i + 1:]
            - torch.einsum("kp,kpj->kj", data[:, i, :i], data[:, :i, i + 1:])
        ) / data[:, i, i].unsqueeze(1)
    diag = torch.diagonal(data, dim1=1, dim2=2).abs()
    return torch.log(diag).sum(dim=1)


def compute_constants(r: torch.Tensor,
                      diag_only: bool = False) -> tuple[torch.Tensor, torch.Tensor]:
    """Rinv + per-cluster constant (constants_kernel, gaussian_kernel.cu:196-243).

    constant = -D/2*ln(2*pi) - 0.5*ln|R| (natural log on the GPU path).
    """
    k, d, _ = r.shape
    if diag_only:
        diag = torch.diagonal(r, dim1=1, dim2=2)
        logdet = torch.log(diag.prod(dim=1))
        rinv = torch.diag_embed(1.0 / diag)
    else:
        rinv, logdet = lu_invert_nopivot(r)
    constant = -d * 0.5 * LOG_2PI - 0.5 * logdet
    return rinv, constant


def compute_pi(n_c: torch.Tensor) -> torch.Tensor:
    """pi = N / sum(N), floored at 1e-10 for N < 0.5
    (compute_pi, gaussian_kernel.cu:172-193, with the K>256 indexing bug
    fixed — SURVEY §2.6 #3)."""
    total = n_c.sum()
    pi = n_c / total
    return torch.where(n_c < 0.5, torch.full_like(pi, 1e-10), pi)
