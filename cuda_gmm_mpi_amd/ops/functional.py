"""Device-dispatched EM ops: torch-CPU reference vs HIP/CDNA4 kernels.

The engine calls only these. CPU tensors -> ``cpu_reference`` (golden
semantics); CUDA tensors -> hand-written gfx950 kernels via the extension
(loud failure when unbuilt; no eager fallback).

GEMM-shaped cold ops (mean numerators = W @ [X;1]^T) go through rocBLAS via
torch.matmul, per the MI355X design rules (library GEMMs for plain GEMMs,
custom kernels for the fused hot ops).
"""
from __future__ import annotations

import torch

from . import cpu_reference as cpu
from .backend import hip_ext


def estep_logw(x: torch.Tensor, means: torch.Tensor, rinv: torch.Tensor,
               constant: torch.Tensor, pi: torch.Tensor,
               diag_only: bool = False,
               out: torch.Tensor | None = None) -> torch.Tensor:
    """logw [K, N]; x may be fp32 or bf16 (bf16 reads, fp32 accumulate)."""
    if x.is_cuda:
        k, n = means.shape[0], x.shape[1]
        if out is None:
            out = torch.empty((k, n), dtype=torch.float32, device=x.device)
        hip_ext().estep_logw(
            x, means, rinv, constant, torch.log(pi), out, bool(diag_only)
        )
        return out
    logw = cpu.estep_logw(x.float(), means, rinv, constant, pi, diag_only)
    if out is not None:
        out.copy_(logw)
        return out
    return logw


def estep_posteriors(logw: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """In-place posteriors on CUDA; returns (w, likelihood scalar tensor)."""
    if logw.is_cuda:
        nblocks = 1024
        partial = torch.zeros(nblocks, dtype=torch.float32, device=logw.device)
        hip_ext().estep_posteriors(logw, partial)
        return logw, partial.sum()
    w, lik = cpu.estep_posteriors(logw)
    logw.copy_(w)  # in-place like the CUDA path: the buffer becomes posteriors
    return logw, lik


def mstep_n_means(x_aug_t: torch.Tensor, w: torch.Tensor
                  ) -> tuple[torch.Tensor, torch.Tensor]:
    """(N [K], mean numerators [K, D]) via one rocBLAS GEMM.

    x_aug_t: [N, D+1] = [X^T | 1] (precomputed once). Fuses the reference's
    mstep_N and mstep_means kernels (gaussian_kernel.cu:522-577) into a
    single skinny GEMM: W @ X_aug^T -> [K, D+1].
    """
    nm = w @ x_aug_t                        # [K, D+1]
    return nm[:, -1].contiguous(), nm[:, :-1].contiguous()


_tri_cache: dict = {}


def _tri_unpack_index(d: int, device) -> torch.Tensor:
    """[D*D] gather index from packed lower-triangle to full matrix."""
    key = (d, str(device))
    idx = _tri_cache.get(key)
    if idx is None:
        full = torch.empty(d, d, dtype=torch.long)
        for i in range(d):
            for j in range(d):
                r, c = (i, j) if i >= j else (j, i)
                full[i, j] = r * (r + 1) // 2 + c
        idx = full.reshape(-1).to(device)
        _tri_cache[key] = idx
    return idx


def mstep_covariance_s(x: torch.Tensor, w: torch.Tensor,
                       out: torch.Tensor | None = None,
                       nchunk: int = 64) -> torch.Tensor:
    """Uncentered weighted second moments S [K, D, D].

    CUDA: custom LDS-tiled kernel (the covariance showpiece) producing
    deterministic per-chunk packed partials, summed + unpacked here; CPU:
    batched torch matmuls.
    """
    if x.is_cuda:
        k = w.shape[0]
        d = x.shape[0]
        n = x.shape[1]
        p = d * (d + 1) // 2
        te = min((16384 - 4 * d) // (d + 1) & ~3, 256)  # matches gmm_ext.hip
        tiles = (n + te - 1) // te
        nchunk = int(min(nchunk, tiles))
        partials = torch.empty((nchunk, k, p), dtype=torch.float32,
                               device=x.device)
        hip_ext().mstep_covariance_partials(x, w, partials)
        packed = partials.sum(dim=0)                      # [K, P]
        idx = _tri_unpack_index(d, x.device)
        s = packed[:, idx].view(k, d, d)
        if out is not None:
            out.copy_(s)
            return out
        return s
    k, n = w.shape
    d = x.shape[0]
    xf = x.float()
    s = out if out is not None else torch.empty((k, d, d), dtype=torch.float32)
    for c in range(k):
        s[c] = (xf * w[c].unsqueeze(0)) @ xf.T
    return s


def constants(r: torch.Tensor, diag_only: bool = False
              ) -> tuple[torch.Tensor, torch.Tensor]:
    """(Rinv [K,D,D], constant [K]) via no-pivot LU + ln|det|."""
    if r.is_cuda:
        k, d, _ = r.shape
        rinv = torch.empty_like(r)
        logdet = torch.empty(k, dtype=torch.float32, device=r.device)
        hip_ext().constants(r, rinv, logdet, bool(diag_only))
        const = -d * 0.5 * cpu.LOG_2PI - 0.5 * logdet
        return rinv, const
    return cpu.compute_constants(r, diag_only)


finalize_means = cpu.finalize_means
finalize_covariance = cpu.finalize_covariance
compute_pi = cpu.compute_pi
