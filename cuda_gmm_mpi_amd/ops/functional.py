"""Device-dispatched EM ops: torch-CPU reference vs HIP/CDNA4 kernels.

The engine calls only these. CPU tensors -> ``cpu_reference`` (golden
semantics); CUDA tensors -> hand-written gfx950 kernels via the extension
(loud failure when unbuilt; no eager fallback).

GEMM-shaped cold ops (mean numerators = W @ [X;1]^T) go through rocBLAS via
torch.matmul, per the MI355X design rules (library GEMMs for plain GEMMs,
custom kernels for the fused hot ops).
"""
from __future__ import annotations

import os

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


def estep_lse(logw: torch.Tensor, lse: torch.Tensor,
              need_lik: bool = True) -> torch.Tensor | None:
    """Per-event log-sum-exp into lse [N] + likelihood scalar (CUDA).

    logw is NOT normalized — the lse-aware M-step kernels apply
    exp(logw - lse) while staging w, deleting the posterior write+read
    round trip of the two-kernel path."""
    nblocks = 1024
    partial = torch.zeros(nblocks, dtype=torch.float32, device=logw.device)
    hip_ext().estep_lse(logw, lse, partial)
    if not need_lik:
        return None
    lik = torch.empty(1, dtype=torch.float32, device=logw.device)
    hip_ext().reduce_scalar(partial, lik)
    return lik


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
                       nchunk: int = 64,
                       lse: torch.Tensor | None = None) -> torch.Tensor:
    """Uncentered weighted second moments S [K, D, D].

    CUDA: custom LDS-tiled kernel (the covariance showpiece) producing
    deterministic per-chunk packed partials, summed + unpacked here; CPU:
    batched torch matmuls.
    """
    if x.is_cuda and x.shape[0] > 128:
        # past the covariance kernel's D cap: per-cluster rocBLAS GEMMs
        # (the exact-fp32 D>128 quadrant; ~K GEMMs of D x N x D)
        k, n = w.shape
        d = x.shape[0]
        if lse is not None and lse.numel() > 0:
            w = torch.exp(w - lse.unsqueeze(0))
        s = out if out is not None else torch.empty(
            (k, d, d), dtype=torch.float32, device=x.device)
        for c in range(k):
            s[c] = (x * w[c].unsqueeze(0)) @ x.T
        return s
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
        if lse is None:
            lse = torch.empty(0, dtype=torch.float32, device=x.device)
        hip_ext().mstep_covariance_partials(x, w, lse, partials)
        packed = torch.empty((k, p), dtype=torch.float32, device=x.device)
        hip_ext().reduce_chunks(partials, packed)         # [K, P]
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


def constants(r: torch.Tensor, means: torch.Tensor | None = None,
              diag_only: bool = False, mfac: torch.Tensor | None = None,
              mfac32: torch.Tensor | None = None,
              pi_add: tuple[torch.Tensor, torch.Tensor] | None = None,
              out: tuple[torch.Tensor, torch.Tensor] | None = None
              ) -> tuple[torch.Tensor, torch.Tensor]:
    """(Rinv [K,D,D], constant [K]) via no-pivot LU + ln|det|.

    On CUDA, when ``mfac`` (bf16 [K,2,32,32]) is given, the kernel also
    emits the fused-E-step factor M = [U | -U mu] (U^T U = Rinv) as bf16
    hi/lo fragment pairs — requires ``means`` and D <= 31.
    """
    k, d, _ = r.shape
    if r.is_cuda and d > 142:
        # the constants/emission kernels stage TWO d x (d|1) fp32 planes
        # + a d-float mu buffer in LDS (race-free LU snapshot /
        # triangular-inverse working set); that crosses gfx950's 160 KB
        # workgroup limit at exactly D = 143. Fall back to the CPU LU
        # (bit-faithful invert_cpu math) — the MFMA factor paths are
        # gated to D <= 142, so no caller needs mfac here.
        if mfac is not None and mfac.numel() > 0:
            raise ValueError("factor emission needs D <= 142 (LDS bound)")
        rinv_c, const_c = cpu.compute_constants(r.detach().cpu(), diag_only)
        rinv = out[0] if out is not None else torch.empty_like(r)
        rinv.copy_(rinv_c.to(r.device))
        const = const_c.to(r.device)
        if pi_add is not None:
            pi_t, add_t = pi_add
            const_t = (out[1] if out is not None
                       else torch.empty(k, dtype=torch.float32,
                                        device=r.device))
            const_t.copy_(const)
            add_t.copy_(const + torch.log(pi_t))
            return rinv, const_t
        if out is not None:
            out[1].copy_(const)
            return rinv, out[1]
        return rinv, const
    if r.is_cuda:
        rinv = out[0] if out is not None else torch.empty_like(r)
        logdet = torch.empty(k, dtype=torch.float32, device=r.device)
        if means is None:
            means = torch.zeros(k, d, dtype=torch.float32, device=r.device)
        if mfac is None:
            mfac = torch.empty(0, dtype=torch.bfloat16, device=r.device)
        if mfac32 is None:
            mfac32 = torch.empty(0, dtype=torch.float32, device=r.device)
        empty = torch.empty(0, dtype=torch.float32, device=r.device)
        if pi_add is not None:
            pi_t, add_t = pi_add
            const_t = (out[1] if out is not None
                       else torch.empty(k, dtype=torch.float32,
                                        device=r.device))
            hip_ext().constants(r, means, pi_t, rinv, logdet, const_t,
                                add_t, mfac, mfac32, bool(diag_only))
            if diag_only:  # diag kernel does not emit constant/add
                cval = -d * 0.5 * cpu.LOG_2PI - 0.5 * logdet
                if out is not None:
                    const_t.copy_(cval)
                else:
                    const_t = cval
                add_t.copy_(cval + torch.log(pi_t))
            return rinv, const_t
        hip_ext().constants(r, means, empty, rinv, logdet, empty, empty,
                            mfac, mfac32, bool(diag_only))
        const = -d * 0.5 * cpu.LOG_2PI - 0.5 * logdet
        if out is not None:
            out[1].copy_(const)
            const = out[1]
        return rinv, const
    rinv_c, const_c = cpu.compute_constants(r, diag_only)
    if out is not None:
        out[0].copy_(rinv_c)
        out[1].copy_(const_c)
        return out[0], out[1]
    return rinv_c, const_c


def estep_fused(z: torch.Tensor, mfac: torch.Tensor, add: torch.Tensor,
                w_out: torch.Tensor, lse: torch.Tensor,
                need_lik: bool = True
                ) -> tuple[torch.Tensor, torch.Tensor | None]:
    """Fused bf16-MFMA E-step (CUDA only): LOG weights into w_out [K,N],
    per-event log-sum-exp into lse [N], returns (w_out, likelihood).
    Online-softmax over 256-event blocks; the lse-aware M-step applies
    exp(w_out - lse) on the fly — posteriors are never materialized."""
    n = z.shape[1]
    nblk = (n + 255) // 256
    # every launched block writes its partial slot: no zero-fill needed
    partial = torch.empty(nblk, dtype=torch.float32, device=z.device)
    hip_ext().estep_fused(z, mfac, add, w_out, lse, partial)
    if not need_lik:
        return w_out, None
    lik = torch.empty(1, dtype=torch.float32, device=z.device)
    hip_ext().reduce_scalar(partial, lik)
    return w_out, lik


def estep_fused_available(device: torch.device, dtype: str, d: int,
                          k: int) -> bool:
    """Fused path gate: D <= 31 (any K — the online-softmax redesign
    removed the K-sized logw LDS buffer)."""
    return device.type == "cuda" and d <= 31


def estep_fused_lds_available(device: torch.device, dtype: str, d: int,
                              k: int) -> bool:
    """v1 lw-in-LDS fast path: writes posteriors directly (the M-step
    then skips the exp at staging, which sits on its VALU-bound critical
    path — measured ~19 us/iter at K=64). LDS-bounded in K."""
    if device.type != "cuda" or d > 31:
        return False
    if dtype == "bf16":
        return 128 * 40 * 2 + 4 * k * 132 <= 64 * 1024   # K <= 104
    return 4 * (128 * 33 + k * 132) <= 64 * 1024         # K <= 85


def estep_fused_lds(z: torch.Tensor, mfac: torch.Tensor, add: torch.Tensor,
                    w_out: torch.Tensor, need_lik: bool = True
                    ) -> tuple[torch.Tensor, torch.Tensor | None]:
    """v1 fused bf16 E-step: POSTERIORS into w_out, likelihood scalar."""
    n = z.shape[1]
    nblk = (n + 127) // 128
    partial = torch.empty(nblk, dtype=torch.float32, device=z.device)
    hip_ext().estep_fused_lds(z, mfac, add, w_out, partial)
    if not need_lik:
        return w_out, None
    lik = torch.empty(1, dtype=torch.float32, device=z.device)
    hip_ext().reduce_scalar(partial, lik)
    return w_out, lik


def estep_fused_f32_lds(z: torch.Tensor, mfac32: torch.Tensor,
                        add: torch.Tensor, w_out: torch.Tensor,
                        need_lik: bool = True
                        ) -> tuple[torch.Tensor, torch.Tensor | None]:
    """v1 exact-f32 fused E-step: POSTERIORS into w_out."""
    n = z.shape[1]
    nblk = (n + 127) // 128
    partial = torch.empty(nblk, dtype=torch.float32, device=z.device)
    hip_ext().estep_fused_f32_lds(z, mfac32, add, w_out, partial)
    if not need_lik:
        return w_out, None
    lik = torch.empty(1, dtype=torch.float32, device=z.device)
    hip_ext().reduce_scalar(partial, lik)
    return w_out, lik


def estep_fused_f32(z: torch.Tensor, mfac32: torch.Tensor, add: torch.Tensor,
                    w_out: torch.Tensor, lse: torch.Tensor,
                    need_lik: bool = True
                    ) -> tuple[torch.Tensor, torch.Tensor | None]:
    """Exact-f32 MFMA fused E-step (CUDA, D <= 31, any K): log weights
    into w_out, per-event log-sum-exp into lse."""
    n = z.shape[1]
    nblk = (n + 255) // 256
    partial = torch.empty(nblk, dtype=torch.float32, device=z.device)
    hip_ext().estep_fused_f32(z, mfac32, add, w_out, lse, partial)
    if not need_lik:
        return w_out, None
    lik = torch.empty(1, dtype=torch.float32, device=z.device)
    hip_ext().reduce_scalar(partial, lik)
    return w_out, lik


def estep_big_available(device: torch.device, dtype: str, d: int) -> bool:
    """Big-D MFMA logw path gate (D <= 142, bf16 or exact-f32 MFMA; the
    factor-emission LDS working set crosses 160 KB at D = 143). The
    engine prefers the fused kernel for D <= 31."""
    return device.type == "cuda" and dtype in ("bf16", "fp32") and 1 <= d <= 142


def mfac_shape(d: int) -> tuple[int, int, int]:
    """Factor tensor shape [2, RT*32, KCT*16] for estep MFMA kernels."""
    rows = ((d + 31) // 32) * 32
    kc = (d + 1 + 15) // 16
    kct = 2 if kc <= 2 else 3 if kc == 3 else 5 if kc <= 5 else 9
    return 2, rows, kct * 16


def estep_logw_big(z: torch.Tensor, mfac: torch.Tensor, add: torch.Tensor,
                   out: torch.Tensor) -> torch.Tensor:
    """Big-D MFMA log-weights into out [K, N] (CUDA, bf16, D > 31)."""
    hip_ext().estep_logw_big(z, mfac, add, out)
    return out


def estep_logw_big_f32(z: torch.Tensor, mfac32: torch.Tensor,
                       add: torch.Tensor, out: torch.Tensor) -> torch.Tensor:
    """Exact-f32 big-D MFMA log-weights into out [K, N] (CUDA, D > 31)."""
    hip_ext().estep_logw_big_f32(z, mfac32, add, out)
    return out


def split_bf16_planes(x: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """Persistent hi/lo bf16 split of fp32 data (x = hi + lo + O(2^-18 x))."""
    hi = x.to(torch.bfloat16)
    lo = (x - hi.to(torch.float32)).to(torch.bfloat16)
    return hi.contiguous(), lo.contiguous()


def mstep_moments(x: torch.Tensor, w: torch.Tensor,
                  nchunk: int | None = None,
                  precision: str = "fp32",
                  x_split: tuple[torch.Tensor, torch.Tensor] | None = None,
                  lse: torch.Tensor | None = None) -> torch.Tensor:
    """Fused augmented sufficient statistics, packed lower triangle of
    T_c = sum_e w_ce [x;1][x;1]^T per cluster: [K, Dp*(Dp+1)/2] with layout
    [S_tri (D rows) | mean numerators (row D) | N (corner)] — the single
    all-reduce payload of the M-step.

    CUDA fast path (D <= 31): one f32-MFMA kernel (exact fp32 fmaf chain).
    Otherwise composed from the covariance kernel + rocBLAS/torch ops.
    """
    k, n = w.shape
    d = x.shape[0]
    dp = d + 1
    pp = dp * (dp + 1) // 2
    p = d * (d + 1) // 2
    if x.is_cuda and d <= 31:
        tiles = (n + 127) // 128
        if nchunk is None:
            # enough chunk-parallelism to fill 8 XCDs x 32 CUs, capped so
            # the partial buffer stays <= 64 MB (measured sweeps: bf16x3
            # fastest at ~256 chunks, fp32 at ~512, K=64 N=1M)
            target = 256 if precision == "bf16x3" else 512
            nchunk = max(1, min(target, (64 << 20) // (4 * k * pp)))
        nchunk = int(min(nchunk, tiles))
        partials = torch.empty((nchunk, k, pp), dtype=torch.float32,
                               device=x.device)
        lse_t = (lse if lse is not None
                 else torch.empty(0, dtype=torch.float32, device=x.device))
        if precision == "bf16x3":
            if x_split is None:
                x_split = split_bf16_planes(x)
            hip_ext().mstep_moments_b16(x_split[0], x_split[1], w, lse_t,
                                        partials)
        else:
            hip_ext().mstep_moments(x, w, lse_t, partials)
        out = torch.empty((k, pp), dtype=torch.float32, device=x.device)
        hip_ext().reduce_chunks(partials, out)
        return out
    if x.is_cuda and precision == "bf16x3" and d <= 159:
        tiles = (n + 63) // 64  # MBB_BK
        if nchunk is None and os.environ.get("GMM_MOMENTS_NCHUNK"):
            nchunk = int(os.environ["GMM_MOMENTS_NCHUNK"])
        if nchunk is None:
            # big-D packed rows are large (Pp ~ 8k at D=128): a small byte
            # cap starves chunk-parallelism (7 chunks = 448 blocks at
            # K=256). The partial buffer is cheap in 288 GB and its sum is
            # ~90 us/GB; keep >= 8 XCDs x 32 CUs of blocks instead.
            nchunk = max(1, min(256, (512 << 20) // (4 * k * pp)))
        nchunk = int(min(nchunk, tiles))
        partials = torch.empty((nchunk, k, pp), dtype=torch.float32,
                               device=x.device)
        lse_t = (lse if lse is not None
                 else torch.empty(0, dtype=torch.float32, device=x.device))
        hip_ext().mstep_moments_big(x, w, lse_t, partials)
        out = torch.empty((k, pp), dtype=torch.float32, device=x.device)
        hip_ext().reduce_chunks(partials, out)
        return out
    packed = torch.empty((k, pp), dtype=torch.float32, device=x.device)
    if x.is_cuda:
        if lse is not None:
            # fallback path mixes rocBLAS GEMMs that need materialized
            # posteriors. MUST be non-destructive: w holds the E-step's
            # logw, and an in-place normalize would corrupt the first
            # graph replay (the eager capture-warmup iteration would
            # leave posteriors where the replayed ops expect logw).
            w = torch.exp(w - lse.unsqueeze(0))
        s = mstep_covariance_s(x, w)        # custom kernel, [K, D, D]
        mean_num = w @ x.T                  # rocBLAS
        n_c = w.sum(dim=1)
    else:
        n_c, mean_num, s = cpu.mstep_sufficient_stats(x.float(), w)
    tri = torch.tril_indices(d, d, device=x.device)
    packed[:, :p] = s[:, tri[0], tri[1]]
    packed[:, p:p + d] = mean_num
    packed[:, -1] = n_c
    return packed


def moments_views(packed: torch.Tensor, d: int
                  ) -> tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """(N [K], mean numerators [K,D], S [K,D,D]) from the packed layout."""
    k = packed.shape[0]
    p = d * (d + 1) // 2
    n_c = packed[:, -1]
    mean_num = packed[:, p:p + d]
    idx = _tri_unpack_index(d, packed.device)
    s = packed[:, :p][:, idx].view(k, d, d)
    return n_c, mean_num, s


finalize_means = cpu.finalize_means
finalize_covariance = cpu.finalize_covariance
compute_pi = cpu.compute_pi
