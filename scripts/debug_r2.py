#!/usr/bin/env python3
"""Round-2 isolation: big-D path NaN at K=256/D=128 fp32, config-4
slowdown, and constants kernel cost decomposition."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from cuda_gmm_mpi_amd.engine import build_engine
from cuda_gmm_mpi_amd.ops import functional as F
from cuda_gmm_mpi_amd.utils.config import GmmConfig


def timeit(f, iters=50, warm=3):
    for _ in range(warm):
        f()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        f()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def nan_isolation():
    rng = np.random.default_rng(47)
    data = rng.standard_normal((60000, 128)).astype(np.float32) * 10
    for variant in ("default", "nographs", "valu", "valu-nographs"):
        cfg = GmmConfig(num_clusters=256, target_num_clusters=256,
                        min_iters=1, max_iters=1)
        eng = build_engine(data, cfg, device="cuda")
        if "nographs" in variant:
            eng.use_graphs = False
        if "valu" in variant:
            eng.use_big_estep = False
            eng.mfac32 = None
        liks = []
        lik = eng._reduce_likelihood(eng._estep(256))
        liks.append(lik)
        for i in range(4):
            eng.em_iteration(256)
            liks.append(float(eng._lik_dev.item()))
        print(f"[nan] {variant}: {['%.6e' % v for v in liks]}", flush=True)


def flagship_timing():
    rng = np.random.default_rng(2)
    d, n, k = 24, 1_000_000, 64
    x = torch.from_numpy(
        rng.standard_normal((d, n)).astype(np.float32)).cuda()
    xb = x.to(torch.bfloat16)
    xs = F.split_bf16_planes(x)
    w = torch.rand(k, n, device="cuda")
    lse = torch.zeros(n, device="cuda")
    means = torch.randn(k, d, device="cuda")
    r = torch.eye(d, device="cuda").expand(k, d, d).contiguous() * 3
    mfac = torch.empty(k, 2, 32, 32, dtype=torch.bfloat16, device="cuda")
    rinv, const = F.constants(r, means, False, mfac)
    add = const + float(np.log(1.0 / k))
    wo = torch.empty(k, n, device="cuda")
    print("[f] estep_fused bf16  : %.3f ms"
          % timeit(lambda: F.estep_fused(xb, mfac, add, wo, lse), 50),
          flush=True)
    lw = torch.randn(k, n, device="cuda") * 3
    print("[f] moments_b16 +lse  : %.3f ms"
          % timeit(lambda: F.mstep_moments(x, lw, precision="bf16x3",
                                           x_split=xs, lse=lse), 30),
          flush=True)
    print("[f] moments_b16 plain : %.3f ms"
          % timeit(lambda: F.mstep_moments(x, w, precision="bf16x3",
                                           x_split=xs), 30), flush=True)
    mfac32 = torch.empty(k, 32, 32, dtype=torch.float32, device="cuda")
    rinv, const = F.constants(r, means, False, mfac, mfac32)
    print("[f] estep_fused_f32   : %.3f ms"
          % timeit(lambda: F.estep_fused_f32(x, mfac32, add, wo, lse), 30),
          flush=True)
    print("[f] moments_f32 +lse  : %.3f ms"
          % timeit(lambda: F.mstep_moments(x, lw, precision="fp32",
                                           lse=lse), 20), flush=True)


def big_path_timing():
    rng = np.random.default_rng(1)
    d, n, k = 128, 500_000, 256
    x = torch.from_numpy(
        rng.standard_normal((d, n)).astype(np.float32)).cuda()
    xb = x.to(torch.bfloat16)
    w = torch.rand(k, n, device="cuda")
    logw = torch.randn(k, n, device="cuda") * 5
    lse = torch.empty(n, device="cuda")
    mfac = torch.empty(k, *F.mfac_shape(d), dtype=torch.bfloat16,
                       device="cuda")
    mfac32 = torch.empty(k, *F.mfac_shape(d)[1:], dtype=torch.float32,
                         device="cuda")
    means = torch.randn(k, d, device="cuda")
    r = (torch.eye(d, device="cuda").expand(k, d, d).contiguous() * 3)
    rinv, const = F.constants(r, means, False, mfac, mfac32)
    add = const + np.log(1.0 / k)
    out = torch.empty(k, n, device="cuda")

    print("[t] estep_logw_big bf16 : %.3f ms"
          % timeit(lambda: F.estep_logw_big(xb, mfac, add, out), 20), flush=True)
    print("[t] estep_logw_big f32  : %.3f ms"
          % timeit(lambda: F.estep_logw_big_f32(x, mfac32, add, out), 10),
          flush=True)
    lw2 = logw.clone()
    print("[t] estep_lse K=256     : %.3f ms"
          % timeit(lambda: F.estep_lse(logw, lse), 20), flush=True)
    print("[t] estep_posteriors    : %.3f ms"
          % timeit(lambda: F.estep_posteriors(lw2), 20), flush=True)
    print("[t] moments_big no lse  : %.3f ms"
          % timeit(lambda: F.mstep_moments(x, w, precision="bf16x3"), 10),
          flush=True)
    print("[t] moments_big +lse    : %.3f ms"
          % timeit(lambda: F.mstep_moments(x, w, precision="bf16x3",
                                           lse=lse), 10), flush=True)


def constants_decomposition():
    k, d = 64, 24
    means = torch.randn(k, d, device="cuda")
    a = torch.randn(k, d, d, device="cuda")
    r = a @ a.transpose(1, 2) + d * torch.eye(d, device="cuda")
    mfac = torch.empty(k, 2, 32, 32, dtype=torch.bfloat16, device="cuda")
    mfac32 = torch.empty(k, 32, 32, dtype=torch.float32, device="cuda")
    pi = torch.full((k,), 1.0 / k, device="cuda")
    add = torch.empty(k, device="cuda")
    from cuda_gmm_mpi_amd.ops.backend import hip_ext
    ext = hip_ext()
    rinv = torch.empty_like(r)
    logdet = torch.empty(k, device="cuda")
    const = torch.empty(k, device="cuda")
    empty_b = torch.empty(0, dtype=torch.bfloat16, device="cuda")
    empty_f = torch.empty(0, dtype=torch.float32, device="cuda")

    def lu_only():
        ext.constants(r, means, pi, rinv, logdet, const, add, empty_b,
                      empty_f, False)

    def lu_emit():
        ext.constants(r, means, pi, rinv, logdet, const, add, mfac,
                      mfac32, False)

    def emit_only():
        ext.emit_factors(rinv, means, mfac, mfac32)

    print("[c] constants LU only      : %.1f us" % (timeit(lu_only, 200) * 1e3), flush=True)
    print("[c] constants LU + emit    : %.1f us" % (timeit(lu_emit, 200) * 1e3), flush=True)
    print("[c] emit_factors only      : %.1f us" % (timeit(emit_only, 200) * 1e3), flush=True)
    # K=256 D=128 (config 4 shape)
    k, d = 256, 128
    means = torch.randn(k, d, device="cuda")
    a = torch.randn(k, d, d, device="cuda")
    r = a @ a.transpose(1, 2) + d * torch.eye(d, device="cuda")
    mfac = torch.empty(k, *F.mfac_shape(d), dtype=torch.bfloat16,
                       device="cuda")
    rinv = torch.empty_like(r)
    logdet = torch.empty(k, device="cuda")
    const = torch.empty(k, device="cuda")
    pi = torch.full((k,), 1.0 / k, device="cuda")
    add = torch.empty(k, device="cuda")

    def lu_emit_big():
        ext.constants(r, means, pi, rinv, logdet, const, add, mfac,
                      empty_f, False)

    print("[c] constants K256 D128    : %.1f us" % (timeit(lu_emit_big, 30) * 1e3), flush=True)


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    if which in ("all", "nan"):
        nan_isolation()
    if which in ("all", "timing"):
        big_path_timing()
    if which in ("all", "flagship"):
        flagship_timing()
    if which in ("all", "constants"):
        constants_decomposition()
