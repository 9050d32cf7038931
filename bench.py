#!/usr/bin/env python3
"""Flagship benchmark: EM iterations/sec at K=64, D=24, N=1M events per GPU
(BASELINE.json metric; config 2 at 1 GPU, config 3 at 8).

One step = one full EM iteration: M-step sufficient statistics + fused
RCCL all-reduce + covariance/constants finalize + E-step + likelihood
reduce — nothing skipped.

Reports BOTH precisions per run: the headline JSON
line is the --dtype mode (default bf16, sanctioned by BASELINE config 2);
its "modes" field carries the exact-fp32 measurement (fp32 E-step + fp32
MFMA moments — the reference's precision throughout, gaussian_kernel.cu)
taken back-to-back in the same process on the same data.

Run directly (1 GPU) or under torch.distributed.run with one rank per GPU:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
if REPO not in sys.path:
    sys.path.insert(0, REPO)

import numpy as np  # noqa: E402
import torch  # noqa: E402


def time_mode(dtype: str, data, device: str, k: int, steps: int,
              warmup: int, world: int, profile: bool) -> dict:
    """Build an engine in the given precision and time `steps` iterations
    bracketed by barrier + synchronize on both sides; MAX over ranks."""
    from cuda_gmm_mpi_amd.engine import build_engine
    from cuda_gmm_mpi_amd.parallel import dist as pdist
    from cuda_gmm_mpi_amd.utils.config import GmmConfig

    has_gpu = device == "cuda"
    cfg = GmmConfig(
        num_clusters=k, target_num_clusters=k,
        estep_dtype=("bf16" if dtype == "bf16" else "fp32"),
        mstep_precision=("bf16x3" if dtype == "bf16" else "fp32"),
    )
    engine = build_engine(data, cfg, device=device)
    if profile:
        engine.use_graphs = False  # per-bucket hipEvent timers need eager

    def sync():
        pdist.barrier()
        if has_gpu:
            torch.cuda.synchronize()

    # initial E-step so steps start from a valid posterior state
    engine._reduce_likelihood(engine._estep(k))
    for _ in range(warmup):
        engine.em_iteration(k)
    sync()
    t0 = time.perf_counter()
    for _ in range(steps):
        engine.em_iteration(k)
    sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks == min iterations/sec (slowest rank's time).
    # NCCL/RCCL reduces device tensors only.
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device="cuda" if has_gpu else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())
    report = engine.profile.report(0, 0) if profile else None
    likelihood = float(engine._lik_dev.item())  # sanity: finite math ran
    del engine
    if has_gpu:
        torch.cuda.empty_cache()
    return {
        "value": steps / elapsed,
        "ms_per_step": elapsed / steps * 1e3,
        "likelihood_finite": bool(np.isfinite(likelihood)),
        "profile": report,
    }


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=None)
    p.add_argument("--warmup", type=int, default=None)
    p.add_argument("--events-per-gpu", type=int, default=None)
    p.add_argument("--dims", type=int, default=24)
    p.add_argument("--clusters", type=int, default=64)
    p.add_argument("--dtype", choices=["bf16", "fp32"], default="bf16")
    p.add_argument("--modes", choices=["both", "single"], default=None,
                   help="'both' (GPU default) also measures the other "
                        "precision back-to-back and reports it in the "
                        "'modes' field of the JSON line")
    p.add_argument("--profile", action="store_true")
    args = p.parse_args()

    from cuda_gmm_mpi_amd.parallel import dist as pdist
    from cuda_gmm_mpi_amd.utils.synthetic import make_blobs

    rank, local_rank, world = pdist.init_process_group()
    has_gpu = torch.cuda.is_available()
    device = "cuda" if has_gpu else "cpu"
    if not has_gpu and args.events_per_gpu is None:
        args.events_per_gpu = 20_000  # CPU smoke only; GPU runs use 1M
    events_per_gpu = args.events_per_gpu or 1_000_000
    steps = args.steps if args.steps is not None else (300 if has_gpu else 10)
    warmup = (args.warmup if args.warmup is not None
              else (20 if has_gpu else 2))
    modes = args.modes or ("both" if has_gpu else "single")
    n_total = events_per_gpu * world
    k, d = args.clusters, args.dims

    # deterministic synthetic FCS-shaped data, identical on every rank
    # (no dataset downloads; random-init mixture per BASELINE.json)
    data, _ = make_blobs(n_total, d, k, seed=1234)

    main_dtype = args.dtype if has_gpu else "fp32"
    res = time_mode(main_dtype, data, device, k, steps, warmup, world,
                    args.profile)
    extra = {}
    if modes == "both" and has_gpu:
        other = "fp32" if main_dtype == "bf16" else "bf16"
        extra[other] = time_mode(other, data, device, k, steps, warmup,
                                 world, False)

    if rank == 0:
        out = {
            "metric": "em_iterations_per_sec",
            "value": res["value"],
            "unit": "iters/s",
            "n_gpus": world,
            "steps": steps,
            "warmup": warmup,
            "ms_per_step": res["ms_per_step"],
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": main_dtype,
            "data": "synthetic",
            "modes": {
                dt: {"value": m["value"], "ms_per_step": m["ms_per_step"],
                     "dtype": dt,
                     "note": ("exact fp32 throughout (the reference's "
                              "precision)" if dt == "fp32" else
                              "bf16 E-step reads + bf16x3 split-precision "
                              "moments, fp32 accumulate")}
                for dt, m in {main_dtype: res, **extra}.items()
            },
            "config": {
                "model": "gmm_em",
                "n_events": n_total,
                "events_per_gpu": events_per_gpu,
                "dims": d,
                "clusters": k,
                "parallelism": f"dp{world}",
                "device": device,
                "note": ("bf16 E-step data reads, fp32 accumulate; "
                         "split-precision bf16x3 M-step moments; fp32 "
                         "constants/finalize (BASELINE config 2/3); "
                         "'modes.fp32' is the exact-fp32 measurement"
                         if main_dtype == "bf16" else
                         "exact fp32 E-step and M-step (reference "
                         "precision)"),
            },
        }
        print(json.dumps(out))
        if args.profile and res["profile"]:
            print(res["profile"], file=sys.stderr)
    pdist.destroy()
    return 0


if __name__ == "__main__":
    sys.exit(main())
