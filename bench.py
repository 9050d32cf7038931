#!/usr/bin/env python3
"""Flagship benchmark: EM iterations/sec at K=64, D=24, N=1M events per GPU
(BASELINE.json metric; config 2 at 1 GPU, config 3 at 8).

One step = one full EM iteration: M-step sufficient statistics + fused
RCCL all-reduce + covariance/constants finalize + E-step (bf16 reads,
fp32 accumulate) + likelihood reduce — nothing skipped.

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


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--events-per-gpu", type=int, default=None)
    p.add_argument("--dims", type=int, default=24)
    p.add_argument("--clusters", type=int, default=64)
    p.add_argument("--dtype", choices=["bf16", "fp32"], default="bf16")
    p.add_argument("--profile", action="store_true")
    args = p.parse_args()

    from cuda_gmm_mpi_amd.engine import build_engine
    from cuda_gmm_mpi_amd.parallel import dist as pdist
    from cuda_gmm_mpi_amd.utils.config import GmmConfig
    from cuda_gmm_mpi_amd.utils.synthetic import make_blobs

    rank, local_rank, world = pdist.init_process_group()
    has_gpu = torch.cuda.is_available()
    device = "cuda" if has_gpu else "cpu"
    if not has_gpu and args.events_per_gpu is None:
        args.events_per_gpu = 20_000  # CPU smoke only; GPU runs use 1M
    events_per_gpu = args.events_per_gpu or 1_000_000
    n_total = events_per_gpu * world
    k, d = args.clusters, args.dims

    # deterministic synthetic FCS-shaped data, identical on every rank
    # (no dataset downloads; random-init mixture per BASELINE.json)
    data, _ = make_blobs(n_total, d, k, seed=1234)

    cfg = GmmConfig(
        num_clusters=k, target_num_clusters=k,
        estep_dtype=("bf16" if args.dtype == "bf16" else "fp32"),
        mstep_precision=("bf16x3" if args.dtype == "bf16" else "fp32"),
    )
    engine = build_engine(data, cfg, device=device)
    if args.profile:
        # per-bucket hipEvent timers need the eager path
        engine.use_graphs = False

    def sync():
        pdist.barrier()
        if has_gpu:
            torch.cuda.synchronize()

    # initial E-step so steps start from a valid posterior state
    engine._reduce_likelihood(engine._estep(k))
    for _ in range(args.warmup):
        engine.em_iteration(k)
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        engine.em_iteration(k)
    sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks == min iterations/sec (use the slowest rank's time).
    # NCCL/RCCL reduces device tensors only.
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device="cuda" if has_gpu else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    iters_per_sec = args.steps / elapsed
    if rank == 0:
        out = {
            "metric": "em_iterations_per_sec",
            "value": iters_per_sec,
            "unit": "iters/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            # CPU smoke runs compute in fp32 regardless of the request
            "dtype": args.dtype if has_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": "gmm_em",
                "n_events": n_total,
                "events_per_gpu": events_per_gpu,
                "dims": d,
                "clusters": k,
                "parallelism": f"dp{world}",
                "device": device,
                "note": ("bf16 E-step data reads, fp32 accumulate; "
                         "split-precision bf16x3 M-step moments; "
                         "fp32 constants/finalize (BASELINE config 2/3)"),
            },
        }
        print(json.dumps(out))
        if args.profile:
            print(engine.profile.report(rank, local_rank), file=sys.stderr)
    pdist.destroy()
    return 0


if __name__ == "__main__":
    sys.exit(main())
