#!/usr/bin/env python3
"""Run the BASELINE.json measurement configs and emit one JSON line each.

  1  K=4,  D=2,   N=10k   CPU reference path (plumbing)
  2  K=64, D=24,  N=1M    one GPU, bf16 E-step      (bench.py default)
  3  K=64, D=24,  N=8M    8 GPUs (driver runs this via bench.py --gpus 8)
  4  K=256,D=128, N=4M    MFMA covariance stress (D>31 fallback paths)
  5  K=100->20, D=21, N=2M  full MDL outer loop + inversion path

Usage: python scripts/run_configs.py [1 2 4 5] [--scale F] [--iters N]
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from cuda_gmm_mpi_amd.engine import build_engine  # noqa: E402
from cuda_gmm_mpi_amd.utils.config import GmmConfig  # noqa: E402
from cuda_gmm_mpi_amd.utils.synthetic import make_blobs  # noqa: E402


def time_em(data, cfg, device, k, steps, warmup):
    eng = build_engine(data, cfg, device=device)
    eng._reduce_likelihood(eng._estep(k))
    for _ in range(warmup):
        eng.em_iteration(k)
    if device == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        eng.em_iteration(k)
    if device == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return steps / dt, eng


def main():
    p = argparse.ArgumentParser()
    p.add_argument("configs", nargs="*", type=int, default=[1, 2, 4, 5])
    p.add_argument("--scale", type=float, default=1.0,
                   help="event-count scale factor (for quick checks)")
    p.add_argument("--iters", type=int, default=None,
                   help="override EM iterations for config 5")
    args = p.parse_args()
    configs = args.configs or [1, 2, 4, 5]
    dev = "cuda" if torch.cuda.is_available() else "cpu"

    def emit(cid, payload):
        payload.update({"config": cid})
        print(json.dumps(payload), flush=True)

    if 1 in configs:
        data, _ = make_blobs(int(10_000 * args.scale), 2, 4, seed=1)
        cfg = GmmConfig(num_clusters=4, target_num_clusters=4,
                        min_iters=25, max_iters=25)
        t0 = time.perf_counter()
        eng = build_engine(data, cfg, device="cpu")
        lik = eng.run_em(4)
        emit(1, {"device": "cpu", "likelihood": lik,
                 "seconds": time.perf_counter() - t0})

    if 2 in configs:
        n = int(1_000_000 * args.scale)
        data, _ = make_blobs(n, 24, 64, seed=1234)
        cfg = GmmConfig(num_clusters=64, target_num_clusters=64,
                        estep_dtype="bf16", mstep_precision="bf16x3")
        ips, _ = time_em(data, cfg, dev, 64, 30, 5)
        emit(2, {"device": dev, "n": n, "em_iters_per_sec": ips})

    if 4 in configs:
        n = int(4_000_000 * args.scale)
        data, _ = make_blobs(n, 128, 64, seed=7)
        cfg = GmmConfig(num_clusters=256, target_num_clusters=256,
                        estep_dtype="bf16", mstep_precision="bf16x3")
        ips, eng = time_em(data, cfg, dev, 256, 5, 2)
        emit(4, {"device": dev, "n": n, "em_iters_per_sec": ips,
                 "big_estep": eng.use_big_estep})

    rc = 0
    if 5 in configs:
        # 100 equal-weight, separation-checked clusters: elimination
        # cannot mass-kill, the merge path steps K down one at a time and
        # the sweep LANDS on the target K=20 (the reference's save-target
        # path, gaussian.cu:839) — asserted below, nonzero exit otherwise
        from cuda_gmm_mpi_amd.utils.synthetic import make_supported_blobs
        n = int(2_000_000 * args.scale)
        # unit-spread (|det R| ~ 1): the reference's log10 merge-constant
        # quirk is then magnitude-neutral and cannot trigger mass die-off
        # (measured: scale-250 data at D=21 shifts merged constants by
        # ~+38 nats, the merged cluster swallows the next E-step and the
        # sweep jumps past the target)
        data, _ = make_supported_blobs(n, 21, 100, seed=11,
                                       scale=10.0, spread=1.0)
        iters = args.iters if args.iters is not None else 100
        cfg = GmmConfig(num_clusters=100, target_num_clusters=20,
                        min_iters=iters, max_iters=iters,
                        estep_dtype="bf16", mstep_precision="bf16x3")
        t0 = time.perf_counter()
        eng = build_engine(data, cfg, device=dev)
        res = eng.sweep()
        dt = time.perf_counter() - t0
        total_iters = eng.total_em_iterations
        reached = res.num_clusters == 20
        emit(5, {"device": dev, "n": n, "seconds": dt,
                 "final_k": res.num_clusters,
                 "target_reached": reached,
                 "min_rissanen": res.min_rissanen,
                 "ks_swept": len(res.rissanen_by_k),
                 "em_iterations_total": total_iters,
                 "em_iters_per_sec_incl_merges": total_iters / dt})
        if not reached:
            print(f"config 5 FAILED to land on K=20 (final_k="
                  f"{res.num_clusters})", file=sys.stderr)
            rc = 1
    return rc


if __name__ == "__main__":
    sys.exit(main())
