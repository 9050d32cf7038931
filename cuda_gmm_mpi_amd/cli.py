"""Command-line driver, argument-compatible with the reference
(``gmm num_clusters infile outfile [target_num_clusters]``,
gaussian.cu:1111-1178) plus runtime flags for every compile-time knob.

Multi-GPU: run under ``torchrun --nproc-per-node N`` (one process per GPU,
RCCL over xGMI) or with ``--gpus N`` which self-launches torchrun-style
workers on one node.
"""
from __future__ import annotations

import argparse
import sys

import numpy as np
import torch

from .engine import build_engine
from .parallel import dist as pdist
from .utils import io as gio
from .utils.config import GmmConfig, MAX_CLUSTERS
from .utils.timers import Profile


def make_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(
        prog="gmm",
        description="MI355X-native GMM EM clustering with MDL order reduction",
    )
    p.add_argument("num_clusters", type=int,
                   help="The number of starting clusters")
    p.add_argument("infile", help="FCS data file (.bin binary or CSV)")
    p.add_argument("outfile", help="Clustering results output file stem")
    p.add_argument("target_num_clusters", type=int, nargs="?", default=0,
                   help="A desired number of clusters. Must be less than or "
                        "equal to num_clusters")
    p.add_argument("--min-iters", type=int, default=100)
    p.add_argument("--max-iters", type=int, default=100)
    p.add_argument("--covariance-dynamic-range", type=float, default=1e3)
    p.add_argument("--diag-only", action="store_true")
    p.add_argument("--no-bug-compat", dest="bug_compat", action="store_false",
                   help="use corrected ln determinant in the merge path "
                        "instead of the reference's log10 quirk")
    p.add_argument("--no-output", dest="enable_output", action="store_false",
                   help="skip .summary/.results content (files still created "
                        "like the reference)")
    p.add_argument("--no-results", dest="write_results", action="store_false",
                   help="write only the .summary file")
    p.add_argument("--print", dest="enable_print", action="store_true")
    p.add_argument("--verbose", action="store_true",
                   help="per-iteration likelihood lines (DEBUG-style)")
    p.add_argument("--estep-dtype", choices=["fp32", "bf16"], default="fp32")
    p.add_argument("--mstep-precision", choices=["fp32", "bf16x3"],
                   default="fp32",
                   help="M-step sufficient-statistics precision (bf16x3 = "
                        "split-precision MFMA, ~1e-5 relative, ~1.5x faster)")
    p.add_argument("--no-center", dest="center_data", action="store_false")
    p.add_argument("--device", default=None, choices=["cpu", "cuda"],
                   help="cpu | cuda (default: cuda when available)")
    p.add_argument("--profile", action="store_true",
                   help="print the per-GPU timing report (gaussian.cu:967)")
    p.add_argument("--nearest-target", action="store_true",
                   help="if empty-cluster elimination jumps past the target "
                        "K, keep the closest completed K instead of the "
                        "reference's first-saved model")
    p.add_argument("--checkpoint-dir", default=None,
                   help="directory for MDL-sweep checkpoints; an existing "
                        "checkpoint there is resumed automatically")
    p.add_argument("--metrics-out", default=None,
                   help="write machine-readable run metrics (JSON) here")
    p.add_argument("--gpus", type=int, default=None,
                   help="spawn N single-GPU worker processes on this node "
                        "(one rank per GPU over RCCL)")
    p.add_argument("--striped-results", action="store_true",
                   help="each rank writes its own contiguous "
                        "<out>.results.<rank> stripe (no gather; "
                        "concatenating stripes in rank order equals the "
                        "single gathered file)")
    p.add_argument("--scatter-input", action="store_true",
                   help="only rank 0 reads the input file; event shards "
                        "are scattered point-to-point and seeding stats "
                        "broadcast (no shared filesystem needed; the "
                        "reference MPI_Bcasts the whole dataset, "
                        "gaussian.cu:193-200)")
    return p


def config_from_args(args) -> GmmConfig:
    cfg = GmmConfig(
        num_clusters=args.num_clusters,
        target_num_clusters=args.target_num_clusters,
        min_iters=args.min_iters, max_iters=args.max_iters,
        covariance_dynamic_range=args.covariance_dynamic_range,
        diag_only=args.diag_only, bug_compat=args.bug_compat,
        enable_print=args.enable_print, enable_output=args.enable_output,
        estep_dtype=args.estep_dtype, center_data=args.center_data,
        verbose=args.verbose,
        mstep_precision=args.mstep_precision,
        checkpoint_dir=args.checkpoint_dir,
        nearest_target=args.nearest_target,
    )
    cfg.validate()
    return cfg


def run_clustering(data: np.ndarray | None, cfg: GmmConfig, outfile: str,
                   device: str, write_results: bool = True,
                   profile_report: bool = False,
                   scatter_input: bool = False,
                   striped_results: bool = False) -> dict:
    """Full pipeline on already-initialized process group. Returns a result
    dict (rank 0) with num_clusters / rissanen / likelihood.

    With scatter_input, only rank 0 passes data (others None): shards are
    scattered and seeding stats broadcast (parallel.dist.distribute_input).
    """
    rank, local_rank, world = pdist.rank(), 0, pdist.world_size()
    if device == "cuda":
        local_rank = torch.cuda.current_device()
    import time as _time
    prof = Profile(device)
    t_start = _time.perf_counter()
    shard = None
    if scatter_input:
        from .engine import build_engine_sharded
        full = (torch.from_numpy(np.ascontiguousarray(data, np.float32))
                if rank == 0 else None)
        shard, n_tot, mean, var, seed_means = pdist.distribute_input(
            full, cfg.num_clusters)
        engine = build_engine_sharded(shard, cfg, n_tot, mean, var,
                                      seed_means, device=device,
                                      profile=prof)
    else:
        engine = build_engine(data, cfg, device=device, profile=prof)
    if profile_report:
        # graph replay bypasses the per-bucket hipEvent timers
        engine.use_graphs = False
    result = engine.sweep()

    # memberships first: recompute_memberships needs the saved means in the
    # engine's internally-centered frame, so it must run before de-centering
    w_shard = engine.recompute_memberships(result.state)

    # de-center the saved model for output. clone() is load-bearing: on CPU
    # .to("cpu") returns aliased tensors and the += would corrupt the saved
    # state in place.
    out_state = result.state.to("cpu").clone(with_memberships=False)
    out_state.means += engine.center.cpu().unsqueeze(0)
    memberships = None
    if striped_results:
        # SURVEY §5 long-context plan: per-rank file stripes — shard-local
        # posteriors and data rows, no gather. Stripes are contiguous in
        # rank order, each ends with a newline, so
        # `cat out.results.0 out.results.1 ...` == the gathered file.
        if cfg.enable_output and write_results:
            s0, e0 = pdist.shard_bounds(engine.n_total, world, rank)
            local = (shard.numpy() if shard is not None
                     else np.asarray(data[s0:e0], dtype=np.float32))
            gio.write_results(f"{outfile}.results.{rank}", local,
                              w_shard.cpu().numpy())
        if rank == 0:
            gio.write_summary(outfile + ".summary", out_state,
                              cfg.enable_output)
    else:
        memberships = engine.gather_memberships(w_shard)
        if rank == 0:
            gio.write_summary(outfile + ".summary", out_state,
                              cfg.enable_output)
            if cfg.enable_output and write_results and memberships is not None:
                gio.write_results(outfile + ".results", data, memberships)

    if profile_report:
        print(engine.profile.report(rank, local_rank))

    elapsed = _time.perf_counter() - t_start
    return {
        "num_clusters": result.num_clusters,
        "rissanen": result.min_rissanen,
        "likelihood": result.likelihood,
        "rissanen_by_k": result.rissanen_by_k,
        "state": out_state,
        "memberships": memberships,
        "seconds": elapsed,
        "total_em_iterations": engine.total_em_iterations,
        "em_iterations_per_sec": (engine.total_em_iterations / elapsed
                                  if elapsed > 0 else 0.0),
        "n_events": engine.n_total,
    }


def main(argv=None) -> int:
    if argv is None:
        argv = sys.argv[1:]
    # self-launch N local ranks when asked and not already inside a world
    from .parallel.launcher import launch_workers, strip_gpus_arg
    try:
        rest, ngpus = strip_gpus_arg(list(argv))
    except ValueError as e:
        print(f"{e}\n")
        return 1
    if ngpus and ngpus > 1 and "WORLD_SIZE" not in __import__("os").environ:
        return launch_workers(rest, ngpus)
    try:
        args = make_parser().parse_args(rest)
    except SystemExit:
        # reference prints usage and returns 1 on bad arguments
        # (gaussian.cu:1111-1166)
        return 1
    if not (1 <= args.num_clusters <= MAX_CLUSTERS):
        print("Invalid number of starting clusters\n")
        return 1
    if args.target_num_clusters > args.num_clusters:
        print("target_num_clusters must be less than equal to num_clusters\n")
        return 4
    try:
        cfg = config_from_args(args)
    except ValueError as e:
        print(str(e))
        return 1

    device = args.device
    if device is None:
        device = "cuda" if torch.cuda.is_available() else "cpu"

    rank, local_rank, world = pdist.init_process_group()
    # outfile writability check BEFORE any compute — the reference's own
    # check exists but is commented out (gaussian.cu:1135-1141, dead
    # "return 3" path); revived here with its exit code. Probed on rank 0
    # only (the writing rank) and broadcast so no rank hangs.
    out_err = 0
    if rank == 0:
        try:
            with open(args.outfile + ".summary", "a"):
                pass
        except OSError as e:
            print(f"Unable to create output file. ({e})\n")
            out_err = 1
    if world > 1:
        flag = torch.tensor([out_err], dtype=torch.long)
        pdist.broadcast_(flag)
        out_err = int(flag.item())
    if out_err:
        pdist.destroy()
        return 3
    if cfg.enable_print:
        import socket
        dev_name = (torch.cuda.get_device_name(0)
                    if device == "cuda" else "cpu")
        print(f"Rank {rank} of {world} on {socket.gethostname()} "
              f"using {dev_name}")
    try:
        scatter = args.scatter_input and world > 1
        data = None
        read_err = 0
        if not scatter or rank == 0:
            try:
                data = gio.read_data(args.infile)
            except (OSError, ValueError) as e:
                print(f"Invalid infile. ({e})\n")
                if not scatter:
                    return 2
                read_err = 1
        if scatter:
            # rank 0's read outcome gates everyone (no hang on bad input)
            flag = torch.tensor([read_err], dtype=torch.long)
            pdist.broadcast_(flag)
            if int(flag.item()):
                return 2
        if rank == 0 and data is not None and not np.isfinite(data).all():
            # faithful behavior propagates them (the reference's atof
            # parses "nan"/"inf"); warn so garbage results are explicable
            bad = int((~np.isfinite(data)).sum())
            print(f"WARNING: input contains {bad} non-finite values; "
                  "results will be degenerate", file=sys.stderr)
        result = run_clustering(
            data, cfg, args.outfile, device,
            write_results=args.write_results, profile_report=args.profile,
            scatter_input=scatter, striped_results=args.striped_results,
        )
        if rank == 0 and args.metrics_out:
            import json
            with open(args.metrics_out, "w") as f:
                json.dump({
                    "num_clusters": result["num_clusters"],
                    "rissanen": result["rissanen"],
                    "likelihood": result["likelihood"],
                    "rissanen_by_k": {
                        str(kk): v
                        for kk, v in result["rissanen_by_k"].items()
                    },
                    "world_size": world,
                    "device": device,
                    "seconds": result["seconds"],
                    "total_em_iterations": result["total_em_iterations"],
                    "em_iterations_per_sec":
                        result["em_iterations_per_sec"],
                    "n_events": result["n_events"],
                }, f, indent=2)
        if rank == 0 and args.enable_print:
            print(f"Ideal clusters: {result['num_clusters']} "
                  f"(rissanen {result['rissanen']:.4f})")
    finally:
        pdist.destroy()
    return 0


if __name__ == "__main__":
    sys.exit(main())
