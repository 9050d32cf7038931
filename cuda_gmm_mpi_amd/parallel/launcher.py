"""Single-node multi-GPU launcher: one process per GPU over RCCL.

Replaces the reference's mpirun + OpenMP fan-out (gaussian.cu:289-298)
with a torchrun-style fork/exec spawner — no MPI dependency. Used by
``gmm --gpus N``; `torchrun --nproc-per-node N` works identically since
workers read the standard RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* env vars.
"""
from __future__ import annotations

import os
import socket
import subprocess
import sys


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def launch_workers(argv: list[str], nproc: int,
                   master_addr: str = "127.0.0.1") -> int:
    """Re-exec this program across ``nproc`` local ranks; returns the max
    exit code. The child command is ``sys.executable -m cuda_gmm_mpi_amd.cli
    <argv minus --gpus>``."""
    port = free_port()
    procs = []
    for rank in range(nproc):
        env = dict(os.environ)
        env.update({
            "RANK": str(rank),
            "LOCAL_RANK": str(rank),
            "WORLD_SIZE": str(nproc),
            "MASTER_ADDR": master_addr,
            "MASTER_PORT": str(port),
        })
        cmd = [sys.executable, "-m", "cuda_gmm_mpi_amd.cli"] + argv
        procs.append(subprocess.Popen(cmd, env=env))
    rc = 0
    for p in procs:
        p.wait()
        rc = max(rc, p.returncode or 0)
    return rc


def strip_gpus_arg(argv: list[str]) -> tuple[list[str], int | None]:
    """Remove --gpus N from argv; returns (rest, n or None)."""
    out = []
    n = None
    i = 0
    while i < len(argv):
        a = argv[i]
        if a == "--gpus":
            if i + 1 >= len(argv):
                raise ValueError("--gpus requires a value")
            n = int(argv[i + 1])
            i += 2
            continue
        if a.startswith("--gpus="):
            n = int(a.split("=", 1)[1])
            i += 1
            continue
        out.append(a)
        i += 1
    return out, n
