"""RCCL de-risk tests on a 1-GPU box (VERDICT r1 task 1).

A real 1-rank RCCL communicator is initialized (WORLD_SIZE=1 under a
launcher env) so NCCL/RCCL init, device-buffer collectives, the
CPU-tensor control-plane collectives (cpu:gloo,cuda:nccl) and hipGraph
capture around a live RCCL all-reduce all execute on the single GPU the
box has — making the driver's 8-GPU SCALE run a formality.

Everything runs in subprocesses so NCCL state never leaks into the
pytest process.
"""
import json
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_py(script: str, env_extra: dict, timeout=420) -> str:
    env = dict(os.environ)
    for var in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR",
                "MASTER_PORT"):
        env.pop(var, None)
    env.update(env_extra)
    out = subprocess.run([sys.executable, "-c", script],
                         capture_output=True, text=True, timeout=timeout,
                         env=env, cwd=REPO)
    assert out.returncode == 0, (out.stdout[-1500:], out.stderr[-3000:])
    return out.stdout


WORLD1_ENV = {
    "RANK": "0", "LOCAL_RANK": "0", "WORLD_SIZE": "1",
    "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29661",
}


def test_world1_rccl_communicator_and_graph_capture():
    """World-1 RCCL: init + device all-reduce + CPU-tensor broadcast
    (gloo sub-backend) + full EM through the dist code path, with hipGraph
    capture of the iteration INCLUDING the live RCCL all-reduce. The
    dist run must agree with the plain single-process run (a world-1
    all-reduce is the identity)."""
    script = r"""
import json, os, torch
from cuda_gmm_mpi_amd.engine import build_engine
from cuda_gmm_mpi_amd.parallel import dist as pdist
from cuda_gmm_mpi_amd.utils.config import GmmConfig
from cuda_gmm_mpi_amd.utils.synthetic import make_blobs
import torch.distributed as dist

data, _ = make_blobs(30000, 24, 8, seed=91)
cfg = dict(num_clusters=8, target_num_clusters=8, min_iters=6, max_iters=6,
           estep_dtype="bf16", mstep_precision="bf16x3")

# plain run first (no process group): the baseline trajectory
eng0 = build_engine(data, GmmConfig(**cfg), device="cuda")
lik0 = eng0.run_em(8)
del eng0

rank, local_rank, world = pdist.init_process_group()
assert dist.is_initialized() and world == 1
assert torch.cuda.current_device() == local_rank  # set_device binding

# device-buffer RCCL all-reduce + CPU-tensor collective through the
# cpu:gloo,cuda:nccl group (the ADVICE r1 crash path, now fixed)
t = torch.ones(1024, device="cuda")
dist.all_reduce(t)
assert torch.all(t == 1.0)
c = torch.full((4,), 7, dtype=torch.long)
dist.broadcast(c, src=0)
assert torch.all(c == 7)

eng = build_engine(data, GmmConfig(**cfg), device="cuda")
lik1 = eng.run_em(8)
captured = any(bool(g) for g in eng._graphs.values())
print(json.dumps({"lik0": lik0, "lik1": lik1, "captured": captured,
                  "graphs": len(eng._graphs)}))
pdist.destroy()
"""
    out = run_py(script, WORLD1_ENV)
    res = json.loads([ln for ln in out.splitlines() if ln.startswith("{")][0])
    # identical math: world-1 all-reduce is the identity
    assert res["lik1"] == pytest.approx(res["lik0"], rel=1e-5)
    # hipGraph capture succeeded with the RCCL collective inside
    assert res["captured"], f"graph capture with RCCL failed: {res}"


def test_world1_torchrun_bench_contract():
    """The driver's exact torchrun launch form at world 1 on the GPU:
    WORLD_SIZE is set by the launcher, so bench.py initializes RCCL and
    its collectives (likelihood all-reduce, MAX-over-ranks elapsed)
    execute over the real communicator."""
    env = dict(os.environ)
    for var in ("RANK", "WORLD_SIZE", "LOCAL_RANK"):
        env.pop(var, None)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "1", "--master-addr", "127.0.0.1",
         "--master-port", "29662", "bench.py", "--gpus", "1",
         "--steps", "5", "--warmup", "2", "--events-per-gpu", "100000",
         "--modes", "single"],
        capture_output=True, text=True, timeout=420, env=env, cwd=REPO)
    assert out.returncode == 0, out.stderr[-3000:]
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, out.stdout
    d = json.loads(lines[0])
    assert d["n_gpus"] == 1 and d["value"] > 0
    assert d["dtype"] == "bf16"


def test_world1_cli_gpu_under_launcher_env(tmp_path):
    """Full CLI pipeline on cuda with a live world-1 communicator: the
    control-plane broadcasts, EM collectives, merge broadcast and the
    membership gather all run through torch.distributed."""
    import numpy as np
    from cuda_gmm_mpi_amd.utils import io as gio
    from cuda_gmm_mpi_amd.utils.synthetic import make_blobs
    data, _ = make_blobs(20000, 8, 4, seed=17)
    binpath = str(tmp_path / "d.bin")
    gio.write_bin(binpath, data)
    out = str(tmp_path / "o")
    script = (
        "import sys; from cuda_gmm_mpi_amd.cli import main; "
        f"sys.exit(main(['6', {binpath!r}, {out!r}, '3', "
        "'--min-iters', '4', '--max-iters', '4', '--device', 'cuda']))"
    )
    env = dict(WORLD1_ENV)
    env["MASTER_PORT"] = "29663"
    run_py(script, env)
    summary = open(out + ".summary").read()
    assert summary.count("Cluster #") == 3
    lines = open(out + ".results").read().splitlines()
    assert len(lines) == 20000
    memb = np.array([float(v) for v in lines[0].split("\t")[1].split(",")])
    assert abs(memb.sum() - 1.0) < 1e-3
