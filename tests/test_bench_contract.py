"""The driver depends on bench.py's exact CLI + one-line JSON contract."""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_bench(args, env_extra=None, timeout=600):
    env = dict(os.environ)
    env.pop("WORLD_SIZE", None)
    if env_extra:
        env.update(env_extra)
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py")] + args,
        capture_output=True, text=True, timeout=timeout, env=env, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, f"expected ONE json line, got: {out.stdout!r}"
    return json.loads(lines[0])


def check_contract(d, n_gpus):
    assert d["metric"] == "em_iterations_per_sec"
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["n_gpus"] == n_gpus
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert abs(d["value"] * d["ms_per_step"] / 1e3 - 1.0) < 1e-6
    assert d["data"] == "synthetic"
    assert d["config"]["parallelism"] == f"dp{n_gpus}"
    for key in ("steps", "warmup", "dtype", "vs_baseline"):
        assert key in d


@pytest.mark.timeout(600)
def test_bench_single_process_cpu():
    d = run_bench(["--steps", "2", "--warmup", "1",
                   "--events-per-gpu", "3000"])
    check_contract(d, 1)
    assert d["config"]["n_events"] == 3000


@pytest.mark.timeout(900)
def test_bench_torchrun_world2_cpu():
    """The driver's exact launch form, world 2 on gloo."""
    env = dict(os.environ)
    env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29733", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--events-per-gpu", "2000"],
        capture_output=True, text=True, timeout=600, env=env, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-3000:]
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, out.stdout
    d = json.loads(lines[0])
    check_contract(d, 2)
    assert d["config"]["n_events"] == 4000  # weak scaling: 2000/rank
