"""Multi-process CPU tests: gloo backend, world_size 2.

Verifies the distributed path (sharding + fused all-reduce + merge
broadcast) is equivalent to single-process execution — the CI stand-in for
the 8-GPU RCCL path (single- vs multi-GPU equivalence).
"""
import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from cuda_gmm_mpi_amd.utils.synthetic import make_blobs


def _worker(rank, world, port, fn_name, out_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    import torch.distributed as dist
    from cuda_gmm_mpi_amd.parallel import dist as pdist
    pdist.init_process_group(backend="gloo")
    try:
        result = globals()[fn_name]()
        if rank == 0:
            out_q.put(result)
    finally:
        if dist.is_initialized():  # CLI fns destroy the group themselves
            dist.destroy_process_group()


def run_world(world, fn_name, port):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_worker, args=(r, world, port, fn_name, q))
        for r in range(world)
    ]
    for p in procs:
        p.start()
    result = q.get(timeout=300)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    return result


def _fit_em():
    from cuda_gmm_mpi_amd.engine import build_engine
    from cuda_gmm_mpi_amd.utils.config import GmmConfig
    data, _ = make_blobs(2002, 3, 3, seed=17)  # non-divisible N: remainder path
    # covariance_dynamic_range huge => avgvar ~ 0: the reference's
    # G*avgvar regularization is world-size dependent BY DESIGN
    # (SURVEY 2.6 #5), so neutralize it for the equivalence check.
    cfg = GmmConfig(num_clusters=3, target_num_clusters=3,
                    min_iters=8, max_iters=8,
                    covariance_dynamic_range=1e15)
    eng = build_engine(data, cfg, device="cpu")
    lik = eng.run_em(3)
    return {
        "lik": lik,
        "means": eng.state.means.numpy().copy(),
        "R": eng.state.R.numpy().copy(),
        "N": eng.state.N.numpy().copy(),
        "pi": eng.state.pi.numpy().copy(),
    }


def _fit_sweep():
    from cuda_gmm_mpi_amd.engine import build_engine
    from cuda_gmm_mpi_amd.utils.config import GmmConfig
    data, _ = make_blobs(1501, 2, 3, seed=23)
    cfg = GmmConfig(num_clusters=5, target_num_clusters=2,
                    min_iters=4, max_iters=4,
                    covariance_dynamic_range=1e15)
    eng = build_engine(data, cfg, device="cpu")
    res = eng.sweep()
    w = eng.recompute_memberships(res.state)
    gathered = eng.gather_memberships(w)
    return {
        "k": res.num_clusters,
        "rissanen": res.min_rissanen,
        "means": res.state.means.numpy().copy(),
        "memberships": None if gathered is None else gathered,
    }


@pytest.mark.timeout(300)
def test_world2_em_matches_single_process():
    single = _run_single("_fit_em")
    multi = run_world(2, "_fit_em", port=29811)
    assert multi["lik"] == pytest.approx(single["lik"], rel=1e-4)
    np.testing.assert_allclose(multi["N"], single["N"], rtol=1e-3)
    np.testing.assert_allclose(multi["means"], single["means"],
                               rtol=1e-3, atol=1e-3)
    np.testing.assert_allclose(multi["R"], single["R"], rtol=2e-2, atol=2e-2)
    np.testing.assert_allclose(multi["pi"], single["pi"], rtol=1e-3)


@pytest.mark.timeout(300)
def test_world2_sweep_and_membership_gather():
    single = _run_single("_fit_sweep")
    multi = run_world(2, "_fit_sweep", port=29812)
    assert multi["k"] == single["k"]
    assert multi["rissanen"] == pytest.approx(single["rissanen"], rel=1e-3)
    np.testing.assert_allclose(multi["means"], single["means"],
                               rtol=1e-3, atol=1e-3)
    assert multi["memberships"].shape == single["memberships"].shape
    np.testing.assert_allclose(
        multi["memberships"], single["memberships"], rtol=1e-2, atol=1e-3)
    # gathered posteriors normalized per event
    np.testing.assert_allclose(
        multi["memberships"].sum(axis=0),
        np.ones(multi["memberships"].shape[1]), rtol=1e-3)


def _run_single(fn_name):
    # run in a clean spawned process so no dist state leaks
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    p = ctx.Process(target=_single_worker, args=(fn_name, q))
    p.start()
    result = q.get(timeout=300)
    p.join(timeout=120)
    assert p.exitcode == 0
    return result


def _single_worker(fn_name, q):
    for var in ("RANK", "WORLD_SIZE", "LOCAL_RANK"):
        os.environ.pop(var, None)
    q.put(globals()[fn_name]())


def _fit_diag_sweep():
    from cuda_gmm_mpi_amd.engine import build_engine
    from cuda_gmm_mpi_amd.utils.config import GmmConfig
    data, _ = make_blobs(2000, 3, 3, seed=55)
    cfg = GmmConfig(num_clusters=5, target_num_clusters=2,
                    min_iters=4, max_iters=4, diag_only=True)
    eng = build_engine(data, cfg, device="cpu")
    res = eng.sweep()
    return {"k": res.num_clusters, "rissanen": res.min_rissanen,
            "means": res.state.means.numpy().copy(),
            "R": res.state.R.numpy().copy()}


@pytest.mark.timeout(300)
def test_world2_diag_sweep_with_merges_matches_single():
    """DIAG_ONLY MDL sweep (5 -> 2, multiple merges) at world 2: merged
    full-inverse params broadcast from rank 0 must reproduce the
    single-process trajectory exactly (diag quirk-#8 corner under
    sharding)."""
    single = _run_single("_fit_diag_sweep")
    multi = run_world(2, "_fit_diag_sweep", port=29817)
    assert multi["k"] == single["k"]
    assert multi["rissanen"] == pytest.approx(single["rissanen"], rel=1e-3)
    np.testing.assert_allclose(multi["means"], single["means"],
                               rtol=1e-3, atol=1e-3)
    np.testing.assert_allclose(multi["R"], single["R"], rtol=2e-2,
                               atol=2e-2)


def _fit_world4():
    from cuda_gmm_mpi_amd.engine import build_engine
    from cuda_gmm_mpi_amd.utils.config import GmmConfig
    data, _ = make_blobs(1003, 3, 3, seed=37)  # odd N: remainder shards
    cfg = GmmConfig(num_clusters=4, target_num_clusters=2,
                    min_iters=3, max_iters=3,
                    covariance_dynamic_range=1e15)
    eng = build_engine(data, cfg, device="cpu")
    res = eng.sweep()
    return {"k": res.num_clusters, "rissanen": res.min_rissanen,
            "means": res.state.means.numpy().copy()}


@pytest.mark.timeout(300)
def test_world4_sweep_matches_single():
    single = _run_single("_fit_world4")
    multi = run_world(4, "_fit_world4", port=29813)
    assert multi["k"] == single["k"]
    assert multi["rissanen"] == pytest.approx(single["rissanen"], rel=1e-3)
    np.testing.assert_allclose(multi["means"], single["means"],
                               rtol=2e-3, atol=2e-3)


def _ck_cfg(**kw):
    from cuda_gmm_mpi_amd.utils.config import GmmConfig
    return GmmConfig(num_clusters=6, min_iters=3, max_iters=3,
                     covariance_dynamic_range=1e15, **kw)


def _resume_data():
    data, _ = make_blobs(1801, 2, 3, seed=41)
    return data


def _fit_ck_phase1():
    """World-1 sweep 6 -> 4, leaving checkpoints behind."""
    from cuda_gmm_mpi_amd.engine import build_engine
    cfg = _ck_cfg(target_num_clusters=4,
                  checkpoint_dir=os.environ["GMM_TEST_CKDIR"])
    eng = build_engine(_resume_data(), cfg, device="cpu")
    eng.sweep()
    return True


def _fit_ck_resume():
    """Resume the phase-1 checkpoint down to 2 (any world size)."""
    from cuda_gmm_mpi_amd.engine import build_engine
    cfg = _ck_cfg(target_num_clusters=2,
                  checkpoint_dir=os.environ["GMM_TEST_CKDIR"])
    eng = build_engine(_resume_data(), cfg, device="cpu")
    res = eng.sweep()
    return {"k": res.num_clusters, "rissanen": res.min_rissanen,
            "means": res.state.means.numpy().copy()}


def _fit_ck_direct():
    from cuda_gmm_mpi_amd.engine import build_engine
    eng = build_engine(_resume_data(), _ck_cfg(target_num_clusters=2),
                       device="cpu")
    res = eng.sweep()
    return {"k": res.num_clusters, "rissanen": res.min_rissanen,
            "means": res.state.means.numpy().copy()}


@pytest.mark.timeout(300)
def test_world2_resume_from_world1_checkpoint(tmp_path):
    """Checkpoints hold model parameters, not shards, so a sweep
    checkpointed at world 1 resumes under world 2 (re-sharded data) and
    matches a direct single-process run (avgvar ridge neutralized — it is
    world-size dependent by reference design)."""
    os.environ["GMM_TEST_CKDIR"] = str(tmp_path / "ck")
    try:
        _run_single("_fit_ck_phase1")
        multi = run_world(2, "_fit_ck_resume", port=29814)
        single = _run_single("_fit_ck_direct")
    finally:
        os.environ.pop("GMM_TEST_CKDIR", None)
    assert multi["k"] == single["k"]
    assert multi["rissanen"] == pytest.approx(single["rissanen"], rel=1e-3)
    np.testing.assert_allclose(multi["means"], single["means"],
                               rtol=1e-3, atol=1e-3)


def _cli_scatter(flag=True):
    """Run the CLI with/without --scatter-input; return the .summary text."""
    from cuda_gmm_mpi_amd.cli import main
    from cuda_gmm_mpi_amd.utils import io as gio
    work = os.environ["GMM_TEST_SCATTER_DIR"]
    binpath = os.path.join(work, "data.bin")
    out = os.path.join(work, f"out_{'sc' if flag else 'fs'}_{os.environ.get('RANK', '0')}")
    argv = ["4", binpath, out, "3", "--min-iters", "4", "--max-iters", "4",
            "--device", "cpu", "--no-results"]
    if flag:
        argv.append("--scatter-input")
    rc = main(argv)
    assert rc == 0
    if os.environ.get("RANK", "0") == "0":
        return open(out + ".summary").read()
    return None


def _cli_scatter_on():
    return _cli_scatter(True)


def _cli_scatter_off():
    return _cli_scatter(False)


def _cli_scatter_missing():
    """Bad input path under --scatter-input must fail on ALL ranks, not
    hang the non-root ranks in distribute_input."""
    from cuda_gmm_mpi_amd.cli import main
    work = os.environ["GMM_TEST_SCATTER_DIR"]
    rc = main(["3", os.path.join(work, "nope.bin"),
               os.path.join(work, "x"), "--device", "cpu",
               "--scatter-input"])
    return rc


@pytest.mark.timeout(300)
def test_scatter_input_matches_shared_fs(tmp_path):
    """--scatter-input (rank-0 read + shard scatter + stats broadcast)
    produces byte-identical output to the every-rank-reads path."""
    from cuda_gmm_mpi_amd.utils import io as gio
    data, _ = make_blobs(1507, 3, 4, seed=53)
    os.environ["GMM_TEST_SCATTER_DIR"] = str(tmp_path)
    try:
        gio.write_bin(str(tmp_path / "data.bin"), data)
        sc = run_world(2, "_cli_scatter_on", port=29815)
        fs = run_world(2, "_cli_scatter_off", port=29816)
    finally:
        os.environ.pop("GMM_TEST_SCATTER_DIR", None)
    assert sc is not None and sc == fs
    assert sc.count("Cluster #") == 3


@pytest.mark.timeout(300)
def test_scatter_input_missing_file_no_hang(tmp_path):
    os.environ["GMM_TEST_SCATTER_DIR"] = str(tmp_path)
    try:
        rc = run_world(2, "_cli_scatter_missing", port=29817)
    finally:
        os.environ.pop("GMM_TEST_SCATTER_DIR", None)
    assert rc == 2


def _cli_results(striped):
    from cuda_gmm_mpi_amd.cli import main
    work = os.environ["GMM_TEST_SCATTER_DIR"]
    out = os.path.join(work, "st" if striped else "ga")
    argv = ["3", os.path.join(work, "data.bin"), out, "3",
            "--min-iters", "3", "--max-iters", "3", "--device", "cpu"]
    if striped:
        argv += ["--striped-results", "--scatter-input"]
    assert main(argv) == 0
    return True


def _cli_results_striped():
    return _cli_results(True)


def _cli_results_gathered():
    return _cli_results(False)


@pytest.mark.timeout(300)
def test_striped_results_concat_equals_gathered(tmp_path):
    """--striped-results: concatenating per-rank stripes in rank order is
    byte-identical to the single gathered .results file (here combined
    with --scatter-input: no gather AND no shared-fs read)."""
    from cuda_gmm_mpi_amd.utils import io as gio
    data, _ = make_blobs(1207, 3, 3, seed=61)
    os.environ["GMM_TEST_SCATTER_DIR"] = str(tmp_path)
    try:
        gio.write_bin(str(tmp_path / "data.bin"), data)
        run_world(2, "_cli_results_striped", port=29818)
        run_world(2, "_cli_results_gathered", port=29819)
    finally:
        os.environ.pop("GMM_TEST_SCATTER_DIR", None)
    stripes = b"".join(
        open(tmp_path / f"st.results.{r}", "rb").read() for r in range(2))
    gathered = open(tmp_path / "ga.results", "rb").read()
    assert stripes == gathered
    assert (open(tmp_path / "st.summary").read()
            == open(tmp_path / "ga.summary").read())


@pytest.mark.timeout(300)
def test_scatter_input_world4(tmp_path):
    """Scatter with 3 concurrent isends + uneven tail shard (N=1205 over
    4 ranks: 301+301+301+302 per the reference split) matches shared-fs."""
    from cuda_gmm_mpi_amd.utils import io as gio
    data, _ = make_blobs(1205, 2, 3, seed=67)
    os.environ["GMM_TEST_SCATTER_DIR"] = str(tmp_path)
    try:
        gio.write_bin(str(tmp_path / "data.bin"), data)
        sc = run_world(4, "_cli_scatter_on", port=29820)
        fs = run_world(4, "_cli_scatter_off", port=29821)
    finally:
        os.environ.pop("GMM_TEST_SCATTER_DIR", None)
    assert sc is not None and sc == fs


def _fit_ck_resume_rank0_only():
    """Resume where ONLY rank 0 can access the checkpoint store — any
    checkpoint I/O from a non-root rank trips an AssertionError."""
    from cuda_gmm_mpi_amd.utils import checkpoint as ckmod
    if os.environ.get("RANK") != "0":
        def boom(*a, **k):
            raise AssertionError("non-root rank touched the checkpoint dir")
        ckmod.load_sweep_checkpoint = boom
        ckmod.save_sweep_checkpoint = boom
    return _fit_ck_resume()


@pytest.mark.timeout(300)
def test_world2_resume_rank0_only_checkpoint(tmp_path):
    """Checkpoint resume is rank-0-authoritative (broadcast): works when
    non-root ranks cannot read the checkpoint directory at all (the
    --scatter-input / rank-0-only-filesystem deployment)."""
    os.environ["GMM_TEST_CKDIR"] = str(tmp_path / "ck")
    try:
        _run_single("_fit_ck_phase1")
        multi = run_world(2, "_fit_ck_resume_rank0_only", port=29822)
        single = _run_single("_fit_ck_direct")
    finally:
        os.environ.pop("GMM_TEST_CKDIR", None)
    assert multi["k"] == single["k"]
    assert multi["rissanen"] == pytest.approx(single["rissanen"], rel=1e-3)


def _fit_die_off():
    """Mass cluster die-off mid-sweep under distribution: the survivor
    stop (new_k < 2) must keep both ranks in lockstep."""
    from cuda_gmm_mpi_amd.engine import build_engine
    from cuda_gmm_mpi_amd.utils.config import GmmConfig
    data, _ = make_blobs(40, 1, 2, seed=1)
    cfg = GmmConfig(num_clusters=8, target_num_clusters=0,
                    min_iters=2, max_iters=2,
                    covariance_dynamic_range=1e15)
    res = build_engine(data, cfg, device="cpu").sweep()
    return {"k": res.num_clusters, "riss": res.min_rissanen}


@pytest.mark.timeout(300)
def test_world2_mass_die_off_lockstep():
    single = _run_single("_fit_die_off")
    multi = run_world(2, "_fit_die_off", port=29823)
    assert multi["k"] == single["k"]
    assert np.isfinite(multi["riss"])


def _gather_bytes():
    """Byte-exact rank-0 gather: shard w filled with the global event index
    so the reassembled [K, N] must equal the analytic pattern EXACTLY."""
    from cuda_gmm_mpi_amd.engine import build_engine
    from cuda_gmm_mpi_amd.parallel import dist as pdist
    from cuda_gmm_mpi_amd.utils.config import GmmConfig
    n, k = 1003, 3
    data, _ = make_blobs(n, 2, 2, seed=5)
    cfg = GmmConfig(num_clusters=k, target_num_clusters=k,
                    min_iters=1, max_iters=1)
    eng = build_engine(data, cfg, device="cpu")
    s, e = pdist.shard_bounds(n, eng.world, eng.rank)
    w = (torch.arange(s, e, dtype=torch.float32).unsqueeze(0)
         + 1000.0 * torch.arange(k, dtype=torch.float32).unsqueeze(1))
    g = eng.gather_memberships(w)
    if eng.rank != 0:
        assert g is None  # non-root allocates/returns nothing
        return None
    expect = (np.arange(n, dtype=np.float32)[None, :]
              + 1000.0 * np.arange(k, dtype=np.float32)[:, None])
    return {"exact": bool((g == expect).all()), "shape": list(g.shape)}


@pytest.mark.timeout(300)
def test_world3_gather_memberships_byte_exact():
    """The point-to-point rank-0 gather (no padded all_gather) reassembles
    uneven shards byte-exactly in event order."""
    res = run_world(3, "_gather_bytes", port=29824)
    assert res["shape"] == [3, 1003]
    assert res["exact"]
