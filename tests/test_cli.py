import numpy as np
import pytest

from cuda_gmm_mpi_amd.cli import main
from cuda_gmm_mpi_amd.utils import io as gio
from cuda_gmm_mpi_amd.utils.synthetic import make_blobs


@pytest.fixture
def csv_file(tmp_path):
    data, _ = make_blobs(600, 2, 3, seed=4)
    path = tmp_path / "data.csv"
    with open(path, "w") as f:
        f.write("c0,c1\n")
        for row in data:
            f.write(f"{row[0]:.6f},{row[1]:.6f}\n")
    return str(path), data


def test_cli_end_to_end_csv(tmp_path, csv_file):
    path, data = csv_file
    out = str(tmp_path / "out")
    rc = main([
        "3", path, out, "3", "--min-iters", "5", "--max-iters", "5",
        "--device", "cpu",
    ])
    assert rc == 0
    summary = open(out + ".summary").read()
    assert summary.count("Cluster #") == 3
    results = open(out + ".results").read().splitlines()
    assert len(results) == 600
    first = results[0].split("\t")
    assert len(first) == 2
    assert len(first[0].split(",")) == 2
    memb = np.array([float(v) for v in first[1].split(",")])
    assert len(memb) == 3
    assert memb.sum() == pytest.approx(1.0, abs=1e-4)


@pytest.mark.parametrize("striped", [False, True])
def test_cli_results_membership_content_on_offset_data(tmp_path, striped):
    """Regression: on CPU runs .to("cpu") aliased the saved
    state and the de-centering += shifted means before memberships were
    recomputed — a correct 200/200 split became 400/0. Assert actual
    assignment content on well-separated data with a nonzero mean."""
    rng = np.random.default_rng(7)
    a = rng.normal(loc=50.0, scale=0.5, size=(200, 2))
    b = rng.normal(loc=80.0, scale=0.5, size=(200, 2))
    data = np.vstack([a, b]).astype(np.float32)
    binpath = str(tmp_path / "sep.bin")
    gio.write_bin(binpath, data)
    out = str(tmp_path / "sep")
    argv = ["2", binpath, out, "2", "--min-iters", "8", "--max-iters", "8",
            "--device", "cpu"]
    if striped:
        argv.append("--striped-results")
    rc = main(argv)
    assert rc == 0
    results_path = out + (".results.0" if striped else ".results")
    lines = open(results_path).read().splitlines()
    assert len(lines) == 400
    assign = [int(np.argmax([float(v) for v in ln.split("\t")[1].split(",")]))
              for ln in lines]
    first, second = assign[:200], assign[200:]
    assert len(set(first)) == 1 and len(set(second)) == 1
    assert first[0] != second[0]


def test_cli_bin_input(tmp_path):
    data, _ = make_blobs(500, 3, 2, seed=6)
    binpath = str(tmp_path / "d.bin")
    gio.write_bin(binpath, data)
    out = str(tmp_path / "o")
    rc = main(["2", binpath, out, "2", "--min-iters", "4", "--max-iters", "4",
               "--device", "cpu", "--no-results"])
    assert rc == 0
    assert open(out + ".summary").read().count("Cluster #") == 2


def test_cli_sweep_with_mdl(tmp_path, csv_file):
    path, _ = csv_file
    out = str(tmp_path / "sweep")
    rc = main(["5", path, out, "2", "--min-iters", "3", "--max-iters", "3",
               "--device", "cpu", "--no-results"])
    assert rc == 0
    assert open(out + ".summary").read().count("Cluster #") == 2


def test_cli_invalid_cluster_count(tmp_path, csv_file):
    path, _ = csv_file
    assert main(["0", path, str(tmp_path / "x")]) == 1
    assert main(["513", path, str(tmp_path / "x")]) == 1


def test_cli_target_exceeds_start(tmp_path, csv_file):
    path, _ = csv_file
    assert main(["3", path, str(tmp_path / "x"), "5"]) == 4


def test_cli_missing_infile(tmp_path):
    assert main(["3", str(tmp_path / "nope.csv"), str(tmp_path / "x"),
                 "--device", "cpu"]) == 2


def test_cli_no_output_creates_empty_summary(tmp_path, csv_file):
    path, _ = csv_file
    out = str(tmp_path / "noout")
    rc = main(["2", path, out, "2", "--min-iters", "2", "--max-iters", "2",
               "--device", "cpu", "--no-output"])
    assert rc == 0
    assert open(out + ".summary").read() == ""


def test_cli_gpus_launcher_cpu_world2(tmp_path, csv_file, monkeypatch):
    """--gpus 2 spawns two gloo ranks on CPU and produces one output."""
    monkeypatch.delenv("WORLD_SIZE", raising=False)
    path, _ = csv_file
    out = str(tmp_path / "mp")
    rc = main(["3", path, out, "3", "--min-iters", "3", "--max-iters", "3",
               "--device", "cpu", "--no-results", "--gpus", "2"])
    assert rc == 0
    assert open(out + ".summary").read().count("Cluster #") == 3


def test_strip_gpus_arg():
    from cuda_gmm_mpi_amd.parallel.launcher import strip_gpus_arg
    rest, n = strip_gpus_arg(["3", "in.csv", "out", "--gpus", "4", "--print"])
    assert rest == ["3", "in.csv", "out", "--print"] and n == 4
    rest, n = strip_gpus_arg(["3", "x", "y", "--gpus=8"])
    assert rest == ["3", "x", "y"] and n == 8
    rest, n = strip_gpus_arg(["3", "x", "y"])
    assert n is None


def test_cli_bad_args_exit_code():
    # reference prints usage and returns 1 (gaussian.cu:1162-1165)
    assert main([]) == 1
    assert main(["notanumber", "x", "y"]) == 1


def test_cli_no_bug_compat_and_verbose(tmp_path, csv_file, capsys):
    """--no-bug-compat (corrected merge constant) + --verbose progress."""
    path, _ = csv_file
    out = str(tmp_path / "fix")
    rc = main(["5", path, out, "2", "--min-iters", "3", "--max-iters", "3",
               "--device", "cpu", "--no-results", "--no-bug-compat",
               "--verbose"])
    assert rc == 0
    assert open(out + ".summary").read().count("Cluster #") == 2
    logged = capsys.readouterr().out
    assert "clusters" in logged or "likelihood" in logged.lower()


def test_cli_single_cluster(tmp_path, csv_file):
    """K=1 end to end: no merging possible, one cluster summary."""
    path, _ = csv_file
    out = str(tmp_path / "k1")
    rc = main(["1", path, out, "1", "--min-iters", "3", "--max-iters", "3",
               "--device", "cpu", "--no-results"])
    assert rc == 0
    summary = open(out + ".summary").read()
    assert summary.count("Cluster #") == 1
    assert "Probability: 1.0" in summary


def test_cli_profile_report(tmp_path, csv_file, capsys):
    """--profile prints the reference-shaped per-GPU timing report
    (gaussian.cu:967) after the run."""
    path, _ = csv_file
    out = str(tmp_path / "prof")
    rc = main(["2", path, out, "2", "--min-iters", "2", "--max-iters", "2",
               "--device", "cpu", "--no-results", "--profile"])
    assert rc == 0
    text = capsys.readouterr().out
    assert "E-step Kernel:" in text and "M-step Kernel:" in text
    assert "Consts Kernel:" in text


def test_cli_unwritable_outfile_exits_early(tmp_path, csv_file):
    """Unwritable output path returns 3 BEFORE any compute (the
    reference's outfile check is commented out, gaussian.cu:1135-1141;
    revived with its dead exit code)."""
    path, _ = csv_file
    rc = main(["3", path, "/nonexistent_dir/out", "3", "--device", "cpu"])
    assert rc == 3


def test_cli_warns_on_nonfinite_input(tmp_path, capsys):
    """NaN/Inf in the input propagate (reference atof semantics) but
    produce a rank-0 warning at load time."""
    path = tmp_path / "bad.csv"
    path.write_text("a,b\n1.0,2.0\nnan,3.0\n4.0,inf\n")
    rc = main(["2", str(path), str(tmp_path / "o"), "2", "--min-iters", "1",
               "--max-iters", "1", "--device", "cpu", "--no-results"])
    assert rc == 0
    assert "non-finite" in capsys.readouterr().err


def test_python_dash_m_package_entry(tmp_path, csv_file):
    """`python -m cuda_gmm_mpi_amd` runs the same CLI."""
    import subprocess
    import sys
    path, _ = csv_file
    out = str(tmp_path / "m")
    r = subprocess.run(
        [sys.executable, "-m", "cuda_gmm_mpi_amd", "2", path, out, "2",
         "--min-iters", "1", "--max-iters", "1", "--device", "cpu",
         "--no-results"],
        capture_output=True, timeout=240)
    assert r.returncode == 0, r.stderr.decode()[-400:]
    assert open(out + ".summary").read().count("Cluster #") == 2
