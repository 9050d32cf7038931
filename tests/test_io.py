import numpy as np
import pytest

from cuda_gmm_mpi_amd.models.state import GmmState
from cuda_gmm_mpi_amd.utils import io as gio


def test_bin_roundtrip(tmp_path, rng):
    data = rng.standard_normal((37, 5)).astype(np.float32)
    path = str(tmp_path / "x.bin")
    gio.write_bin(path, data)
    # layout: int32 nevents, int32 ndims, float32 payload (readData.cpp:35-47)
    raw = np.fromfile(path, dtype=np.int32, count=2)
    assert raw[0] == 37 and raw[1] == 5
    back = gio.read_data(path)
    np.testing.assert_array_equal(back, data)


def test_bin_extension_dispatch(tmp_path, rng):
    # dispatch compares the LAST 3 chars with "bin" (readData.cpp:28)
    data = rng.standard_normal((4, 2)).astype(np.float32)
    path = str(tmp_path / "weird.mybin")
    gio.write_bin(path, data)
    back = gio.read_data(path)
    np.testing.assert_array_equal(back, data)


def test_csv_header_skip_and_commas(tmp_path):
    path = str(tmp_path / "d.csv")
    with open(path, "w") as f:
        f.write("chan1,chan2,chan3\n")       # header dropped (readData.cpp:84)
        f.write("1.5,2.0,3.25\n")
        f.write("\n")                         # blank lines skipped
        f.write("-1.0,0.5,7\n")
    data = gio.read_data(path)
    assert data.shape == (2, 3)
    np.testing.assert_allclose(data, [[1.5, 2.0, 3.25], [-1.0, 0.5, 7.0]])


def test_csv_empty_raises(tmp_path):
    path = str(tmp_path / "e.csv")
    open(path, "w").close()
    with pytest.raises(ValueError):
        gio.read_csv(path)


def test_summary_format(tmp_path):
    import torch
    st = GmmState.empty(2, 3)
    st.pi[:] = torch.tensor([0.25, 0.75])
    st.N[:] = torch.tensor([10.0, 30.0])
    st.means[0] = torch.tensor([1.0, 2.0, 3.0])
    st.means[1] = torch.tensor([4.5555, 5.0, 6.0])
    st.R[0] = torch.eye(3)
    st.R[1] = 2 * torch.eye(3)
    path = str(tmp_path / "out.summary")
    gio.write_summary(path, st, enable_output=True)
    text = open(path).read()
    # exact reference layout (gaussian.cu:1035-1037, 1180-1197)
    assert text.startswith("Cluster #0\nProbability: 0.250000\nN: 10.000000\n")
    assert "Means: 1.000 2.000 3.000 \n" in text
    assert "\nR Matrix:\n1.000 0.000 0.000 \n0.000 1.000 0.000 \n" in text
    assert "Means: 4.556 5.000 6.000 \n" in text  # %.3f rounding
    assert text.count("Cluster #") == 2
    assert text.endswith("\n\n")


def test_summary_disabled_output_creates_empty_file(tmp_path):
    st = GmmState.empty(2, 3)
    path = str(tmp_path / "out.summary")
    gio.write_summary(path, st, enable_output=False)
    assert open(path).read() == ""


def test_results_format(tmp_path):
    data = np.array([[1.0, 2.0], [3.5, 4.0]], dtype=np.float32)
    w = np.array([[0.9, 0.25], [0.1, 0.75]], dtype=np.float32)  # [K=2, N=2]
    path = str(tmp_path / "out.results")
    gio.write_results(path, data, w)
    lines = open(path).read().splitlines()
    # event values comma-joined, tab, memberships comma-joined
    # (gaussian.cu:1046-1056)
    assert lines[0] == "1.000000,2.000000\t0.900000,0.100000"
    assert lines[1] == "3.500000,4.000000\t0.250000,0.750000"


def test_csv_atof_junk_fallback(tmp_path):
    """C atof semantics on malformed cells (readData.cpp:108): parse the
    longest valid leading prefix, 0.0 on pure junk — via the faithful
    parser fallback (pandas rejects the junk column)."""
    path = tmp_path / "junk.csv"
    path.write_text("a,b\n1.5xyz,2.0\nfoo,3e2\n-0.25,nan4\n")
    data = gio.read_csv(str(path))
    np.testing.assert_allclose(
        data, [[1.5, 2.0], [0.0, 300.0], [-0.25, float("nan")]],
        equal_nan=True)


def test_csv_pandas_and_faithful_agree(tmp_path):
    """Well-formed files parse identically through the fast pandas path
    and the quirk-faithful fallback."""
    path = tmp_path / "ok.csv"
    rows = "\n".join(f"{i * 0.5},{-i},{i ** 2}" for i in range(1, 30))
    path.write_text("h1,h2,h3\n" + rows + "\n")
    fast = gio.read_csv(str(path))
    faithful = gio._read_csv_faithful(str(path))
    np.testing.assert_array_equal(fast, faithful)
