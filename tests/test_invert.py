import math

import numpy as np
import pytest

from cuda_gmm_mpi_amd.ops.invert import (
    gmm_constant, invert_cpu, invert_matrix_pivot,
)


def spd(rng, n, scale=1.0):
    a = rng.standard_normal((n, n))
    return (a @ a.T + n * np.eye(n)).astype(np.float32) * scale


@pytest.mark.parametrize("n", [1, 2, 3, 8, 24])
def test_invert_cpu_matches_numpy(rng, n):
    a = spd(rng, n)
    inv, _ = invert_cpu(a, base10=False)
    np.testing.assert_allclose(inv, np.linalg.inv(a), rtol=2e-3, atol=2e-4)


@pytest.mark.parametrize("n", [2, 5, 24])
def test_invert_cpu_logdet_bases(rng, n):
    a = spd(rng, n)
    _, ld_ln = invert_cpu(a, base10=False)
    _, ld_10 = invert_cpu(a, base10=True)
    sign, ref = np.linalg.slogdet(a.astype(np.float64))
    assert sign > 0
    assert ld_ln == pytest.approx(ref, rel=1e-3, abs=1e-3)
    # the reference's log10 variant (invert_matrix.cpp:61)
    assert ld_10 == pytest.approx(ref / math.log(10.0), rel=1e-3, abs=1e-3)


def test_invert_cpu_d1():
    inv, ld = invert_cpu(np.array([[4.0]], dtype=np.float32), base10=False)
    assert inv[0, 0] == pytest.approx(0.25)
    assert ld == pytest.approx(math.log(4.0))


@pytest.mark.parametrize("n", [2, 6, 21])
def test_pivot_variant(rng, n):
    a = spd(rng, n)
    inv, det = invert_matrix_pivot(a)
    np.testing.assert_allclose(inv, np.linalg.inv(a), rtol=1e-3, atol=1e-4)
    assert det == pytest.approx(float(np.linalg.det(a.astype(np.float64))),
                                rel=1e-3)


def test_pivot_handles_row_swaps(rng):
    # a matrix that no-pivot LU would mangle (zero leading pivot)
    a = np.array([[0.0, 1.0], [1.0, 0.0]], dtype=np.float32)
    inv, det = invert_matrix_pivot(a)
    np.testing.assert_allclose(inv, a)  # its own inverse
    assert det == pytest.approx(-1.0)


def test_gmm_constant():
    d = 24
    ld = 3.7
    assert gmm_constant(ld, d) == pytest.approx(
        -d * 0.5 * math.log(2 * math.pi) - 0.5 * ld
    )
