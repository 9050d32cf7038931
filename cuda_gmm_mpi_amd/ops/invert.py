"""CPU matrix inversion paths, numerically faithful to the reference.

- ``invert_cpu``: in-place LU (no pivoting) + inversion, vectorized numpy;
  same elimination order as invert_matrix.cpp:25-101. The determinant is
  accumulated in **log10** (invert_matrix.cpp:61) when ``base10=True`` — the
  reference then consumes that value as if it were a natural log in the
  merge constant (gaussian.cu:1249), which is quirk #2 in SURVEY.md §2.6.
  Pass ``base10=False`` for the corrected natural-log determinant (this is
  what the GPU ``invert`` device path uses, gaussian_kernel.cu:139).

- ``invert_matrix_pivot``: LU with partial pivoting (ludcmp/lubksb,
  Numerical-Recipes style, invert_matrix.cpp:108-309). Declared-but-unused
  in the reference; provided here as a tested utility.
"""
from __future__ import annotations

import math

import numpy as np


def invert_cpu(a: np.ndarray, base10: bool = True) -> tuple[np.ndarray, float]:
    """LU-decompose (no pivoting) and invert ``a``; returns (inv, log_det).

    ``log_det`` is log10|det| when base10 (reference behavior) else ln|det|.
    fp32 arithmetic like the reference. Does not modify the input.
    """
    data = np.array(a, dtype=np.float32, copy=True)
    n = data.shape[0]
    assert data.shape == (n, n)
    if n == 1:
        # the reference's 1x1 special case uses logf (natural log) even
        # though its n>=2 path accumulates log10 (invert_matrix.cpp:39
        # vs :61) — mirror that regardless of base10
        log_det = float(np.log(data[0, 0]))
        return np.array([[1.0 / data[0, 0]]], dtype=np.float32), log_det

    # normalize row 0 (invert_matrix.cpp:42)
    data[0, 1:] /= data[0, 0]
    # Doolittle-style in-place LU, no pivoting (invert_matrix.cpp:43-58)
    for i in range(1, n):
        # column i of L
        data[i:, i] -= data[i:, :i] @ data[:i, i]
        if i == n - 1:
            continue
        # row i of U
        data[i, i + 1:] = (data[i, i + 1:] - data[i, :i] @ data[:i, i + 1:]) / data[i, i]
    data = data.astype(np.float32)

    diag = np.abs(np.diag(data)).astype(np.float64)
    if base10:
        log_det = float(np.sum(np.log10(diag)))
    else:
        log_det = float(np.sum(np.log(diag)))

    # invert L (unit upper part is stored above diag) — invert_matrix.cpp:65-74
    for i in range(n):
        for j in range(i, n):
            x = np.float32(1.0)
            if i != j:
                x = np.float32(-(data[j, i:j] @ data[i:j, i]))
            data[j, i] = x / data[j, j]
    # invert U — invert_matrix.cpp:75-82
    for i in range(n):
        for j in range(i + 1, n):
            k = np.arange(i, j)
            terms = data[k, j] * np.where(k == i, np.float32(1.0), data[i, k])
            data[i, j] = -np.sum(terms, dtype=np.float32)
    # final multiply U^-1 * L^-1 — invert_matrix.cpp:83-89
    out = np.empty_like(data)
    for i in range(n):
        for j in range(n):
            k0 = max(i, j)
            k = np.arange(k0, n)
            terms = np.where(k == j, np.float32(1.0), data[j, k]) * data[k, i]
            out[j, i] = np.sum(terms, dtype=np.float32)
    return out, log_det


def log_det_lu_nopivot(a: np.ndarray, base10: bool = False) -> float:
    """Log|det| via the same no-pivot LU as invert_cpu, without inverting."""
    _, log_det = invert_cpu(a, base10=base10)
    return log_det


def invert_matrix_pivot(a: np.ndarray) -> tuple[np.ndarray, float]:
    """LU with partial pivoting + back substitution; returns (inv, det).

    Faithful reimplementation of the Bouman "cluster" path
    (invert_matrix.cpp:108-309): ludcmp with implicit scaling and partial
    pivoting, lubksb per unit column. Returns the determinant (linear, not
    log). Raises ZeroDivisionError on singular input like ludcmp's d=0 path.
    """
    n = a.shape[0]
    lu = np.array(a, dtype=np.float64, copy=True)
    indx = np.zeros(n, dtype=np.int64)
    d = 1.0
    vv = np.empty(n)
    big = np.max(np.abs(lu), axis=1)
    if np.any(big == 0.0):
        raise ZeroDivisionError("Singular matrix in routine ludcmp")
    vv = 1.0 / big
    for j in range(n):
        for i in range(j):
            lu[i, j] -= lu[i, :i] @ lu[:i, j]
        big = 0.0
        imax = j
        for i in range(j, n):
            lu[i, j] -= lu[i, :j] @ lu[:j, j]
            dum = vv[i] * abs(lu[i, j])
            if dum >= big:
                big = dum
                imax = i
        if j != imax:
            lu[[imax, j]] = lu[[j, imax]]
            d = -d
            vv[imax] = vv[j]
        indx[j] = imax
        if lu[j, j] == 0.0:
            lu[j, j] = 1e-20
        if j != n - 1:
            lu[j + 1:, j] /= lu[j, j]
    det = d * float(np.prod(np.diag(lu)))

    inv = np.empty((n, n))
    for col in range(n):
        b = np.zeros(n)
        b[col] = 1.0
        # lubksb
        ii = -1
        for i in range(n):
            ip = indx[i]
            s = b[ip]
            b[ip] = b[i]
            if ii >= 0:
                s -= lu[i, ii:i] @ b[ii:i]
            elif s != 0.0:
                ii = i
            b[i] = s
        for i in range(n - 1, -1, -1):
            b[i] = (b[i] - lu[i, i + 1:] @ b[i + 1:]) / lu[i, i]
        inv[:, col] = b
    return inv.astype(np.float32), det


def gmm_constant(log_det: float, num_dimensions: int) -> float:
    """-D/2*ln(2*pi) - 0.5*log_det (gaussian_kernel.cu:241, gaussian.cu:1249).

    Note: callers in bug-compat mode pass a log10 determinant here exactly
    like the reference merge path does.
    """
    return float(-num_dimensions * 0.5 * math.log(2.0 * math.pi) - 0.5 * log_det)
