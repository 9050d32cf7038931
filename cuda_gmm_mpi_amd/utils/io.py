"""Data readers and result writers, format-compatible with the reference.

Readers (readData.cpp):
 - dispatch on extension: names ending in "bin" -> binary, else CSV
   (readData.cpp:25-33 compares the last 3 characters with "bin").
 - BIN layout: int32 nevents, int32 ndims, float32[nevents*ndims] row-major
   (readData.cpp:35-47).
 - CSV: comma-delimited (readData.cpp:77,101 — despite README.txt:68 claiming
   space-delimited); the FIRST line is treated as a header and dropped
   (readData.cpp:84); blank lines are skipped (readData.cpp:61).

Writers (gaussian.cu:998-1061, 1180-1197):
 - <out>.summary: per-cluster "Cluster #c / Probability / N / Means /
   R Matrix" blocks with %.3f params.
 - <out>.results: one line per event: "d0,d1,...\tm0,m1,...\n" with %f.
"""
from __future__ import annotations

import struct

import numpy as np


def read_data(path: str) -> np.ndarray:
    """Reads FCS-shaped data; returns float32 array [num_events, num_dims]."""
    if path[-3:] == "bin":
        return read_bin(path)
    return read_csv(path)


def read_bin(path: str) -> np.ndarray:
    with open(path, "rb") as f:
        header = f.read(8)
        if len(header) < 8:
            raise ValueError(f"Truncated BIN file: {path}")
        nevents, ndims = struct.unpack("<ii", header)
        if nevents <= 0 or ndims <= 0:
            raise ValueError(f"Invalid BIN header in {path}: {nevents}x{ndims}")
        data = np.fromfile(f, dtype=np.float32, count=nevents * ndims)
    if data.size != nevents * ndims:
        raise ValueError(f"Truncated BIN payload in {path}")
    return data.reshape(nevents, ndims)


def write_bin(path: str, data: np.ndarray) -> None:
    data = np.ascontiguousarray(data, dtype=np.float32)
    nevents, ndims = data.shape
    with open(path, "wb") as f:
        f.write(struct.pack("<ii", nevents, ndims))
        data.tofile(f)


def read_csv(path: str) -> np.ndarray:
    """CSV reader; fast pandas path for well-formed files, quirk-faithful
    fallback (atof semantics) otherwise."""
    try:
        import pandas as pd
        df = pd.read_csv(path, header=0, sep=",", skip_blank_lines=True,
                         dtype=np.float32)
        if df.shape[0] > 0 and df.shape[1] > 0 and not df.isna().any().any():
            return np.ascontiguousarray(df.to_numpy(dtype=np.float32))
    except Exception:  # noqa: BLE001 — fall back to the faithful parser
        pass
    return _read_csv_faithful(path)


def _read_csv_faithful(path: str) -> np.ndarray:
    with open(path, "r") as f:
        lines = [ln for ln in (line.strip("\n") for line in f) if ln != ""]
    if not lines:
        raise ValueError(f"Empty CSV file: {path}")
    num_dims = len(lines[0].split(","))
    lines = lines[1:]  # first line assumed to be a header (readData.cpp:84)
    num_events = len(lines)
    if num_events == 0:
        raise ValueError(f"CSV file has a header but no data rows: {path}")
    data = np.empty((num_events, num_dims), dtype=np.float32)
    for i, ln in enumerate(lines):
        parts = ln.split(",")
        if len(parts) < num_dims:
            raise ValueError(
                f"Inconsistent number of dimensions at data row {i} of {path}"
            )
        # atof semantics (readData.cpp:108): parse leading float, 0.0 on junk
        for j in range(num_dims):
            try:
                data[i, j] = float(parts[j])
            except ValueError:
                data[i, j] = _atof(parts[j])
    return data


def _atof(s: str) -> float:
    """C atof: parse the longest valid leading prefix, else 0.0."""
    s = s.strip()
    best = 0.0
    for end in range(len(s), 0, -1):
        try:
            best = float(s[:end])
            return best
        except ValueError:
            continue
    return best


def format_cluster_block(c: int, pi: float, n: float, means: np.ndarray,
                         r: np.ndarray) -> str:
    """One cluster's .summary block (writeCluster, gaussian.cu:1180-1197).

    Byte-layout: "Probability: %f\n", "N: %f\n", "Means: " + D*"%.3f " + "\n",
    "\nR Matrix:\n", D rows of D*"%.3f " each ending "\n".
    """
    d = means.shape[0]
    parts = [f"Probability: {pi:f}\n", f"N: {n:f}\n", "Means: "]
    parts.append("".join(f"{means[i]:.3f} " for i in range(d)))
    parts.append("\n")
    parts.append("\nR Matrix:\n")
    r = r.reshape(d, d)
    for i in range(d):
        parts.append("".join(f"{r[i, j]:.3f} " for j in range(d)))
        parts.append("\n")
    return "".join(parts)


def write_summary(path: str, state, enable_output: bool = True) -> None:
    """Writes <out>.summary (gaussian.cu:1015-1040).

    The file is always created; cluster blocks are written only when output
    is enabled, matching ENABLE_OUTPUT semantics.
    """
    with open(path, "w") as f:
        if enable_output:
            for c in range(state.num_clusters):
                f.write(f"Cluster #{c}\n")
                f.write(
                    format_cluster_block(
                        c, float(state.pi[c]), float(state.N[c]),
                        np.asarray(state.means[c]), np.asarray(state.R[c]),
                    )
                )
                f.write("\n\n")


def write_results(path: str, data_by_event: np.ndarray,
                  memberships: np.ndarray,
                  chunk: int = 65536) -> None:
    """Writes <out>.results (gaussian.cu:1042-1059).

    Per event: comma-joined %f data values, a tab, comma-joined %f
    memberships (cluster-major array indexed [c, e]). Vectorized chunked
    formatting — the reference's per-value fprintf is minutes at N=1M.
    """
    n_events, n_dims = data_by_event.shape
    with open(path, "w") as f:
        for s in range(0, n_events, chunk):
            e = min(s + chunk, n_events)
            left = _csv_block(data_by_event[s:e])
            right = _csv_block(memberships[:, s:e].T)
            f.write("\n".join(a + "\t" + b for a, b in zip(left, right)))
            f.write("\n")


def _csv_block(arr: np.ndarray) -> list[str]:
    """Rows of ``arr`` as %f comma-joined strings (C printf %f == .6f)."""
    flat = np.char.mod("%f", arr.astype(np.float64))
    return [",".join(row) for row in flat]
