"""MDL model-order reduction: cluster merge math on the host (rank 0).

Faithful to gaussian.cu:1203-1263 and the empty-cluster elimination /
exhaustive pair scan of gaussian.cu:857-907. All arithmetic in fp32 numpy
to track the reference's float math.

Quirk preserved (default ``bug_compat=True``): the merged
cluster's constant uses ``invert_cpu``'s **log10** determinant consumed as
if natural log (invert_matrix.cpp:61 -> gaussian.cu:1249), while unmerged
clusters carry ln-based GPU constants. The first E-step after a merge
consumes this mixed state (quirk #8), which this module reproduces by
writing the tainted constant/Rinv back into the merged slot.
"""
from __future__ import annotations

import dataclasses
import math

import numpy as np

from ..ops.invert import invert_cpu


@dataclasses.dataclass
class HostClusters:
    """Host-side fp32 mirror of the cluster params (no memberships)."""
    N: np.ndarray          # [K]
    pi: np.ndarray         # [K]
    constant: np.ndarray   # [K]
    avgvar: np.ndarray     # [K]
    means: np.ndarray      # [K, D]
    R: np.ndarray          # [K, D, D]
    Rinv: np.ndarray       # [K, D, D]

    @property
    def num_clusters(self) -> int:
        return int(self.N.shape[0])

    @property
    def num_dimensions(self) -> int:
        return int(self.means.shape[1])

    def copy_cluster(self, dest: int, src: int) -> None:
        """copy_cluster (gaussian.cu:1255-1263); memberships not copied,
        matching the reference's open question at gaussian.cu:1263."""
        self.N[dest] = self.N[src]
        self.pi[dest] = self.pi[src]
        self.constant[dest] = self.constant[src]
        self.avgvar[dest] = self.avgvar[src]
        self.means[dest] = self.means[src]
        self.R[dest] = self.R[src]
        self.Rinv[dest] = self.Rinv[src]

    def truncated(self, k: int) -> "HostClusters":
        return HostClusters(
            N=self.N[:k], pi=self.pi[:k], constant=self.constant[:k],
            avgvar=self.avgvar[:k], means=self.means[:k], R=self.R[:k],
            Rinv=self.Rinv[:k],
        )


@dataclasses.dataclass
class MergedCluster:
    N: float
    pi: float
    constant: float
    avgvar: float
    means: np.ndarray
    R: np.ndarray
    Rinv: np.ndarray


def add_clusters(c: HostClusters, c1: int, c2: int,
                 bug_compat: bool = True) -> MergedCluster:
    """Merge two clusters (add_clusters, gaussian.cu:1210-1253).

    Weighted means; covariance = weighted sum of (R_i + shift outer
    products); pi/N additive; constant recomputed through invert_cpu with
    the log10 determinant when bug_compat (reference behavior), natural log
    otherwise.
    """
    d = c.num_dimensions
    f32 = np.float32
    n1, n2 = f32(c.N[c1]), f32(c.N[c2])
    wt1 = f32(n1 / (n1 + n2))
    wt2 = f32(1.0) - wt1

    means = (wt1 * c.means[c1] + wt2 * c.means[c2]).astype(f32)

    r = np.empty((d, d), dtype=f32)
    # upper triangle incl. diagonal, mirrored (gaussian.cu:1222-1235)
    for i in range(d):
        dm1 = means[i] - c.means[c1, i]
        dm2 = means[i] - c.means[c2, i]
        j = np.arange(i, d)
        contrib1 = (dm1 * (means[j] - c.means[c1, j]) + c.R[c1, i, j]) * wt1
        contrib2 = (dm2 * (means[j] - c.means[c2, j]) + c.R[c2, i, j]) * wt2
        r[i, j] = contrib1 + contrib2
        r[j, i] = r[i, j]

    rinv, log_det = invert_cpu(r, base10=bug_compat)
    constant = f32(-d * 0.5 * math.log(2.0 * math.pi) - 0.5 * log_det)
    return MergedCluster(
        N=float(n1 + n2), pi=float(c.pi[c1] + c.pi[c2]),
        constant=float(constant), avgvar=float(c.avgvar[c1]),
        means=means, R=r, Rinv=rinv,
    )


def cluster_distance(c: HostClusters, c1: int, c2: int,
                     bug_compat: bool = True) -> tuple[float, MergedCluster]:
    """Merge distance (gaussian.cu:1203-1208):
    N1*const1 + N2*const2 - N12*const12."""
    merged = add_clusters(c, c1, c2, bug_compat=bug_compat)
    dist = (
        float(c.N[c1]) * float(c.constant[c1])
        + float(c.N[c2]) * float(c.constant[c2])
        - merged.N * merged.constant
    )
    return dist, merged


def eliminate_empty_clusters(c: HostClusters) -> int:
    """Drop clusters with N < 0.5 by left-compaction (gaussian.cu:866-874).
    Returns the new cluster count."""
    k = c.num_clusters
    i = k - 1
    while i >= 0:
        if c.N[i] < 0.5:
            for j in range(i, k - 1):
                c.copy_cluster(j, j + 1)
            k -= 1
        i -= 1
    return k


def reduce_order(c: HostClusters, bug_compat: bool = True) -> tuple[int, int, int]:
    """One MDL order-reduction step (gaussian.cu:860-907): eliminate empty
    clusters, exhaustively scan all pairs for the minimum merge distance,
    merge the closest pair in place, compact.

    Returns (new_num_clusters, merged_c1, merged_c2) where indices refer to
    the post-elimination numbering. ``c`` is modified in place; the caller
    then uses ``c.truncated(new_k)``.
    """
    k = eliminate_empty_clusters(c)
    min_c1, min_c2 = 0, 1
    min_distance = 0.0
    best: MergedCluster | None = None
    for c1 in range(k):
        for c2 in range(c1 + 1, k):
            distance, merged = cluster_distance(c, c1, c2, bug_compat=bug_compat)
            if (c1 == 0 and c2 == 1) or distance < min_distance:
                min_distance = distance
                min_c1, min_c2 = c1, c2
                best = merged
    assert best is not None, "reduce_order called with < 2 clusters"
    # write merged into slot min_c1 (copy_cluster(clusters, min_c1, scratch))
    c.N[min_c1] = best.N
    c.pi[min_c1] = best.pi
    c.constant[min_c1] = best.constant
    c.avgvar[min_c1] = best.avgvar
    c.means[min_c1] = best.means
    c.R[min_c1] = best.R
    c.Rinv[min_c1] = best.Rinv
    # compact out slot min_c2 (gaussian.cu:903-907)
    for i in range(min_c2, k - 1):
        c.copy_cluster(i, i + 1)
    return k - 1, min_c1, min_c2


def reduce_order_batched(c: HostClusters, bug_compat: bool = True,
                         device: str = "cpu") -> tuple[int, int, int]:
    """Batched MDL order-reduction step — same decision rule as
    ``reduce_order`` but the O(K^2) pair scan runs as ONE vectorized
    batch (merged covariances via numpy broadcasting; log-determinants
    via the batched no-pivot LU, optionally on the GPU).

    The distance (gaussian.cu:1203-1208) needs only the merged constant,
    i.e. the log-determinant — the full inverse is computed (faithfully,
    via invert_cpu) only for the winning pair. In bug_compat mode the
    base-10 determinant is ln/ln(10): it differs from the reference's
    per-pivot log10 accumulation only in the last-bit rounding of the
    sum, so a decision could flip only on exact ties.
    """
    import torch

    from ..ops.cpu_reference import lu_logdet_nopivot

    k = eliminate_empty_clusters(c)
    if k < 2:
        raise ValueError("reduce_order needs >= 2 clusters")
    d = c.num_dimensions
    f32 = np.float32
    iu, ju = np.triu_indices(k, 1)
    n1 = c.N[iu].astype(f32)
    n2 = c.N[ju].astype(f32)
    wt1 = (n1 / (n1 + n2)).astype(f32)[:, None]
    wt2 = (f32(1.0) - wt1).astype(f32)
    m1 = c.means[iu]
    m2 = c.means[ju]
    means_m = (wt1 * m1 + wt2 * m2).astype(f32)           # [P, D]
    d1 = means_m - m1
    d2 = means_m - m2
    r_m = (
        wt1[:, :, None] * (d1[:, :, None] * d1[:, None, :] + c.R[iu])
        + wt2[:, :, None] * (d2[:, :, None] * d2[:, None, :] + c.R[ju])
    ).astype(f32)
    t = torch.from_numpy(r_m).to(device)
    logdet = lu_logdet_nopivot(t).cpu().numpy().astype(np.float64)
    if bug_compat:
        logdet = logdet / math.log(10.0)
    const_m = -d * 0.5 * math.log(2.0 * math.pi) - 0.5 * logdet
    dist = (
        n1.astype(np.float64) * c.constant[iu]
        + n2.astype(np.float64) * c.constant[ju]
        - (n1 + n2).astype(np.float64) * const_m
    )
    p = int(np.argmin(dist))  # first minimum, like the reference's strict <
    min_c1, min_c2 = int(iu[p]), int(ju[p])
    best = add_clusters(c, min_c1, min_c2, bug_compat=bug_compat)
    c.N[min_c1] = best.N
    c.pi[min_c1] = best.pi
    c.constant[min_c1] = best.constant
    c.avgvar[min_c1] = best.avgvar
    c.means[min_c1] = best.means
    c.R[min_c1] = best.R
    c.Rinv[min_c1] = best.Rinv
    for i in range(min_c2, k - 1):
        c.copy_cluster(i, i + 1)
    return k - 1, min_c1, min_c2
