"""Deterministic cluster seeding.

Reference behavior: the GPU seed kernel
(gaussian_kernel.cu:269-328) computes avgvar on the master GPU's shard and
sets R=I, pi=1/K, then the host ``seed_clusters`` (gaussian.cu:108-123)
overwrites means and N from evenly-strided events of the FULL dataset, and
the result is broadcast. What survives the overwrite: R=I, pi=1/K, avgvar,
and the host means/N.

MI355X-native version: deterministic and GPU-count independent — the
strided means come from the full dataset (exactly the host formula) and
avgvar is computed from full-data sums reduced across shards, so 1-GPU and
8-GPU runs seed identically.
"""
from __future__ import annotations

import torch

from .state import GmmState


def seed_means_host(data_by_event: torch.Tensor, num_clusters: int) -> torch.Tensor:
    """Evenly-strided event means (host seed_clusters, gaussian.cu:108-123).

    means[c] = data[(int)(c * seed)] with seed = (N-1)/(K-1).
    data_by_event: [N, D].
    """
    n = data_by_event.shape[0]
    if num_clusters > 1:
        seed = (n - 1.0) / (num_clusters - 1.0)
    else:
        seed = 0.0
    idx = torch.tensor(
        [int(c * seed) for c in range(num_clusters)], dtype=torch.long
    )
    return data_by_event[idx].to(torch.float32)


def seed_state(
    state: GmmState,
    seed_means: torch.Tensor,   # [K, D] from seed_means_host on full data
    total_variance_per_dim: torch.Tensor,  # [D] full-data per-dim variance
    num_events_total: int,
    covariance_dynamic_range: float,
) -> None:
    """Populate a GmmState in place with the reference's surviving seed
    values: means (strided events), N = N_total/K (integer division,
    gaussian.cu:118), pi = 1/K, R = I, avgvar = mean(var)/CDR
    (gaussian_kernel.cu:316-326)."""
    k, d = state.num_clusters, state.num_dimensions
    state.means.copy_(seed_means.to(state.means.device))
    state.N.fill_(float(num_events_total // k))
    state.pi.fill_(1.0 / k)
    avgvar = float(total_variance_per_dim.mean()) / covariance_dynamic_range
    state.avgvar.fill_(avgvar)
    eye = torch.eye(d, dtype=torch.float32, device=state.R.device)
    state.R.copy_(eye.expand(k, d, d))
