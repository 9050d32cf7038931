"""GMM cluster state — the MI355X-native equivalent of ``clusters_t``
(gaussian.h:62-76): SoA of fp32 torch tensors, device-resident.

Unlike the reference (which keeps one host copy per GPU plus a device copy
and round-trips every M-step stage), this state lives on ONE device and the
sufficient statistics are reduced in place with RCCL.
"""
from __future__ import annotations

import dataclasses

import torch


@dataclasses.dataclass
class GmmState:
    """Parameters of a K-cluster, D-dimensional GMM.

    memberships are cluster-major [K, N_shard] like the reference
    (memberships[c*num_events+e], gaussian.h:75) and always shard-local.
    """
    N: torch.Tensor          # [K]
    pi: torch.Tensor         # [K]
    constant: torch.Tensor   # [K] log normalizer
    avgvar: torch.Tensor     # [K] diagonal regularizer
    means: torch.Tensor      # [K, D]
    R: torch.Tensor          # [K, D, D]
    Rinv: torch.Tensor       # [K, D, D]
    memberships: torch.Tensor | None = None  # [K, N_shard]

    @property
    def num_clusters(self) -> int:
        return int(self.N.shape[0])

    @property
    def num_dimensions(self) -> int:
        return int(self.means.shape[1])

    @property
    def device(self) -> torch.device:
        return self.N.device

    @staticmethod
    def empty(num_clusters: int, num_dimensions: int,
              device: torch.device | str = "cpu") -> "GmmState":
        k, d = num_clusters, num_dimensions
        f = dict(dtype=torch.float32, device=device)
        return GmmState(
            N=torch.zeros(k, **f), pi=torch.zeros(k, **f),
            constant=torch.zeros(k, **f), avgvar=torch.zeros(k, **f),
            means=torch.zeros(k, d, **f), R=torch.zeros(k, d, d, **f),
            Rinv=torch.zeros(k, d, d, **f), memberships=None,
        )

    def clone(self, with_memberships: bool = True) -> "GmmState":
        return GmmState(
            N=self.N.clone(), pi=self.pi.clone(), constant=self.constant.clone(),
            avgvar=self.avgvar.clone(), means=self.means.clone(),
            R=self.R.clone(), Rinv=self.Rinv.clone(),
            memberships=(
                self.memberships.clone()
                if (with_memberships and self.memberships is not None) else None
            ),
        )

    def shrink(self, num_clusters: int) -> "GmmState":
        """View of the first ``num_clusters`` clusters (the sweep loop
        narrows K in place like the reference's num_clusters countdown)."""
        return GmmState(
            N=self.N[:num_clusters], pi=self.pi[:num_clusters],
            constant=self.constant[:num_clusters],
            avgvar=self.avgvar[:num_clusters],
            means=self.means[:num_clusters], R=self.R[:num_clusters],
            Rinv=self.Rinv[:num_clusters],
            memberships=(
                self.memberships[:num_clusters]
                if self.memberships is not None else None
            ),
        )

    def to(self, device) -> "GmmState":
        return GmmState(
            N=self.N.to(device), pi=self.pi.to(device),
            constant=self.constant.to(device), avgvar=self.avgvar.to(device),
            means=self.means.to(device), R=self.R.to(device),
            Rinv=self.Rinv.to(device),
            memberships=(
                self.memberships.to(device)
                if self.memberships is not None else None
            ),
        )

    def param_vector(self) -> torch.Tensor:
        """Flat fp32 vector of all parameters (for one-shot broadcast after
        a merge — replaces the reference's 7 separate MPI_Bcasts,
        gaussian.cu:918-924)."""
        return torch.cat([
            self.N, self.pi, self.constant, self.avgvar,
            self.means.reshape(-1), self.R.reshape(-1), self.Rinv.reshape(-1),
        ])

    def load_param_vector(self, v: torch.Tensor) -> None:
        k, d = self.num_clusters, self.num_dimensions
        off = 0
        for t, numel in (
            (self.N, k), (self.pi, k), (self.constant, k), (self.avgvar, k),
            (self.means, k * d), (self.R, k * d * d), (self.Rinv, k * d * d),
        ):
            t.copy_(v[off:off + numel].view_as(t))
            off += numel
