"""Device-resident EM + MDL order-reduction engine.

MI355X-native redesign of the reference driver (gaussian.cu:128-1106):

 - one process per GPU; events sharded across ranks (remainder handled
   correctly — fixes SURVEY §2.6 #4);
 - all sufficient statistics live on the device; ONE fused in-place RCCL
   all-reduce of {N, mean numerators, second moments} per EM iteration
   (+ one scalar likelihood reduce) replaces the reference's 7 host
   round-trips and 4 staged MPI_Allreduce calls;
 - covariance is finalized as (S - N mu mu^T + G*avgvar*I)/N, algebraically
   identical to the reference's centered per-event sums because S is
   mean-independent — this is what makes the single fused reduce possible;
 - the data is internally centered by the global mean (translation
   invariant math; output means get the center added back), eliminating
   the fp32 cancellation that the uncentered second-moment would suffer;
 - MDL outer loop: rank 0 merges on host with reference-faithful fp32
   math (including the log10/ln determinant quirk under bug_compat),
   broadcasts one fused parameter vector (vs 7 MPI_Bcasts).
"""
from __future__ import annotations

import dataclasses

import numpy as np
import torch

from .models.merge import HostClusters, reduce_order_batched
from .models.seed import seed_means_host, seed_state
from .models.state import GmmState
from .ops import functional as F
from .parallel import dist as pdist
from .utils.config import GmmConfig, em_epsilon, rissanen_score
from .utils.timers import Profile


@dataclasses.dataclass
class SweepResult:
    """Best-MDL model of the sweep (rank 0) plus bookkeeping."""
    state: GmmState                  # saved best params (host copies OK)
    num_clusters: int
    min_rissanen: float
    likelihood: float
    rissanen_by_k: dict[int, float]


class EmEngine:
    """EM on one event shard; collectives across the world."""

    def __init__(self, data_shard_by_event: torch.Tensor, config: GmmConfig,
                 num_events_total: int, center: torch.Tensor,
                 seed_means: torch.Tensor, var_per_dim: torch.Tensor,
                 device: torch.device | str = "cpu",
                 profile: Profile | None = None):
        self.cfg = config
        self.device = torch.device(device)
        self.rank = pdist.rank()
        self.world = pdist.world_size()
        self.n_total = num_events_total
        self.profile = profile or Profile(self.device)

        shard = data_shard_by_event.to(self.device, torch.float32)
        self.n_shard = int(shard.shape[0])
        self.d = int(shard.shape[1])
        self.center = center.to(self.device, torch.float32)       # [D]
        shard = shard - self.center.unsqueeze(0)
        # dimension-major resident copy for the quadratic-form kernels
        self.x = shard.T.contiguous()                              # [D, n]
        if config.estep_dtype == "bf16" and self.device.type == "cuda":
            self.x_estep: torch.Tensor = self.x.to(torch.bfloat16)
        else:
            self.x_estep = self.x
        # persistent hi/lo bf16 planes for the split-precision M-step
        self.x_split = (
            F.split_bf16_planes(self.x)
            if (config.mstep_precision == "bf16x3"
                and self.device.type == "cuda")
            else None
        )

        k0 = config.num_clusters
        # MFMA E-step paths (D <= 31: fused online-softmax, bf16 or
        # exact-f32, any K; 31 < D <= 142: big-D logw, bf16 or exact-f32);
        # otherwise the VALU kernels. DIAG_ONLY routes through the same
        # factor path: a diagonal Rinv yields a diagonal Cholesky factor,
        # so q = ||Uz+u0||^2 equals the diagonal quadratic form exactly.
        self.use_fused_estep = F.estep_fused_available(
            self.device, config.estep_dtype, self.d, k0,
        )
        # v1 lw-in-LDS variant writes posteriors directly — preferred
        # when K fits its LDS budget (the lse-aware M-step's staging exp
        # is on its VALU-bound critical path; measured ~19 us/iter)
        self.use_lds_estep = F.estep_fused_lds_available(
            self.device, config.estep_dtype, self.d, k0,
        )
        self.use_big_estep = F.estep_big_available(
            self.device, config.estep_dtype, self.d,
        ) and not self.use_fused_estep
        need_fac = self.use_fused_estep or self.use_big_estep
        need_f32_fac = need_fac and config.estep_dtype == "fp32"
        # the bf16 factor plane always exists when factors are emitted (it
        # is what triggers emission in the constants kernel); the f32 plane
        # is added for the exact-f32 MFMA kernels
        self.mfac = (
            torch.empty(k0, *F.mfac_shape(self.d), dtype=torch.bfloat16,
                        device=self.device)
            if need_fac else None
        )
        self.mfac32 = (
            torch.empty(k0, *F.mfac_shape(self.d)[1:], dtype=torch.float32,
                        device=self.device)
            if need_f32_fac else None
        )

        # persistent constant+ln(pi) buffer for the fused E-step kernels
        self._add = (
            torch.empty(k0, dtype=torch.float32, device=self.device)
            if self.device.type == "cuda" else None
        )

        self.state = GmmState.empty(k0, self.d, self.device)
        seed_state(
            self.state, seed_means - center.unsqueeze(0), var_per_dim,
            num_events_total, config.covariance_dynamic_range,
        )
        # constants for the seeded R=I state (constants_kernel after seeding,
        # gaussian.cu:404); also fills the add buffer
        self._update_constants(self.state)
        self._refresh_rinv(k0)  # state.Rinv valid from the start

        # membership / logw buffer, cluster-major [K, n_shard]. On CUDA
        # this holds LOG weights after the E-step, with the per-event
        # log-sum-exp in self._lse — the lse-aware M-step kernels apply
        # exp(logw - lse) on the fly, so posteriors are never materialized
        # during the sweep (use .posteriors() for normalized values).
        self.w = torch.empty(k0, self.n_shard, dtype=torch.float32,
                             device=self.device)
        self._lse = (
            torch.empty(self.n_shard, dtype=torch.float32, device=self.device)
            if self.device.type == "cuda" else None
        )
        self.epsilon = em_epsilon(self.d, num_events_total)
        self.likelihood = 0.0
        # hipGraph capture of the EM iteration (launch-overhead elimination);
        # per-K graphs, lazily captured. Disable with use_graphs=False or
        # GMM_NO_GRAPHS=1 (e.g. when per-bucket profiling is wanted).
        self.use_graphs = True
        self._graphs: dict[int, object] = {}
        self._lik_dev = torch.zeros(1, dtype=torch.float32, device=self.device)
        self.total_em_iterations = 0  # across the whole sweep (all Ks)


    def _refresh_mfac(self, k: int) -> None:
        """Re-emit the fused-E-step factors from the CURRENT covariance R
        (Cholesky of R, stable for any PD covariance) without touching
        the constants — after an MDL merge or a checkpoint resume the
        merged cluster's constant comes from the host path and must feed
        the next E-step unchanged, but the factor tables
        must match the compacted state."""
        if self.mfac is None and self.mfac32 is None:
            return
        from .ops.backend import hip_ext
        st = self.state.shrink(k)
        r_src = st.R
        if self.cfg.diag_only:
            # DIAG parity through a merge: the reference's diag E-step
            # reads diag(inv(R_merged_full)) — the host merge computed
            # the FULL-matrix inverse and quirk #8 carries it into the
            # next E-step — which is NOT inv(diag(R_full)). Emit the
            # factors from the diagonal matrix with exactly that
            # inverse; for an already-diagonal R (seed, resume, every
            # iteration after the first M-step) this is identical to R.
            d_inv = st.Rinv.diagonal(dim1=-2, dim2=-1)
            r_src = torch.diag_embed(1.0 / d_inv)
        empty_b = torch.empty(0, dtype=torch.bfloat16, device=self.device)
        empty_f = torch.empty(0, dtype=torch.float32, device=self.device)
        hip_ext().emit_factors(
            r_src.contiguous(), st.means.contiguous(),
            self.mfac[:k] if self.mfac is not None else empty_b,
            self.mfac32[:k] if self.mfac32 is not None else empty_f,
            empty_f, empty_f, empty_f,  # constants NOT recomputed (quirk #8)
        )

    def _update_constants(self, st: GmmState) -> None:
        k = st.num_clusters
        with self.profile.time("constants"):
            mfac = self.mfac[:k] if self.mfac is not None else None
            mfac32 = self.mfac32[:k] if self.mfac32 is not None else None
            pi_add = (
                (st.pi, self._add[:k]) if self._add is not None else None
            )
            if self._lazy_rinv and mfac is not None and pi_add is not None:
                # factor-path iterations never read Rinv: one emission
                # kernel produces the Cholesky factors AND the constants
                # (ln|R| = 2 sum ln diag L, fp-equivalent to the LU det);
                # the reference-faithful LU Rinv is refreshed once per K
                # (_refresh_rinv) for the merge/output path. Saves the
                # ~47 us LU kernel on every EM iteration.
                from .ops.backend import hip_ext
                empty_f = torch.empty(0, dtype=torch.float32,
                                      device=self.device)
                hip_ext().emit_factors(
                    st.R, st.means, mfac,
                    mfac32 if mfac32 is not None else empty_f,
                    st.pi, st.constant, self._add[:k],
                )
            else:
                # written in place into the state (no copy-back kernels)
                F.constants(st.R, st.means, self.cfg.diag_only,
                            mfac, mfac32, pi_add=pi_add,
                            out=(st.Rinv, st.constant))
        self.profile.count("constants")

    @property
    def _lazy_rinv(self) -> bool:
        """Factor-path iterations skip the LU kernel (Rinv refreshed per
        K). A PROPERTY of the live dispatch flags: tests (and debugging)
        flip use_fused_estep/use_big_estep after construction, and the
        VALU E-step they force DOES read Rinv every iteration."""
        return (self.device.type == "cuda"
                and (self.use_fused_estep or self.use_big_estep)
                and not self.cfg.diag_only)

    def _refresh_rinv(self, k: int) -> None:
        """Reference-faithful no-pivot-LU Rinv for the merge/output/
        checkpoint consumers (the factor-path iterations keep it lazy)."""
        if not self._lazy_rinv:
            return
        from .ops.backend import hip_ext
        st = self.state.shrink(k)
        empty_f = torch.empty(0, dtype=torch.float32, device=self.device)
        empty_b = torch.empty(0, dtype=torch.bfloat16, device=self.device)
        logdet = torch.empty(k, dtype=torch.float32, device=self.device)
        hip_ext().constants(st.R.contiguous(), st.means.contiguous(),
                            empty_f, st.Rinv, logdet, empty_f, empty_f,
                            empty_b, empty_f, False)

    def _sync_add(self, k: int) -> None:
        """Recompute constant+ln(pi) after host-side param loads (merge /
        resume) where constants must NOT be recomputed (quirk #8)."""
        if self._add is not None:
            st = self.state.shrink(k)
            self._add[:k] = st.constant + torch.log(st.pi)

    # ------------------------------------------------------------------ EM

    def _estep(self, k: int, need_lik: bool = True) -> torch.Tensor | None:
        """E-step into self.w[:k]; returns the shard-partial likelihood
        tensor, or None with ``need_lik=False`` (the dead-epsilon loop
        only reads the likelihood after its FINAL iteration — the
        reference computes-and-discards it every iteration)."""
        st = self.state.shrink(k)
        with self.profile.time("e_step"):
            self._w_is_logw = False
            if self.use_fused_estep and self.use_lds_estep:
                add = self._add[:k]
                if self.mfac32 is not None:
                    w, lik = F.estep_fused_f32_lds(self.x_estep,
                                                   self.mfac32[:k], add,
                                                   self.w[:k], need_lik)
                else:
                    w, lik = F.estep_fused_lds(self.x_estep, self.mfac[:k],
                                               add, self.w[:k], need_lik)
            elif self.use_fused_estep:
                self._w_is_logw = True
                add = self._add[:k]
                if self.mfac32 is not None:
                    w, lik = F.estep_fused_f32(self.x_estep, self.mfac32[:k],
                                               add, self.w[:k], self._lse,
                                               need_lik)
                else:
                    w, lik = F.estep_fused(self.x_estep, self.mfac[:k], add,
                                           self.w[:k], self._lse, need_lik)
            elif self.use_big_estep:
                add = self._add[:k]
                if self.mfac32 is not None:
                    logw = F.estep_logw_big_f32(self.x_estep,
                                                self.mfac32[:k], add,
                                                self.w[:k])
                    self._w_is_logw = True
                else:
                    logw = F.estep_logw_big(self.x_estep, self.mfac[:k],
                                            add, self.w[:k])
                lik = F.estep_lse(logw, self._lse, need_lik)
                self._w_is_logw = True
            elif self.device.type == "cuda":
                logw = F.estep_logw(
                    self.x_estep, st.means, st.Rinv, st.constant, st.pi,
                    self.cfg.diag_only, out=self.w[:k],
                )
                lik = F.estep_lse(logw, self._lse, need_lik)
                self._w_is_logw = True
            else:
                logw = F.estep_logw(
                    self.x_estep, st.means, st.Rinv, st.constant, st.pi,
                    self.cfg.diag_only, out=self.w[:k],
                )
                w, lik = F.estep_posteriors(logw)
        self.profile.count("regroup")
        return lik

    def _finish_likelihood(self, lik_part: torch.Tensor) -> None:
        """All-reduce the shard likelihood into the persistent device
        scalar (graph-capturable: no host sync)."""
        with self.profile.time("comm"):
            t = lik_part.reshape(1)
            pdist.all_reduce_(t)
            self._lik_dev.copy_(t)

    def _reduce_likelihood(self, lik_part: torch.Tensor) -> float:
        self._finish_likelihood(lik_part)
        return float(self._lik_dev.item())

    def _iteration_body(self, k: int, need_lik: bool = True) -> None:
        """One EM iteration; with need_lik the reduced likelihood lands in
        the persistent device scalar."""
        self._mstep(k)
        lik = self._estep(k, need_lik)
        if need_lik and lik is not None:
            self._finish_likelihood(lik)

    def _can_graph(self) -> bool:
        import os
        if not self.use_graphs or os.environ.get("GMM_NO_GRAPHS"):
            return False
        if self.device.type != "cuda":
            return False
        if self.world > 1:
            # RCCL collectives inside hipGraph capture are unexercised in
            # this environment (single-GPU boxes); the eager path costs
            # ~3% and cannot hang. Opt in with GMM_DIST_GRAPHS=1.
            return bool(os.environ.get("GMM_DIST_GRAPHS"))
        return True

    def _mstep(self, k: int) -> None:
        """M-step: fused moments, ONE all-reduce, finalize params + constants.

        The packed buffer [K, (D+1)(D+2)/2] carries S, the mean numerators
        and N in one contiguous payload (functional.mstep_moments layout) —
        a single in-place RCCL all-reduce replaces the reference's staged
        N / means / R reductions (gaussian.cu:545-686).
        """
        st = self.state.shrink(k)
        with self.profile.time("m_step"):
            packed = F.mstep_moments(
                self.x, self.w[:k], precision=self.cfg.mstep_precision,
                x_split=self.x_split,
                lse=self._lse if getattr(self, "_w_is_logw", False) else None)
        with self.profile.time("comm"):
            pdist.all_reduce_(packed)
        lazy = (self._lazy_rinv and self._add is not None
                and self.mfac is not None)
        if lazy:
            # ONE kernel: finalize (N/means/R/pi, reference rules) fused
            # with the factor + constants emission — R stays L1-hot and
            # the iteration drops a launch; Rinv stays lazy (per K)
            from .ops.backend import hip_ext
            with self.profile.time("m_step"):
                empty_f = torch.empty(0, dtype=torch.float32,
                                      device=self.device)
                hip_ext().mstep_finalize_emit(
                    packed, st.avgvar, self.world, st.N, st.means, st.R,
                    st.pi, st.constant, self._add[:k], self.mfac[:k],
                    self.mfac32[:k] if self.mfac32 is not None else empty_f,
                )
            self.profile.count("params")
            self.profile.count("constants")
            return
        with self.profile.time("m_step"):
            if self.device.type == "cuda":
                # one kernel: N, means, R, pi from the packed moments with
                # the reference's exact finalize rules
                from .ops.backend import hip_ext
                hip_ext().mstep_finalize(
                    packed, st.avgvar, self.world, st.N, st.means, st.R,
                    st.pi, bool(self.cfg.diag_only),
                )
            else:
                n_c, mean_num, s = F.moments_views(packed, self.d)
                st.N.copy_(n_c)
                st.means.copy_(F.finalize_means(n_c, mean_num))
                st.R.copy_(F.finalize_covariance(
                    n_c, st.means, s, st.avgvar, self.world,
                    self.cfg.diag_only,
                ))
        self.profile.count("params")
        self._update_constants(st)
        if self.device.type != "cuda":
            st.pi.copy_(F.compute_pi(st.N))

    def run_em(self, k: int) -> float:
        """Full EM at fixed K (the inner loop of gaussian.cu:479-755).

        Returns the final global log-likelihood.
        """
        cfg = self.cfg
        if cfg.min_iters >= cfg.max_iters and not cfg.verbose:
            # epsilon is dead (reference default MIN_ITERS == MAX_ITERS,
            # gaussian.h:26-27): the loop runs exactly min_iters times no
            # matter what the likelihood does. Keep the scalar on device,
            # read it ONCE per K, and skip the likelihood reduce chain on
            # every iteration but the last (the reference computes and
            # discards it; the final model and likelihood are identical).
            self._estep(k, need_lik=False)
            for _ in range(max(0, cfg.min_iters - 1)):
                self.em_iteration(k, need_lik=False)
            if cfg.min_iters >= 1:
                self.em_iteration(k, need_lik=True)
            else:
                self._finish_likelihood(self._estep(k))
            lik = float(self._lik_dev.item())
            self.likelihood = lik
            self.total_em_iterations += cfg.min_iters
            return lik
        lik = self._reduce_likelihood(self._estep(k))
        iters = 0
        change = self.epsilon * 2
        while iters < cfg.min_iters or (
            abs(change) > self.epsilon and iters < cfg.max_iters
        ):
            old_lik = lik
            self.em_iteration(k)
            lik = float(self._lik_dev.item())
            change = lik - old_lik
            iters += 1
            if cfg.verbose and self.rank == 0:
                print(f"[K={k}] iter {iters}: likelihood {lik:e} "
                      f"(change {change:e})")
        self.likelihood = lik
        self.total_em_iterations += iters
        return lik

    def em_iteration(self, k: int, need_lik: bool = True) -> None:
        """One EM iteration: M-step + all-reduce + constants + E-step +
        likelihood reduce. On GPU the whole sequence replays as one
        hipGraph (captured lazily per K; the capturing call runs the
        iteration eagerly so iteration counts stay exact).

        (Requires a prior _estep so self.w holds posteriors.)
        """
        if self._can_graph():
            key = (k, need_lik)
            g = self._graphs.get(key)
            if g is None:
                self._capture_iteration(k, need_lik)
                return  # the capture's eager warmup WAS this iteration
            if g:
                g.replay()
                return
        self._iteration_body(k, need_lik)

    def _capture_iteration(self, k: int, need_lik: bool = True):
        """Run one eager iteration (counts), then capture the graph."""
        self._iteration_body(k, need_lik)
        try:
            self.profile.paused = True
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._iteration_body(k, need_lik)
            self._graphs[(k, need_lik)] = g
        except Exception:  # noqa: BLE001 — graphs are an optimization only
            self._graphs[(k, need_lik)] = False
        finally:
            self.profile.paused = False
        return None

    # ------------------------------------------------- MDL sweep / merging

    def _bcast_checkpoint(self, ck: dict | None) -> dict | None:
        """Rank-0-authoritative checkpoint resume. Only rank 0 reads the
        checkpoint directory — it is the only rank that writes checkpoints,
        and on a rank-0-only filesystem (--scatter-input deployments) the
        only rank that can have one. The payload is broadcast as tensors
        so every rank resumes identically (a per-rank read could diverge:
        rank 0 resuming at k' while others start at k0 deadlocks the
        collectives)."""
        dev = self.device
        if self.rank == 0:
            found = int(bool(ck and ck.get("state") is not None))
            hdr_vals = [
                found,
                int(ck["k"]) if found else 0,
                int(ck["best_k"]) if found else 0,
                0 if (found and ck.get("best") is not None) else 1,
                len(ck["rissanen_by_k"]) if found else 0,
            ]
            hdr = torch.tensor(hdr_vals, dtype=torch.int64, device=dev)
        else:
            hdr = torch.zeros(5, dtype=torch.int64, device=dev)
        pdist.broadcast_(hdr)
        found, k, best_k, best_none, nk = (int(x) for x in hdr.cpu())
        if not found:
            return None
        if self.rank == 0:
            scal = torch.tensor([ck["min_rissanen"], ck["best_lik"]],
                                dtype=torch.float64, device=dev)
        else:
            scal = torch.zeros(2, dtype=torch.float64, device=dev)
        pdist.broadcast_(scal)
        keys = torch.zeros(max(nk, 1), dtype=torch.int64, device=dev)
        vals = torch.zeros(max(nk, 1), dtype=torch.float64, device=dev)
        if self.rank == 0 and nk:
            kk = sorted(ck["rissanen_by_k"])
            keys[:nk] = torch.tensor(kk, dtype=torch.int64)
            vals[:nk] = torch.tensor([ck["rissanen_by_k"][x] for x in kk],
                                     dtype=torch.float64)
        pdist.broadcast_(keys)
        pdist.broadcast_(vals)

        def bcast_state(st_src, kc):
            size = 4 * kc + kc * self.d + 2 * kc * self.d * self.d
            if self.rank == 0:
                vec = st_src.param_vector().to(dev)
            else:
                vec = torch.zeros(size, dtype=torch.float32, device=dev)
            pdist.broadcast_(vec)
            if self.rank == 0:
                return st_src
            st = GmmState.empty(kc, self.d)
            st.load_param_vector(vec.cpu())
            return st

        state = bcast_state(ck["state"] if self.rank == 0 else None, k)
        best = (None if best_none
                else bcast_state(ck["best"] if self.rank == 0 else None,
                                 best_k))
        return {
            "k": k, "state": state, "best": best, "best_k": best_k,
            "min_rissanen": float(scal[0].item()),
            "best_lik": float(scal[1].item()),
            "rissanen_by_k": {
                int(keys[i].item()): float(vals[i].item())
                for i in range(nk)
            },
        }

    def _host_clusters(self, k: int) -> HostClusters:
        st = self.state.shrink(k)
        return HostClusters(
            N=st.N.cpu().numpy().copy(),
            pi=st.pi.cpu().numpy().copy(),
            constant=st.constant.cpu().numpy().copy(),
            avgvar=st.avgvar.cpu().numpy().copy(),
            means=st.means.cpu().numpy().copy(),
            R=st.R.cpu().numpy().copy(),
            Rinv=st.Rinv.cpu().numpy().copy(),
        )

    def _load_host_clusters(self, hc: HostClusters, k: int) -> None:
        st = self.state.shrink(k)
        st.N.copy_(torch.from_numpy(np.ascontiguousarray(hc.N[:k])))
        st.pi.copy_(torch.from_numpy(np.ascontiguousarray(hc.pi[:k])))
        st.constant.copy_(torch.from_numpy(np.ascontiguousarray(hc.constant[:k])))
        st.avgvar.copy_(torch.from_numpy(np.ascontiguousarray(hc.avgvar[:k])))
        st.means.copy_(torch.from_numpy(np.ascontiguousarray(hc.means[:k])))
        st.R.copy_(torch.from_numpy(np.ascontiguousarray(hc.R[:k])))
        st.Rinv.copy_(torch.from_numpy(np.ascontiguousarray(hc.Rinv[:k])))

    def sweep(self) -> SweepResult:
        """MDL model-order sweep K0 -> stop (gaussian.cu:479-960).

        With ``cfg.checkpoint_dir`` set, every completed K (post-merge)
        writes a parameter checkpoint; an existing checkpoint is resumed
        from automatically.
        """
        cfg = self.cfg
        k = cfg.num_clusters
        stop = cfg.stop_number
        best_state: GmmState | None = None
        best_k = k
        min_rissanen = float("inf")
        best_lik = 0.0
        riss_by_k: dict[int, float] = {}

        if cfg.checkpoint_dir:
            from .utils.checkpoint import load_sweep_checkpoint
            ck = (load_sweep_checkpoint(cfg.checkpoint_dir)
                  if self.rank == 0 else None)
            if self.world > 1:
                ck = self._bcast_checkpoint(ck)
            if (ck and ck["state"] is not None
                    and ck["state"].means.shape[1] != self.d):
                # a checkpoint for a different dataset shape: loading it
                # would crash mid-copy with an opaque shape error
                if self.rank == 0:
                    import warnings
                    warnings.warn(
                        f"checkpoint dimensionality "
                        f"D={int(ck['state'].means.shape[1])} does not "
                        f"match the dataset D={self.d}; ignoring it and "
                        f"sweeping fresh")
                ck = None
            if ck and ck["state"] is not None and ck["k"] < stop:
                # the checkpoint's sweep position is already BELOW this
                # run's target: it was written under a different target
                # (checkpoints lag one merge, and best-state snapshots
                # follow the OLD target's save rule), so the model this
                # run asks for may not exist in it. Resuming would
                # silently return a stale snapshot — start fresh instead.
                if self.rank == 0:
                    import warnings
                    warnings.warn(
                        f"checkpoint at K={ck['k']} is below the requested "
                        f"target K={stop}; ignoring it and sweeping fresh")
                ck = None
            if ck and ck["state"] is not None and ck["k"] <= k:
                k = ck["k"]
                st = ck["state"].to(self.device)
                self._load_host_clusters(
                    HostClusters(
                        N=st.N.cpu().numpy(), pi=st.pi.cpu().numpy(),
                        constant=st.constant.cpu().numpy(),
                        avgvar=st.avgvar.cpu().numpy(),
                        means=st.means.cpu().numpy(), R=st.R.cpu().numpy(),
                        Rinv=st.Rinv.cpu().numpy(),
                    ), k)
                # regenerate the fused-E-step factors for the resumed
                # params without recomputing the saved constants
                self._refresh_mfac(k)
                self._sync_add(k)
                best_state = ck["best"]
                best_k = ck["best_k"]
                min_rissanen = ck["min_rissanen"]
                best_lik = ck["best_lik"]
                riss_by_k = dict(ck["rissanen_by_k"])

        while k >= stop:
            lik = self.run_em(k)
            # lazy-Rinv paths: make state.Rinv reference-faithful before
            # it is saved, merged, checkpointed or broadcast
            self._refresh_rinv(k)
            riss = rissanen_score(lik, k, self.d, self.n_total)
            riss_by_k[k] = riss
            if cfg.enable_print and self.rank == 0:
                print(f"\nRissanen Score: {riss:e}")

            save = (
                k == cfg.num_clusters
                or (riss < min_rissanen and cfg.target_num_clusters == 0)
                or k == cfg.target_num_clusters
                # nearest-target mode: keep updating while above the target
                # so a jumped-over target yields the closest completed K
                or (cfg.nearest_target and cfg.target_num_clusters > 0
                    and k >= cfg.target_num_clusters)
            )
            if save:
                min_rissanen = riss
                best_k = k
                best_lik = lik
                best_state = self.state.shrink(k).clone(with_memberships=False)

            if k <= stop:
                break

            # ---- order reduction on rank 0, fused param broadcast
            with self.profile.time("reduce"):
                new_k = k
                if self.rank == 0:
                    from .models.merge import eliminate_empty_clusters
                    hc = self._host_clusters(k)
                    k_live = eliminate_empty_clusters(hc)
                    if k_live >= 2:
                        new_k, _, _ = reduce_order_batched(
                            hc, bug_compat=cfg.bug_compat,
                            device=str(self.device))
                    else:
                        # elimination left < 2 clusters: nothing to merge
                        # (the reference's pair scan would read
                        # uninitialized merge state here); stop reducing
                        # at the survivor count
                        new_k = k_live
                if self.world > 1:
                    # NCCL/RCCL broadcasts device tensors only
                    nk = torch.tensor([new_k], dtype=torch.int64,
                                      device=self.device)
                    pdist.broadcast_(nk, src=0)
                    new_k = int(nk.item())
                if self.rank == 0 and new_k >= 1:
                    self._load_host_clusters(hc.truncated(new_k), new_k)
                if self.world > 1 and new_k >= 1:
                    with self.profile.time("comm"):
                        vec = self.state.shrink(new_k).param_vector().contiguous()
                        pdist.broadcast_(vec, src=0)
                    if self.rank != 0:
                        self.state.shrink(new_k).load_param_vector(vec)
            # the merged/compacted Rinv needs fresh E-step factors; the
            # constants stay as the host merge path produced them
            if new_k >= 1:
                self._refresh_mfac(new_k)
                self._sync_add(new_k)
            self.profile.count("reduce")
            k = new_k
            if cfg.checkpoint_dir and self.rank == 0:
                from .utils.checkpoint import save_sweep_checkpoint
                save_sweep_checkpoint(
                    cfg.checkpoint_dir, self.state.shrink(k), k,
                    best_state, best_k, min_rissanen, best_lik, riss_by_k)

        assert best_state is not None
        if cfg.enable_print and self.rank == 0:
            print(f"\nFinal rissanen score was: {min_rissanen:f}, "
                  f"with {best_k} clusters.")
        return SweepResult(
            state=best_state, num_clusters=best_k,
            min_rissanen=min_rissanen, likelihood=best_lik,
            rissanen_by_k=riss_by_k,
        )

    def posteriors(self, k: int) -> torch.Tensor:
        """Normalized posteriors [k, n_shard] for the current E-step state
        (paths that emit log weights normalize with the lse)."""
        if self._lse is None or not getattr(self, "_w_is_logw", False):
            return self.w[:k]
        return torch.exp(self.w[:k] - self._lse.unsqueeze(0))

    # ------------------------------------------------------------- output

    def recompute_memberships(self, saved: GmmState) -> torch.Tensor:
        """Shard posteriors [K, n_shard] for the saved best model.

        Memberships never leave their shard during the sweep; for the
        .results file they are regenerated from the saved parameters (the
        final E-step of the best K is a pure function of those parameters)
        and gathered once.
        """
        k = saved.num_clusters
        st = saved.to(self.device)
        logw = F.estep_logw(
            self.x_estep, st.means, st.Rinv, st.constant, st.pi,
            self.cfg.diag_only,
        )
        w, _ = F.estep_posteriors(logw)
        return w

    def gather_memberships(self, w_shard: torch.Tensor) -> np.ndarray | None:
        """Gather per-event posteriors to rank 0 only, via exact-size
        point-to-point sends (reference analog: the hand-rolled
        MPI_Send/Recv gather, gaussian.cu:783-823).

        Non-root ranks send their [K, n_shard] shard and allocate nothing;
        rank 0 streams one shard-sized device buffer at a time into the
        host [K, N_total] output — no padded all_gather, no full [K, N]
        materialization anywhere but the writing rank's host buffer.
        Returns [K, N_total] on rank 0, None elsewhere.
        """
        k = int(w_shard.shape[0])
        if self.world == 1:
            return w_shard.cpu().numpy()
        if self.rank != 0:
            torch.distributed.send(w_shard.contiguous(), dst=0)
            return None
        out = np.empty((k, self.n_total), dtype=np.float32)
        s0, e0 = pdist.shard_bounds(self.n_total, self.world, 0)
        out[:, s0:e0] = w_shard.cpu().numpy()
        for r in range(1, self.world):
            s, e = pdist.shard_bounds(self.n_total, self.world, r)
            buf = torch.empty(k, e - s, dtype=torch.float32,
                              device=w_shard.device)
            torch.distributed.recv(buf, src=r)
            out[:, s:e] = buf.cpu().numpy()
        return out


def build_engine(data_by_event: np.ndarray | torch.Tensor, config: GmmConfig,
                 device: torch.device | str = "cpu",
                 profile: Profile | None = None) -> EmEngine:
    """Build an engine from rank-consistent full data.

    Every rank passes the same full dataset (CLI: rank 0 broadcasts it
    first — see cli.run_clustering); each rank keeps only its shard.
    Seeding statistics (strided means, per-dim variance, global center)
    are computed from the full data so results are world-size independent.
    """
    if isinstance(data_by_event, np.ndarray):
        data = torch.from_numpy(np.ascontiguousarray(data_by_event, np.float32))
    else:
        data = data_by_event.to(torch.float32).cpu()
    n, d = data.shape
    config.validate()

    mean = data.double().mean(dim=0)
    var = data.double().pow(2).mean(dim=0) - mean * mean
    center = mean.float() if config.center_data else torch.zeros(d)
    seed_means = seed_means_host(data, config.num_clusters)

    r, w = pdist.rank(), pdist.world_size()
    s, e = pdist.shard_bounds(n, w, r)
    return EmEngine(
        data[s:e], config, n, center, seed_means, var.float(),
        device=device, profile=profile,
    )


def build_engine_sharded(shard: torch.Tensor, config: GmmConfig,
                         n_total: int, mean: torch.Tensor,
                         var: torch.Tensor, seed_means: torch.Tensor,
                         device: torch.device | str = "cpu",
                         profile: Profile | None = None) -> EmEngine:
    """Build an engine from a pre-distributed shard (rank-0-read input:
    parallel.dist.distribute_input). `mean`/`var`/`seed_means` are the
    full-data statistics broadcast from rank 0, so results are identical
    to the shared-filesystem `build_engine` path."""
    config.validate()
    d = shard.shape[1]
    center = mean.float() if config.center_data else torch.zeros(d)
    return EmEngine(
        shard.to(torch.float32), config, n_total, center, seed_means,
        var.float(), device=device, profile=profile,
    )
