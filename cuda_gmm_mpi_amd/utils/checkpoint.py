"""Checkpoint / resume for the MDL sweep.

The reference keeps only an in-RAM best-model snapshot
(gaussian.cu:262-275, 839-851); here each completed K writes a small .npz
(cluster params + sweep bookkeeping) so an interrupted sweep resumes at
the next K. Parameters are KB-scale, so checkpointing is effectively free.
"""
from __future__ import annotations

import os

import numpy as np

from ..models.state import GmmState


def checkpoint_path(directory: str) -> str:
    return os.path.join(directory, "gmm_sweep.npz")


def save_sweep_checkpoint(directory: str, state: GmmState, k: int,
                          best: GmmState | None, best_k: int,
                          min_rissanen: float, best_lik: float,
                          riss_by_k: dict[int, float]) -> None:
    os.makedirs(directory, exist_ok=True)
    payload = {
        "k": np.int64(k),
        "best_k": np.int64(best_k),
        "min_rissanen": np.float64(min_rissanen),
        "best_lik": np.float64(best_lik),
        "riss_keys": np.array(sorted(riss_by_k), dtype=np.int64),
        "riss_vals": np.array(
            [riss_by_k[kk] for kk in sorted(riss_by_k)], dtype=np.float64),
    }
    for prefix, st in (("cur", state), ("best", best)):
        if st is None:
            continue
        payload.update({
            f"{prefix}_N": st.N.cpu().numpy(),
            f"{prefix}_pi": st.pi.cpu().numpy(),
            f"{prefix}_constant": st.constant.cpu().numpy(),
            f"{prefix}_avgvar": st.avgvar.cpu().numpy(),
            f"{prefix}_means": st.means.cpu().numpy(),
            f"{prefix}_R": st.R.cpu().numpy(),
            f"{prefix}_Rinv": st.Rinv.cpu().numpy(),
        })
    # np.savez appends .npz when missing — keep the suffix on the temp name
    tmp = os.path.join(directory, "gmm_sweep.tmp.npz")
    np.savez(tmp, **payload)
    os.replace(tmp, checkpoint_path(directory))


def _load_state(z, prefix: str) -> GmmState | None:
    key = f"{prefix}_N"
    if key not in z:
        return None
    import torch
    st = GmmState.empty(int(z[key].shape[0]), int(z[f"{prefix}_means"].shape[1]))
    for name in ("N", "pi", "constant", "avgvar", "means", "R", "Rinv"):
        getattr(st, name).copy_(torch.from_numpy(z[f"{prefix}_{name}"]))
    return st


def load_sweep_checkpoint(directory: str) -> dict | None:
    """Load the sweep checkpoint; a missing file returns None and an
    unreadable/corrupt one WARNS and returns None (the sweep starts
    fresh) — checkpoint writes are atomic (os.replace), so corruption
    here means external damage, not an interrupted writer."""
    path = checkpoint_path(directory)
    if not os.path.exists(path):
        return None
    try:
        z = np.load(path)
        riss = {int(k): float(v)
                for k, v in zip(z["riss_keys"], z["riss_vals"])}
        return {
            "k": int(z["k"]),
            "best_k": int(z["best_k"]),
            "min_rissanen": float(z["min_rissanen"]),
            "best_lik": float(z["best_lik"]),
            "rissanen_by_k": riss,
            "state": _load_state(z, "cur"),
            "best": _load_state(z, "best"),
        }
    except Exception as e:  # noqa: BLE001 — corrupt/alien file
        import sys
        print(f"WARNING: ignoring unreadable sweep checkpoint {path}: {e}",
              file=sys.stderr)
        return None
