"""Dispatch between the CPU reference ops and the HIP/CDNA4 extension.

Policy (deliberate): on a CUDA/ROCm device the hand-written gfx950 kernels
are THE compute path — if the extension is missing or fails to import, GPU
ops raise instead of silently falling back to eager torch. CPU tensors use
the torch fp32 reference implementations (`cpu_reference`).
"""
from __future__ import annotations

import importlib

_ext = None
_ext_error: Exception | None = None


def _load_extension():
    global _ext, _ext_error
    if _ext is not None or _ext_error is not None:
        return _ext
    try:
        _ext = importlib.import_module("cuda_gmm_mpi_amd.ops._gmm_hip")
    except Exception as e:  # noqa: BLE001
        _ext_error = e
        _ext = None
    return _ext


def hip_ext():
    """The compiled extension module, or raise with a loud message."""
    ext = _load_extension()
    if ext is None:
        raise RuntimeError(
            "cuda_gmm_mpi_amd HIP extension (_gmm_hip) is not built. "
            "GPU execution requires the hand-written gfx950 kernels — there "
            "is no eager fallback by design. Build with: "
            "PYTORCH_ROCM_ARCH=gfx950 python build_ext.py "
            f"(import error: {_ext_error})"
        )
    return ext


def has_hip_ext() -> bool:
    return _load_extension() is not None
