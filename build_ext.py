#!/usr/bin/env python3
"""Direct hipcc build of the gfx950 extension — no hipify pass, no CUDA
compat machinery: the sources are native HIP and are compiled as-is.

Produces cuda_gmm_mpi_amd/ops/_gmm_hip.so in-tree (travels with repo
snapshots to GPU boxes).
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

REPO = os.path.dirname(os.path.abspath(__file__))
SRC = os.path.join(REPO, "cuda_gmm_mpi_amd", "ops", "hip", "gmm_ext.hip")
OUT = os.path.join(REPO, "cuda_gmm_mpi_amd", "ops", "_gmm_hip.so")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def torch_paths():
    import torch
    from torch.utils import cpp_extension as ce
    return ce.include_paths(), ce.library_paths(), torch._C._GLIBCXX_USE_CXX11_ABI


def build(verbose: bool = True, asan: bool = False) -> str:
    """asan=True adds host-side AddressSanitizer to the extension build
    (SURVEY §5 race/sanitizer row; device code is unaffected)."""
    includes, libdirs, cxx11_abi = torch_paths()
    cmd = [
        HIPCC, "-O3", "-std=c++17", "-fPIC", "-shared",
        f"--offload-arch={ARCH}",
        f"-D_GLIBCXX_USE_CXX11_ABI={1 if cxx11_abi else 0}",
        "-DUSE_ROCM=1", "-D__HIP_PLATFORM_AMD__=1",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-DTORCH_EXTENSION_NAME=_gmm_hip",
        "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1",
        "-Wno-unused-result",
        SRC, "-o", OUT,
    ]
    for inc in includes + [sysconfig.get_paths()["include"]]:
        cmd.append(f"-I{inc}")
    if asan:
        cmd += ["-fsanitize=address", "-shared-libasan"]
    for lib in libdirs:
        cmd += [f"-L{lib}", f"-Wl,-rpath,{lib}"]
    cmd += ["-ltorch", "-ltorch_cpu", "-ltorch_hip", "-lc10", "-lc10_hip",
            "-ltorch_python", "-lamdhip64"]
    if verbose:
        print(" ".join(cmd))
    subprocess.run(cmd, check=True)
    return OUT


if __name__ == "__main__":
    import sys as _sys
    build(asan="--asan" in _sys.argv)
    print(f"built {OUT}")
    sys.exit(0)
