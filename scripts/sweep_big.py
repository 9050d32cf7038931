import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch, time
from cuda_gmm_mpi_amd.ops import functional as F
from cuda_gmm_mpi_amd.ops.backend import hip_ext
rng = np.random.default_rng(1)
d, n, k = 128, 500_000, 256
x = torch.from_numpy(rng.standard_normal((d, n)).astype(np.float32)).cuda()
w = torch.rand(k, n, device="cuda")
def t(f, iters=10):
    for _ in range(2): f()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): f()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters * 1e3
for nc in (16, 31, 62, 124, 248):
    ms = t(lambda nc=nc: F.mstep_moments(x, w, nchunk=nc, precision="bf16x3"))
    print(f"moments_big nchunk={nc}: {ms:.3f} ms", flush=True)
