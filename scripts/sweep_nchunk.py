import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch, time
from cuda_gmm_mpi_amd.ops import functional as F
rng = np.random.default_rng(0)
d, n, k = 24, 1000000, 64
x = torch.from_numpy(rng.standard_normal((d,n)).astype(np.float32)).cuda()
w = torch.rand(k, n, device="cuda")
xs = F.split_bf16_planes(x)
def t(f, iters=30):
    for _ in range(3): f()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(iters): f()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/iters*1e3
for nc in (128, 192, 256, 384, 512, 768):
    ms = t(lambda nc=nc: F.mstep_moments(x, w, nchunk=nc, precision="bf16x3", x_split=xs))
    print(f"b16 nchunk={nc}: {ms:.3f} ms")
