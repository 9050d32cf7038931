import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch, time
from cuda_gmm_mpi_amd.ops import functional as F
rng = np.random.default_rng(0)
d, n, k = 24, 1000000, 64
x = torch.from_numpy(rng.standard_normal((d,n)).astype(np.float32)).cuda()
w = torch.rand(k, n, device="cuda")
def t(f, iters=30):
    for _ in range(3): f()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(iters): f()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/iters*1e3
def m32(): return F.mstep_moments(x, w)
xs = F.split_bf16_planes(x)
def mb16(): return F.mstep_moments(x, w, precision="bf16x3", x_split=xs)
print("moments fp32 : %.3f ms" % t(m32))
print("moments b16x3: %.3f ms" % t(mb16))
means = torch.randn(k, d, device="cuda")
r = torch.eye(d, device="cuda").expand(k,d,d).contiguous()*3
mfac = torch.empty(k,2,32,32,dtype=torch.bfloat16,device="cuda")
rinv, const = F.constants(r, means, False, mfac)
add = const + torch.log(torch.full((k,),1.0/k,device="cuda"))
xb = x.to(torch.bfloat16); wo = torch.empty(k,n,device="cuda")
lse = torch.empty(n, device="cuda")
def es(): return F.estep_fused(xb, mfac, add, wo, lse)
print("estep_fused  : %.3f ms" % t(es))
