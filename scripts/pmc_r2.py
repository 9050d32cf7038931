#!/usr/bin/env python3
"""Minimal kernel exercise for rocprofv3 --pmc counter collection (r2).

Runs each hot kernel a handful of times on its benchmark shape so the
counter pass attributes cleanly. Run under:
  rocprofv3 --pmc LDSBankConflict MfmaUtil VALUBusy -d DIR -- python ...
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from cuda_gmm_mpi_amd.ops import functional as F

rng = np.random.default_rng(0)

# flagship shapes: K=64, D=24, N=1M
d, n, k = 24, 1_000_000, 64
x = torch.from_numpy(rng.standard_normal((d, n)).astype(np.float32)).cuda()
xb = x.to(torch.bfloat16)
xs = F.split_bf16_planes(x)
w = torch.rand(k, n, device="cuda")
lse = torch.zeros(n, device="cuda")
means = torch.randn(k, d, device="cuda")
r = torch.eye(d, device="cuda").expand(k, d, d).contiguous() * 3
mfac = torch.empty(k, 2, 32, 32, dtype=torch.bfloat16, device="cuda")
rinv, const = F.constants(r, means, False, mfac)
add = const + float(np.log(1.0 / k))
wo = torch.empty(k, n, device="cuda")

for _ in range(4):
    F.estep_fused(xb, mfac, add, wo, lse)
for _ in range(4):
    F.mstep_moments(x, w, precision="bf16x3", x_split=xs, lse=lse)
torch.cuda.synchronize()

# config-4 shapes: K=256, D=128, N=250k slice
d2, n2, k2 = 128, 250_000, 256
x2 = torch.from_numpy(rng.standard_normal((d2, n2)).astype(np.float32)).cuda()
x2b = x2.to(torch.bfloat16)
w2 = torch.rand(k2, n2, device="cuda")
lse2 = torch.zeros(n2, device="cuda")
means2 = torch.randn(k2, d2, device="cuda")
r2 = torch.eye(d2, device="cuda").expand(k2, d2, d2).contiguous() * 3
mfac2 = torch.empty(k2, *F.mfac_shape(d2), dtype=torch.bfloat16,
                    device="cuda")
rinv2, const2 = F.constants(r2, means2, False, mfac2)
add2 = const2 + float(np.log(1.0 / k2))
out2 = torch.empty(k2, n2, device="cuda")
for _ in range(3):
    F.estep_logw_big(x2b, mfac2, add2, out2)
for _ in range(3):
    F.mstep_moments(x2, w2, precision="bf16x3", lse=lse2)
torch.cuda.synchronize()
print("pmc exercise done")
