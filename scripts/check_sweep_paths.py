import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from cuda_gmm_mpi_amd.engine import build_engine
from cuda_gmm_mpi_amd.utils.config import GmmConfig
from cuda_gmm_mpi_amd.utils.synthetic import make_blobs

data, _ = make_blobs(200000, 21, 100, seed=11)
for name, ed, mp in (("fp32", "fp32", "fp32"), ("bf16", "bf16", "bf16x3")):
    cfg = GmmConfig(num_clusters=100, target_num_clusters=20,
                    min_iters=20, max_iters=20, estep_dtype=ed,
                    mstep_precision=mp)
    eng = build_engine(data, cfg, device="cuda")
    res = eng.sweep()
    ks = sorted(res.rissanen_by_k)
    print(name, "ks_swept:", len(ks), "min_k:", ks[0], "max_k:", ks[-1],
          "final:", res.num_clusters)
