import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from cuda_gmm_mpi_amd.engine import build_engine
from cuda_gmm_mpi_amd.utils.config import GmmConfig
from cuda_gmm_mpi_amd.utils.synthetic import make_blobs

data, _ = make_blobs(20000, 6, 4, seed=31)
cfg = GmmConfig(num_clusters=4, target_num_clusters=4, min_iters=15, max_iters=15)

def run(graphs):
    if not graphs:
        import os; os.environ["GMM_NO_GRAPHS"]="1"
    else:
        import os; os.environ.pop("GMM_NO_GRAPHS", None)
    eng = build_engine(data, cfg, device="cuda")
    lik = eng._reduce_likelihood(eng._estep(4))
    liks=[lik]
    states=[]
    for i in range(6):
        eng.em_iteration(4)
        liks.append(float(eng._lik_dev.item()))
        states.append((eng.state.N.cpu().numpy().copy(),
                       eng.state.means.cpu().numpy().copy()))
    return liks, states

lik_e, st_e = run(False)
lik_g, st_g = run(True)
for i,(a,b) in enumerate(zip(lik_e, lik_g)):
    print(i, a, b, "DIFF" if a!=b else "")
for i in range(6):
    dn = np.abs(st_e[i][0]-st_g[i][0]).max()
    dm = np.abs(st_e[i][1]-st_g[i][1]).max()
    print("iter", i, "dN", dn, "dmeans", dm)
