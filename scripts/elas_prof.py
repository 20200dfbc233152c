import os, sys, time
sys.path.insert(0, "/root/repo")
import torch
import amgcl_amd as am
from amgcl_amd.backend import make_backend
from amgcl_amd.generators import elasticity3d, rigid_body_modes
from amgcl_amd.profiler import prof

hip = make_backend("hip")
n = 64
t0 = time.perf_counter()
Ah, bh, coords = elasticity3d(n)
B = rigid_body_modes(coords)
print("fixture:", time.perf_counter()-t0)
prm = {"precond": {"class": "amg", "block_value": 3,
                   "relax": {"type": "chebyshev"},
                   "coarsening": {"type": "smoothed_aggregation",
                                  "nullspace_raw": B, "block_size": 3,
                                  "estimate_spectral_radius": True,
                                  "power_iters": 10}},
       "solver": {"type": "cg", "tol": 1e-6, "maxiter": 500}}
import copy
for it in range(2):
    Ad = hip.matrix(Ah)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    s = am.make_solver(Ad, copy.deepcopy(prm), backend=hip)
    torch.cuda.synchronize()
    print(f"setup[{it}]: {time.perf_counter()-t0:.3f}")
print(prof.report())
