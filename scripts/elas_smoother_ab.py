"""Elasticity (64^3 nodes, RBM nullspace, BSR(3) levels): chebyshev vs the
block-valued ILU(0) smoother, measured on one MI355X."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import numpy as np
import torch

import amgcl_amd as am
from amgcl_amd.generators import elasticity3d, rigid_body_modes


def main():
    from amgcl_amd.backend import make_backend

    hip = make_backend("hip")
    Ah, bh, coords = elasticity3d(64)
    B = rigid_body_modes(coords)
    A = hip.matrix(Ah)  # device input -> device block/nullspace setup
    b = hip.from_host(bh)
    # exactly BENCH_configs_r02's config #3, with the smoother swapped
    for relax in ({"type": "chebyshev"},
                  {"type": "block_ilu0", "block_size": 3}):
        prm = {"precond": {"class": "amg", "block_value": 3,
                           "coarsening": {"type": "smoothed_aggregation",
                                          "block_size": 3, "nullspace_raw": B,
                                          "estimate_spectral_radius": True,
                                          "power_iters": 10},
                           "relax": relax},
               "solver": {"type": "cg", "tol": 1e-6, "maxiter": 300}}
        t0 = time.perf_counter()
        s = am.make_solver(A, prm, backend=hip)
        torch.cuda.synchronize()
        t1 = time.perf_counter()
        x, it, r = s(b)
        torch.cuda.synchronize()
        t2 = time.perf_counter()
        name = relax["type"]
        print(f"{name:12s} setup {t1-t0:.3f} solve {t2-t1:.3f} "
              f"iters {it} resid {r:.1e}")


if __name__ == "__main__":
    main()
