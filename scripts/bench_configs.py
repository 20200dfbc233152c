#!/usr/bin/env python3
"""Measured evidence for the non-flagship BASELINE.json configurations on one
MI355X (the flagship 512^3 CG config is bench.py):

  #2  Poisson 256^3 fp64, BiCGStab + SA/SPAI0
  #3  3D linear elasticity, CG + SA + rigid-body nullspace + BSR blocks
  #5  Schur pressure correction (stabilized saddle-point system), FGMRES

Prints one JSON line per config.
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

import amgcl_amd as am
from amgcl_amd.backend import make_backend


def run(name, A, b, prm, hip):
    import copy
    import math

    am.make_solver(A, copy.deepcopy(prm), backend=hip)  # warmup (cold allocs)
    hip.synchronize()
    t0 = time.perf_counter()
    solve = am.make_solver(A, copy.deepcopy(prm), backend=hip)
    hip.synchronize()
    t1 = time.perf_counter()
    solve(b)  # warm
    hip.synchronize()
    ts = time.perf_counter()
    x, it, res = solve(b)
    hip.synchronize()
    te = time.perf_counter()
    r = hip.vector(A.nrows)
    hip.residual(b if not isinstance(b, np.ndarray) else hip.from_host(b),
                 solve.system_matrix(), x, r)
    true_rel = math.sqrt(hip.dot(r, r)) / math.sqrt(hip.dot(b, b))
    print(json.dumps({"config": name, "unknowns": A.nrows, "setup_s": round(t1 - t0, 4),
                      "solve_s": round(te - ts, 4), "iters": it, "resid": res,
                      "true_rel_resid": true_rel}), flush=True)


def main():
    hip = make_backend("hip")

    # config #2: 256^3 BiCGStab
    from amgcl_amd.backend.hip_setup import poisson3d_device

    A = poisson3d_device(256)
    g = torch.Generator(device="cuda").manual_seed(3)
    b = torch.randn(256**3, dtype=torch.float64, device="cuda", generator=g)
    run("poisson 256^3 BiCGStab+SA/SPAI0 fp64",
        A, b, {"solver": {"type": "bicgstab", "tol": 1e-6, "maxiter": 200}}, hip)
    del A, b

    # config #3: elasticity + RBM + BSR
    from amgcl_amd.generators import elasticity3d, rigid_body_modes

    n = int(os.environ.get("ELAS_N", "48"))
    Ah, bh, coords = elasticity3d(n)
    B = rigid_body_modes(coords)
    bd = hip.from_host(bh)
    Ad = hip.matrix(Ah)  # device input -> device block/nullspace setup
    run(f"elasticity {n}^3 nodes CG+SA(esr)+RBM+Chebyshev+BSR(3)", Ad, bd,
        {"precond": {"class": "amg", "block_value": 3,
                     "relax": {"type": "chebyshev"},
                     "coarsening": {"type": "smoothed_aggregation",
                                    "nullspace_raw": B, "block_size": 3,
                                    "estimate_spectral_radius": True,
                                    "power_iters": 10}},
         "solver": {"type": "cg", "tol": 1e-6, "maxiter": 500}}, hip)

    # config #5-class: Schur pressure correction on a stabilized saddle system
    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "tests"))
    from test_composite_precond import stokes_like

    K, pmask = stokes_like(int(os.environ.get("STOKES_N", "48")))
    rng = np.random.default_rng(0)
    bk = hip.from_host(rng.standard_normal(K.nrows))
    run("schur pressure correction FGMRES", K, bk,
        {"precond": {"class": "schur_pressure_correction", "pmask_raw": pmask,
                     "usolver": {"precond": {"class": "relaxation", "type": "spai0"},
                                 "solver": {"type": "cg", "tol": 1e-2, "maxiter": 8}},
                     "psolver": {"precond": {"class": "amg"},
                                 "solver": {"type": "cg", "tol": 1e-2, "maxiter": 8}}},
         "solver": {"type": "fgmres", "tol": 1e-6, "maxiter": 200}}, hip)


if __name__ == "__main__":
    main()
