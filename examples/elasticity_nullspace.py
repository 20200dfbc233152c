#!/usr/bin/env python3
"""Elasticity with rigid-body-mode nullspace and block-valued storage
(reference analogue: tutorial/Nullspace, CoupCons3D block values)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import sys

import amgcl_amd as am
from amgcl_amd.generators import elasticity3d, rigid_body_modes


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 16
    backend = sys.argv[2] if len(sys.argv) > 2 else "cpu"
    A, b, coords = elasticity3d(n)
    B = rigid_body_modes(coords)
    prm = {
        "precond": {"class": "amg",
                    "relax": {"type": "chebyshev"},
                    "coarsening": {"type": "smoothed_aggregation",
                                   "nullspace_raw": B, "block_size": 3,
                                   "estimate_spectral_radius": True,
                                   "power_iters": 10}},
        "solver": {"type": "cg", "tol": 1e-8, "maxiter": 500},
    }
    if backend == "hip":
        prm["precond"]["block_value"] = 3  # BSR level storage
        prm["precond"]["keep_host_matrices"] = True
    solve = am.make_solver(A, prm, backend=backend)
    x, iters, resid = solve(b)
    print(f"unknowns: {A.nrows}  iters: {iters}  resid: {resid:.3e}")


if __name__ == "__main__":
    main()
