#!/usr/bin/env python3
"""Minimal flagship example: AMG-preconditioned CG on 3D Poisson
(reference analogue: examples/solver.cpp, tutorial/poisson3Db)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import sys

import amgcl_amd as am


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 64
    backend = sys.argv[2] if len(sys.argv) > 2 else "cpu"
    A, b = am.poisson3d(n, rhs="random")
    solve = am.make_solver(
        A,
        {"precond": {"class": "amg",
                     "coarsening": {"type": "smoothed_aggregation"},
                     "relax": {"type": "spai0"}},
         "solver": {"type": "cg", "tol": 1e-8}},
        backend=backend,
    )
    print(solve)
    x, iters, resid = solve(b)
    print(f"iters: {iters}  resid: {resid:.3e}")


if __name__ == "__main__":
    main()
