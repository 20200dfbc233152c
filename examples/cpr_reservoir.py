"""CPR (constrained pressure residual) on a reservoir-style block system.

Reference analogue: examples/cpr.cpp / cpr_drs.cpp. Unknowns are interleaved
per cell (pressure first); the quasi-IMPES pressure matrix is handled by AMG
and a SPAI0 smoother sweeps the full system.
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np
import scipy.sparse as sp

import amgcl_amd as am
from amgcl_amd.matrix import CSR


def main(n=16, drs=False):
    Ap, _ = am.poisson3d(n)
    a = Ap.to_scipy()
    nv = a.shape[0]
    C = sp.csr_matrix(np.array([[1.0, 0.2], [0.3, 1.0]]))
    K = (sp.kron(a, C) + sp.kron(sp.identity(nv), 0.5 * sp.identity(2))).tocsr()
    K.sort_indices()
    b = np.random.default_rng(7).standard_normal(K.shape[0])
    solve = am.make_solver(
        CSR.from_scipy(K),
        {"precond": {"class": "cpr", "block_size": 2, "drs": drs},
         "solver": {"type": "bicgstab", "tol": 1e-8, "maxiter": 200}})
    x, iters, resid = solve(b)
    rel = np.linalg.norm(b - K @ x) / np.linalg.norm(b)
    print(f"cpr{'_drs' if drs else ''}: {iters} iters, true rel resid {rel:.2e}")


if __name__ == "__main__":
    main(drs="--drs" in sys.argv)
