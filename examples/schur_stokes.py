#!/usr/bin/env python3
"""Schur pressure-correction field split on a stabilized saddle-point system
(reference analogue: tutorial/Stokes, examples/schur_pressure_correction)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import sys

import numpy as np
import scipy.sparse as sp

import amgcl_amd as am


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 24
    backend = sys.argv[2] if len(sys.argv) > 2 else "cpu"
    Ap, _ = am.poisson3d(n)
    a = Ap.to_scipy()
    nv = a.shape[0]
    B = 0.1 * (sp.identity(nv) - sp.diags(np.ones(nv - 1), 1)).tocsr()
    K = sp.bmat([[a, B], [B.T, a + sp.identity(nv)]], format="csr")
    pmask = np.zeros(2 * nv, dtype=bool)
    pmask[nv:] = True
    b = np.random.default_rng(0).standard_normal(2 * nv)
    solve = am.make_solver(
        am.CSR.from_scipy(K.tocsr()),
        {"precond": {"class": "schur_pressure_correction", "pmask_raw": pmask,
                     "usolver": {"precond": {"class": "amg"},
                                 "solver": {"type": "preonly"}},
                     "psolver": {"precond": {"class": "amg"},
                                 "solver": {"type": "preonly"}}},
         "solver": {"type": "fgmres", "tol": 1e-8, "maxiter": 200}},
        backend=backend,
    )
    x, iters, resid = solve(b)
    print(f"unknowns: {2 * nv}  outer iters: {iters}  resid: {resid:.3e}")


if __name__ == "__main__":
    main()
