"""Schur pressure correction and CPR composite preconditioners."""
import numpy as np
import pytest
import scipy.sparse as sp

import amgcl_amd as am
from amgcl_amd.matrix import CSR


def stokes_like(n=10):
    """Stabilized saddle-point-ish system: velocity Poisson coupled with a
    stabilized pressure block through a local discrete-gradient-like operator
    (local coupling keeps the Schur complement sparse, like real Stokes)."""
    Ap, _ = am.poisson3d(n)
    a = Ap.to_scipy()
    nv = a.shape[0]
    # discrete gradient-ish stencil: B = 0.1 (I - shift_x)
    B = 0.1 * (sp.identity(nv) - sp.diags(np.ones(nv - 1), 1)).tocsr()
    K = sp.bmat([[a, B], [B.T, a + sp.identity(nv)]], format="csr")
    K.sort_indices()
    pmask = np.zeros(2 * nv, dtype=bool)
    pmask[nv:] = True
    return CSR.from_scipy(K), pmask


def test_schur_pressure_correction():
    K, pmask = stokes_like(8)
    rng = np.random.default_rng(0)
    b = rng.standard_normal(K.nrows)
    s = am.make_solver(
        K,
        {
            "precond": {
                "class": "schur_pressure_correction",
                "pmask_raw": pmask,
                "usolver": {"precond": {"class": "relaxation", "type": "spai0"},
                            "solver": {"type": "cg", "tol": 1e-2, "maxiter": 8}},
                "psolver": {"precond": {"class": "amg", "coarse_enough": 500},
                            "solver": {"type": "cg", "tol": 1e-2, "maxiter": 8}},
            },
            "solver": {"type": "fgmres", "tol": 1e-8, "maxiter": 100},
        },
    )
    x, iters, resid = s(b)
    assert resid < 1e-7
    assert np.linalg.norm(b - K @ x) / np.linalg.norm(b) < 1e-6
    assert iters < 60


def block_reservoir(n=8, b=2):
    """Interleaved block system: pressure + saturation per cell, pressure
    block elliptic, saturation hyperbolic-ish (diagonal dominant)."""
    rng = np.random.default_rng(7)
    Ap, _ = am.poisson3d(n)
    a = Ap.to_scipy()
    nc = a.shape[0]
    blocks = []
    eye = sp.identity(nc)
    off = sp.random(nc, nc, density=0.001, random_state=rng, format="csr")
    off.data = 0.05 * rng.standard_normal(off.nnz)
    K = sp.bmat([[a, off], [off.T, 8.0 * eye + 0.5 * a]], format="csr")
    # interleave: unknown = cell*b + comp
    perm = np.arange(2 * nc).reshape(2, nc).T.ravel()
    K = K[perm][:, perm].tocsr()
    K.sort_indices()
    return CSR.from_scipy(K)


@pytest.mark.parametrize("kind", ["cpr", "cpr_drs"])
def test_cpr(kind):
    K = block_reservoir(8)
    rng = np.random.default_rng(1)
    b = rng.standard_normal(K.nrows)
    s = am.make_solver(
        K,
        {
            "precond": {"class": kind, "block_size": 2,
                        "pprecond": {"class": "amg", "coarse_enough": 400}},
            "solver": {"type": "fgmres", "tol": 1e-8, "maxiter": 200},
        },
    )
    x, iters, resid = s(b)
    assert resid < 1e-7
    assert np.linalg.norm(b - K @ x) / np.linalg.norm(b) < 1e-6


def test_shared_memory_deflation():
    A, b = am.poisson3d(14, rhs="random")
    rng = np.random.default_rng(2)
    Z = np.stack([np.ones(A.nrows), rng.random(A.nrows)], axis=1)
    s = am.make_solver(
        A,
        {"precond": {"class": "deflation", "Z_raw": Z,
                     "inner": {"class": "amg", "coarse_enough": 400}},
         "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}},
    )
    x, it, res = s(b)
    assert res < 1e-7
    assert np.linalg.norm(b - A @ x) / np.linalg.norm(b) < 1e-6


def test_cpr_quasi_impes_weights():
    """Quasi-IMPES decoupling (amgcl/preconditioner/cpr.hpp): the per-cell
    weights are the first row of the inverted diagonal block, so w^T D = e_p^T
    and the non-pressure couplings cancel within each cell."""
    from amgcl_amd.precond.cpr import CPR

    K = block_reservoir(6)
    m = K.to_scipy().tocsr()
    b = 2
    nc = K.nrows // b
    w = CPR._quasi_impes_weights(m, nc, b).reshape(nc, b)
    for i in [0, 3, nc - 1]:
        D = m[i * b : (i + 1) * b, i * b : (i + 1) * b].toarray()
        np.testing.assert_allclose(w[i] @ D, np.eye(b)[0], atol=1e-10)
