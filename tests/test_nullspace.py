"""Near-nullspace (rigid body modes) + pointwise aggregation for elasticity
(parity: amgcl tutorial Nullspace.rst, coarsening/rigid_body_modes.hpp,
pointwise_aggregates.hpp)."""
import numpy as np

import amgcl_amd as am
from amgcl_amd.generators import elasticity3d, rigid_body_modes


def test_rbm_improves_elasticity_convergence():
    A, b, coords = elasticity3d(8)
    B = rigid_body_modes(coords)
    assert B.shape == (A.nrows, 6)
    prm0 = {"precond": {"class": "amg", "coarse_enough": 500},
            "solver": {"type": "cg", "tol": 1e-8, "maxiter": 1000}}
    x0, it0, r0 = am.make_solver(A, prm0)(b)
    prm1 = {"precond": {"class": "amg", "coarse_enough": 500,
                        "relax": {"type": "chebyshev"},
                        "coarsening": {"type": "smoothed_aggregation",
                                       "nullspace_raw": B, "block_size": 3,
                                       "estimate_spectral_radius": True,
                                       "power_iters": 10}},
            "solver": {"type": "cg", "tol": 1e-8, "maxiter": 1000}}
    x1, it1, r1 = am.make_solver(A, prm1)(b)
    assert r1 < 1e-7
    assert np.linalg.norm(b - A @ x1) / np.linalg.norm(b) < 1e-6
    assert it1 < 0.6 * it0  # RBM must substantially cut iterations


def test_rbm_2d_shape():
    coords = np.random.default_rng(0).random((50, 2))
    B = rigid_body_modes(coords)
    assert B.shape == (100, 3)
    # rotation column: (-y, x)
    assert np.allclose(B[0::2, 2], -coords[:, 1])
    assert np.allclose(B[1::2, 2], coords[:, 0])


def test_pointwise_matrix_structure():
    from amgcl_amd import _core

    A, b, coords = elasticity3d(6)
    pp, pc, pv = _core.pointwise_matrix(A.nrows, A.ptr, A.col, A.val, 3)
    npts = A.nrows // 3
    assert len(pp) == npts + 1
    assert np.all(np.asarray(pv) >= 0)
