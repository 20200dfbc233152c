"""BASELINE configs #3 (elasticity/BSR/nullspace) and #5 (Schur field-split)
as first-class GPU tests (these were previously
measured only by scripts/bench_configs.py, outside the -m gpu suite).

Iteration classes asserted against the reference tutorials
(/root/reference/docs/tutorial/Nullspace.rst, Stokes.rst shapes).
"""
import numpy as np
import pytest

import amgcl_amd as am

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def hip():
    import torch

    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from amgcl_amd.backend import make_backend

    return make_backend("hip")


def test_config3_elasticity_bsr_nullspace(hip):
    """Config #3: 3D linear elasticity, CG + SA with rigid-body nullspace,
    BSR(3) block storage on the device (tutorial: Nullspace/Serena class)."""
    from amgcl_amd.generators import elasticity3d, rigid_body_modes

    n = 24
    Ah, bh, coords = elasticity3d(n)
    B = rigid_body_modes(coords)
    bd = hip.from_host(bh)
    Ad = hip.matrix(Ah)  # device input: block/nullspace setup runs on-GPU
    s = am.make_solver(
        Ad,
        {"precond": {"class": "amg", "block_value": 3,
                     "relax": {"type": "chebyshev"},
                     "coarsening": {"type": "smoothed_aggregation",
                                    "nullspace_raw": B, "block_size": 3,
                                    "estimate_spectral_radius": True,
                                    "power_iters": 10}},
         "solver": {"type": "cg", "tol": 1e-6, "maxiter": 300}},
        backend=hip)
    x, iters, resid = s(bd)
    assert resid < 1e-6
    # nullspace-aware SA keeps elasticity in the few-dozen-iteration class
    assert iters < 80, iters
    xh = hip.to_host(x)
    assert np.linalg.norm(bh - Ah @ xh) / np.linalg.norm(bh) < 1e-5


def test_elasticity_device_vs_host_setup(hip):
    """The device block/nullspace coarsening (pointwise aggregation +
    batched-QR tentative + filtered smoothing, all on-GPU) solves the same
    elasticity problem in the same iteration class as the host engine."""
    from amgcl_amd.generators import elasticity3d, rigid_body_modes

    n = 16
    Ah, bh, coords = elasticity3d(n)
    B = rigid_body_modes(coords)
    bd = hip.from_host(bh)
    prm = {"precond": {"class": "amg", "coarse_enough": 600,
                       "relax": {"type": "chebyshev"},
                       "coarsening": {"type": "smoothed_aggregation",
                                      "nullspace_raw": B, "block_size": 3}},
           "solver": {"type": "cg", "tol": 1e-8, "maxiter": 400}}
    import copy

    s_host = am.make_solver(Ah, copy.deepcopy(prm), backend=hip)
    x1, it1, r1 = s_host(bd)
    s_dev = am.make_solver(hip.matrix(Ah), copy.deepcopy(prm), backend=hip)
    x2, it2, r2 = s_dev(bd)
    assert r1 < 1e-8 and r2 < 1e-8
    # different QR bases (MGS vs batched Householder) and aggregation
    # engines: same class, not identical counts
    assert abs(it1 - it2) <= max(5, int(0.3 * it1)), (it1, it2)
    xh = hip.to_host(x2)
    assert np.linalg.norm(bh - Ah @ xh) / np.linalg.norm(bh) < 1e-7


def test_config5_schur_pressure_correction(hip):
    """Config #5: Schur pressure-correction field split on a stabilized
    saddle-point system, FGMRES outer (tutorial: Stokes class)."""
    import sys
    import os

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from test_composite_precond import stokes_like

    K, pmask = stokes_like(16)
    rng = np.random.default_rng(4)
    bh = rng.standard_normal(K.nrows)
    bd = hip.from_host(bh)
    s = am.make_solver(
        K,
        {"precond": {"class": "schur_pressure_correction",
                     "pmask_raw": pmask,
                     "usolver": {"precond": {"class": "relaxation",
                                             "type": "spai0"},
                                 "solver": {"type": "preonly"}},
                     "psolver": {"precond": {"class": "amg",
                                             "coarse_enough": 500},
                                 "solver": {"type": "preonly"}}},
         "solver": {"type": "fgmres", "tol": 1e-6, "maxiter": 200}},
        backend=hip)
    x, iters, resid = s(bd)
    assert resid < 1e-6
    # field-split keeps the saddle system in the few-dozen class (the
    # unpreconditioned/naive AMG would take hundreds)
    assert iters < 60, iters
    xh = hip.to_host(x)
    assert np.linalg.norm(bh - K @ xh) / np.linalg.norm(bh) < 1e-5
