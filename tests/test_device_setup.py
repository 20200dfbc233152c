"""Device-side setup engine vs the host engine (equivalence tests).

The device kernels implement the exact same deterministic algorithms as the
host engine (same aggregation hash keys), so hierarchies must match
structurally; SpGEMM values match to reduction-order roundoff.
"""
import numpy as np
import pytest

import amgcl_amd as am
from amgcl_amd import _core
from amgcl_amd.matrix import CSR

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def hip():
    import torch

    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from amgcl_amd.backend import make_backend

    return make_backend("hip")


def to_scipy_dev(D):
    import scipy.sparse as sp

    m = sp.csr_matrix(
        (D.val.cpu().numpy(), D.col.cpu().numpy(), D.ptr.cpu().numpy()),
        shape=(D.nrows, D.ncols),
    )
    m.sort_indices()
    return m


def test_poisson_device_matches_host(hip):
    from amgcl_amd.backend.hip_setup import poisson3d_device

    n = 12
    Ad = poisson3d_device(n)
    A, _ = am.poisson3d(n)
    diff = abs(to_scipy_dev(Ad) - A.to_scipy())
    assert diff.nnz == 0 or diff.max() < 1e-14


def test_device_aggregates_match_host_parallel(hip):
    from amgcl_amd.backend import hip_setup
    from amgcl_amd.backend.hip import DeviceCSR

    A, _ = am.poisson3d(16)
    naggr_h, id_h, strong_h = _core.aggregates_parallel(
        A.nrows, A.ptr, A.col, A.val, 0.08
    )
    Ad = DeviceCSR(A, hip.device)
    naggr_d, id_d, strong_d = hip_setup.aggregates(Ad, 0.08)
    assert naggr_d == naggr_h
    np.testing.assert_array_equal(id_d.cpu().numpy(), np.asarray(id_h))
    np.testing.assert_array_equal(strong_d.cpu().numpy(), np.asarray(strong_h))


def test_device_prolongation_and_galerkin_match_host(hip):
    from amgcl_amd.backend import hip_setup
    from amgcl_amd.backend.hip import DeviceCSR
    from amgcl_amd.matrix import galerkin

    A, _ = am.poisson3d(12)
    Ad = DeviceCSR(A, hip.device)
    naggr, ids, strong = hip_setup.aggregates(Ad, 0.08)
    P = hip_setup.smoothed_prolongation(Ad, strong, ids, naggr, 2.0 / 3.0)
    ph, pc, pv = _core.smoothed_prolongation(
        A.nrows, A.ptr, A.col, A.val,
        strong.cpu().numpy(), ids.cpu().numpy().astype(np.int32), naggr, 2.0 / 3.0,
    )
    P_h = CSR(A.nrows, naggr, ph, pc, pv)
    diff = abs(to_scipy_dev(P) - P_h.to_scipy())
    assert diff.nnz == 0 or diff.max() < 1e-13

    R = hip_setup.transpose(P)
    diff = abs(to_scipy_dev(R) - P_h.transpose().to_scipy())
    assert diff.nnz == 0 or diff.max() < 1e-13

    Ac = galerkin(R, Ad, P)
    Ac_h = P_h.transpose() @ (A @ P_h)
    d = to_scipy_dev(Ac) - Ac_h.to_scipy()
    assert abs(d).max() < 1e-11


def test_device_spgemm_random(hip):
    import scipy.sparse as sp

    from amgcl_amd.backend import hip_setup
    from amgcl_amd.backend.hip import DeviceCSR

    rng = np.random.default_rng(3)
    a = sp.random(300, 200, density=0.05, random_state=rng, format="csr")
    a.data = rng.standard_normal(a.nnz)
    b = sp.random(200, 250, density=0.05, random_state=rng, format="csr")
    b.data = rng.standard_normal(b.nnz)
    Ad = DeviceCSR(CSR.from_scipy(a), hip.device)
    Bd = DeviceCSR(CSR.from_scipy(b), hip.device)
    Cd = hip_setup.spgemm(Ad, Bd)
    diff = abs(to_scipy_dev(Cd) - (a @ b).tocsr())
    assert diff.nnz == 0 or diff.max() < 1e-12


def test_full_device_setup_solve(hip):
    """End-to-end: device-generated A, device setup, device solve."""
    import torch

    from amgcl_amd.backend.hip_setup import poisson3d_device

    n = 48
    A = poisson3d_device(n)
    g = torch.Generator(device="cuda").manual_seed(7)
    b = torch.randn(n**3, dtype=torch.float64, device="cuda", generator=g)
    solve = am.make_solver(
        A, {"solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}}, backend=hip
    )
    x, iters, resid = solve(b)
    assert resid < 1e-8
    assert iters <= 30
    # true residual via tested kernels
    r = hip.vector(n**3)
    hip.residual(b, solve.system_matrix(), x, r)
    import math

    rel = math.sqrt(hip.dot(r, r)) / math.sqrt(hip.dot(b, b))
    assert rel < 1e-7


def test_device_setup_iteration_parity_with_host(hip):
    """Device-built and host-built hierarchies give identical iterations."""
    A, b = am.poisson3d(32, rhs="random")
    prm = {"solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}}
    s_host = am.make_solver(A, prm, backend=hip)
    x1, it_host, _ = s_host(b)

    from amgcl_amd.backend.hip import DeviceCSR

    Ad = DeviceCSR(A, hip.device)
    s_dev = am.make_solver(Ad, prm, backend=hip)
    x2, it_dev, _ = s_dev(b)
    # host greedy vs device MIS aggregation may differ slightly
    assert abs(it_dev - it_host) <= 3


def test_mixed_precision_hierarchy(hip):
    """fp32 hierarchy under fp64 CG: converges to fp64-level tolerance
    (parity: reference mixed_precision.cpp / Serena tutorial)."""
    import math

    import torch

    A, b = am.poisson3d(48, rhs="random")
    prm64 = {"solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}}
    s64 = am.make_solver(A, prm64, backend=hip)
    x64, it64, _ = s64(b)
    prm32 = {"precond": {"class": "amg", "precision": "mixed"},
             "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}}
    s32 = am.make_solver(A, prm32, backend=hip)
    assert s32.P._mixed and s32.P.levels[0].A.val.dtype == torch.float32
    x32, it32, res32 = s32(b)
    assert res32 < 1e-8
    r = b - A @ hip.to_host(x32)
    assert np.linalg.norm(r) / np.linalg.norm(b) < 1e-7
    assert it32 <= it64 + 4  # fp32 hierarchy costs at most a few iterations


def test_bsr_matches_csr(hip):
    """Block (BSR) kernels reproduce scalar CSR numerics."""
    import torch

    from amgcl_amd.backend.hip import DeviceBSR, DeviceCSR
    from amgcl_amd.generators import elasticity3d

    A, b, _ = elasticity3d(6)
    Ad = DeviceCSR(A, hip.device)
    Ab = DeviceBSR(A, 3, hip.device)
    x = torch.rand(A.nrows, dtype=torch.float64, device=hip.device)
    y1 = hip.vector(A.nrows)
    y2 = hip.vector(A.nrows)
    hip.spmv(1.3, Ad, x, 0.0, y1)
    hip.spmv(1.3, Ab, x, 0.0, y2)
    assert (y1 - y2).abs().max().item() < 1e-12
    bd = hip.from_host(b)
    hip.residual(bd, Ad, x, y1)
    hip.residual(bd, Ab, x, y2)
    assert (y1 - y2).abs().max().item() < 1e-12


def test_block_value_amg_solve(hip):
    """Elasticity with RBM nullspace + BSR-stored levels (config #3 class)."""
    from amgcl_amd.generators import elasticity3d, rigid_body_modes

    A, b, coords = elasticity3d(8)
    B = rigid_body_modes(coords)
    prm = {"precond": {"class": "amg", "coarse_enough": 500, "block_value": 3,
                       "keep_host_matrices": True,
                       "relax": {"type": "chebyshev"},
                       "coarsening": {"type": "smoothed_aggregation",
                                      "nullspace_raw": B, "block_size": 3,
                                      "estimate_spectral_radius": True,
                                      "power_iters": 10}},
           "solver": {"type": "cg", "tol": 1e-8, "maxiter": 300}}
    s = am.make_solver(A, prm, backend=hip)
    from amgcl_amd.backend.hip import DeviceBSR

    assert isinstance(s.P.levels[0].A, DeviceBSR)
    x, iters, resid = s(b)
    assert resid < 1e-8
    r = b - A @ hip.to_host(x)
    assert np.linalg.norm(r) / np.linalg.norm(b) < 1e-6


def test_multicolor_gauss_seidel_gpu(hip):
    """GPU multicolor GS (exceeds the reference, which is CPU-only for GS):
    must match the convergence class of the serial CPU sweeps."""
    A, b = am.poisson3d(24, rhs="random")
    prm = {"precond": {"class": "amg", "relax": {"type": "gauss_seidel"},
                       "coarse_enough": 500},
           "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}}
    s_gpu = am.make_solver(A, prm, backend=hip)
    x, it_gpu, res = s_gpu(b)
    assert res < 1e-8
    r = b - A @ hip.to_host(x)
    assert np.linalg.norm(r) / np.linalg.norm(b) < 1e-7
    s_cpu = am.make_solver(A, prm)
    _, it_cpu, _ = s_cpu(b)
    assert it_gpu <= it_cpu + 5


def test_rebuild_on_device_hierarchy(hip):
    """amg.rebuild with device-built levels (reuses device transfer ops)."""
    from amgcl_amd.backend.hip import DeviceCSR
    from amgcl_amd.backend.hip_setup import poisson3d_device

    A = poisson3d_device(32)
    import torch

    g = torch.Generator(device="cuda").manual_seed(5)
    b = torch.randn(32**3, dtype=torch.float64, device="cuda", generator=g)
    s = am.make_solver(A, {"solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}},
                       backend=hip)
    x, it0, _ = s(b)
    A2 = DeviceCSR.from_tensors(A.nrows, A.ncols, A.ptr, A.col, A.val * 2.0)
    s.P.rebuild(A2)
    s._native = None  # rebuilt levels -> rebuild the native driver or bypass
    x2 = hip.vector(A.nrows)
    it2, res2 = s.S(s.P, b, x2)
    assert res2 < 1e-8
    r = hip.vector(A.nrows)
    hip.residual(b, A2, x2, r)
    import math

    assert math.sqrt(hip.dot(r, r)) / math.sqrt(hip.dot(b, b)) < 1e-7


def test_dist_amg_on_device(hip):
    """DistAMG hierarchy with device-resident levels (single-rank gloo:
    exchange paths no-op, kernels and replicated tail run on the GPU)."""
    import os

    import torch
    import torch.distributed as dist

    import amgcl_amd as am
    from amgcl_amd.parallel.dist_amg import DistAMG
    from amgcl_amd.parallel.dist_backend import DistBackend

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29551")
    created = False
    if not dist.is_initialized():
        dist.init_process_group("gloo", rank=0, world_size=1)
        created = True
    try:
        A, b = am.poisson3d(24, rhs="random")
        backend = DistBackend(hip)
        Ad = backend.matrix(A)
        for cross in (False, True):
            amg = DistAMG(Ad, {"coarse_enough": 300, "repart_threshold": 800,
                               "coarsening": {"cross_rank": cross}}, backend)
            bd = hip.from_host(b)
            x = hip.vector(A.nrows)
            r = hip.vector(A.nrows)
            xa = hip.vector(A.nrows)
            res = []
            for _ in range(6):
                backend.residual(bd, Ad, xa, r)
                res.append(float(r.norm()))
                amg.apply(r, x)
                xa += x
            assert res[-1] < 0.05 * res[0], (cross, res)
    finally:
        if created:
            dist.destroy_process_group()


def test_device_spgemm_huge_rows(hip):
    """SpGEMM rows whose A row exceeds the LDS staging capacity (BIGROW=256)
    take the serial-A-entry path — must match the host product exactly."""
    import scipy.sparse as sp

    from amgcl_amd import _core
    from amgcl_amd.backend import hip_setup
    from amgcl_amd.matrix import CSR

    rng = np.random.default_rng(41)
    n, m, k = 300, 2000, 350
    A = sp.random(n, m, density=0.02, random_state=rng, format="lil")
    A[0, :800] = rng.standard_normal(800)   # alen 800 >> BIGROW
    A[5, :300] = rng.standard_normal(300)   # alen 300 > BIGROW
    A = A.tocsr()
    A.sort_indices()
    B = sp.random(m, k, density=0.02, random_state=rng, format="csr")
    B.data = rng.standard_normal(B.nnz)
    B.sort_indices()
    ref = (A @ B).tocsr()
    ref.sort_indices()

    Ad = hip.matrix(CSR.from_scipy(A.tocsr()))
    Bd = hip.matrix(CSR.from_scipy(B))
    Cd = hip_setup.spgemm(Ad, Bd, sort=True)
    C = hip_setup.download(Cd).to_scipy()
    assert C.nnz == ref.nnz
    np.testing.assert_allclose(C.toarray(), ref.toarray(), rtol=1e-12, atol=1e-13)


def test_rebuild_on_device_with_sell(hip):
    """amg.rebuild(A') on a device-resident hierarchy (time-dependent
    problems; amgcl/amg.hpp:250-269): transfers are reused, level values
    and SELL images are refreshed, and the rebuilt solver converges on the
    scaled operator."""
    import torch

    from amgcl_amd.backend.hip_setup import poisson3d_device

    A = poisson3d_device(48)
    s = am.make_solver(
        A, {"precond": {"class": "amg", "coarse_enough": 500,
                        "sell_min_rows": 1},   # force SELL on every level
            "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}},
        backend=hip)
    g = torch.Generator(device="cuda").manual_seed(1)
    b = torch.randn(48**3, dtype=torch.float64, device="cuda", generator=g)
    x1, it1, r1 = s(b)
    assert r1 < 1e-8

    # new coefficients, same pattern: A' = 2A
    A2 = poisson3d_device(48)
    A2.val.mul_(2.0)
    s.rebuild(A2)  # also refreshes the native driver's level pointers
    x2, it2, r2 = s(b)
    assert r2 < 1e-8
    assert abs(it1 - it2) <= 2
    # x2 should be ~x1/2
    xa = hip.to_host(x1) / 2.0
    xb = hip.to_host(x2)
    assert np.linalg.norm(xa - xb) / np.linalg.norm(xa) < 1e-6


def test_device_strip_generator_and_split():
    """poisson3d_device_strip + split_strip_torch on the GPU match the host
    generator + C++ split exactly (the single-box validation of the
    device-resident distributed setup; multi-rank semantics are covered by
    test_distributed_gpu on a >1-GPU node)."""
    import torch

    from amgcl_amd import _core
    from amgcl_amd.backend import hip_setup

    n, world, rank = 32, 3, 1
    ntot = n**3
    row_beg = rank * ntot // world
    row_end = (rank + 1) * ntot // world
    dstrip = hip_setup.poisson3d_device_strip(n, row_beg, row_end)
    # host reference strip over the same rows
    hstrip = hip_setup.download(dstrip)
    A_host, _, rb, re_ = am.poisson3d_strip(n, rank, world, rhs=None)
    # partitions may differ; compare on the intersection via to_scipy
    import scipy.sparse as sp

    full, _ = am.poisson3d(n)
    ref = full.to_scipy()[row_beg:row_end]
    assert abs(hstrip.to_scipy() - ref).max() == 0.0

    ref_split = _core.split_strip(row_end - row_beg, row_beg, row_end,
                                  hstrip.ptr, hstrip.col, hstrip.val)
    got = hip_setup.split_strip_torch(dstrip.ptr, dstrip.col, dstrip.val,
                                      row_beg, row_end)
    for a, b in zip(ref_split, got):
        np.testing.assert_array_equal(np.asarray(a), b.cpu().numpy())

    # weak-mode box extent
    dbox = hip_setup.poisson3d_device_strip(n, 0, n * n * 2 * n, nz=2 * n)
    from amgcl_amd.generators import poisson3d_box_strip

    box, _, _, _ = poisson3d_box_strip(n, n, 2 * n, 0, 1, rhs=None)
    assert abs(hip_setup.download(dbox).to_scipy() - box.to_scipy()).max() == 0.0
