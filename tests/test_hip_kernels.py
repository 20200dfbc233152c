"""Numerics tests for the hand-written gfx950 kernels vs the CPU fp64
reference (_core / numpy). All tests require a GPU (@pytest.mark.gpu)."""
import numpy as np
import pytest
import scipy.sparse as sp

import amgcl_amd as am
from amgcl_amd.matrix import CSR

pytestmark = pytest.mark.gpu


def rand_csr(rng, n, m, density=0.05):
    a = sp.random(n, m, density=density, random_state=rng, format="csr")
    a.data = rng.standard_normal(a.nnz)
    # ensure nonempty rows for stability of subw variants
    return CSR(n, m, a.indptr, a.indices, a.data)


@pytest.fixture(scope="module")
def hip():
    import torch

    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from amgcl_amd.backend import make_backend

    return make_backend("hip")


@pytest.mark.parametrize("subw", [1, 2, 4, 8, 16, 32, 64])
def test_spmv_variants(hip, subw):
    rng = np.random.default_rng(subw)
    A = rand_csr(rng, 500, 400, 0.05)
    x = rng.standard_normal(400)
    y0 = rng.standard_normal(500)
    Ad = hip.matrix(A)
    Ad.subw = subw
    xd, yd = hip.from_host(x), hip.from_host(y0)
    hip.spmv(1.7, Ad, xd, 0.3, yd)
    ref = y0.copy()
    A.spmv(1.7, x, 0.3, ref)
    np.testing.assert_allclose(hip.to_host(yd), ref, rtol=1e-12, atol=1e-12)
    # beta = 0 path
    hip.spmv(2.0, Ad, xd, 0.0, yd)
    ref2 = np.zeros(500)
    A.spmv(2.0, x, 0.0, ref2)
    np.testing.assert_allclose(hip.to_host(yd), ref2, rtol=1e-12, atol=1e-12)


def test_residual_and_relax(hip):
    rng = np.random.default_rng(7)
    A, _ = am.poisson3d(12)
    x = rng.standard_normal(A.nrows)
    b = rng.standard_normal(A.nrows)
    m = rng.random(A.nrows) + 0.5
    Ad = hip.matrix(A)
    xd, bd = hip.from_host(x), hip.from_host(b)
    rd = hip.vector(A.nrows)
    hip.residual(bd, Ad, xd, rd)
    ref = np.empty(A.nrows)
    A.residual(b, x, ref)
    np.testing.assert_allclose(hip.to_host(rd), ref, rtol=1e-12, atol=1e-12)

    md = hip.from_host(m)
    td = hip.vector(A.nrows)
    x2 = hip.from_host(x)
    hip.relax_diag(Ad, md, bd, x2, td)
    np.testing.assert_allclose(hip.to_host(td), m * ref, rtol=1e-12, atol=1e-12)
    np.testing.assert_allclose(hip.to_host(x2), x + m * ref, rtol=1e-12, atol=1e-12)


def test_vector_ops(hip):
    rng = np.random.default_rng(11)
    n = 10001
    x, y, z, m = (rng.standard_normal(n) for _ in range(4))
    xd, yd, zd, md = (hip.from_host(v) for v in (x, y, z, m))

    hip.axpby(1.5, xd, -0.5, yd)
    np.testing.assert_allclose(hip.to_host(yd), 1.5 * x - 0.5 * y, rtol=1e-13)
    hip.axpbypcz(0.3, xd, 0.7, yd, -1.1, zd)
    np.testing.assert_allclose(
        hip.to_host(zd), 0.3 * x + 0.7 * (1.5 * x - 0.5 * y) - 1.1 * z, rtol=1e-12
    )
    hip.vmul(2.0, md, xd, 0.0, zd)
    np.testing.assert_allclose(hip.to_host(zd), 2.0 * m * x, rtol=1e-13)
    hip.clear(zd)
    assert np.all(hip.to_host(zd) == 0.0)
    hip.copy(xd, zd)
    np.testing.assert_allclose(hip.to_host(zd), x)


def test_dot_and_dot2(hip):
    rng = np.random.default_rng(13)
    n = 1 << 20
    x, y = rng.standard_normal(n), rng.standard_normal(n)
    xd, yd = hip.from_host(x), hip.from_host(y)
    assert abs(hip.dot(xd, yd) - np.dot(x, y)) < 1e-7 * n**0.5
    d1, d2 = hip.dot2(xd, yd, xd, xd)
    assert abs(d1 - np.dot(x, y)) < 1e-7 * n**0.5
    assert abs(d2 - np.dot(x, x)) < 1e-7 * n**0.5


def test_gather_scatter(hip):
    import torch

    rng = np.random.default_rng(17)
    x = rng.standard_normal(1000)
    idx = rng.choice(1000, size=300, replace=False).astype(np.int32)
    xd = hip.from_host(x)
    idxd = torch.from_numpy(idx).to(hip.device)
    buf = hip.vector(300)
    hip.gather(xd, idxd, buf)
    np.testing.assert_allclose(hip.to_host(buf), x[idx])
    y = hip.vector(1000)
    hip.scatter(buf, idxd, y)
    ref = np.zeros(1000)
    ref[idx] = x[idx]
    np.testing.assert_allclose(hip.to_host(y), ref)


def test_coarse_gemv_solver(hip):
    rng = np.random.default_rng(19)
    n = 257
    a = rng.standard_normal((n, n)) + n * np.eye(n)
    A = CSR.from_dense(a)
    solver = hip.coarse_solver(A)
    f = rng.standard_normal(n)
    fd, ud = hip.from_host(f), hip.vector(n)
    solver(fd, ud)
    np.testing.assert_allclose(hip.to_host(ud), np.linalg.solve(a, f), rtol=1e-9, atol=1e-9)


def test_hip_solve_end_to_end(hip):
    """Full AMG-preconditioned solve on GPU matches the CPU backend's
    iteration count and converges to the true residual."""
    A, b = am.poisson3d(48, rhs="random")
    prm = {"solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}}
    s_cpu = am.make_solver(A, prm)
    x_cpu, it_cpu, res_cpu = s_cpu(b)
    s_hip = am.make_solver(A, prm, backend=hip)
    x_hip, it_hip, res_hip = s_hip(b)
    assert res_hip < 1e-8
    assert abs(it_hip - it_cpu) <= 2
    r = b - A @ hip.to_host(x_hip)
    assert np.linalg.norm(r) / np.linalg.norm(b) < 1e-7


def test_hip_bicgstab_solve(hip):
    A, b = am.poisson3d(32, rhs="random")
    s = am.make_solver(
        A, {"solver": {"type": "bicgstab", "tol": 1e-8, "maxiter": 100}}, backend=hip
    )
    x, iters, resid = s(b)
    assert resid < 1e-8
    r = b - A @ hip.to_host(x)
    assert np.linalg.norm(r) / np.linalg.norm(b) < 1e-7


def test_native_library_is_loaded(hip):
    """The .so with hand-written kernels must actually be mapped in-process."""
    maps = open("/proc/self/maps").read()
    assert "libamghip.so" in maps


def test_f32_kernels_match_f64(hip):
    """fp32 kernel variants (mixed-precision hierarchy) vs fp64 reference."""
    import torch

    from amgcl_amd.backend.hip import DeviceCSR

    rng = np.random.default_rng(23)
    A, _ = am.poisson3d(10)
    Ad = DeviceCSR(A, hip.device)
    A32 = DeviceCSR.from_tensors(Ad.nrows, Ad.ncols, Ad.ptr, Ad.col,
                                 Ad.val.to(torch.float32))
    x = rng.standard_normal(A.nrows)
    x64 = hip.from_host(x)
    x32 = x64.to(torch.float32)
    y64 = hip.vector(A.nrows)
    y32 = hip.vector(A.nrows, torch.float32)
    hip.spmv(1.0, Ad, x64, 0.0, y64)
    hip.spmv(1.0, A32, x32, 0.0, y32)
    assert (y32.double() - y64).abs().max().item() < 1e-4
    b64 = hip.from_host(rng.standard_normal(A.nrows))
    b32 = b64.to(torch.float32)
    r32 = hip.vector(A.nrows, torch.float32)
    hip.residual(b32, A32, x32, r32)
    r64 = hip.vector(A.nrows)
    hip.residual(b64, Ad, x64, r64)
    assert (r32.double() - r64).abs().max().item() < 1e-4
    # axpby/vmul/dot f32
    hip.axpby(1.5, x32, -0.5, r32)
    assert torch.isfinite(r32).all()
    d = hip.dot(x32, x32)
    assert abs(d - float(x64.dot(x64))) < 1e-3
    # casts
    z64 = hip.vector(A.nrows)
    hip.cast(x32, z64)
    assert (z64 - x32.double()).abs().max().item() == 0.0


def test_blkdiag_vmul_and_as_block(hip):
    """Block-diagonal matvec kernel vs numpy einsum, and the as_block
    smoother end-to-end on the HIP backend."""
    import torch

    rng = np.random.default_rng(29)
    for B in (2, 3, 4):
        nb = 1000
        M = rng.standard_normal((nb, B, B))
        x = rng.standard_normal(nb * B)
        yd = hip.vector(nb * B)
        hip.blkdiag_vmul(B, hip.from_host(M.ravel()), hip.from_host(x), yd)
        ref = np.einsum("bij,bj->bi", M, x.reshape(nb, B)).ravel()
        np.testing.assert_allclose(hip.to_host(yd), ref, rtol=1e-12, atol=1e-12)

    import scipy.sparse as sp

    A0, _ = am.poisson3d(12)
    C = np.full((2, 2), 0.15) + 0.85 * np.eye(2)
    Ab = sp.kron(A0.to_scipy(), sp.csr_matrix(C), format="csr")
    A = CSR.from_scipy(Ab)
    b = np.random.default_rng(3).standard_normal(A.nrows)
    prm = {"precond": {"class": "amg", "coarse_enough": 400,
                       "relax": {"type": "as_block", "block_size": 2,
                                 "base": {"type": "spai0"}}},
           "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}}
    s_cpu = am.make_solver(A, prm)
    _, it_cpu, _ = s_cpu(b)
    s_hip = am.make_solver(A, prm, backend=hip)
    x, it_hip, resid = s_hip(b)
    assert resid < 1e-8
    assert abs(it_hip - it_cpu) <= 2
    assert np.linalg.norm(b - A @ hip.to_host(x)) / np.linalg.norm(b) < 1e-7


def test_device_dense_coarse_inverse(hip):
    """GPU-resident densify + torch.linalg.inv path (no host round-trip)."""
    from amgcl_amd.backend.hip import DeviceCSR, DeviceDenseSolver

    rng = np.random.default_rng(31)
    n = 173
    a = rng.standard_normal((n, n)) + n * np.eye(n)
    A = CSR.from_dense(a)
    solver = DeviceDenseSolver.from_device(DeviceCSR(A, hip.device), hip)
    f = rng.standard_normal(n)
    fd, ud = hip.from_host(f), hip.vector(n)
    solver(fd, ud)
    np.testing.assert_allclose(hip.to_host(ud), np.linalg.solve(a, f),
                               rtol=1e-9, atol=1e-9)


def test_sell_kernels_match_csr(hip):
    """SELL-64 spmv/residual/relax match the CSR kernels and the CPU
    reference bitwise-tolerance (1e-12), incl. ragged rows + padding."""
    rng = np.random.default_rng(11)
    # ragged: mix dense-ish and nearly-empty rows across slice boundaries
    A = rand_csr(rng, 1000, 1000, 0.03)
    x = rng.standard_normal(1000)
    b = rng.standard_normal(1000)
    m = rng.random(1000) + 0.5
    Ad = hip.matrix(A)
    Ad.build_sell()
    assert Ad.nslice == (1000 + 63) // 64
    xd, bd, md = hip.from_host(x), hip.from_host(b), hip.from_host(m)
    yd = hip.vector(1000)
    hip.spmv(1.3, Ad, xd, 0.0, yd)
    ref = np.zeros(1000)
    A.spmv(1.3, x, 0.0, ref)
    np.testing.assert_allclose(hip.to_host(yd), ref, rtol=1e-12, atol=1e-12)
    # beta != 0
    y0 = rng.standard_normal(1000)
    yd = hip.from_host(y0)
    hip.spmv(1.3, Ad, xd, 0.4, yd)
    ref = y0.copy()
    A.spmv(1.3, x, 0.4, ref)
    np.testing.assert_allclose(hip.to_host(yd), ref, rtol=1e-12, atol=1e-12)
    # residual
    rd = hip.vector(1000)
    hip.residual(bd, Ad, xd, rd)
    refr = b - A @ x
    np.testing.assert_allclose(hip.to_host(rd), refr, rtol=1e-12, atol=1e-12)
    # fused diagonal relax: x += M(b - Ax)
    x2 = hip.from_host(x)
    td = hip.vector(1000)
    hip.relax_diag(Ad, md, bd, x2, td)
    refx = x + m * (b - A @ x)
    np.testing.assert_allclose(hip.to_host(x2), refx, rtol=1e-12, atol=1e-12)


def test_sell_in_amg_solve(hip):
    """Full AMG solve with forced SELL conversion on every eligible level
    gives the same iteration count/solution class as the CSR path."""
    A, b = am.poisson3d(24, rhs="random")
    s_csr = am.make_solver(
        A, {"precond": {"class": "amg", "sell": "off", "coarse_enough": 300},
            "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}}, backend=hip)
    s_sell = am.make_solver(
        A, {"precond": {"class": "amg", "sell": "auto", "sell_min_rows": 1,
                        "sell_min_mean": 0.0, "coarse_enough": 300},
            "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}}, backend=hip)
    x1, it1, r1 = s_csr(b)
    x2, it2, r2 = s_sell(b)
    # summation order differs (lane-serial vs shfl-tree), so allow 1 iter
    assert abs(it1 - it2) <= 1
    assert r2 < 1e-8
    xa = hip.to_host(x2) if not isinstance(x2, np.ndarray) else x2
    assert np.linalg.norm(b - A @ xa) / np.linalg.norm(b) < 1e-7


def test_ilu0_exact_gpu_sptrsv(hip):
    """Exact level-scheduled GPU triangular solve (cooperative kernel)
    reproduces the serial host ILU0 application to 1e-12 — the reference's
    vendor-SpSV exact path (amgcl/relaxation/rocsparse_ilu0.hpp:225-300)
    rebuilt without a vendor analysis object."""
    from amgcl_amd.relaxation.ilu0 import ILU0
    from amgcl_amd.backend import make_backend

    A, _ = am.poisson3d(16)
    rng = np.random.default_rng(5)
    z = rng.standard_normal(A.nrows)

    cpu = make_backend("cpu")
    r_cpu = ILU0(A, {"solve_serial": True}, cpu)
    x_cpu = z.copy()
    r_cpu._solve_serial(x_cpu)

    r_gpu = ILU0(A, {"solve": "exact"}, hip)
    zd = hip.from_host(z)
    r_gpu._solve_exact(zd)
    np.testing.assert_allclose(hip.to_host(zd), x_cpu, rtol=1e-12, atol=1e-12)


def test_ilu0_exact_vs_jacobi_iterations(hip):
    """On an anisotropic problem the exact triangular solve must be at
    least as strong per iteration as the approximate Jacobi variant."""
    A, b = am.poisson3d(24, rhs="random", anisotropy=100.0)

    def solve(solve_kind):
        s = am.make_solver(
            A, {"precond": {"class": "relaxation",
                            "type": "ilu0", "solve": solve_kind},
                "solver": {"type": "bicgstab", "tol": 1e-8, "maxiter": 500}},
            backend=hip)
        x, iters, resid = s(b)
        assert resid < 1e-8
        return iters

    it_exact = solve("exact")
    it_jac = solve("jacobi")
    assert it_exact <= it_jac + 1, (it_exact, it_jac)


def test_bsr_mfma_spmv_matches(hip):
    """The v_mfma_f64_16x16x4_f64 BSR SpMV variant (B=4) matches the
    unrolled block kernel and the scalar reference to 1e-12 — the
    correctness half of the VERDICT-mandated MFMA head-to-head (the
    performance half is scripts/bench_blocks.py -> profiles/)."""
    import torch

    from amgcl_amd.backend._hiplib import check, lib
    from amgcl_amd.backend.hip import DeviceBSR, _stream

    rng = np.random.default_rng(9)
    nb = 301  # not a multiple of 4: exercises the tail group
    B = 4
    a = sp.random(nb, nb, density=0.04, random_state=rng, format="csr")
    a.data[:] = 1.0
    a = a + sp.identity(nb)
    a = a.tocsr()
    # expand to BSR with random blocks via kron
    dense_blocks = rng.standard_normal((a.nnz, B, B))
    m = sp.bsr_matrix((dense_blocks, a.indices, a.indptr), shape=(nb * B, nb * B)).tocsr()
    m.sort_indices()
    A = CSR(nb * B, nb * B, m.indptr, m.indices, m.data)
    Ad = DeviceBSR(A, B, hip.device)
    x = rng.standard_normal(nb * B)
    y0 = rng.standard_normal(nb * B)
    xd = hip.from_host(x)
    ref = m @ x

    # beta=0
    yd = hip.from_host(np.zeros(nb * B))
    check(lib().amg_bsr_spmv_mfma4_f64(Ad.nbrows, Ad.ptr.data_ptr(), Ad.col.data_ptr(),
                                       Ad.val.data_ptr(), xd.data_ptr(), 1.0, 0.0,
                                       yd.data_ptr(), _stream()), "mfma4")
    np.testing.assert_allclose(hip.to_host(yd), ref, rtol=1e-12, atol=1e-12)
    # alpha/beta
    yd = hip.from_host(y0)
    check(lib().amg_bsr_spmv_mfma4_f64(Ad.nbrows, Ad.ptr.data_ptr(), Ad.col.data_ptr(),
                                       Ad.val.data_ptr(), xd.data_ptr(), 1.7, 0.3,
                                       yd.data_ptr(), _stream()), "mfma4")
    np.testing.assert_allclose(hip.to_host(yd), 1.7 * ref + 0.3 * y0,
                               rtol=1e-12, atol=1e-12)


def test_sell_sigma_sorted_matches(hip):
    """Sigma-sorted SELL (row-length-sorted slices with a window-local
    permutation) matches the CSR kernels on a large ragged matrix where the
    sort actually engages (pad > 1.08, n > 65536)."""
    rng = np.random.default_rng(17)
    n = 140_000
    lens = rng.integers(0, 64, size=n)  # includes EMPTY rows
    ptr = np.zeros(n + 1, dtype=np.int64)
    np.cumsum(lens, out=ptr[1:])
    nnz = int(ptr[-1])
    col = rng.integers(0, n, size=nnz).astype(np.int32)
    val = rng.standard_normal(nnz)
    A = CSR(n, n, ptr, col, val)
    Ad = hip.matrix(A)
    Ad.build_sell(sigma=8192)
    assert Ad.srows is not None, "sigma sort should engage on this shape"
    x = rng.standard_normal(n)
    b = rng.standard_normal(n)
    m = rng.random(n) + 0.5
    xd, bd, md = hip.from_host(x), hip.from_host(b), hip.from_host(m)
    yd = hip.vector(n)
    hip.spmv(1.0, Ad, xd, 0.0, yd)
    ref = A.to_scipy() @ x
    np.testing.assert_allclose(hip.to_host(yd), ref, rtol=1e-12, atol=1e-10)
    rd = hip.vector(n)
    hip.residual(bd, Ad, xd, rd)
    np.testing.assert_allclose(hip.to_host(rd), b - ref, rtol=1e-12, atol=1e-10)
    x2 = hip.from_host(x)
    td = hip.vector(n)
    hip.relax_diag(Ad, md, bd, x2, td)
    np.testing.assert_allclose(hip.to_host(x2), x + m * (b - ref),
                               rtol=1e-12, atol=1e-10)


def test_bsr_from_device_matches_host(hip):
    """Device CSR->BSR conversion (torch ops) produces the same BSR arrays
    as the host converter (_core.csr_to_bsr)."""
    from amgcl_amd.backend.hip import DeviceBSR

    rng = np.random.default_rng(23)
    nb, B = 120, 3
    a = sp.random(nb, nb, density=0.08, random_state=rng, format="csr")
    a = (a + sp.identity(nb)).tocsr()
    a.data[:] = 1.0
    blocks = rng.standard_normal((a.nnz, B, B))
    # zero a few entries inside blocks so scalar CSR has partial blocks
    blocks[rng.random(a.nnz) < 0.3, 0, 1] = 0.0
    m = sp.bsr_matrix((blocks, a.indices, a.indptr),
                      shape=(nb * B, nb * B)).tocsr()
    m.eliminate_zeros()
    m.sort_indices()
    A = CSR(nb * B, nb * B, m.indptr, m.indices, m.data)
    d_host = DeviceBSR(A, B, hip.device)
    d_dev = DeviceBSR.from_device(hip.matrix(A), B, hip.device)
    assert d_host.nbrows == d_dev.nbrows
    np.testing.assert_array_equal(d_host.ptr.cpu().numpy(),
                                  d_dev.ptr.cpu().numpy())
    np.testing.assert_array_equal(d_host.col.cpu().numpy(),
                                  d_dev.col.cpu().numpy())
    np.testing.assert_allclose(d_host.val.cpu().numpy(),
                               d_dev.val.cpu().numpy(), atol=0, rtol=0)
