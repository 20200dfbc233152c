"""I/O round-trips, adapters, rebuild, CLI smoke."""
import os

import numpy as np
import pytest
import scipy.sparse as sp

import amgcl_amd as am
from amgcl_amd import io
from amgcl_amd.adapter import Reordered, ScaledProblem, zero_copy
from amgcl_amd.matrix import CSR


def rand_csr(rng, n, density=0.1, spd=True):
    a = sp.random(n, n, density=density, random_state=rng, format="csr")
    a.data = rng.standard_normal(a.nnz)
    if spd:
        a = (a + a.T).tocsr()
        a = (a + sp.identity(n) * (abs(a).sum(axis=1).max() + 1)).tocsr()
    a.sort_indices()
    return CSR(n, n, a.indptr, a.indices, a.data)


def test_mm_roundtrip(tmp_path):
    rng = np.random.default_rng(0)
    A = rand_csr(rng, 40)
    p = tmp_path / "a.mtx"
    io.mm_write(str(p), A)
    B = io.mm_read(str(p))
    assert abs(B.to_scipy() - A.to_scipy()).max() < 1e-14
    v = rng.standard_normal(40)
    io.mm_write(str(tmp_path / "v.mtx"), v)
    v2 = io.mm_read(str(tmp_path / "v.mtx")).ravel()
    np.testing.assert_allclose(v2, v)


def test_mm_symmetric_expansion(tmp_path):
    p = tmp_path / "s.mtx"
    p.write_text(
        "%%MatrixMarket matrix coordinate real symmetric\n"
        "3 3 4\n1 1 2.0\n2 1 -1.0\n2 2 2.0\n3 3 1.0\n"
    )
    A = io.mm_read(str(p))
    d = A.to_dense()
    assert d[0, 1] == -1.0 and d[1, 0] == -1.0


def test_binary_roundtrip_and_strip(tmp_path):
    rng = np.random.default_rng(1)
    A = rand_csr(rng, 50)
    p = str(tmp_path / "a.bin")
    io.write_crs(p, A)
    B = io.read_crs(p)
    assert abs(B.to_scipy() - A.to_scipy()).max() < 1e-15
    S = io.read_crs(p, row_beg=10, row_end=30)
    assert S.nrows == 20
    assert abs(S.to_scipy() - A.to_scipy()[10:30]).max() < 1e-15
    x = rng.standard_normal(50)
    io.write_dense(str(tmp_path / "x.bin"), x)
    np.testing.assert_allclose(io.read_dense(str(tmp_path / "x.bin")), x)


def test_zero_copy_no_copy():
    A, _ = am.poisson3d(8)
    Z = zero_copy(A.nrows, A.ptr, A.col, A.val)
    assert Z.val is A.val  # same buffer
    assert Z.nnz == A.nnz


def test_reorder_solves_same_system():
    A, b = am.poisson3d(10, rhs="random")
    R = Reordered(A)
    s = am.make_solver(R.A, {"solver": {"type": "cg", "tol": 1e-10, "maxiter": 100}})
    y, iters, resid = s(R.forward(b))
    x = R.inverse(y)
    assert np.linalg.norm(b - A @ x) / np.linalg.norm(b) < 1e-8
    # bandwidth actually reduced
    row_of = np.repeat(np.arange(R.A.nrows), np.diff(R.A.ptr))
    bw_new = int(np.abs(row_of - R.A.col).max())
    row_of0 = np.repeat(np.arange(A.nrows), np.diff(A.ptr))
    bw_old = int(np.abs(row_of0 - A.col).max())
    assert bw_new <= bw_old


def test_scaled_problem():
    A, b = am.poisson3d(10, rhs="random")
    S = ScaledProblem(A)
    s = am.make_solver(S.A, {"solver": {"type": "cg", "tol": 1e-10, "maxiter": 100}})
    y, iters, resid = s(S.scale_rhs(b))
    x = S.unscale_x(y)
    assert np.linalg.norm(b - A @ x) / np.linalg.norm(b) < 1e-8
    assert np.allclose(np.asarray(S.A.diagonal()), 1.0)


def test_rebuild_reuses_hierarchy():
    A, b = am.poisson3d(12, rhs="random")
    s = am.make_solver(A, {"solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}})
    x, it0, _ = s(b)
    # scale the matrix: same structure, new values
    A2 = CSR(A.nrows, A.ncols, A.ptr, A.col, A.val * 2.0)
    s.P.rebuild(A2)
    x2 = s.backend.vector(A.nrows)
    it2, resid2 = s.S(s.P, b, x2)
    assert resid2 < 1e-8
    assert np.linalg.norm(b - A2 @ x2) / np.linalg.norm(b) < 1e-7
    assert abs(it2 - it0) <= 2


def test_cli_smoke(tmp_path, capsys):
    from amgcl_amd.cli import main

    rc = main(["--poisson", "12", "-p", "solver.type=cg", "-p", "solver.tol=1e-8",
               "-o", str(tmp_path / "x.bin")])
    assert rc == 0
    out = capsys.readouterr().out
    assert "iters:" in out
    x = io.read_dense(str(tmp_path / "x.bin"))
    assert x.shape[0] == 12**3


def test_cli_mm_with_reorder_scale(tmp_path):
    from amgcl_amd.cli import main

    rng = np.random.default_rng(3)
    A = rand_csr(rng, 60)
    io.mm_write(str(tmp_path / "a.mtx"), A)
    rc = main(["-A", str(tmp_path / "a.mtx"), "--reorder", "--scale",
               "-p", "solver.type=bicgstab"])
    assert rc == 0


def test_complex_adapter():
    import scipy.sparse as sp

    from amgcl_amd.adapter import complex_to_real, real_to_complex

    rng = np.random.default_rng(4)
    n = 60
    m = sp.random(n, n, density=0.1, random_state=rng, format="csr")
    m = m + m.T + sp.identity(n) * 8.0
    m = m.astype(np.complex128)
    m.data = m.data + 1j * 0.2 * rng.standard_normal(m.nnz)
    m = (m + m.conj().T).tocsr()  # hermitian
    z = rng.standard_normal(n) + 1j * rng.standard_normal(n)
    b = m @ z
    Ar, br = complex_to_real(m, b)
    s = am.make_solver(Ar, {"solver": {"type": "bicgstab", "tol": 1e-10,
                                       "maxiter": 500}})
    xr, it, res = s(br)
    x = real_to_complex(xr)
    assert np.linalg.norm(m @ x - b) / np.linalg.norm(b) < 1e-7


def test_crs_builder():
    from amgcl_amd.adapter import CrsBuilder

    n = 50

    def row(i):
        cols, vals = [i], [2.0]
        if i > 0:
            cols.append(i - 1)
            vals.append(-1.0)
        if i + 1 < n:
            cols.append(i + 1)
            vals.append(-1.0)
        order = np.argsort(cols)
        return np.asarray(cols)[order], np.asarray(vals)[order]

    A = CrsBuilder(n, row).matrix
    b = np.ones(n)
    s = am.make_solver(A, {"precond": {"class": "relaxation", "type": "ilu0"},
                           "solver": {"type": "cg", "tol": 1e-10, "maxiter": 200}})
    x, it, res = s(b)
    assert np.linalg.norm(b - A @ x) < 1e-8


def test_cli_end_to_end(tmp_path):
    """The solver CLI (reference examples/solver.cpp): file input, key=value
    params, reorder + scaling, binary solution output, format conversion."""
    import json
    import subprocess
    import sys

    import amgcl_amd as am
    from amgcl_amd import io

    A, b = am.poisson3d(10, rhs="random")
    io.mm_write(str(tmp_path / "A.mtx"), A)
    io.mm_write(str(tmp_path / "b.mtx"), np.asarray(b).reshape(-1, 1))
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "-m", "amgcl_amd.cli",
         "-A", str(tmp_path / "A.mtx"), "-f", str(tmp_path / "b.mtx"),
         "-p", "solver.type=bicgstab", "-p", "solver.tol=1e-8",
         "-p", "precond.coarse_enough=200",
         "--reorder", "--scale", "-o", str(tmp_path / "x.bin")],
        capture_output=True, text=True, cwd=repo, timeout=300)
    assert r.returncode == 0, r.stderr[-1500:]
    x = io.read_dense(str(tmp_path / "x.bin")).ravel()
    assert np.linalg.norm(b - A @ x) / np.linalg.norm(b) < 1e-6
    # conversion path
    r = subprocess.run(
        [sys.executable, "-m", "amgcl_amd.cli", "--convert",
         str(tmp_path / "A.mtx"), str(tmp_path / "A.bin")],
        capture_output=True, text=True, cwd=repo, timeout=120)
    assert r.returncode == 0
    B = io.read_crs(str(tmp_path / "A.bin"))
    assert B.nnz == A.nnz


def test_mm_pattern_integer_complex(tmp_path):
    """MatrixMarket field variants (reference io/mm.hpp handles real,
    integer, pattern, and complex banners)."""
    from amgcl_amd import io

    p = tmp_path / "pat.mtx"
    p.write_text(
        "%%MatrixMarket matrix coordinate pattern general\n"
        "3 3 4\n1 1\n2 2\n3 3\n1 3\n")
    A = io.mm_read(str(p))
    assert A.nnz == 4 and A.val[0] == 1.0

    q = tmp_path / "int.mtx"
    q.write_text(
        "%%MatrixMarket matrix coordinate integer general\n"
        "2 2 3\n1 1 4\n2 2 5\n1 2 -3\n")
    B = io.mm_read(str(q))
    assert B.to_scipy()[0, 1] == -3.0 and B.to_scipy()[1, 1] == 5.0

    c = tmp_path / "cplx.mtx"
    c.write_text(
        "%%MatrixMarket matrix coordinate complex general\n"
        "2 2 2\n1 1 1.0 2.0\n2 2 3.0 -1.0\n")
    C = io.mm_read(str(c))
    assert np.iscomplexobj(np.asarray(C.val))
    assert np.asarray(C.val)[0] == 1.0 + 2.0j


def test_binary_reference_layout(tmp_path):
    """Files in the upstream amgcl binary layout (size_t n, int64 ptr/col,
    float64 val, no magic — amgcl/io/binary.hpp, examples/mm2bin.cpp) load
    directly, including strip reads."""
    rng = np.random.default_rng(3)
    A = rand_csr(rng, 40)
    p = str(tmp_path / "ref.bin")
    with open(p, "wb") as f:
        np.array([A.nrows], dtype=np.int64).tofile(f)
        A.ptr.astype(np.int64).tofile(f)
        A.col.astype(np.int64).tofile(f)
        A.val.astype(np.float64).tofile(f)
    B = io.read_crs(p)
    assert abs(B.to_scipy() - A.to_scipy()).max() < 1e-15
    S = io.read_crs(p, row_beg=5, row_end=25)
    assert S.nrows == 20
    assert abs(S.to_scipy() - A.to_scipy()[5:25]).max() < 1e-15


def test_unblock_bsr_and_as_scalar():
    """Block-valued (BSR) input path (parity: amgcl/coarsening/as_scalar.hpp
    + adapter/block_matrix.hpp direction): unblock a BSR matrix to scalar
    CSR exactly, and run AMG through the registry's as_scalar wrapper."""
    import scipy.sparse as sp

    from amgcl_amd.coarsening.as_scalar import unblock_bsr

    rng = np.random.default_rng(0)
    nb, B = 40, 3
    a = sp.random(nb, nb, density=0.1, random_state=rng, format="csr")
    a = (a + sp.identity(nb)).tocsr()
    a.data[:] = 1.0
    blocks = rng.standard_normal((a.nnz, B, B))
    m = sp.bsr_matrix((blocks, a.indices, a.indptr), shape=(nb * B, nb * B)).tocsr()
    A = unblock_bsr(nb, B, a.indptr, a.indices, blocks)
    assert abs(A.to_scipy() - m).max() < 1e-15

    Ah, b = am.poisson3d(16, rhs="ones")
    s = am.make_solver(
        Ah, {"precond": {"class": "amg", "coarse_enough": 300,
                         "coarsening": {"type": "as_scalar",
                                        "base": {"type": "smoothed_aggregation"}}},
             "solver": {"type": "cg", "tol": 1e-8}})
    x, it, r = s(b)
    assert r < 1e-8
    assert np.linalg.norm(b - Ah @ x) / np.linalg.norm(b) < 1e-7


def test_make_solver_rebuild_refreshes():
    """MakeSolver.rebuild: new coefficients, reused transfers, correct
    solutions afterwards (the preconditioner-level rebuild alone would
    leave a native driver with stale pointers on the GPU backend)."""
    A, b = am.poisson3d(12, rhs="ones")
    s = am.make_solver(A, {"precond": {"class": "amg", "coarse_enough": 200},
                           "solver": {"type": "cg", "tol": 1e-9, "maxiter": 100}})
    x1, it1, _ = s(b)
    A2 = am.matrix.CSR(A.nrows, A.ncols, A.ptr, A.col, 2.0 * A.val)
    s.rebuild(A2)
    x2, it2, _ = s(b)
    np.testing.assert_allclose(x2, x1 / 2.0, rtol=1e-7, atol=1e-10)
