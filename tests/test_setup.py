"""Unit tests for the host setup engine (_core) against scipy references."""
import numpy as np
import pytest
import scipy.sparse as sp

import amgcl_amd as am
from amgcl_amd import _core
from amgcl_amd.matrix import CSR


def rand_csr(rng, n, m, density=0.05):
    a = sp.random(n, m, density=density, random_state=rng, format="csr")
    a.data = rng.standard_normal(a.nnz)
    return CSR(n, m, a.indptr, a.indices, a.data), a


def test_poisson3d_matches_kron_construction():
    n = 8
    A, b = am.poisson3d(n)
    # scipy reference: 3D Laplacian via kron sums, h=1
    one = sp.identity(n)
    t = sp.diags([-1, 2, -1], [-1, 0, 1], shape=(n, n))
    ref = (
        sp.kron(sp.kron(one, one), t)
        + sp.kron(sp.kron(one, t), one)
        + sp.kron(sp.kron(t, one), one)
    ).tocsr()
    ours = A.to_scipy()
    assert (abs(ours - ref)).max() < 1e-14
    assert b.shape == (n**3,)


def test_transpose_matches_scipy():
    rng = np.random.default_rng(0)
    A, a = rand_csr(rng, 60, 37)
    T = A.transpose()
    diff = abs(T.to_scipy() - a.T.tocsr())
    assert diff.nnz == 0 or diff.max() < 1e-14


def test_spgemm_matches_scipy():
    rng = np.random.default_rng(1)
    A, a = rand_csr(rng, 40, 55)
    B, b = rand_csr(rng, 55, 33)
    C = A @ B
    diff = abs(C.to_scipy() - (a @ b).tocsr())
    assert diff.nnz == 0 or diff.max() < 1e-12
    # rows sorted
    for i in range(C.nrows):
        row = C.col[C.ptr[i] : C.ptr[i + 1]]
        assert np.all(np.diff(row) > 0)


def test_spmv_residual_match_numpy():
    rng = np.random.default_rng(2)
    A, a = rand_csr(rng, 70, 70, 0.1)
    x = rng.standard_normal(70)
    y = rng.standard_normal(70)
    y2 = y.copy()
    A.spmv(1.5, x, 0.25, y2)
    assert np.allclose(y2, 1.5 * (a @ x) + 0.25 * y)
    r = np.empty(70)
    A.residual(y, x, r)
    assert np.allclose(r, y - a @ x)


def test_aggregates_cover_all_connected_nodes():
    A, _ = am.poisson3d(8)
    naggr, ids, strong = _core.aggregates(A.nrows, A.ptr, A.col, A.val, 0.08)
    ids = np.asarray(ids)
    assert naggr > 0
    assert ids.max() == naggr - 1
    # Poisson has no Dirichlet-isolated rows -> every node aggregated
    assert (ids >= 0).all()
    # every aggregate non-empty
    assert len(np.unique(ids)) == naggr


def test_galerkin_matches_scipy_triple_product():
    A, _ = am.poisson3d(6)
    naggr, ids, strong = _core.aggregates(A.nrows, A.ptr, A.col, A.val, 0.08)
    pp, pc, pv = _core.smoothed_prolongation(
        A.nrows, A.ptr, A.col, A.val, strong, ids, naggr, 2.0 / 3.0
    )
    P = CSR(A.nrows, naggr, pp, pc, pv)
    R = P.transpose()
    Ac = am.galerkin(R, A, P)
    ref = (P.to_scipy().T @ A.to_scipy() @ P.to_scipy()).tocsr()
    diff = abs(Ac.to_scipy() - ref)
    assert diff.nnz == 0 or diff.max() < 1e-12


def test_ilu0_factorization_reproduces_dense_ilu():
    A, _ = am.poisson3d(4)
    lu, dia = _core.ilu0_factor(A.nrows, A.ptr, A.col, A.val)
    # solve with it: must reduce residual substantially for diagonally dominant A
    rng = np.random.default_rng(3)
    b = rng.standard_normal(A.nrows)
    z = b.copy()
    _core.ilu0_solve(A.nrows, A.ptr, A.col, lu, dia, z)
    r = b - A @ z
    assert np.linalg.norm(r) < 0.5 * np.linalg.norm(b)


def test_unknown_param_raises():
    A, b = am.poisson3d(8)
    with pytest.raises(Exception, match="unknown"):
        am.make_solver(A, {"precond": {"class": "amg", "coarsening": {"type": "smoothed_aggregation", "bogus": 1}}})
    with pytest.raises(Exception, match="unknown"):
        am.make_solver(A, {"bogus": {}})


def test_splu_coarse_solver():
    """Alternative sparse-LU coarse solve (parity: solver/eigen.hpp class of
    alternative direct coarse solvers) matches the dense-inverse iterations."""
    import amgcl_amd as am

    A, b = am.poisson3d(16, rhs="random")
    prm = {"solver": {"type": "cg", "tol": 1e-8, "maxiter": 100},
           "precond": {"class": "amg", "coarse_enough": 800}}
    _, it_dense, r_dense = am.make_solver(A, prm)(b)
    prm["precond"]["direct_solver"] = "splu"
    _, it_splu, r_splu = am.make_solver(A, prm)(b)
    assert r_splu < 1e-8
    assert it_splu == it_dense


def test_pre_cycles_and_tuple_input():
    """pre_cycles > 1 (multiple cycles per apply) and the crs_tuple input
    adapter (reference adapter/crs_tuple.hpp)."""
    import amgcl_amd as am

    A, b = am.poisson3d(12, rhs="random")
    s2 = am.make_solver(
        (A.ptr, A.col, A.val),  # tuple input
        {"precond": {"class": "amg", "pre_cycles": 2, "coarse_enough": 200},
         "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}})
    x, it2, r2 = s2(b)
    assert r2 < 1e-8
    s1 = am.make_solver(A, {"precond": {"class": "amg", "coarse_enough": 200},
                            "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}})
    _, it1, _ = s1(b)
    assert it2 <= it1  # the stronger preconditioner cannot need more iterations


def test_lagged_preconditioner():
    """Solve with a matrix different from the setup matrix (reference
    make_solver.hpp:116 — time stepping with a lagged hierarchy)."""
    import amgcl_amd as am
    import numpy as np

    A, b = am.poisson3d(12, rhs="random")
    s = am.make_solver(A, {"solver": {"type": "cg", "tol": 1e-8, "maxiter": 200},
                           "precond": {"class": "amg", "coarse_enough": 200}})
    A2 = am.CSR(A.nrows, A.ncols, A.ptr, A.col, 1.3 * np.asarray(A.val))
    x, iters, resid = s(b, A=A2)
    assert resid < 1e-8
    assert np.linalg.norm(b - A2 @ x) / np.linalg.norm(b) < 1e-7


def test_hierarchy_quality_poisson64():
    """Pin operator/grid complexity and per-level nnz decay for 64^3 Poisson
    against the reference's printed hierarchy class (amgcl/amg.hpp:561-598
    prints these tables; SA on 3D Poisson gives oc ~1.3-1.6, gc ~1.1), so
    aggregation regressions (MIS vs greedy) are caught numerically."""
    A, b = am.poisson3d(64, rhs="ones")
    s = am.make_solver(
        A, {"precond": {"class": "amg"},
            "solver": {"type": "cg", "tol": 1e-8, "maxiter": 60}})
    amg = s.P
    rows = [l.rows for l in amg.levels]
    nnzs = [l.nnz for l in amg.levels]
    oc = sum(nnzs) / nnzs[0]
    gc = sum(rows) / rows[0]
    assert 1.25 <= oc <= 1.65, f"operator complexity {oc} out of SA class"
    assert 1.05 <= gc <= 1.20, f"grid complexity {gc} out of SA class"
    # coarsening ratio: every level at least ~6x smaller (3D SA aggregates
    # have ~8+ points each)
    for a, c in zip(rows, rows[1:]):
        assert c * 6 <= a, (rows, "insufficient coarsening")
    # iteration class: reference solves this problem class in ~15-25 CG
    # iterations (docs/benchmarks.rst Poisson tables)
    x, iters, resid = s(b)
    assert resid < 1e-8
    assert iters <= 25


def test_parallel_mis_vs_greedy_hierarchy_quality():
    """The deterministic parallel MIS(2) aggregation must produce a
    hierarchy of the same complexity class as the reference-semantics greedy
    pass (drift was previously unasserted)."""
    from amgcl_amd import _core

    A, _ = am.poisson3d(32)
    eps = 0.08
    ng, idg, _ = _core.aggregates(A.nrows, A.ptr, A.col, A.val, eps)
    np_, idp, _ = _core.aggregates_parallel(A.nrows, A.ptr, A.col, A.val, eps)
    # MIS(2) roots are sparser than greedy seeds, so its aggregates are
    # systematically bigger (measured: 3067 vs 4192 on 32^3 => mean size
    # 10.7 vs 7.8).  Pin the relationship and the size class; the
    # iteration-count impact is pinned by test_hierarchy_quality_poisson64
    # and the device-setup tests.
    assert abs(ng - np_) <= 0.35 * max(ng, np_), (ng, np_)
    for cnt in (ng, np_):
        mean_size = A.nrows / cnt
        assert 6.0 <= mean_size <= 14.0, (ng, np_, mean_size)
    # both partitions: all nodes assigned, aggregate sizes sane (<= 3^3+ring)
    for ids, cnt in ((idg, ng), (idp, np_)):
        ids = np.asarray(ids)
        assigned = ids[ids >= 0]
        assert len(assigned) == A.nrows  # no isolated nodes on Poisson
        sizes = np.bincount(assigned, minlength=cnt)
        assert sizes.min() >= 1
        assert sizes.max() <= 40


def test_skyline_lu_coarse_solver():
    """Skyline (profile) LU after CM reorder matches a dense solve to 1e-9
    and works as the AMG coarsest-level solver (parity:
    amgcl/solver/skyline_lu.hpp:85, the reference's default coarse solver)."""
    from amgcl_amd.backend.cpu import SkylineCoarseSolver

    A, b = am.poisson3d(8, rhs="random")
    s = SkylineCoarseSolver(A)
    x = np.empty(A.nrows)
    s(b, x)
    ref = np.linalg.solve(A.to_dense(), b)
    assert np.abs(x - ref).max() < 1e-9
    # profile storage is far below dense n^2
    assert s.bytes() < A.nrows ** 2 * 8 / 4

    A, b = am.poisson3d(20, rhs="ones")
    sol = am.make_solver(
        A, {"precond": {"class": "amg", "direct_solver": "skyline",
                        "coarse_enough": 500},
            "solver": {"type": "cg", "tol": 1e-8}})
    x, it, r = sol(b)
    assert r < 1e-8 and it < 20


def test_skyline_nonsymmetric():
    """Skyline LU on a nonsymmetric (convection-diffusion-like) matrix."""
    rng = np.random.default_rng(3)
    import scipy.sparse as sp

    n = 60
    d = 2.5 + rng.random(n)
    a = sp.diags([d, -1.2 * np.ones(n - 1), -0.4 * np.ones(n - 1),
                  0.3 * np.ones(n - 5)], [0, -1, 1, 5], format="csr")
    A = am.matrix.CSR(n, n, a.indptr, a.indices, a.data)
    b = rng.standard_normal(n)
    from amgcl_amd.backend.cpu import SkylineCoarseSolver

    s = SkylineCoarseSolver(A)
    x = np.empty(n)
    s(b, x)
    ref = np.linalg.solve(a.toarray(), b)
    assert np.abs(x - ref).max() < 1e-9
