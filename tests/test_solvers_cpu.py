"""Combinatorial convergence matrix on the CPU backend.

Mirrors the reference's test strategy (tests/test_solver.hpp:111-210): loop
coarsenings x smoothers x Krylov solvers through the runtime interface on a
3D Poisson problem and assert the relative residual. Sizes are kept small so
the suite runs in seconds on the CPU-only container.
"""
import numpy as np
import pytest

import amgcl_amd as am

COARSENING = ["smoothed_aggregation", "aggregation", "ruge_stuben"]
RELAX = ["spai0", "damped_jacobi", "chebyshev", "gauss_seidel", "ilu0"]
RELAX_EXTRA = ["spai1", "iluk", "ilup", "ilut", "ilu0_chow_patel"]
SOLVERS = ["cg", "bicgstab", "gmres", "richardson"]
SOLVERS_EXTRA = ["bicgstabl", "fgmres", "lgmres", "idrs"]

N = 16  # 4096 unknowns


@pytest.fixture(scope="module")
def problem():
    A, b = am.poisson3d(N, rhs="random")
    return A, b


@pytest.mark.parametrize("coarsening", COARSENING)
@pytest.mark.parametrize("relax", RELAX)
@pytest.mark.parametrize("solver", SOLVERS)
def test_convergence_matrix(problem, coarsening, relax, solver):
    A, b = problem
    maxiter = 300 if solver == "richardson" else 100
    s = am.make_solver(
        A,
        {
            "precond": {
                "class": "amg",
                "coarsening": {"type": coarsening},
                "relax": {"type": relax},
                "coarse_enough": 500,
            },
            "solver": {"type": solver, "tol": 1e-8, "maxiter": maxiter},
        },
    )
    x, iters, resid = s(b)
    assert resid < 1e-6, f"{coarsening}/{relax}/{solver}: resid={resid} iters={iters}"
    r = b - A @ x
    assert np.linalg.norm(r) / np.linalg.norm(b) < 1e-6
    assert iters < maxiter


@pytest.mark.parametrize("relax", RELAX)
def test_smoother_as_preconditioner(problem, relax):
    """Each smoother standalone (parity: tests/test_solver.hpp test_rap)."""
    A, b = problem
    s = am.make_solver(
        A,
        {
            "precond": {"class": "relaxation", "type": relax},
            "solver": {"type": "bicgstab", "tol": 1e-8, "maxiter": 1000},
        },
    )
    x, iters, resid = s(b)
    assert resid < 1e-6


def test_cg_iteration_count_parity(problem):
    """SA + SPAI0 + CG on Poisson must stay in the reference's iteration
    class (reference: 12-24 its on comparable Poisson problems)."""
    A, b = problem
    s = am.make_solver(A, {"solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}})
    x, iters, resid = s(b)
    assert iters <= 30
    assert resid < 1e-8


def test_dummy_preconditioner(problem):
    A, b = problem
    s = am.make_solver(
        A,
        {
            "precond": {"class": "dummy"},
            "solver": {"type": "cg", "tol": 1e-8, "maxiter": 2000},
        },
    )
    x, iters, resid = s(b)
    assert resid < 1e-7


def test_w_cycle_and_pre_post_counts(problem):
    A, b = problem
    s = am.make_solver(
        A,
        {
            "precond": {"class": "amg", "ncycle": 2, "npre": 2, "npost": 2},
            "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100},
        },
    )
    x, iters, resid = s(b)
    assert resid < 1e-8
    assert iters <= 20


def test_zero_rhs(problem):
    A, _ = problem
    s = am.make_solver(A, {"solver": {"type": "cg"}})
    x, iters, resid = s(np.zeros(A.nrows))
    assert iters == 0
    assert np.all(np.asarray(x) == 0)


def test_preonly_nested():
    A, b = am.poisson3d(12, rhs="random")
    s = am.make_solver(A, {"solver": {"type": "preonly"}})
    x, iters, resid = s(b)
    assert iters == 1
    assert resid < 1.0


@pytest.mark.parametrize("solver", SOLVERS_EXTRA)
def test_extra_solvers(problem, solver):
    A, b = problem
    s = am.make_solver(
        A,
        {"precond": {"class": "amg", "coarse_enough": 500},
         "solver": {"type": solver, "tol": 1e-8, "maxiter": 200}},
    )
    x, iters, resid = s(b)
    r = b - A @ x
    assert np.linalg.norm(r) / np.linalg.norm(b) < 1e-6


def test_ruge_stuben_iteration_class(problem):
    A, b = problem
    s = am.make_solver(
        A,
        {"precond": {"class": "amg", "coarsening": {"type": "ruge_stuben"},
                     "coarse_enough": 500},
         "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}},
    )
    x, iters, resid = s(b)
    assert resid < 1e-8 and iters <= 20


@pytest.mark.parametrize("relax", RELAX_EXTRA)
def test_extra_smoothers(problem, relax):
    A, b = problem
    s = am.make_solver(
        A,
        {"precond": {"class": "amg", "relax": {"type": relax}, "coarse_enough": 500},
         "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}},
    )
    x, iters, resid = s(b)
    assert resid < 1e-7
    assert np.linalg.norm(b - A @ x) / np.linalg.norm(b) < 1e-6


def test_anisotropic_poisson():
    """Anisotropy exercises the strong-connection filtering
    (reference fixture supports anisotropy, tests/sample_problem.hpp)."""
    A, b = am.poisson3d(16, anisotropy=0.25, rhs="random")
    s = am.make_solver(
        A, {"precond": {"class": "amg", "coarse_enough": 500},
            "solver": {"type": "cg", "tol": 1e-8, "maxiter": 200}})
    x, iters, resid = s(b)
    assert resid < 1e-8
    assert np.linalg.norm(b - A @ x) / np.linalg.norm(b) < 1e-7
    assert iters < 100


def test_smoothed_aggr_emin_nonsymmetric():
    """Energy-minimizing SA with separately smoothed R on a nonsymmetric
    convection-diffusion system (parity: coarsening/smoothed_aggr_emin.hpp)."""
    import scipy.sparse as sp

    A, b = am.poisson3d(14, rhs="random")
    m = A.to_scipy()
    n = m.shape[0]
    conv = 0.4 * (sp.diags(np.ones(n - 1), 1) - sp.diags(np.ones(n - 1), -1))
    K = (m + conv).tocsr()
    K.sort_indices()
    Ak = am.CSR.from_scipy(K)
    s = am.make_solver(
        Ak, {"precond": {"class": "amg", "coarsening": {"type": "smoothed_aggr_emin"},
                         "coarse_enough": 400},
             "solver": {"type": "bicgstab", "tol": 1e-8, "maxiter": 200}})
    x, iters, resid = s(b)
    assert resid < 1e-7
    assert np.linalg.norm(b - K @ x) / np.linalg.norm(b) < 1e-6
    assert iters < 40


def test_ilup_fill_pattern(problem):
    """ILU(p)'s factor pattern is the symbolic power A^(k+1)
    (reference: relaxation/ilup.hpp symb_product loop)."""
    from amgcl_amd.backend import make_backend
    from amgcl_amd.relaxation import ILU0, ILUP

    A, _ = problem
    cpu = make_backend("cpu")
    s0 = ILU0(A, {}, cpu)
    s1 = ILUP(A, {"k": 1}, cpu)
    assert s1.col.size > s0.col.size  # strictly more fill than ILU(0)
    # k=0 degenerates to ILU(0)
    s2 = ILUP(A, {"k": 0}, cpu)
    assert s2.col.size == s0.col.size
    np.testing.assert_allclose(s2.lu, s0.lu)


def _block_problem(n=10, bsize=2, eps=0.1):
    """Poisson x identity with intra-block coupling: block-structured SPD."""
    import scipy.sparse as sp

    from amgcl_amd.matrix import CSR

    A, _ = am.poisson3d(n)
    C = np.full((bsize, bsize), eps) + (1.0 - eps) * np.eye(bsize)
    Ab = sp.kron(A.to_scipy(), sp.csr_matrix(C), format="csr")
    rng = np.random.default_rng(5)
    return CSR.from_scipy(Ab), rng.standard_normal(Ab.shape[0])


@pytest.mark.parametrize("base", ["spai0", "damped_jacobi"])
def test_as_block_smoother(base):
    """as_block wrapper: block-valued base smoother over the scalar matrix
    (reference: relaxation/as_block.hpp)."""
    A, b = _block_problem()
    s = am.make_solver(
        A,
        {"precond": {"class": "amg", "coarse_enough": 500,
                     "relax": {"type": "as_block",
                               "block_size": 2, "base": {"type": base}}},
         "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}},
    )
    x, iters, resid = s(b)
    assert resid < 1e-7
    assert np.linalg.norm(b - A @ x) / np.linalg.norm(b) < 1e-6


def test_as_block_beats_scalar_on_coupled_blocks():
    """With strong intra-block coupling the block-diagonal inverse must not
    be slower than the scalar diagonal smoother."""
    A, b = _block_problem(n=10, bsize=3, eps=0.4)
    prm = {"solver": {"type": "cg", "tol": 1e-8, "maxiter": 200}}
    s_blk = am.make_solver(
        A, {**prm, "precond": {"class": "amg", "coarse_enough": 500,
                               "relax": {"type": "as_block", "block_size": 3,
                                         "base": {"type": "damped_jacobi"}}}})
    _, it_blk, r_blk = s_blk(b)
    assert r_blk < 1e-8
    s_sc = am.make_solver(
        A, {**prm, "precond": {"class": "amg", "coarse_enough": 500,
                               "relax": {"type": "damped_jacobi"}}})
    _, it_sc, r_sc = s_sc(b)
    assert it_blk <= it_sc + 2


def test_level_scheduled_ilu_solve_matches_serial():
    """OpenMP level-scheduled triangular sweeps are bitwise identical to the
    serial sweeps (parity: relaxation/detail/ilu_solve.hpp level scheduling)."""
    from amgcl_amd.backend import make_backend
    from amgcl_amd.relaxation import ILU0

    A, b = am.poisson3d(24, rhs="random")
    cpu = make_backend("cpu")
    s_ser = ILU0(A, {"solve_serial": True}, cpu)
    s_par = ILU0(A, {"solve_serial": False}, cpu)
    assert s_par._levels is not None
    z1, z2 = b.copy(), b.copy()
    s_ser._solve_serial(z1)
    s_par._solve_serial(z2)
    np.testing.assert_array_equal(z1, z2)   # bitwise


def test_multicolor_cpu_gauss_seidel():
    """Deterministic CPU-parallel GS (multicolor sweep): converges in the
    serial class and is reproducible across runs."""
    A, b = am.poisson3d(16, rhs="random")
    prm = {"precond": {"class": "amg", "coarse_enough": 500,
                       "relax": {"type": "gauss_seidel", "serial": False}},
           "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}}
    x1, it1, r1 = am.make_solver(A, prm)(b)
    x2, it2, r2 = am.make_solver(A, prm)(b)
    assert r1 < 1e-8 and it1 == it2
    np.testing.assert_array_equal(x1, x2)   # deterministic
    prm_ser = {"precond": {"class": "amg", "coarse_enough": 500,
                           "relax": {"type": "gauss_seidel", "serial": True}},
               "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}}
    _, it_ser, _ = am.make_solver(A, prm_ser)(b)
    assert it1 <= it_ser + 4  # colored ordering stays in the serial class


def test_ns_search_finds_nullspace_vector():
    """ns_search (reference tests/test_solver_ns_builtin.cpp): with a zero
    right-hand side and a nonzero initial guess, the solver iterates toward a
    null-space vector of a singular operator instead of returning x = 0."""
    import scipy.sparse as sp

    from amgcl_amd.matrix import CSR

    # singular pure-Neumann graph Laplacian on a 1D chain: nullspace = const
    n = 200
    main = np.full(n, 2.0)
    main[0] = main[-1] = 1.0
    L = sp.diags([np.full(n - 1, -1.0), main, np.full(n - 1, -1.0)],
                 [-1, 0, 1]).tocsr()
    L.sort_indices()
    A = CSR.from_scipy(L)
    rng = np.random.default_rng(3)
    x0 = rng.standard_normal(n)
    s = am.make_solver(
        A, {"precond": {"class": "relaxation", "type": "damped_jacobi"},
            "solver": {"type": "cg", "tol": 1e-10, "maxiter": 2000,
                       "ns_search": True}})
    x, iters, resid = s(np.zeros(n), x=x0.copy())
    x = np.asarray(x)
    assert np.linalg.norm(x) > 1e-6          # NOT the trivial solution
    assert np.linalg.norm(L @ x) < 1e-6 * np.linalg.norm(x)  # in the nullspace
    # the found vector is (up to scale) the constant vector
    xn = x / np.linalg.norm(x)
    const = np.ones(n) / np.sqrt(n)
    assert min(np.linalg.norm(xn - const), np.linalg.norm(xn + const)) < 1e-4


def test_block_ilu0_defining_property():
    """Block ILU(0): (L·U) restricted to the BLOCK pattern equals A
    (parity: amgcl/relaxation/ilu0.hpp over static_matrix values)."""
    from amgcl_amd import _core

    b = 3
    A0, _ = am.poisson3d(6)
    n = A0.nrows
    bp, bc, bv = _core.csr_to_bsr(n, A0.ptr, A0.col, A0.val, b)
    bp, bc = np.asarray(bp), np.asarray(bc)
    nb = n // b
    lu, dia = _core.block_ilu0_factor(nb, b, bp, bc, np.asarray(bv))
    lu, dia = np.asarray(lu).reshape(-1, b, b), np.asarray(dia)
    L = np.eye(n)
    U = np.zeros((n, n))
    mask = np.zeros((n, n), bool)
    for i in range(nb):
        for j in range(bp[i], bp[i + 1]):
            c = bc[j]
            mask[i * b:(i + 1) * b, c * b:(c + 1) * b] = True
            if j < dia[i]:
                L[i * b:(i + 1) * b, c * b:(c + 1) * b] = lu[j]
            elif j == dia[i]:
                U[i * b:(i + 1) * b, c * b:(c + 1) * b] = np.linalg.inv(lu[j])
            else:
                U[i * b:(i + 1) * b, c * b:(c + 1) * b] = lu[j]
    err = np.abs((L @ U - A0.to_scipy().toarray()) * mask).max()
    assert err < 1e-10
    # the serial sweeps apply exactly M^-1 = (LU)^-1
    rng = np.random.default_rng(0)
    r = rng.standard_normal(n)
    z = r.copy()
    _core.block_ilu0_solve(nb, b, bp, bc, np.asarray(lu).ravel(), dia, z)
    np.testing.assert_allclose(z, np.linalg.solve(L @ U, r), atol=1e-10)


def test_block_ilu0_converges_on_block_system():
    """block_ilu0 as the AMG smoother on a coupled block system (elasticity
    shape, 3 dofs/node — the CoupCons3D tutorial configuration class)."""
    from amgcl_amd.generators import elasticity3d

    A, b, coords = elasticity3d(6)
    s = am.make_solver(A, {"precond": {"class": "amg", "coarse_enough": 400,
                                       "relax": {"type": "block_ilu0",
                                                 "block_size": 3}},
                           "solver": {"type": "cg", "tol": 1e-8,
                                      "maxiter": 200}})
    x, iters, resid = s(b)
    assert resid < 1e-8
    true = np.linalg.norm(b - A.to_scipy() @ x) / np.linalg.norm(b)
    assert true < 1e-7
    assert iters < 80
