"""Native complex-valued solves (parity: amgcl/value_type/complex.hpp,
tests/test_solver_complex.cpp — the reference instantiates the same solver
templates over std::complex; here the CPU backend, CSR algebra, SA
coarsening, SPAI0/Jacobi smoothers and Krylov solvers all run over
complex128 natively).  The 2x2-real expansion adapter remains the route for
the HIP backend and is cross-checked here.
"""
import numpy as np
import pytest
import scipy.sparse as sp

import amgcl_amd as am
from amgcl_amd.matrix import CSR


def helmholtz(n=16, shift=0.4 + 0.35j):
    A0, _ = am.poisson3d(n)
    m = A0.to_scipy().astype(np.complex128) + shift * sp.identity(n**3)
    m = m.tocsr()
    m.sort_indices()
    return CSR.from_scipy(m), m


@pytest.mark.parametrize("solver", ["bicgstab", "gmres", "fgmres", "lgmres", "richardson"])
def test_native_complex_solve(solver):
    A, m = helmholtz(14)
    rng = np.random.default_rng(0)
    b = rng.standard_normal(A.nrows) + 1j * rng.standard_normal(A.nrows)
    s = am.make_solver(
        A, {"precond": {"class": "amg", "coarse_enough": 400},
            "solver": {"type": solver, "tol": 1e-8, "maxiter": 200}})
    x, iters, resid = s(b)
    assert resid < 1e-8
    assert np.iscomplexobj(x)
    assert np.linalg.norm(b - m @ x) / np.linalg.norm(b) < 1e-7
    assert iters < 60


def test_native_complex_matches_real_expansion():
    """The native complex route and the 2x2-real expansion adapter agree on
    the solution (the adapter doubles nnz and changes the spectrum, so
    iteration counts may differ; the solution must not)."""
    from amgcl_amd.adapter import complex_to_real, real_to_complex

    A, m = helmholtz(12)
    rng = np.random.default_rng(1)
    b = rng.standard_normal(A.nrows) + 1j * rng.standard_normal(A.nrows)

    s1 = am.make_solver(
        A, {"precond": {"class": "amg", "coarse_enough": 300},
            "solver": {"type": "bicgstab", "tol": 1e-10, "maxiter": 300}})
    x1, it1, _ = s1(b)

    Ar, br = complex_to_real(A, b)
    s2 = am.make_solver(
        Ar, {"precond": {"class": "amg", "coarse_enough": 600},
             "solver": {"type": "bicgstab", "tol": 1e-10, "maxiter": 300}})
    x2r, it2, _ = s2(br)
    x2 = real_to_complex(x2r)
    np.testing.assert_allclose(x1, x2, rtol=1e-6, atol=1e-8)


def test_complex_spai0_and_jacobi_smoothers():
    A, m = helmholtz(10)
    rng = np.random.default_rng(2)
    b = rng.standard_normal(A.nrows) + 1j * rng.standard_normal(A.nrows)
    for relax in ("spai0", "damped_jacobi"):
        s = am.make_solver(
            A, {"precond": {"class": "amg", "coarse_enough": 300,
                            "relax": {"type": relax}},
                "solver": {"type": "bicgstab", "tol": 1e-8, "maxiter": 200}})
        x, iters, resid = s(b)
        assert resid < 1e-8
        assert np.linalg.norm(b - m @ x) / np.linalg.norm(b) < 1e-7


def test_complex_adjoint_transpose():
    """CSR.transpose applies the adjoint for complex values (backend
    parity: builtin transpose uses math::adjoint)."""
    rng = np.random.default_rng(3)
    a = sp.random(30, 20, density=0.2, random_state=rng, format="csr")
    a.data = rng.standard_normal(a.nnz) + 1j * rng.standard_normal(a.nnz)
    A = CSR.from_scipy(a)
    T = A.transpose()
    assert abs(T.to_scipy() - a.conj().T).max() < 1e-15


def test_cli_complex_mm(tmp_path, capsys):
    """CLI end-to-end on a complex MatrixMarket system: the front door
    auto-selects the complex128 CPU backend (reference: examples/solver.cpp
    compiled for std::complex)."""
    from amgcl_amd import cli, io

    A, m = helmholtz(8)
    p = str(tmp_path / "helm.mtx")
    io.mm_write(p, A)
    rc = cli.main(["-A", p, "-p", "solver.type=bicgstab",
                   "-p", "solver.tol=1e-8",
                   "-p", "precond.coarse_enough=200"])
    assert rc in (0, None)
    out = capsys.readouterr().out
    assert "iters:" in out
    resid = float([l for l in out.splitlines() if l.startswith("error:")][0]
                  .split()[1])
    assert resid < 1e-8


@pytest.mark.gpu
@pytest.mark.parametrize("solver", ["bicgstab", "gmres"])
def test_native_complex_on_hip(solver):
    """Native complex128 solves on the HIP backend: hand-written double2
    kernels for the whole solve phase (host complex setup, per-level move —
    the reference's own layout for its device backends over std::complex)."""
    import torch

    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    A, m = helmholtz(14)
    rng = np.random.default_rng(0)
    b = rng.standard_normal(A.nrows) + 1j * rng.standard_normal(A.nrows)
    s = am.make_solver(
        A, {"precond": {"class": "amg", "coarse_enough": 400},
            "solver": {"type": solver, "tol": 1e-8, "maxiter": 200}},
        backend="hip")
    from amgcl_amd.backend.hip import HipBackend

    assert isinstance(s.backend, HipBackend)
    assert s.backend.dtype == torch.complex128
    x, iters, resid = s(b)
    assert resid < 1e-8
    xh = s.backend.to_host(x)
    assert np.iscomplexobj(xh)
    assert np.linalg.norm(b - m @ xh) / np.linalg.norm(b) < 1e-7
    assert iters < 60

    # matches the CPU-native complex solve
    s2 = am.make_solver(
        A, {"precond": {"class": "amg", "coarse_enough": 400},
            "solver": {"type": solver, "tol": 1e-8, "maxiter": 200}})
    x2, it2, r2 = s2(b)
    assert abs(iters - it2) <= 2, (iters, it2)


def test_complex_unsupported_components_fail_loudly():
    """Real-only components reject complex systems with clear errors
    instead of silently discarding imaginary parts."""
    A, m = helmholtz(14)
    b = np.ones(A.nrows, dtype=complex)
    for prm, err in (
        ({"coarsening": {"type": "ruge_stuben"}}, "ruge_stuben"),
        ({"relax": {"type": "ilu0"}}, "ILU"),
        ({"coarsening": {"type": "smoothed_aggr_emin"}}, "emin"),
    ):
        with pytest.raises((ValueError, TypeError)):
            am.make_solver(A, {"precond": {"class": "amg", "coarse_enough": 300,
                                           **prm},
                               "solver": {"type": "bicgstab"}})


def test_complex_chebyshev_smoother():
    A, m = helmholtz(14)
    rng = np.random.default_rng(5)
    b = rng.standard_normal(A.nrows) + 1j * rng.standard_normal(A.nrows)
    s = am.make_solver(A, {"precond": {"class": "amg", "coarse_enough": 300,
                                       "relax": {"type": "chebyshev"}},
                           "solver": {"type": "bicgstab", "tol": 1e-8,
                                      "maxiter": 300}})
    x, it, r = s(b)
    assert r < 1e-8
    assert np.linalg.norm(b - m @ x) / np.linalg.norm(b) < 1e-7
