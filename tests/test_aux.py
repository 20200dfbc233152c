"""Auxiliary subsystems: tracing/profiler, failure detection,
verbose solver output, hierarchy report, runtime registries."""
import os

import numpy as np
import pytest

import amgcl_amd as am
from amgcl_amd.matrix import CSR


def test_profiler_scopes_and_report():
    from amgcl_amd.profiler import Profiler

    p = Profiler("t")
    with p.scope("outer"):
        with p.scope("inner"):
            pass
    rep = p.report()
    assert "outer" in rep and "inner" in rep and "[t]" in rep


def test_zero_pivot_ilu_raises():
    """Failure detection: singular pivot aborts the factorization loudly
    (reference throws in ilu0.hpp when the diagonal vanishes)."""
    import scipy.sparse as sp

    # 2x2 with a zero pivot after elimination: [[1, 1], [1, 1]]
    A = CSR.from_scipy(sp.csr_matrix(np.array([[1.0, 1.0], [1.0, 1.0]])))
    from amgcl_amd import _core

    with pytest.raises(RuntimeError, match="pivot"):
        _core.ilu0_factor(A.nrows, A.ptr, A.col, A.val)


def test_empty_aggregation_raises():
    """A diagonal matrix has no strong connections: the aggregation must
    refuse to build an empty level instead of looping."""
    import scipy.sparse as sp

    from amgcl_amd import _core

    A = CSR.from_scipy(sp.identity(50, format="csr"))
    with pytest.raises(RuntimeError, match="empty level"):
        _core.aggregates_parallel(A.nrows, A.ptr, A.col, A.val, 0.08)


def test_verbose_solver_prints_iterations(capsys):
    A, b = am.poisson3d(10, rhs="random")
    s = am.make_solver(A, {"solver": {"type": "cg", "tol": 1e-8,
                                      "maxiter": 100, "verbose": True},
                           "precond": {"class": "amg", "coarse_enough": 200}})
    s(b)
    out = capsys.readouterr().out
    assert out.strip()  # per-iteration residual lines


def test_hierarchy_report_complexities():
    A, _ = am.poisson3d(16)
    s = am.make_solver(A, {"precond": {"class": "amg", "coarse_enough": 300}})
    rep = str(s)
    assert "operator complexity" in rep
    assert "grid complexity" in rep
    assert "unknowns" in rep


def test_runtime_registries_reject_unknown():
    A, b = am.poisson3d(8)
    with pytest.raises(ValueError, match="unknown"):
        am.make_solver(A, {"solver": {"type": "does_not_exist"}})
    with pytest.raises(ValueError, match="unknown"):
        am.make_solver(A, {"precond": {"class": "amg",
                                       "relax": {"type": "nope"}}})


def test_missing_hip_lib_fails_loudly(monkeypatch, tmp_path):
    """GPU ops must never fall back silently: a missing kernel library is a
    hard error (round-end 'native code not loaded' guard)."""
    import amgcl_amd.backend._hiplib as hl

    monkeypatch.setattr(hl, "_LIB", None)
    monkeypatch.setattr(hl.os.path, "exists", lambda p: False)

    def no_build():
        return None

    import amgcl_amd.build as bld

    monkeypatch.setattr(bld, "build_hip_lib", no_build)
    with pytest.raises(RuntimeError, match="libamghip.so not found"):
        hl.lib()


@pytest.mark.parametrize("name,args", [
    ("poisson", ["32"]),
    ("mixed_precision", ["48"]),
    ("schur_stokes", ["12"]),
    ("cpr_reservoir", []),
    ("complex_helmholtz", []),
    ("elasticity_nullspace", ["10"]),
])
def test_examples_run(name, args):
    """Every Python example executes end-to-end on CPU at a small size —
    catches example bit-rot (the reference ships 25 buildable examples;
    these are their working equivalents)."""
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(root, "examples", f"{name}.py"), *args],
        capture_output=True, text=True, timeout=420, cwd=root)
    assert out.returncode == 0, out.stdout + out.stderr


def test_distributed_example_runs():
    """examples/distributed_solver.py under torchrun (2 ranks, gloo on CPU) —
    the reference's mpi_solver example equivalent."""
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29641",
         os.path.join(root, "examples", "distributed_solver.py"), "24"],
        capture_output=True, text=True, timeout=420, cwd=root)
    assert out.returncode == 0, out.stdout + out.stderr


@pytest.mark.parametrize("prm,msg", [
    ({"precond": {"class": "amg", "precision": "bogus"}}, "precision"),
    ({"precond": {"class": "amg", "block_value": -3}}, "block_value"),
    ({"precond": {"class": "amg", "ncycle": 0}}, "ncycle"),
    ({"precond": {"class": "amg", "npre": -1}}, "npre"),
    ({"precond": {"class": "amg", "max_levels": 0}}, "max_levels"),
    ({"precond": {"class": "amg", "direct_solver": "nope"}}, "direct_solver"),
])
def test_degenerate_params_rejected(prm, msg):
    """Degenerate parameter values fail loudly at construction instead of
    producing a silently misconfigured hierarchy."""
    A, b = am.poisson3d(8)
    with pytest.raises(ValueError, match=msg):
        am.make_solver(A, prm)


@pytest.mark.parametrize("prm,msg", [
    ({"solver": {"type": "gmres", "M": 0}}, "restart M"),
    ({"solver": {"type": "fgmres", "M": 0}}, "restart M"),
    ({"solver": {"type": "lgmres", "M": 0}}, "restart M"),
    ({"solver": {"type": "idrs", "s": 0}}, "shadow-space"),
    ({"precond": {"class": "amg", "coarse_enough": 100,
                  "relax": {"type": "chebyshev", "degree": 0}}},
     "degree"),  # coarse_enough < n so a smoothed level actually exists
])
def test_degenerate_solver_params_rejected(prm, msg):
    """gmres(M=0) previously looped forever (zero Krylov steps per restart);
    these all fail loudly now."""
    A, b = am.poisson3d(8)
    with pytest.raises(ValueError, match=msg):
        am.make_solver(A, prm)


def test_truncated_mm_rejected(tmp_path):
    p = str(tmp_path / "t.mtx")
    with open(p, "w") as f:
        f.write("%%MatrixMarket matrix coordinate real general\n3 3 5\n1 1 2.0\n")
    from amgcl_amd import io

    with pytest.raises(ValueError, match="truncated"):
        io.mm_read(p)


def test_cpr_zero_block_size_rejected():
    A, _ = am.poisson3d(8)
    with pytest.raises(ValueError, match="block_size"):
        am.make_solver(A, {"precond": {"class": "cpr", "block_size": 0}})


def test_non_square_rejected():
    import scipy.sparse as sp

    from amgcl_amd.matrix import CSR

    ns = sp.random(6, 4, density=0.5, format="csr")
    with pytest.raises(ValueError, match="square"):
        am.make_solver(CSR.from_scipy(ns), {})


def test_wrong_size_rhs_rejected():
    """A mismatched rhs previously returned (iters=0, resid=nan) silently."""
    A, _ = am.poisson3d(8)
    s = am.make_solver(A, {})
    with pytest.raises(ValueError, match="rhs"):
        s(np.ones(100))


def test_truncated_binary_rejected(tmp_path):
    from amgcl_amd import io

    A, _ = am.poisson3d(6)
    p = str(tmp_path / "a.bin")
    io.write_crs(p, A)
    data = open(p, "rb").read()
    q = str(tmp_path / "trunc.bin")
    open(q, "wb").write(data[: len(data) // 2])
    with pytest.raises(ValueError, match="truncated"):
        io.read_crs(q)
