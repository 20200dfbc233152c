"""Synthetic problem generators (test/benchmark fixtures).

poisson3d mirrors the reference fixture (tests/sample_problem.hpp:11): 7-point
stencil on an n^3 unit cube, Dirichlet boundaries via truncated stencils,
optional anisotropy. Used by the combinatorial convergence tests and by
bench.py (BASELINE.json: synthetic 7-pt 3D Poisson, random RHS).
"""
import numpy as np

from . import _core
from .matrix import CSR


def poisson3d(n, anisotropy=1.0, rhs="ones"):
    ptr, col, val = _core.poisson3d(int(n), float(anisotropy))
    A = CSR(n**3, n**3, ptr, col, val)
    if rhs == "ones":
        b = np.ones(n**3, dtype=np.float64)
    elif rhs == "random":
        b = np.random.default_rng(42).standard_normal(n**3)
    else:
        b = None
    return A, b


def poisson3d_strip(n, rank, nranks, rhs="ones"):
    """Row strip [row_beg, row_end) of the global n^3 Poisson matrix, for
    distributed tests (parity: examples/mpi/mpi_solver.cpp:47 assembles the
    local strip per rank). Returns (A_strip, b_strip, row_beg, row_end) where
    A_strip has global column indices."""
    n3 = n**3
    row_beg = (n3 * rank) // nranks
    row_end = (n3 * (rank + 1)) // nranks
    ptr, col, val = _core.poisson3d_strip(int(n), int(row_beg), int(row_end))
    strip = CSR(row_end - row_beg, n3, ptr, col, val)
    if rhs == "ones":
        b = np.ones(row_end - row_beg, dtype=np.float64)
    else:
        b = np.random.default_rng(42 + rank).standard_normal(row_end - row_beg)
    return strip, b, row_beg, row_end
