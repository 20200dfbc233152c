"""Matrix adapters (parity: amgcl/adapter/).

- zero_copy:        wrap raw ptr/col/val arrays without copying
  (adapter/zero_copy.hpp:44)
- reorder:          apply a (reverse) Cuthill-McKee permutation to the system
  (adapter/reorder.hpp:48, reorder/cuthill_mckee.hpp:79)
- scaled_problem:   symmetric diagonal scaling D^-1/2 A D^-1/2
  (adapter/scaled_problem.hpp:61)
- block_matrix:     view a scalar CSR as a block-valued one
  (adapter/block_matrix.hpp:44) — see blockval module
"""
from collections import deque

import numpy as np

from .matrix import CSR


def zero_copy(n, ptr, col, val):
    """Non-owning CSR view over user arrays (no copy when dtypes match)."""
    a = CSR.__new__(CSR)
    a.nrows = int(n)
    a.ncols = int(n)
    a.ptr = np.asarray(ptr, dtype=np.int32)
    a.col = np.asarray(col, dtype=np.int32)
    a.val = np.asarray(val, dtype=np.float64)
    return a


def cuthill_mckee(A: CSR, reverse=True):
    """(Reverse) Cuthill-McKee ordering: returns permutation `perm` such that
    row perm[k] of A becomes row k of the reordered matrix."""
    n = A.nrows
    degree = np.diff(A.ptr)
    visited = np.zeros(n, dtype=bool)
    perm = np.empty(n, dtype=np.int64)
    pos = 0
    order = np.argsort(degree, kind="stable")
    for seed in order:
        if visited[seed]:
            continue
        visited[seed] = True
        q = deque([seed])
        while q:
            v = q.popleft()
            perm[pos] = v
            pos += 1
            nbrs = A.col[A.ptr[v] : A.ptr[v + 1]]
            nbrs = [c for c in nbrs if not visited[c]]
            nbrs.sort(key=lambda c: degree[c])
            for c in nbrs:
                visited[c] = True
                q.append(c)
    if reverse:
        perm = perm[::-1].copy()
    return perm


class Reordered:
    """Reordered system: solves P A P^T (P x) = P b (parity: adapter/reorder)."""

    def __init__(self, A: CSR, reverse=True):
        self.perm = cuthill_mckee(A, reverse)
        self.iperm = np.empty_like(self.perm)
        self.iperm[self.perm] = np.arange(A.nrows)
        m = A.to_scipy()[self.perm][:, self.perm].tocsr()
        m.sort_indices()
        self.A = CSR(A.nrows, A.ncols, m.indptr, m.indices, m.data)

    def forward(self, v):
        return np.asarray(v)[self.perm]

    def inverse(self, v):
        return np.asarray(v)[self.iperm]


class ScaledProblem:
    """Symmetric diagonal scaling: solve (D^-1/2 A D^-1/2) y = D^-1/2 b,
    x = D^-1/2 y (parity: adapter/scaled_problem.hpp:61)."""

    def __init__(self, A: CSR):
        d = np.asarray(A.diagonal())
        if np.any(d <= 0):
            raise ValueError("scaled_problem needs positive diagonal")
        self.dsqrt_inv = 1.0 / np.sqrt(d)
        row_of = np.repeat(np.arange(A.nrows), np.diff(A.ptr))
        val = A.val * self.dsqrt_inv[row_of] * self.dsqrt_inv[A.col]
        self.A = CSR(A.nrows, A.ncols, A.ptr, A.col, val)

    def scale_rhs(self, b):
        return np.asarray(b) * self.dsqrt_inv

    def unscale_x(self, y):
        return np.asarray(y) * self.dsqrt_inv


def complex_to_real(A, rhs=None):
    """View a complex system as its 2x2-real expansion
    (parity: amgcl/adapter/complex.hpp:45): each complex entry a+bi becomes
    the block [[a, -b], [b, a]]; complex unknown z = u+vi becomes (u, v).
    Returns (A_real CSR, rhs_real) — solve, then recombine with
    real_to_complex()."""
    import scipy.sparse as sp

    if isinstance(A, CSR):
        m = A.to_scipy().astype(np.complex128)
    else:
        m = sp.csr_matrix(A, dtype=np.complex128)
    re, im = m.real.tocoo(), m.imag.tocoo()
    rows = np.concatenate([2 * re.row, 2 * re.row, 2 * re.row + 1, 2 * re.row + 1])
    cols = np.concatenate([2 * re.col, 2 * re.col + 1, 2 * re.col, 2 * re.col + 1])
    vals = np.concatenate([re.data, np.zeros_like(re.data), np.zeros_like(re.data),
                           re.data])
    rows = np.concatenate([rows, 2 * im.row, 2 * im.row + 1])
    cols = np.concatenate([cols, 2 * im.col + 1, 2 * im.col])
    vals = np.concatenate([vals, -im.data, im.data])
    n2 = 2 * m.shape[0]
    mr = sp.coo_matrix((vals, (rows, cols)), shape=(n2, 2 * m.shape[1])).tocsr()
    mr.sum_duplicates()
    mr.sort_indices()
    A_real = CSR.from_scipy(mr)
    if rhs is None:
        return A_real, None
    rhs = np.asarray(rhs, dtype=np.complex128)
    b = np.empty(n2)
    b[0::2], b[1::2] = rhs.real, rhs.imag
    return A_real, b


def real_to_complex(x):
    """Recombine the 2x2-real-expanded solution into complex."""
    x = np.asarray(x)
    return x[0::2] + 1j * x[1::2]


class CrsBuilder:
    """Matrix-free row-generator assembly
    (parity: amgcl/adapter/crs_builder.hpp:105): build a CSR from a callable
    row(i) -> (cols, vals)."""

    def __init__(self, n, row_func):
        ptr = np.zeros(n + 1, dtype=np.int64)
        cols, vals = [], []
        for i in range(n):
            c, v = row_func(i)
            cols.append(np.asarray(c, dtype=np.int32))
            vals.append(np.asarray(v, dtype=np.float64))
            ptr[i + 1] = ptr[i] + len(c)
        self.matrix = CSR(n, n, ptr, np.concatenate(cols), np.concatenate(vals))
