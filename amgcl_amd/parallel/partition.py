"""Geometric domain partitioning (recursive coordinate bisection).

Parity: amgcl/mpi/partition/{ptscotch,parmetis}.hpp provide graph
partitioners for the initial decomposition and coarse-level repartitioning.
Neither library ships in a ROCm image, and for the PDE problems this
framework targets the classic geometric alternative — recursive coordinate
bisection over the node coordinates — produces comparable subdomain quality
(compact boxes, low halo surface) deterministically and with no extra
dependency. Coarse-level repartitioning itself is handled by DistAMG's
replicated tail (see parallel/dist_amg.py).

Usage (before building the distributed solver):

    part  = rcb_partition(coords, world)            # part id per node
    perm, sizes = partition_permutation(part)       # contiguous strips
    A_perm = permute_system(A, perm)                # global reorder
    # rank r owns rows [sum(sizes[:r]), sum(sizes[:r+1])) of A_perm
"""
import numpy as np

from ..matrix import CSR


def rcb_partition(coords, nparts):
    """Recursive coordinate bisection: split the widest axis at the weighted
    median until `nparts` parts remain. Handles any nparts >= 1 (splits
    proportionally for non-powers-of-two). Deterministic."""
    coords = np.asarray(coords, dtype=np.float64)
    if coords.ndim == 1:
        coords = coords[:, None]
    n = coords.shape[0]
    part = np.zeros(n, dtype=np.int32)

    def split(idx, lo, hi):
        count = hi - lo
        if count <= 1 or len(idx) == 0:
            part[idx] = lo
            return
        left = count // 2
        frac = left / count
        c = coords[idx]
        widths = c.max(axis=0) - c.min(axis=0)
        ax = int(np.argmax(widths))
        order = np.argsort(c[:, ax], kind="stable")
        k = int(round(frac * len(idx)))
        split(idx[order[:k]], lo, lo + left)
        split(idx[order[k:]], lo + left, hi)

    split(np.arange(n, dtype=np.int64), 0, nparts)
    return part


def partition_permutation(part):
    """Permutation making each part a contiguous row strip (stable within a
    part) plus the per-part sizes."""
    part = np.asarray(part)
    perm = np.argsort(part, kind="stable").astype(np.int64)
    sizes = np.bincount(part, minlength=int(part.max()) + 1).tolist()
    return perm, sizes


def permute_system(A: CSR, perm, b=None):
    """Symmetric permutation P A P^T (+ permuted rhs)."""
    iperm = np.empty_like(perm)
    iperm[perm] = np.arange(len(perm))
    m = A.to_scipy()[perm][:, perm].tocsr()
    m.sort_indices()
    Ap = CSR(A.nrows, A.ncols, m.indptr, m.indices, m.data)
    if b is None:
        return Ap
    return Ap, np.asarray(b)[perm]


def edge_cut(A: CSR, part):
    """Number of nonzeros coupling different parts (halo volume proxy)."""
    part = np.asarray(part)
    row_of = np.repeat(np.arange(A.nrows), np.diff(A.ptr))
    return int(np.sum(part[row_of] != part[np.asarray(A.col)]))
