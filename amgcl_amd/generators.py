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


def poisson3d_box_strip(nx, ny, nz, rank, nranks, rhs="ones"):
    """Row strip of an nx*ny*nz 7-point Poisson box, cut on z-planes when the
    grid divides evenly — the weak-scaling fixture (fixed per-rank cube,
    domain grown along z with the rank count)."""
    ntot = nx * ny * nz
    row_beg = (ntot * rank) // nranks
    row_end = (ntot * (rank + 1)) // nranks
    ptr, col, val = _core.poisson3d_box_strip(int(nx), int(ny), int(nz),
                                              int(row_beg), int(row_end))
    strip = CSR(row_end - row_beg, ntot, ptr, col, val)
    if rhs == "ones":
        b = np.ones(row_end - row_beg, dtype=np.float64)
    else:
        rng = np.random.default_rng(1234 + rank)
        b = rng.standard_normal(row_end - row_beg)
    return strip, b, row_beg, row_end


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


def rigid_body_modes(coords, transpose=False):
    """Near-nullspace block from node coordinates for elasticity
    (parity: amgcl/coarsening/rigid_body_modes.hpp:45): 3 translations +
    3 rotations in 3D (6 columns), 2+1 in 2D. coords: (nnodes, dim);
    unknowns are interleaved (node-major, dim dofs per node)."""
    coords = np.asarray(coords, dtype=np.float64)
    nnodes, dim = coords.shape
    n = nnodes * dim
    if dim == 3:
        k = 6
        B = np.zeros((n, k))
        x, y, z = coords[:, 0], coords[:, 1], coords[:, 2]
        B[0::3, 0] = 1.0
        B[1::3, 1] = 1.0
        B[2::3, 2] = 1.0
        B[0::3, 3], B[1::3, 3] = -y, x          # rotation about z
        B[1::3, 4], B[2::3, 4] = -z, y          # rotation about x
        B[0::3, 5], B[2::3, 5] = z, -x          # rotation about y
    elif dim == 2:
        k = 3
        B = np.zeros((n, k))
        x, y = coords[:, 0], coords[:, 1]
        B[0::2, 0] = 1.0
        B[1::2, 1] = 1.0
        B[0::2, 2], B[1::2, 2] = -y, x
    else:
        raise ValueError("rigid body modes need 2D or 3D coordinates")
    return B


def elasticity3d(n, E=1.0, nu=0.3, rhs="random"):
    """3D linear elasticity on an n^3 node grid of hex elements, node-major
    interleaved dofs (ux, uy, uz per node), Dirichlet on the z=0 face.
    Returns (A CSR, b, coords). Synthetic FEM fixture for the nullspace /
    block-value configurations (BASELINE.json config #3)."""
    import scipy.sparse as sp

    # 8-node hex element stiffness via 2x2x2 Gauss quadrature
    lam = E * nu / ((1 + nu) * (1 - 2 * nu))
    mu = E / (2 * (1 + nu))
    D = np.zeros((6, 6))
    D[:3, :3] = lam
    D[np.arange(3), np.arange(3)] += 2 * mu
    D[3:, 3:] = mu * np.eye(3)
    g = 1.0 / np.sqrt(3.0)
    pts = np.array([[i, j, k] for k in (-g, g) for j in (-g, g) for i in (-g, g)])
    corners = np.array([[i, j, k] for k in (-1, 1) for j in (-1, 1) for i in (-1, 1)],
                       dtype=np.float64)
    Ke = np.zeros((24, 24))
    for gp in pts:
        dN = np.zeros((8, 3))
        for a in range(8):
            xi = corners[a]
            dN[a, 0] = 0.125 * xi[0] * (1 + xi[1] * gp[1]) * (1 + xi[2] * gp[2])
            dN[a, 1] = 0.125 * xi[1] * (1 + xi[0] * gp[0]) * (1 + xi[2] * gp[2])
            dN[a, 2] = 0.125 * xi[2] * (1 + xi[0] * gp[0]) * (1 + xi[1] * gp[1])
        # element is the bi-unit cube: J = I (h=2); scale dropped (h factors
        # only rescale the condition number of the fixture)
        Bm = np.zeros((6, 24))
        for a in range(8):
            bx, by, bz = dN[a]
            c = 3 * a
            Bm[0, c] = bx
            Bm[1, c + 1] = by
            Bm[2, c + 2] = bz
            Bm[3, c], Bm[3, c + 1] = by, bx
            Bm[4, c + 1], Bm[4, c + 2] = bz, by
            Bm[5, c], Bm[5, c + 2] = bz, bx
        Ke += Bm.T @ D @ Bm
    # assemble
    idx = lambda i, j, k: (k * n + j) * n + i
    nel = (n - 1) ** 3
    conn = np.empty((nel, 8), dtype=np.int64)
    e = 0
    for k in range(n - 1):
        for j in range(n - 1):
            for i in range(n - 1):
                conn[e] = [idx(i, j, k), idx(i + 1, j, k), idx(i, j + 1, k),
                           idx(i + 1, j + 1, k), idx(i, j, k + 1), idx(i + 1, j, k + 1),
                           idx(i, j + 1, k + 1), idx(i + 1, j + 1, k + 1)]
                e += 1
    dofs = (conn[:, :, None] * 3 + np.arange(3)).reshape(nel, 24)
    rows = np.repeat(dofs, 24, axis=1).ravel()
    cols = np.tile(dofs, (1, 24)).ravel()
    vals = np.tile(Ke.ravel(), nel)
    N = 3 * n**3
    A = sp.coo_matrix((vals, (rows, cols)), shape=(N, N)).tocsr()
    # Dirichlet on z=0 nodes (clamp): zero rows/cols, unit diagonal
    fixed_nodes = np.where(np.arange(n**3) < n * n)[0]
    fixed = (fixed_nodes[:, None] * 3 + np.arange(3)).ravel()
    mask = np.ones(N, dtype=bool)
    mask[fixed] = False
    di = sp.diags(mask.astype(np.float64))
    A = (di @ A @ di + sp.diags((~mask).astype(np.float64))).tocsr()
    A.sort_indices()
    A.eliminate_zeros()
    ii, jj, kk = np.meshgrid(range(n), range(n), range(n), indexing="ij")
    coords = np.stack([ii.ravel(order="F"), jj.ravel(order="F"),
                       kk.ravel(order="F")], axis=1).astype(np.float64)
    rng = np.random.default_rng(9)
    b = rng.standard_normal(N) if rhs == "random" else np.ones(N)
    b[fixed] = 0.0
    from .matrix import CSR

    return CSR.from_scipy(A), b, coords
