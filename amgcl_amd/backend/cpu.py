"""OpenMP CPU backend.

Parity with the reference 'builtin' backend (amgcl/backend/builtin.hpp:918):
vectors are numpy float64 arrays, matrices are host CSR, SpMV/residual run
through the OpenMP C++ engine, vector math through numpy. This backend is
the numerics reference the HIP kernels are tested against.
"""
import numpy as np

from .. import _core
from ..matrix import CSR
from . import register


@register("cpu")
class CpuBackend:
    name = "cpu"
    device = "cpu"

    def __init__(self, dtype=np.float64):
        if dtype not in (np.float64, np.complex128):
            raise ValueError("cpu backend supports fp64 and complex128")
        self.dtype = dtype

    # --- containers -------------------------------------------------------
    def matrix(self, csr: CSR):
        return csr

    def vector(self, n):
        return np.zeros(n, dtype=self.dtype)

    def from_host(self, a):
        return np.array(a, dtype=self.dtype, copy=True)

    def to_host(self, v):
        return np.asarray(v)

    # --- primitives (amgcl/backend/interface.hpp:253-443) ----------------
    def spmv(self, alpha, A, x, beta, y):
        if np.iscomplexobj(A.val) or np.iscomplexobj(x):
            A.spmv(alpha, x, beta, y)
            return
        _core.spmv(alpha, A.nrows, A.ptr, A.col, A.val, x, beta, y)

    def residual(self, b, A, x, r):
        if np.iscomplexobj(A.val) or np.iscomplexobj(x):
            A.residual(b, x, r)
            return
        _core.residual(A.nrows, A.ptr, A.col, A.val, b, x, r)

    def clear(self, x):
        x.fill(0.0)

    def copy(self, x, y):
        np.copyto(y, x)

    def dot(self, x, y):
        if np.iscomplexobj(x) or np.iscomplexobj(y):
            # adjoint inner product (math::inner_product conjugates the
            # first argument for complex value types)
            return complex(np.vdot(x, y))
        return float(np.dot(x, y))

    def dot2(self, x1, y1, x2, y2):
        return self.dot(x1, y1), self.dot(x2, y2)

    def norm(self, x):
        if np.iscomplexobj(x):
            return float(np.linalg.norm(x))
        return float(np.sqrt(np.dot(x, x)))

    def axpby(self, a, x, b, y):
        # y = a*x + b*y
        if b == 0.0:
            np.multiply(x, a, out=y)
        else:
            y *= b
            y += a * x

    def axpbypcz(self, a, x, b, y, c, z):
        # z = a*x + b*y + c*z
        if c == 0.0:
            np.multiply(x, a, out=z)
        else:
            z *= c
            z += a * x
        z += b * y

    def vmul(self, a, m, x, b, z):
        # z = a*(m ∘ x) + b*z
        if b == 0.0:
            np.multiply(m, x, out=z)
            if a != 1.0:
                z *= a
        else:
            z *= b
            z += a * (m * x)

    def gather(self, x, idx, buf):
        np.take(x, idx, out=buf)

    def scatter(self, buf, idx, x):
        x[idx] = buf

    # --- coarse direct solver --------------------------------------------
    def coarse_solver(self, csr: CSR, kind="dense"):
        if kind == "splu":
            return SpluCoarseSolver(csr, self)
        if kind == "skyline":
            return SkylineCoarseSolver(csr, self)
        if kind != "dense":
            raise ValueError(f"unknown direct_solver '{kind}' "
                             "(dense, skyline, splu)")
        return DenseCoarseSolver(csr, self)

    def synchronize(self):
        pass


class SkylineCoarseSolver:
    """Coarsest-level direct solve by skyline (profile) LU after a
    Cuthill-McKee reorder (parity: amgcl/solver/skyline_lu.hpp:85, the
    reference's default coarse solver): memory O(profile) instead of the
    dense inverse's O(n^2), factorization in native C++ (_core)."""

    def __init__(self, csr: CSR, backend=None):
        from .. import _core
        from ..adapter import Reordered

        self._re = Reordered(csr)
        Ap = self._re.A
        self.n = csr.nrows
        self._fac = _core.skyline_factor(Ap.nrows, Ap.ptr, Ap.col, Ap.val)

    def __call__(self, f, u):
        from .. import _core

        y = _core.skyline_solve(*self._fac, self._re.forward(np.asarray(f)))
        np.copyto(u, self._re.inverse(y))

    def bytes(self):
        sp = self._fac[0]
        return int(sp[-1]) * 16 + self.n * 8


class SpluCoarseSolver:
    """Alternative coarsest-level direct solve: scipy SuperLU factorization
    (parity: solver/eigen.hpp — Eigen SparseLU as alternative coarse solver).
    Preferable to the dense inverse when coarse_enough is large and the
    coarse operator is still very sparse."""

    def __init__(self, csr: CSR, backend=None):
        import scipy.sparse.linalg as spla

        self._lu = spla.splu(csr.to_scipy().tocsc())

    def __call__(self, f, u):
        u[:] = self._lu.solve(np.asarray(f))


class DenseCoarseSolver:
    """Coarsest-level direct solve via a precomputed dense inverse.

    The reference uses a host skyline LU (amgcl/solver/skyline_lu.hpp:85)
    with a D2H/H2D round-trip per cycle on GPU backends (backend/hip.hpp:73-96).
    The MI355X-native design instead inverts the (<= a few thousand rows)
    coarse operator once at setup and applies it as a dense GEMV, which stays
    device-resident on the HIP backend.
    """

    def __init__(self, csr: CSR, backend):
        self.inv = np.linalg.inv(csr.to_scipy().toarray())

    def __call__(self, f, u):
        np.matmul(self.inv, f, out=u)
