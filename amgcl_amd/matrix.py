"""Host CSR matrix (fp64 values, int32 indices) and setup-phase algebra.

The canonical host container mirroring the reference's backend::crs
(amgcl/backend/builtin.hpp:61): raw ptr/col/val arrays, with transpose /
product / diagonal implemented by the OpenMP C++ engine (_core).
The AMG hierarchy is always assembled in this format on the host and then
moved to a compute backend — the reference's core architectural invariant
(docs/design.rst:20-38).
"""
import numpy as np

from . import _core


class CSR:
    __slots__ = ("nrows", "ncols", "ptr", "col", "val")

    def __init__(self, nrows, ncols, ptr, col, val):
        self.nrows = int(nrows)
        self.ncols = int(ncols)
        self.ptr = np.ascontiguousarray(ptr, dtype=np.int32)
        self.col = np.ascontiguousarray(col, dtype=np.int32)
        # complex values are only carried by IO/adapters until
        # adapter.complex_to_real expands them; casting them to fp64 here
        # would silently drop the imaginary part
        dt = np.complex128 if np.iscomplexobj(np.asarray(val)) else np.float64
        self.val = np.ascontiguousarray(val, dtype=dt)

    @property
    def nnz(self):
        return int(self.col.size)

    @property
    def shape(self):
        return (self.nrows, self.ncols)

    def bytes(self):
        return self.ptr.nbytes + self.col.nbytes + self.val.nbytes

    @property
    def is_complex(self):
        return np.iscomplexobj(self.val)

    def diagonal(self):
        if self.is_complex:
            return self.to_scipy().diagonal()
        return _core.diagonal(self.nrows, self.ptr, self.col, self.val)

    def transpose(self):
        """Adjoint transpose (parity: backend::transpose applies
        math::adjoint — for complex values this is the conjugate
        transpose, amgcl/backend/builtin.hpp:348)."""
        if self.is_complex:
            m = self.to_scipy().conj().T.tocsr()
            m.sort_indices()
            return CSR.from_scipy(m)
        tp, tc, tv = _core.transpose(self.nrows, self.ncols, self.ptr, self.col, self.val)
        return CSR(self.ncols, self.nrows, tp, tc, tv)

    def __matmul__(self, other):
        if isinstance(other, CSR):
            if self.ncols != other.nrows:
                raise ValueError("dimension mismatch in CSR product")
            if self.is_complex or other.is_complex:
                m = (self.to_scipy() @ other.to_scipy()).tocsr()
                m.sort_indices()
                return CSR.from_scipy(m)
            cp, cc, cv = _core.spgemm(
                self.nrows, self.ncols, other.ncols,
                self.ptr, self.col, self.val,
                other.ptr, other.col, other.val,
            )
            return CSR(self.nrows, other.ncols, cp, cc, cv)
        if self.is_complex or np.iscomplexobj(other):
            return self.to_scipy() @ np.asarray(other)
        x = np.ascontiguousarray(other, dtype=np.float64)
        y = np.empty(self.nrows, dtype=np.float64)
        _core.spmv(1.0, self.nrows, self.ptr, self.col, self.val, x, 0.0, y)
        return y

    def spmv(self, alpha, x, beta, y):
        if self.is_complex or np.iscomplexobj(x):
            if beta == 0.0:
                np.copyto(y, alpha * (self.to_scipy() @ x))
            else:
                y *= beta
                y += alpha * (self.to_scipy() @ x)
            return
        _core.spmv(alpha, self.nrows, self.ptr, self.col, self.val, x, beta, y)

    def residual(self, b, x, r):
        if self.is_complex or np.iscomplexobj(x):
            np.copyto(r, b - self.to_scipy() @ x)
            return
        _core.residual(self.nrows, self.ptr, self.col, self.val, b, x, r)

    def to_scipy(self):
        import scipy.sparse as sp

        return sp.csr_matrix((self.val, self.col, self.ptr), shape=self.shape)

    @staticmethod
    def from_scipy(m):
        m = m.tocsr()
        return CSR(m.shape[0], m.shape[1], m.indptr, m.indices, m.data)

    @staticmethod
    def from_dense(a):
        a = np.asarray(a, dtype=np.float64)
        nrows, ncols = a.shape
        mask = a != 0.0
        counts = mask.sum(axis=1)
        ptr = np.zeros(nrows + 1, dtype=np.int64)
        np.cumsum(counts, out=ptr[1:])
        col = np.nonzero(mask)[1].astype(np.int32)
        val = a[mask]
        return CSR(nrows, ncols, ptr, col, val)

    def to_dense(self):
        a = np.zeros(self.shape, dtype=np.float64)
        for i in range(self.nrows):
            for j in range(self.ptr[i], self.ptr[i + 1]):
                a[i, self.col[j]] = self.val[j]
        return a

    def __repr__(self):
        return f"CSR({self.nrows}x{self.ncols}, nnz={self.nnz})"


def galerkin(R, A, P):
    """Coarse operator Ac = R*A*P via two SpGEMMs
    (parity: amgcl/coarsening/detail/galerkin.hpp:42).
    Runs on the device (backend/hip_setup.py spgemm) when the operands are
    device-resident.  The association R*(A*P) is a measured choice, not just
    reference parity: (R*A)*P looks cheaper on paper (each A entry inserted
    once into an aggregate row) but its intermediate lives in FINE column
    space, so its rows are fat (union of whole-aggregate neighborhoods →
    wave-per-row hash tier, and LDS overflow past level 0) while A*P's rows
    live in coarse space and stay in the fast small-bin tier.  Measured on
    512^3 Poisson level 0: A*P + R*(AP) = 59+85 ms vs R*A + (RA)*P =
    200+154 ms at identical intermediate nnz (scripts/galerkin_order.py)."""
    if not isinstance(A, CSR):
        from .backend import hip_setup
        from .profiler import prof

        try:
            with prof.scope("galerkin(dev)"):
                return hip_setup.spgemm(R, hip_setup.spgemm(A, P, sort=False))
        except OverflowError:
            Rh = hip_setup.download(R)
            Ah = hip_setup.download(A)
            Ph = hip_setup.download(P)
            return Rh @ (Ah @ Ph)
    return R @ (A @ P)
