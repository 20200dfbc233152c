"""Block-valued ILU(0) smoother.

Parity: amgcl/relaxation/ilu0.hpp:51 instantiated over
``static_matrix<double,B,B>`` (value_type/static_matrix.hpp) — the
reference's block-valued ILU route used by the CoupCons3D tutorial.  The
factorization runs over b×b blocks on the host (`_core.block_ilu0_factor`,
inverted diagonal blocks via Gauss-Jordan = detail/inverse.hpp:45); the
apply is serial block sweeps on the CPU backend and, on the GPU, the
reference's damped-Jacobi iterated approximate triangular solves
(detail/ilu_solve.hpp:44-124) composed from the BSR SpMV and
block-diagonal-multiply kernels (csrc/hip/block.hip).
"""
import numpy as np

from .. import _core
from ..matrix import CSR
from ..params import merge_params


class BlockILU0:
    gpu_supported = True

    @staticmethod
    def defaults():
        return {"block_size": 2, "damping": 1.0,
                "solve_iters": 2, "solve_damping": 0.72}

    def __init__(self, A, prm, backend):
        if not isinstance(A, CSR):
            from ..backend import hip_setup

            A = hip_setup.download(A)
        if A.is_complex:
            raise ValueError("block_ilu0 is real-valued")
        p = merge_params(self.defaults(), prm)
        b = int(p["block_size"])
        if b < 1 or b > 8:
            raise ValueError("block_ilu0 needs 1 <= block_size <= 8")
        if A.nrows % b:
            raise ValueError("matrix size not divisible by block_size")
        self.damping = float(p["damping"])
        self.backend = backend
        self.n = A.nrows
        self.b = b
        nb = A.nrows // b
        self.nb = nb
        bp, bc, bv = _core.csr_to_bsr(A.nrows, A.ptr, A.col, A.val, b)
        bp, bc = np.asarray(bp), np.asarray(bc)
        lu, dia = _core.block_ilu0_factor(nb, b, bp, bc, np.asarray(bv))
        lu, dia = np.asarray(lu), np.asarray(dia)

        if backend.name == "cpu":
            self._serial = True
            self.bptr, self.bcol, self.lu, self.dia = bp, bc, lu, dia
        else:
            self._serial = False
            self.solve_iters = int(p["solve_iters"])
            self.solve_damping = float(p["solve_damping"])
            import torch

            from ..backend.hip import DeviceBSR

            dev = backend.device
            bb = b * b
            blocks = lu.reshape(-1, bb)
            row_of = np.repeat(np.arange(nb, dtype=np.int64), np.diff(bp))
            idx = np.arange(len(bc), dtype=np.int64)
            low = idx < dia[row_of]
            up = idx > dia[row_of]

            def bsr_part(mask):
                P = np.zeros(nb + 1, dtype=np.int32)
                np.cumsum(np.bincount(row_of[mask], minlength=nb), out=P[1:])
                M = DeviceBSR.__new__(DeviceBSR)
                M.nbrows, M.bsize = nb, b
                M.nrows = M.ncols = self.n
                M.ptr = torch.from_numpy(P).to(dev)
                M.col = torch.from_numpy(bc[mask].astype(np.int32)).to(dev)
                M.val = torch.from_numpy(
                    np.ascontiguousarray(blocks[mask].ravel())).to(dev)
                return M

            self.L = bsr_part(low)
            self.U = bsr_part(up)
            # diagonal blocks, already inverted by the factorization
            self.Dinv = torch.from_numpy(
                np.ascontiguousarray(blocks[dia].ravel())).to(dev)
            self._t0 = backend.vector(self.n)
            self._t1 = backend.vector(self.n)
            self._t2 = backend.vector(self.n)
            self._t3 = backend.vector(self.n)

    def _blkdiag(self, x, y):
        from ..backend._hiplib import check, lib
        from ..backend.hip import _stream

        check(lib().amg_blkdiag_vmul_f64(self.nb, self.b, self.Dinv.data_ptr(),
                                         x.data_ptr(), y.data_ptr(), _stream()),
              "blkdiag_vmul")

    def _solve_jacobi(self, z):
        """(I+L) y = z then (D+U') u = y via damped-Jacobi iterations, all
        through the BSR kernels (block twin of ILU0._solve_jacobi).  The
        block-diagonal multiply always uses DISTINCT in/out buffers: blocks
        whose b lanes straddle a wavefront boundary (b not dividing 64)
        would race in-place."""
        b = self.backend
        om = self.solve_damping
        y, s, bu, t3 = self._t0, self._t1, self._t2, self._t3
        b.copy(z, y)
        for _ in range(self.solve_iters):
            b.spmv(-1.0, self.L, y, 0.0, s)
            b.axpby(1.0, z, 1.0, s)
            b.axpby(om, s, 1.0 - om, y)
        b.copy(y, s)                  # s = rhs of the upper solve
        self._blkdiag(s, y)           # y = Dinv rhs (first Jacobi iterate)
        for _ in range(self.solve_iters):
            b.spmv(-1.0, self.U, y, 0.0, bu)
            b.axpby(1.0, s, 1.0, bu)         # bu = rhs - U' y
            self._blkdiag(bu, t3)            # t3 = Dinv (...)
            b.axpby(om, t3, 1.0 - om, y)
        b.copy(y, z)

    def _step(self, A, rhs, x, tmp):
        b = self.backend
        b.residual(rhs, A, x, tmp)
        if self._serial:
            _core.block_ilu0_solve(self.nb, self.b, self.bptr, self.bcol,
                                   self.lu, self.dia, tmp)
        else:
            self._solve_jacobi(tmp)
        b.axpby(self.damping, tmp, 1.0, x)

    def apply_pre(self, A, rhs, x, tmp):
        self._step(A, rhs, x, tmp)

    def apply_post(self, A, rhs, x, tmp):
        self._step(A, rhs, x, tmp)

    def apply(self, A, rhs, x, tmp=None):
        b = self.backend
        if tmp is None:
            tmp = b.vector(self.n)
        b.clear(x)
        self._step(A, rhs, x, tmp)
