"""Block-matrix smoother wrapper.

Parity: amgcl/relaxation/as_block.hpp:46-120 — converts the (scalar) level
matrix to block form, builds the base smoother over block values, and applies
it by reinterpreting the scalar vectors as block vectors. With block values
the base smoothers of the diagonal family produce one BxB matrix per grid
point (relaxation/spai0.hpp:66-78 with value_type = static_matrix:
M_i = A_ii / sum_j ||A_ij||_F^2; damped_jacobi: M_i = omega * inv(A_ii)),
so each application is x_i += M_i (rhs - A x)_i blockwise.

The wrapper supports those block-diagonal bases (spai0, damped_jacobi); the
residual runs on the level matrix in its native (scalar CSR or BSR) layout
and the block-diagonal correction is a dedicated gfx950 kernel
(block.hip:blkdiag_vmul_k).
"""
import numpy as np

from .. import _core
from ..params import merge_params


class AsBlock:
    gpu_supported = True

    @staticmethod
    def defaults():
        return {
            "block_size": 2,
            "base": {},  # {"type": "spai0"|"damped_jacobi", ...}
        }

    def __init__(self, A, prm, backend):
        from ..matrix import CSR

        p = merge_params(self.defaults(), prm, opaque=("base",))
        base = dict(p["base"] or {})
        kind = base.pop("type", "spai0")
        B = int(p["block_size"])
        if kind not in ("spai0", "damped_jacobi"):
            raise ValueError(
                f"as_block base '{kind}' not supported (block-diagonal bases: "
                "spai0, damped_jacobi)"
            )
        if not isinstance(A, CSR):
            from ..backend import hip_setup

            A = hip_setup.download(A)  # block condensation is host-side setup
        if A.nrows % B:
            raise ValueError("as_block: matrix size not divisible by block_size")

        bp, bc, bv = _core.csr_to_bsr(A.nrows, A.ptr, A.col, A.val, B)
        bp, bc = np.asarray(bp), np.asarray(bc)
        bv = np.asarray(bv).reshape(-1, B, B)
        nb = A.nrows // B
        row_of = np.repeat(np.arange(nb, dtype=np.int64), np.diff(bp))
        diag_pos = np.flatnonzero(bc == row_of)
        if diag_pos.size != nb:
            raise ValueError("as_block: missing diagonal block")
        D = bv[diag_pos]
        if kind == "spai0":
            merge_params({}, base)
            den = np.zeros(nb)
            np.add.at(den, row_of, (bv * bv).sum(axis=(1, 2)))
            M = D / den[:, None, None]
        else:
            bprm = merge_params({"damping": 0.72}, base)
            M = float(bprm["damping"]) * np.linalg.inv(D)

        self.backend = backend
        self.bsize = B
        self.n = A.nrows
        if backend.name == "cpu":
            self.M = np.ascontiguousarray(M)
        else:
            if B not in (2, 3, 4):
                raise ValueError("as_block on the HIP backend supports B in {2,3,4}")
            self.M = backend.from_host(M.ravel())
            self._t2 = backend.vector(self.n)

    def _bvmul(self, src, dst):
        """dst = blockdiag(M) src"""
        b = self.backend
        if b.name == "cpu":
            np.einsum("bij,bj->bi", self.M, src.reshape(-1, self.bsize),
                      out=dst.reshape(-1, self.bsize))
        else:
            b.blkdiag_vmul(self.bsize, self.M, src, dst)

    def _step(self, A, rhs, x, tmp):
        b = self.backend
        b.residual(rhs, A, x, tmp)
        if b.name == "cpu":
            x += np.einsum("bij,bj->bi", self.M,
                           tmp.reshape(-1, self.bsize)).ravel()
        else:
            self._bvmul(tmp, self._t2)
            b.axpby(1.0, self._t2, 1.0, x)

    def apply_pre(self, A, rhs, x, tmp):
        self._step(A, rhs, x, tmp)

    def apply_post(self, A, rhs, x, tmp):
        self._step(A, rhs, x, tmp)

    def apply(self, A, rhs, x, tmp=None):
        self._bvmul(rhs, x)
