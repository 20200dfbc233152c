"""ILU(0) smoother.

Parity: amgcl/relaxation/ilu0.hpp:51 (host IKJ factorization) with the
solve strategy of amgcl/relaxation/detail/ilu_solve.hpp: serial sweeps on the
CPU backend; on the GPU backend the triangular solves use the reference's
damped-Jacobi iterated approximate triangular solve (ilu_solve.hpp:57-75,
iters=2, damping=0.72), which runs entirely through backend primitives
(no SpSV analysis phase, GPU-native).
"""
import numpy as np

from .. import _core
from ..matrix import CSR
from ..params import merge_params


class ILU0:
    gpu_supported = True  # via iterated-Jacobi triangular solves

    @staticmethod
    def defaults():
        # solve_serial=None: level-scheduled OpenMP triangular sweeps for
        # large factors (bitwise-identical to serial; parity with
        # relaxation/detail/ilu_solve.hpp:257), serial below the threshold
        # solve="jacobi": the reference's GPU-native iterated approximate
        # triangular solve (ilu_solve.hpp:44-124). solve="exact": exact
        # level-scheduled solve by one cooperative GPU kernel (parity with
        # the vendor-SpSV exact path, amgcl/relaxation/rocsparse_ilu0.hpp).
        return {"damping": 1.0, "solve_iters": 2, "solve_damping": 0.72,
                "solve_serial": None, "solve": "jacobi"}

    def _factor(self, A, p):
        lu, dia = _core.ilu0_factor(A.nrows, A.ptr, A.col, A.val)
        return A.ptr, A.col, lu, dia

    def __init__(self, A, prm, backend):
        from ..matrix import CSR

        if not isinstance(A, CSR):
            from ..backend import hip_setup

            A = hip_setup.download(A)  # ILU factorizations are host-side
        if A.is_complex:
            raise ValueError("ILU smoothers are real-valued; complex systems "
                             "use spai0/damped_jacobi/chebyshev (or the "
                             "2x2-real adapter)")
        p = merge_params(self.defaults(), prm)
        self.damping = float(p["damping"])
        self.backend = backend
        fptr, fcol, lu, dia = self._factor(A, p)
        self.n = A.nrows

        if backend.name == "cpu":
            self._serial = True
            self.ptr, self.col = np.asarray(fptr), np.asarray(fcol)
            self.lu, self.dia = np.asarray(lu), np.asarray(dia)
            import os

            ser = p["solve_serial"]
            if ser is None:
                ser = os.cpu_count() < 4 or self.n < 20_000
            self._levels = None
            if not ser:
                self._dia32 = np.ascontiguousarray(self.dia, dtype=np.int32)
                lp, lr = _core.tri_levels(self.n, self.ptr, self.col,
                                          self._dia32, True)
                up, ur = _core.tri_levels(self.n, self.ptr, self.col,
                                          self._dia32, False)
                self._levels = (np.asarray(lp), np.asarray(lr),
                                np.asarray(up), np.asarray(ur))
        else:
            self._serial = False
            self.solve_iters = int(p["solve_iters"])
            self.solve_damping = float(p["solve_damping"])
            # Split LU into strictly-lower L (unit diagonal implied) and
            # strictly-upper U' with inverted diagonal kept separately.
            lu = np.asarray(lu)
            dia = np.asarray(dia).astype(np.int64)
            ptr = np.asarray(fptr).astype(np.int64)
            col = np.asarray(fcol)
            row_of = np.repeat(np.arange(self.n, dtype=np.int64), np.diff(ptr))
            idx = np.arange(col.size, dtype=np.int64)
            lower = idx < dia[row_of]
            upper = idx > dia[row_of]
            Lp = np.zeros(self.n + 1, dtype=np.int64)
            np.cumsum(np.bincount(row_of[lower], minlength=self.n), out=Lp[1:])
            Up = np.zeros(self.n + 1, dtype=np.int64)
            np.cumsum(np.bincount(row_of[upper], minlength=self.n), out=Up[1:])
            self.L = backend.matrix(CSR(self.n, self.n, Lp, col[lower], lu[lower]))
            self.U = backend.matrix(CSR(self.n, self.n, Up, col[upper], lu[upper]))
            self.Dinv = backend.from_host(lu[dia])  # already inverted
            self._t0 = backend.vector(self.n)
            self._t1 = backend.vector(self.n)
            self._t2 = backend.vector(self.n)
            if str(p["solve"]) not in ("jacobi", "exact"):
                raise ValueError(f"ilu solve must be jacobi|exact, "
                                 f"got '{p['solve']}'")
            self._exact = str(p["solve"]) == "exact"
            if self._exact:
                import torch

                from ..backend._hiplib import lib as _hl

                if not _hl().amg_coop_supported():
                    raise RuntimeError("exact GPU sptrsv needs cooperative launch")
                dia32 = np.ascontiguousarray(dia, dtype=np.int32)
                lp, lr = _core.tri_levels(self.n, fptr, fcol, dia32, True)
                up, ur = _core.tri_levels(self.n, fptr, fcol, dia32, False)
                dev = backend.device
                self._lev = tuple(
                    torch.from_numpy(np.ascontiguousarray(a)).to(dev)
                    for a in (lp, lr, up, ur))
                self._nlev = (len(lp) - 1, len(up) - 1)

    def _solve_serial(self, z):
        if self._levels is not None:
            lp, lr, up, ur = self._levels
            _core.ilu0_solve_parallel(self.n, self.ptr, self.col, self.lu,
                                      self._dia32, lp, lr, up, ur, z)
        else:
            _core.ilu0_solve(self.n, self.ptr, self.col, self.lu, self.dia, z)

    def _solve_exact(self, z):
        """Exact level-scheduled triangular solves on the GPU (one
        cooperative kernel per factor; bitwise semantics of the serial
        sweeps — see kernels.hip sptrsv_levels_k)."""
        from ..backend._hiplib import check, lib
        from ..backend.hip import _stream

        lp, lr, up, ur = self._lev
        check(lib().amg_sptrsv_f64(self._nlev[0], lp.data_ptr(), lr.data_ptr(),
                                   self.L.ptr.data_ptr(), self.L.col.data_ptr(),
                                   self.L.val.data_ptr(), 0, z.data_ptr(), 1,
                                   _stream()), "sptrsv_lower")
        check(lib().amg_sptrsv_f64(self._nlev[1], up.data_ptr(), ur.data_ptr(),
                                   self.U.ptr.data_ptr(), self.U.col.data_ptr(),
                                   self.U.val.data_ptr(), self.Dinv.data_ptr(),
                                   z.data_ptr(), 0, _stream()), "sptrsv_upper")

    def _solve_jacobi(self, z):
        """Damped-Jacobi iterated approximate triangular solves
        (parity: amgcl/relaxation/detail/ilu_solve.hpp:44-124)."""
        b = self.backend
        om = self.solve_damping
        y, s, bu = self._t0, self._t1, self._t2
        # lower: (I + L) y = z
        b.copy(z, y)
        for _ in range(self.solve_iters):
            b.spmv(-1.0, self.L, y, 0.0, s)   # s = -L y
            b.axpby(1.0, z, 1.0, s)           # s = z - L y
            b.axpby(om, s, 1.0 - om, y)
        # upper: (D + U') u = y  ->  u = Dinv (y - U' u)
        b.copy(y, bu)
        b.vmul(1.0, self.Dinv, bu, 0.0, y)
        for _ in range(self.solve_iters):
            b.spmv(-1.0, self.U, y, 0.0, s)
            b.axpby(1.0, bu, 1.0, s)
            b.vmul(1.0, self.Dinv, s, 0.0, s)
            b.axpby(om, s, 1.0 - om, y)
        b.copy(y, z)

    def _step(self, A, rhs, x, tmp):
        b = self.backend
        b.residual(rhs, A, x, tmp)
        if self._serial:
            self._solve_serial(tmp)
        elif getattr(self, "_exact", False):
            self._solve_exact(tmp)
        else:
            self._solve_jacobi(tmp)
        b.axpby(self.damping, tmp, 1.0, x)

    def apply_pre(self, A, rhs, x, tmp):
        self._step(A, rhs, x, tmp)

    def apply_post(self, A, rhs, x, tmp):
        self._step(A, rhs, x, tmp)

    def apply(self, A, rhs, x, tmp=None):
        b = self.backend
        b.copy(rhs, x)
        if self._serial:
            self._solve_serial(x)
        elif getattr(self, "_exact", False):
            self._solve_exact(x)
        else:
            self._solve_jacobi(x)


class ILUK(ILU0):
    """ILU(k) level-of-fill smoother (parity: amgcl/relaxation/iluk.hpp:49)."""

    @staticmethod
    def defaults():
        d = ILU0.defaults()
        d["k"] = 1
        return d

    def _factor(self, A, p):
        fptr, fcol, lu, dia = _core.iluk_factor(A.nrows, A.ptr, A.col, A.val,
                                                int(p["k"]))
        return fptr, fcol, lu, dia


class ILUT(ILU0):
    """ILUT(p, tau) threshold smoother (parity: amgcl/relaxation/ilut.hpp:56)."""

    @staticmethod
    def defaults():
        d = ILU0.defaults()
        d["p"] = 2.0
        d["tau"] = 1e-2
        return d

    def _factor(self, A, p):
        fptr, fcol, lu, dia = _core.ilut_factor(A.nrows, A.ptr, A.col, A.val,
                                                float(p["p"]), float(p["tau"]))
        return fptr, fcol, lu, dia


class ILUP(ILU0):
    """ILU(p) with fill pattern from the symbolic matrix power A^(k+1)
    (parity: amgcl/relaxation/ilup.hpp:120-186 — as opposed to iluk, the
    fill-in positions are the pattern of A multiplied symbolically by itself
    k times; A's values are scattered onto that pattern and an ILU(0)
    factorization runs over it)."""

    @staticmethod
    def defaults():
        d = ILU0.defaults()
        d["k"] = 1
        return d

    def _factor(self, A, p):
        k = int(p["k"])
        if k == 0:
            return super()._factor(A, p)
        import scipy.sparse as sp

        S = sp.csr_matrix(
            (np.ones(A.nnz), np.asarray(A.col), np.asarray(A.ptr)), shape=A.shape
        )
        P = S
        for _ in range(k):
            P = (P @ S).tocsr()
        P.sort_indices()
        # scatter A's values onto the power pattern (zero fill elsewhere);
        # entries not covered by the pattern are dropped, as in the reference
        ptr = P.indptr.astype(np.int32)
        col = P.indices.astype(np.int32)
        val = np.zeros(P.nnz)
        n = np.int64(A.nrows)
        key_p = np.repeat(np.arange(n, dtype=np.int64), np.diff(ptr)) * n + col
        key_a = (np.repeat(np.arange(n, dtype=np.int64), np.diff(A.ptr)) * n
                 + np.asarray(A.col))
        pos = np.searchsorted(key_p, key_a)
        hit = key_p[np.minimum(pos, key_p.size - 1)] == key_a
        val[pos[hit]] = np.asarray(A.val)[hit]
        lu, dia = _core.ilu0_factor(A.nrows, ptr, col, val)
        return ptr, col, lu, dia


class ILU0ChowPatel(ILU0):
    """Fine-grained (Chow-Patel) parallel ILU(0) factorization
    (parity: amgcl/relaxation/ilu0_chow_patel.hpp:87)."""

    @staticmethod
    def defaults():
        d = ILU0.defaults()
        d["sweeps"] = 4
        return d

    def _factor(self, A, p):
        fptr, fcol, lu, dia = _core.ilu0_chow_patel(A.nrows, A.ptr, A.col, A.val,
                                                    int(p["sweeps"]))
        return fptr, fcol, lu, dia
