"""CPR (Constrained Pressure Residual) preconditioner for reservoir-style
block systems.

Parity: amgcl/preconditioner/cpr.hpp:45-561 — unknowns interleaved per cell
(block_size unknowns per cell, pressure first); quasi-IMPES decoupling builds
the scalar pressure matrix App, AMG handles the pressure subsystem and a
global smoother (SPAI0/ILU0) handles the full system:

    p_corr = AMG(App) @ (W^T r);  x += U p_corr;  x += Smooth(r - K x)

cpr_drs (dynamic row sums) reweights the pressure equations by per-cell
dynamic row-sum weights (amgcl/preconditioner/cpr_drs.hpp:46).
"""
import numpy as np

from ..matrix import CSR
from ..params import merge_params


class CPR:
    @staticmethod
    def defaults():
        return {
            "block_size": 2,
            "active_rows": 0,  # 0 = all
            "drs": False,       # dynamic row sums (cpr_drs)
            "drs_eps_ps": 0.02,
            "pprecond": {"class": "amg"},
            "sprecond": {"class": "relaxation", "type": "spai0"},
        }

    def __init__(self, A, prm=None, backend=None):
        from . import make_preconditioner

        if backend is None:
            from ..backend import make_backend

            backend = make_backend("cpu")
        self.backend = backend
        p = merge_params(self.defaults(), prm, opaque=("pprecond", "sprecond"))
        b = int(p["block_size"])
        if b < 1:
            raise ValueError("cpr block_size must be >= 1")
        self.bsize = b

        if not isinstance(A, CSR):
            from ..backend import hip_setup

            A = hip_setup.download(A)
        n = A.nrows
        if n % b:
            raise ValueError("matrix size not divisible by block_size")
        ncells = n // b
        self.ncells = ncells

        # quasi-IMPES weights (amgcl/preconditioner/cpr.hpp): per cell, take the
        # b-by-b diagonal block D and use the first row of inv(D) as the
        # equation weights, so that w^T D = e_p^T and the non-pressure
        # couplings cancel inside the cell.  App = W^T A U restricted to the
        # pressure columns.
        import scipy.sparse as sp

        m = A.to_scipy().tocsr()
        if p["drs"]:
            # dynamic row-sum weights per cell equation (cpr_drs): weight each
            # cell equation by its row sum ratio before summing into pressure
            rs = np.abs(m).sum(axis=1).A.ravel()
            w = 1.0 / np.maximum(rs, float(p["drs_eps_ps"]))
        else:
            w = self._quasi_impes_weights(m, ncells, b)
        # restriction W^T: cell i <- sum_k w[i*b+k] * row(i*b+k)
        rows = np.arange(n) // b
        Wt = sp.csr_matrix((w, (rows, np.arange(n))), shape=(ncells, n))
        # prolongation U: pressure correction goes to the pressure unknown
        pcols = np.arange(ncells) * b
        U = sp.csr_matrix((np.ones(ncells), (pcols, np.arange(ncells))), shape=(n, ncells))
        App = (Wt @ m @ U).tocsr()
        App.sort_indices()

        self.pprecond = make_preconditioner(CSR.from_scipy(App), dict(p["pprecond"]),
                                            backend)
        self.sprecond = make_preconditioner(A, dict(p["sprecond"]), backend)

        bk = backend
        self._A = bk.matrix(A)
        self.Wt = bk.matrix(CSR.from_scipy(Wt))
        self.U = bk.matrix(CSR.from_scipy(U))
        self.rp = bk.vector(ncells)
        self.pc = bk.vector(ncells)
        self.tmp = bk.vector(n)
        self.r2 = bk.vector(n)

    @staticmethod
    def _quasi_impes_weights(m, ncells, b):
        """Per-unknown weights w[i*b+k] = inv(D_i)[0, k] where D_i is cell i's
        diagonal block.  Blocks are gathered via offset diagonals (O(b^2)
        sparse diagonal extractions, no per-cell Python loop)."""
        n = ncells * b
        blocks = np.zeros((ncells, b, b))
        for k in range(b):
            for l in range(b):
                d = l - k
                diag = m.diagonal(d)
                start = k if d >= 0 else l
                vals = diag[start::b]
                blocks[: len(vals), k, l] = vals[:ncells]
        w = np.zeros((ncells, b))
        # invert in bulk; fall back to e_p for singular cells
        try:
            inv = np.linalg.inv(blocks)
            w[:] = inv[:, 0, :]
        except np.linalg.LinAlgError:
            for i in range(ncells):
                try:
                    w[i] = np.linalg.inv(blocks[i])[0]
                except np.linalg.LinAlgError:
                    w[i, 0] = 1.0
        bad = ~np.isfinite(w).all(axis=1)
        if bad.any():
            w[bad] = 0.0
            w[bad, 0] = 1.0
        return w.reshape(n)

    def system_matrix(self):
        return self._A

    def apply(self, rhs, x):
        bk = self.backend
        # pressure correction
        bk.spmv(1.0, self.Wt, rhs, 0.0, self.rp)
        self.pprecond.apply(self.rp, self.pc)
        bk.spmv(1.0, self.U, self.pc, 0.0, x)
        # global smoothing on the updated residual
        bk.residual(rhs, self._A, x, self.r2)
        self.sprecond.apply(self.r2, self.tmp)
        bk.axpby(1.0, self.tmp, 1.0, x)
