"""Schur pressure correction (two-field split) preconditioner.

Parity: amgcl/preconditioner/schur_pressure_correction.hpp:59-635 —
K = [[Kuu, Kup], [Kpu, Kpp]] split by a pressure mask; the Schur complement
is approximated as S = Kpp - Kpu diag(Kuu)^-1 Kup (SIMPLE/SIMPLEC style) and
handled by a nested pressure solver (AMG+Krylov by default), with a nested
velocity solver on Kuu. One application:

    Kuu u* = rhs_u
    S p    = rhs_p - Kpu u*
    x_u    = u* - diag(Kuu)^-1 Kup p ,  x_p = p
"""
import numpy as np

from ..matrix import CSR
from ..params import merge_params


def _split(A: CSR, pmask):
    """Split K into the 2x2 field blocks by boolean pressure mask."""
    import scipy.sparse as sp

    m = A.to_scipy()
    pm = np.asarray(pmask, dtype=bool)
    um = ~pm
    uu = m[um][:, um].tocsr()
    up = m[um][:, pm].tocsr()
    pu = m[pm][:, um].tocsr()
    pp = m[pm][:, pm].tocsr()
    return (CSR.from_scipy(uu), CSR.from_scipy(up),
            CSR.from_scipy(pu), CSR.from_scipy(pp))


class SchurPressureCorrection:
    @staticmethod
    def defaults():
        return {
            "usolver": {"precond": {"class": "relaxation", "type": "spai0"},
                        "solver": {"type": "preonly"}},
            "psolver": {"precond": {"class": "amg"},
                        "solver": {"type": "preonly"}},
            "pmask_raw": None,     # boolean array marking pressure unknowns
            "pmask_pattern": None,  # e.g. "%4" -> every 4th unknown is pressure
            "simplec": True,        # SIMPLEC diagonal (row-sum of |Kuu|)
        }

    def __init__(self, A, prm=None, backend=None):
        from ..make_solver import MakeSolver

        if backend is None:
            from ..backend import make_backend

            backend = make_backend("cpu")
        self.backend = backend
        p = merge_params(self.defaults(), prm, opaque=("usolver", "psolver"))

        if not isinstance(A, CSR):
            from ..backend import hip_setup

            A = hip_setup.download(A)

        if p["pmask_raw"] is not None:
            pmask = np.asarray(p["pmask_raw"], dtype=bool)
        elif p["pmask_pattern"]:
            pat = str(p["pmask_pattern"])
            if not pat.startswith("%"):
                raise ValueError("pmask_pattern must look like '%4'")
            b = int(pat[1:])
            pmask = (np.arange(A.nrows) % b) == (b - 1)
        else:
            raise ValueError("schur needs pmask_raw or pmask_pattern")
        self.pmask = pmask
        self.uidx = np.where(~pmask)[0]
        self.pidx = np.where(pmask)[0]

        Kuu, Kup, Kpu, Kpp = _split(A, pmask)

        # SIMPLE(C) diagonal approximation of Kuu^-1
        if p["simplec"]:
            row_of = np.repeat(np.arange(Kuu.nrows), np.diff(Kuu.ptr))
            dsum = np.zeros(Kuu.nrows)
            np.add.at(dsum, row_of, np.abs(Kuu.val))
            self.dinv_host = 1.0 / dsum
        else:
            self.dinv_host = 1.0 / np.asarray(Kuu.diagonal())

        # S = Kpp - Kpu * D^-1 * Kup
        import scipy.sparse as sp

        D = sp.diags(self.dinv_host)
        S = (Kpp.to_scipy() - Kpu.to_scipy() @ D @ Kup.to_scipy()).tocsr()
        S.sort_indices()
        S = CSR.from_scipy(S)

        self.usolve = MakeSolver(Kuu, dict(p["usolver"]), backend)
        self.psolve = MakeSolver(S, dict(p["psolver"]), backend)

        bk = backend
        self.Kuu = bk.matrix(Kuu)
        self.Kup = bk.matrix(Kup)
        self.Kpu = bk.matrix(Kpu)
        self._A = bk.matrix(A)
        self.dinv = bk.from_host(self.dinv_host)
        nu, npr = len(self.uidx), len(self.pidx)
        self.rhs_u = bk.vector(nu)
        self.rhs_p = bk.vector(npr)
        self.u = bk.vector(nu)
        self.pvec = bk.vector(npr)
        self.tmp_u = bk.vector(nu)
        self.tmp_p = bk.vector(npr)
        import torch

        if bk.name == "hip":
            self.uidx_d = torch.from_numpy(self.uidx.astype(np.int32)).to(bk.device)
            self.pidx_d = torch.from_numpy(self.pidx.astype(np.int32)).to(bk.device)
        else:
            self.uidx_d, self.pidx_d = self.uidx, self.pidx

    def system_matrix(self):
        return self._A

    def apply(self, rhs, x):
        bk = self.backend
        bk.gather(rhs, self.uidx_d, self.rhs_u)
        bk.gather(rhs, self.pidx_d, self.rhs_p)
        # u* = Kuu^-1 rhs_u (approximately, via the nested U solver)
        bk.clear(self.u)
        self.usolve.S(self.usolve.P, self.rhs_u, self.u)
        # rhs_p <- rhs_p - Kpu u*
        bk.spmv(-1.0, self.Kpu, self.u, 1.0, self.rhs_p)
        # S p = rhs_p
        bk.clear(self.pvec)
        self.psolve.S(self.psolve.P, self.rhs_p, self.pvec)
        # u <- u - D^-1 Kup p
        bk.spmv(1.0, self.Kup, self.pvec, 0.0, self.tmp_u)
        bk.vmul(-1.0, self.dinv, self.tmp_u, 1.0, self.u)
        bk.scatter(self.u, self.uidx_d, x)
        bk.scatter(self.pvec, self.pidx_d, x)
