"""Distributed solver bundle (parity: amgcl/mpi/make_solver.hpp:56)."""
import numpy as np

from ..matrix import CSR
from ..solver import make_solver_component
from .dist_backend import DistBackend, DistInnerProduct
from .precond import LocalBlockPrecond


class DistSolver:
    def __init__(self, strip: CSR, prm=None, backend="cpu", group=None):
        if isinstance(backend, str):
            from ..backend import make_backend

            backend = make_backend(backend)
        self.backend = DistBackend(backend, group)
        prm = dict(prm or {})
        for key in prm:
            if key not in ("precond", "solver", "deflation"):
                raise ValueError(f"unknown parameter '{key}'")
        pprm = dict(prm.get("precond") or {})
        if pprm.get("class") == "dist_amg":
            from .dist_amg import DistAMG

            pprm.pop("class")
            self.A = self.backend.matrix(strip)
            self.P = DistAMG(self.A, pprm, self.backend)
        elif pprm.get("class") == "schur_pressure_correction":
            from .schur import DistSchurPressureCorrection

            pprm.pop("class")
            self.P = DistSchurPressureCorrection(strip, pprm, self.backend, group)
            self.A = self.P.A  # the full distributed operator
        elif pprm.get("class") == "cpr":
            from .cpr import DistCPR

            pprm.pop("class")
            self.P = DistCPR(strip, pprm, self.backend, group)
            self.A = self.P.A
        else:
            self.A = self.backend.matrix(strip)
            self.P = LocalBlockPrecond(self.A, pprm or prm.get("precond"),
                                       self.backend)
        self.inner = DistInnerProduct(backend, group)
        self.S = make_solver_component(
            self.A.n_loc, prm.get("solver"), self.backend, self.inner
        )
        self.defl = None
        dprm = prm.get("deflation")
        if dprm:
            from .deflation import ProjectedDistMatrix, SubdomainDeflation

            dprm = dict(dprm)
            self.defl = SubdomainDeflation(
                self.A, self.backend, dprm.pop("type", "constant"),
                dprm.pop("coords_raw", None),
            )
            if dprm:
                raise ValueError(f"unknown deflation parameters {list(dprm)}")
            self._proj = ProjectedDistMatrix(self.A, self.defl)

    def __call__(self, rhs, x=None):
        b = self.backend
        if len(rhs) != self.A.n_loc:
            raise ValueError(f"rhs has {len(rhs)} entries, this rank's strip "
                             f"has {self.A.n_loc} rows")
        if isinstance(rhs, np.ndarray) and b.base.name != "cpu":
            rhs = b.from_host(rhs)
        if x is None:
            x = b.vector(self.A.n_loc)
        if self.defl is None:
            iters, resid = self.S(self.P, rhs, x, A=self.A)
            return x, iters, resid
        # deflated solve (parity: mpi/subdomain_deflation.hpp:476-479):
        # x0 = Z E^-1 Z^T rhs;  solve (P A) y = P (rhs - A x0);  x = x0 +
        # (I - Z E^-1 Z^T A) y
        defl = self.defl
        x0 = b.vector(self.A.n_loc)
        defl.coarse_guess(rhs, x0)
        r = b.vector(self.A.n_loc)
        b.residual(rhs, self.A, x0, r)
        defl.project(r)
        y = x  # reuse the output buffer for the projected iterate
        b.clear(y)
        iters, resid = self.S(self.P, r, y, A=self._proj)
        Ay = b.vector(self.A.n_loc)
        b.spmv(1.0, self.A, y, 0.0, Ay)
        xfinal = defl.post_correct(y, Ay, x0)
        b.copy(xfinal, x)
        return x, iters, resid

    def gather_solution(self, x):
        """All-gather the solution strips; returns the global vector on rank 0
        (None elsewhere)."""
        import torch
        import torch.distributed as dist

        xh = self.backend.to_host(x)
        out = [None] * self.A.world
        dist.all_gather_object(out, xh, group=self.A.group)
        if self.A.rank == 0:
            return np.concatenate(out)
        return None


def make_dist_solver(strip, prm=None, backend="cpu", group=None):
    return DistSolver(strip, prm, backend, group)
