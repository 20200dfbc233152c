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
            if key not in ("precond", "solver"):
                raise ValueError(f"unknown parameter '{key}'")
        self.A = self.backend.matrix(strip)
        self.P = LocalBlockPrecond(self.A, prm.get("precond"), self.backend)
        self.inner = DistInnerProduct(backend, group)
        self.S = make_solver_component(
            self.A.n_loc, prm.get("solver"), self.backend, self.inner
        )

    def __call__(self, rhs, x=None):
        b = self.backend
        if isinstance(rhs, np.ndarray) and b.base.name != "cpu":
            rhs = b.from_host(rhs)
        if x is None:
            x = b.vector(self.A.n_loc)
        iters, resid = self.S(self.P, rhs, x, A=self.A)
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
