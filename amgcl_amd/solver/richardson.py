"""Damped preconditioned Richardson iteration.

Parity: amgcl/solver/richardson.hpp:56.
"""
from .base import SolverBase


class Richardson(SolverBase):
    @staticmethod
    def defaults():
        return {"damping": 1.0}

    def __init__(self, n, prm=None, backend=None, inner_product=None):
        self._init_common(n, prm, backend, inner_product)
        b = self.backend
        self.r = b.vector(n)
        self.s = b.vector(n)

    def solve(self, A, P, rhs, x):
        b = self.backend
        prm = self.prm
        damping = float(prm["damping"])

        norm_rhs = self.norm(rhs)
        if norm_rhs == 0.0:
            if not prm["ns_search"]:
                b.clear(x)
                return 0, 0.0
            norm_rhs = 1.0
        eps = max(prm["tol"] * norm_rhs, prm["abstol"])

        b.residual(rhs, A, x, self.r)
        res = self.norm(self.r)
        iters = 0
        while res > eps and iters < prm["maxiter"]:
            P.apply(self.r, self.s)
            b.axpby(damping, self.s, 1.0, x)
            b.residual(rhs, A, x, self.r)
            res = self.norm(self.r)
            iters += 1
        return iters, res / norm_rhs
