"""Preconditioned conjugate gradients.

Parity: amgcl/solver/cg.hpp:67-258 (4 work vectors r,s,p,q; identical
update order so iteration counts match the reference).
"""
from .base import SolverBase


class CG(SolverBase):
    @staticmethod
    def defaults():
        return {}

    def __init__(self, n, prm=None, backend=None, inner_product=None):
        self._init_common(n, prm, backend, inner_product)
        b = self.backend
        self.r = b.vector(n)
        self.s = b.vector(n)
        self.p = b.vector(n)
        self.q = b.vector(n)

    def solve(self, A, P, rhs, x):
        b = self.backend
        prm = self.prm
        r, s, p, q = self.r, self.s, self.p, self.q

        norm_rhs = self.norm(rhs)
        if norm_rhs == 0.0:
            if not prm["ns_search"]:
                b.clear(x)
                return 0, 0.0
            norm_rhs = 1.0

        b.residual(rhs, A, x, r)
        eps = max(prm["tol"] * norm_rhs, prm["abstol"])

        rho1 = 0.0
        iter_done = 0
        res = self.norm(r)
        while res > eps and iter_done < prm["maxiter"]:
            P.apply(r, s)
            rho2 = rho1
            rho1 = self.inner(r, s)
            if iter_done == 0:
                b.copy(s, p)
            else:
                b.axpby(1.0, s, rho1 / rho2, p)
            b.spmv(1.0, A, p, 0.0, q)
            alpha = rho1 / self.inner(q, p)
            b.axpby(alpha, p, 1.0, x)
            b.axpby(-alpha, q, 1.0, r)
            res = self.norm(r)
            iter_done += 1
            if prm["verbose"] and iter_done % 5 == 0:
                print(f"cg {iter_done}\t{res / norm_rhs:.3e}")

        return iter_done, res / norm_rhs
