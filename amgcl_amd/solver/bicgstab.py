"""Preconditioned BiCGStab with selectable preconditioning side.

Parity: amgcl/solver/bicgstab.hpp:56-304 (7 work vectors, right-side
preconditioning by default, early exit on the s-norm check).
"""
from .base import SolverBase


class BiCGStab(SolverBase):
    @staticmethod
    def defaults():
        return {"pside": "right", "check_after": False}

    def __init__(self, n, prm=None, backend=None, inner_product=None):
        self._init_common(n, prm, backend, inner_product)
        b = self.backend
        for name in ("r", "p", "v", "s", "t", "rh", "T"):
            setattr(self, name, b.vector(n))

    def _pspmv(self, P, A, x, y):
        """y = A (P^-1 x) for right preconditioning, y = P^-1 (A x) for left
        (parity: amgcl/preconditioner/precond_side.hpp)."""
        b = self.backend
        if self.prm["pside"] == "right":
            P.apply(x, self.T)
            b.spmv(1.0, A, self.T, 0.0, y)
        else:
            b.spmv(1.0, A, x, 0.0, self.T)
            P.apply(self.T, y)

    def solve(self, A, P, rhs, x):
        b = self.backend
        prm = self.prm
        left = prm["pside"] == "left"

        norm_rhs = self.norm(rhs)
        if norm_rhs == 0.0:
            if not prm["ns_search"]:
                b.clear(x)
                return 0, 0.0
            norm_rhs = 1.0

        if left:
            b.residual(rhs, A, x, self.rh)
            P.apply(self.rh, self.r)
        else:
            b.residual(rhs, A, x, self.r)
        b.copy(self.r, self.rh)

        eps = max(norm_rhs * prm["tol"], prm["abstol"])
        res = 2 * eps if prm["check_after"] else self.norm(self.r)

        rho1 = rho2 = alpha = omega = 0.0
        iter_done = 0
        first = True
        while res > eps and iter_done < prm["maxiter"]:
            rho2 = rho1
            rho1 = self.inner(self.r, self.rh)
            if first:
                b.copy(self.r, self.p)
                first = False
            else:
                if rho2 == 0.0:
                    raise ZeroDivisionError("zero rho in BiCGStab")
                beta = (rho1 * alpha) / (rho2 * omega)
                b.axpbypcz(1.0, self.r, -beta * omega, self.v, beta, self.p)

            self._pspmv(P, A, self.p, self.v)
            alpha = rho1 / self.inner(self.rh, self.v)
            b.axpby(alpha, self.p if left else self.T, 1.0, x)
            b.axpbypcz(1.0, self.r, -alpha, self.v, 0.0, self.s)

            res = self.norm(self.s)
            if res > eps:
                self._pspmv(P, A, self.s, self.t)
                ts, tt = (
                    self.inner2(self.t, self.s, self.t, self.t)
                    if self.inner2
                    else (self.inner(self.t, self.s), self.inner(self.t, self.t))
                )
                omega = ts / tt
                if omega == 0.0:
                    raise ZeroDivisionError("zero omega in BiCGStab")
                b.axpby(omega, self.s if left else self.T, 1.0, x)
                b.axpbypcz(1.0, self.s, -omega, self.t, 0.0, self.r)
                res = self.norm(self.r)

            iter_done += 1
            if prm["verbose"] and iter_done % 5 == 0:
                print(f"bicgstab {iter_done}\t{res / norm_rhs:.3e}")

        return iter_done, res / norm_rhs
