"""BiCGStab(L) (Sleijpen & Fokkema), default L=2.

Parity: amgcl/solver/bicgstabl.hpp:89 (L, delta, convex combination).
Right-preconditioned: iterates on A*M^-1, recovers x = M^-1 y at the end of
each polynomial step.
"""
import numpy as np

from .base import SolverBase


class BiCGStabL(SolverBase):
    @staticmethod
    def defaults():
        return {"L": 2, "delta": 0.0, "convex": True}

    def __init__(self, n, prm=None, backend=None, inner_product=None):
        self._init_common(n, prm, backend, inner_product)
        b = self.backend
        L = int(self.prm["L"])
        self.L = L
        self.r = [b.vector(n) for _ in range(L + 1)]
        self.u = [b.vector(n) for _ in range(L + 1)]
        self.r0 = b.vector(n)
        self.t = b.vector(n)

    def _pspmv(self, P, A, x, out):
        b = self.backend
        P.apply(x, self.t)
        b.spmv(1.0, A, self.t, 0.0, out)

    def solve(self, A, P, rhs, x):
        b = self.backend
        prm = self.prm
        L = self.L

        norm_rhs = self.norm(rhs)
        if norm_rhs == 0.0:
            if not prm["ns_search"]:
                b.clear(x)
                return 0, 0.0
            norm_rhs = 1.0
        eps = max(prm["tol"] * norm_rhs, prm["abstol"])

        r, u = self.r, self.u
        b.residual(rhs, A, x, r[0])
        b.copy(r[0], self.r0)
        b.clear(u[0])

        rho0, alpha, omega = 1.0, 0.0, 1.0
        res = self.norm(r[0])
        iters = 0
        # x is accumulated in the preconditioned space: we keep x_hat = y with
        # x updated via P at each axpy on u[0]/r[0] directions. Use the
        # standard trick: run on A' = A P and apply P to the final correction
        # directions as they are added.
        while res > eps and iters < prm["maxiter"]:
            rho0 = -omega * rho0
            # Bi-CG part
            for j in range(L):
                rho1 = self.inner(r[j], self.r0)
                if rho0 == 0.0:
                    return iters, res / norm_rhs
                beta = alpha * rho1 / rho0
                rho0 = rho1
                for i in range(j + 1):
                    b.axpby(1.0, r[i], -beta, u[i])
                self._pspmv(P, A, u[j], u[j + 1])
                alpha = rho0 / self.inner(u[j + 1], self.r0)
                for i in range(j + 1):
                    b.axpby(-alpha, u[i + 1], 1.0, r[i])
                self._pspmv(P, A, r[j], r[j + 1])
                # x += alpha * M^-1 u[0]
                P.apply(u[0], self.t)
                b.axpby(alpha, self.t, 1.0, x)
            # MR part: minimize ||r[0] - sum gamma_j r[j]||
            tau = np.zeros((L + 1, L + 1))
            sigma = np.zeros(L + 1)
            gamma_p = np.zeros(L + 1)
            for j in range(1, L + 1):
                for i in range(1, j):
                    tau[i][j] = self.inner(r[j], r[i]) / sigma[i]
                    b.axpby(-tau[i][j], r[i], 1.0, r[j])
                sigma[j] = self.inner(r[j], r[j])
                gamma_p[j] = self.inner(r[0], r[j]) / sigma[j]
            gamma = np.zeros(L + 1)
            gamma[L] = gamma_p[L]
            omega = gamma[L]
            for j in range(L - 1, 0, -1):
                gamma[j] = gamma_p[j] - sum(tau[j][i] * gamma[i] for i in range(j + 1, L + 1))
            gamma_pp = np.zeros(L + 1)
            for j in range(1, L):
                gamma_pp[j] = gamma[j + 1] + sum(
                    tau[j][i] * gamma[i + 1] for i in range(j + 1, L)
                )
            # updates
            P.apply(r[0], self.t)
            b.axpby(gamma[1], self.t, 1.0, x)
            b.axpby(-gamma_p[L], r[L], 1.0, r[0])
            b.axpby(-gamma[L], u[L], 1.0, u[0])
            for j in range(1, L):
                b.axpby(-gamma[j], u[j], 1.0, u[0])
                P.apply(r[j], self.t)
                b.axpby(gamma_pp[j], self.t, 1.0, x)
                b.axpby(-gamma_p[j], r[j], 1.0, r[0])
            res = self.norm(r[0])
            iters += 1
        return iters, res / norm_rhs
