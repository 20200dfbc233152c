"""IDR(s): Induced Dimension Reduction with shadow space of size s.

Parity: amgcl/solver/idrs.hpp:70 (s=4 default, random shadow space,
right preconditioning). Implementation follows the biortho variant of
van Gijzen & Sonneveld (IDR(s) with smoothing options omitted).
"""
import numpy as np

from .base import SolverBase


class IDRS(SolverBase):
    @staticmethod
    def defaults():
        return {"s": 4, "omega": 0.7, "replacement": False}

    def __init__(self, n, prm=None, backend=None, inner_product=None):
        self._init_common(n, prm, backend, inner_product)
        b = self.backend
        s = int(self.prm["s"])
        if s < 1:
            raise ValueError("idrs shadow-space size s must be >= 1")
        self.s = s
        rng = np.random.default_rng(71)
        # random shadow space, orthonormalized with backend ops (modified
        # Gram-Schmidt) so it works at any n and in distributed runs
        self.Pshadow = [b.from_host(rng.standard_normal(n)) for _ in range(s)]
        for j in range(s):
            pj = self.Pshadow[j]
            for i in range(j):
                h = (inner_product or b.dot)(pj, self.Pshadow[i])
                b.axpby(-h, self.Pshadow[i], 1.0, pj)
            nrm = np.sqrt((inner_product or b.dot)(pj, pj))
            b.axpby(1.0 / nrm - 1.0, pj, 1.0, pj)
        self.r = b.vector(n)
        self.v = b.vector(n)
        self.t = b.vector(n)
        self.tmp = b.vector(n)
        self.G = [b.vector(n) for _ in range(s)]
        self.U = [b.vector(n) for _ in range(s)]

    def solve(self, A, P, rhs, x):
        b = self.backend
        prm = self.prm
        s = self.s
        angle = float(self.prm["omega"])

        norm_rhs = self.norm(rhs)
        if norm_rhs == 0.0:
            if not prm["ns_search"]:
                b.clear(x)
                return 0, 0.0
            norm_rhs = 1.0
        eps = max(prm["tol"] * norm_rhs, prm["abstol"])

        b.residual(rhs, A, x, self.r)
        res = self.norm(self.r)
        M = np.eye(s)
        for g in self.G:
            b.clear(g)
        for u in self.U:
            b.clear(u)
        om = 1.0
        iters = 0
        f = np.zeros(s)
        while res > eps and iters < prm["maxiter"]:
            for j in range(s):
                f[j] = self.inner(self.r, self.Pshadow[j])
            for k in range(s):
                # solve lower-triangular M[k:,k:] c = f[k:]
                c = np.linalg.solve(M[k:, k:], f[k:])
                # v = r - sum c_i G[i]
                b.copy(self.r, self.v)
                for i, ci in enumerate(c):
                    b.axpby(-ci, self.G[k + i], 1.0, self.v)
                # U[k] = om * M^-1 v + sum c_i U[i]
                P.apply(self.v, self.tmp)
                b.axpby(om, self.tmp, 0.0, self.t)
                for i, ci in enumerate(c):
                    b.axpby(ci, self.U[k + i], 1.0, self.t)
                b.copy(self.t, self.U[k])
                b.spmv(1.0, A, self.U[k], 0.0, self.G[k])
                # biorthogonalize G[k] against Pshadow[0..k-1]
                for i in range(k):
                    alpha = self.inner(self.G[k], self.Pshadow[i]) / M[i, i]
                    b.axpby(-alpha, self.G[i], 1.0, self.G[k])
                    b.axpby(-alpha, self.U[i], 1.0, self.U[k])
                for i in range(k, s):
                    M[i, k] = self.inner(self.G[k], self.Pshadow[i])
                if M[k, k] == 0.0:
                    return iters, res / norm_rhs
                beta = f[k] / M[k, k]
                b.axpby(-beta, self.G[k], 1.0, self.r)
                b.axpby(beta, self.U[k], 1.0, x)
                res = self.norm(self.r)
                iters += 1
                if res <= eps or iters >= prm["maxiter"]:
                    break
                for i in range(k + 1, s):
                    f[i] -= beta * M[i, k]
                f[k] = 0.0
            if res <= eps or iters >= prm["maxiter"]:
                break
            # dimension-reduction step
            P.apply(self.r, self.tmp)
            b.spmv(1.0, A, self.tmp, 0.0, self.t)
            tr = self.inner(self.t, self.r)
            tt = self.inner(self.t, self.t)
            nr = self.norm(self.r)
            nt = np.sqrt(tt)
            if tt == 0.0:
                break
            om = tr / tt
            rho = abs(tr / (nt * nr)) if nt * nr > 0 else 0.0
            if rho < angle and rho > 0:
                om *= angle / rho
            if om == 0.0:
                break
            b.axpby(om, self.tmp, 1.0, x)
            b.axpby(-om, self.t, 1.0, self.r)
            res = self.norm(self.r)
            iters += 1
        return iters, res / norm_rhs
