"""Flexible GMRES (variable preconditioner).

Parity: amgcl/solver/fgmres.hpp:57 — stores the preconditioned directions
Z_k = M^-1 v_k so the preconditioner may change between iterations.
"""
import math

import numpy as np

from .base import SolverBase


class FGMRES(SolverBase):
    @staticmethod
    def defaults():
        return {"M": 30}

    def __init__(self, n, prm=None, backend=None, inner_product=None):
        self._init_common(n, prm, backend, inner_product)
        b = self.backend
        m = int(self.prm["M"])
        if m < 1:
            raise ValueError("gmres restart M must be >= 1")
        self.m = m
        self.r = b.vector(n)
        self.w = b.vector(n)
        self.v = [b.vector(n) for _ in range(m + 1)]
        self.z = [b.vector(n) for _ in range(m)]
        dt = getattr(backend, "dtype", np.float64)
        dt = (np.complex128 if "complex" in str(dt)
              else (np.float32 if "float32" in str(dt) else np.float64))
        self._cplx = np.issubdtype(np.dtype(dt), np.complexfloating)
        self.H = np.zeros((m + 1, m), dtype=dt)
        self.cs = np.zeros(m + 1)  # real also in the complex rotation
        self.sn = np.zeros(m + 1, dtype=dt)
        self.g = np.zeros(m + 1, dtype=dt)

    def solve(self, A, P, rhs, x):
        b = self.backend
        prm = self.prm
        norm_rhs = self.norm(rhs)
        if norm_rhs == 0.0:
            if not prm["ns_search"]:
                b.clear(x)
                return 0, 0.0
            norm_rhs = 1.0
        eps = max(prm["tol"] * norm_rhs, prm["abstol"])

        iters = 0
        res = 2 * eps
        while iters < prm["maxiter"]:
            b.residual(rhs, A, x, self.r)
            beta = self.norm(self.r)
            res = beta
            if res <= eps:
                break
            b.axpby(1.0 / beta, self.r, 0.0, self.v[0])
            self.g[:] = 0.0
            self.g[0] = beta
            i = -1
            while i + 1 < self.m and iters < prm["maxiter"]:
                i += 1
                P.apply(self.v[i], self.z[i])
                b.spmv(1.0, A, self.z[i], 0.0, self.w)
                for k in range(i + 1):
                    h = self.inner(self.w, self.v[k])
                    self.H[k, i] = h
                    b.axpby(-h, self.v[k], 1.0, self.w)
                hn = self.norm(self.w)
                self.H[i + 1, i] = hn
                if hn != 0.0:
                    b.axpby(1.0 / hn, self.w, 0.0, self.v[i + 1])
                if self._cplx:
                    for k in range(i):
                        h0 = (self.cs[k] * self.H[k, i]
                              + self.sn[k] * self.H[k + 1, i])
                        h1 = (-np.conj(self.sn[k]) * self.H[k, i]
                              + self.cs[k] * self.H[k + 1, i])
                        self.H[k, i], self.H[k + 1, i] = h0, h1
                    cs_, sn_, d = _complex_givens(self.H[i, i], self.H[i + 1, i])
                    if d == 0.0:
                        cs_, sn_ = 1.0, 0.0
                    self.cs[i], self.sn[i] = cs_, sn_
                    self.H[i, i] = d
                    self.H[i + 1, i] = 0.0
                    g0 = cs_ * self.g[i]
                    g1 = -np.conj(sn_) * self.g[i]
                else:
                    for k in range(i):
                        h0 = self.cs[k] * self.H[k, i] + self.sn[k] * self.H[k + 1, i]
                        h1 = -self.sn[k] * self.H[k, i] + self.cs[k] * self.H[k + 1, i]
                        self.H[k, i], self.H[k + 1, i] = h0, h1
                    d = math.hypot(self.H[i, i].real, self.H[i + 1, i].real)
                    self.cs[i] = self.H[i, i].real / d if d else 1.0
                    self.sn[i] = self.H[i + 1, i].real / d if d else 0.0
                    self.H[i, i] = d
                    self.H[i + 1, i] = 0.0
                    g0 = self.cs[i] * self.g[i]
                    g1 = -self.sn[i] * self.g[i]
                self.g[i], self.g[i + 1] = g0, g1
                res = abs(g1)
                iters += 1
                if res <= eps:
                    break
            # x += sum y_k z_k
            y = np.zeros(i + 1, dtype=self.H.dtype)
            for k in range(i, -1, -1):
                y[k] = (self.g[k] - self.H[k, k + 1 : i + 1] @ y[k + 1 : i + 1]) / self.H[k, k]
            for k in range(i + 1):
                b.axpby(y[k], self.z[k], 1.0, x)
            if res <= eps:
                break
        return iters, res / norm_rhs


from .gmres import _complex_givens  # noqa: E402
