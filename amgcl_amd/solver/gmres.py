"""Restarted GMRES(M) with Givens rotations.

Parity: amgcl/solver/gmres.hpp:59-322 (M=30 default, modified Gram-Schmidt
Arnoldi, Givens-rotation least squares, right preconditioning default).
The small Hessenberg system lives on the host; basis vectors on the backend.
"""
import math

import numpy as np

from .base import SolverBase


class GMRES(SolverBase):
    @staticmethod
    def defaults():
        return {"M": 30, "pside": "right"}

    def __init__(self, n, prm=None, backend=None, inner_product=None):
        self._init_common(n, prm, backend, inner_product)
        b = self.backend
        m = int(self.prm["M"])
        if m < 1:
            raise ValueError("gmres restart M must be >= 1")
        self.m = m
        self.r = b.vector(n)
        self.w = b.vector(n)
        self.tmp = b.vector(n)
        self.v = [b.vector(n) for _ in range(m + 1)]
        dt = getattr(backend, "dtype", np.float64)
        dt = (np.complex128 if "complex" in str(dt)
              else (np.float32 if "float32" in str(dt) else np.float64))
        self._cplx = np.issubdtype(np.dtype(dt), np.complexfloating)
        self.H = np.zeros((m + 1, m), dtype=dt)
        self.cs = np.zeros(m + 1)  # real also in the complex rotation
        self.sn = np.zeros(m + 1, dtype=dt)
        self.g = np.zeros(m + 1, dtype=dt)

    def _restart(self, A, P, rhs, x, left):
        b = self.backend
        if left:
            b.residual(rhs, A, x, self.tmp)
            P.apply(self.tmp, self.r)
        else:
            b.residual(rhs, A, x, self.r)
        beta = self.norm(self.r)
        if beta == 0.0:
            return 0.0
        b.axpby(1.0 / beta, self.r, 0.0, self.v[0])
        self.g[:] = 0.0
        self.g[0] = beta
        return beta

    def _iteration(self, A, P, i, left):
        b = self.backend
        if left:
            b.spmv(1.0, A, self.v[i], 0.0, self.tmp)
            P.apply(self.tmp, self.w)
        else:
            P.apply(self.v[i], self.tmp)
            b.spmv(1.0, A, self.tmp, 0.0, self.w)
        # modified Gram-Schmidt
        for k in range(i + 1):
            h = self.inner(self.w, self.v[k])
            self.H[k, i] = h
            b.axpby(-h, self.v[k], 1.0, self.w)
        hnext = self.norm(self.w)
        self.H[i + 1, i] = hnext
        if hnext != 0.0:
            b.axpby(1.0 / hnext, self.w, 0.0, self.v[i + 1])
        # apply existing Givens rotations to column i
        if self._cplx:
            for k in range(i):
                h0 = self.cs[k] * self.H[k, i] + self.sn[k] * self.H[k + 1, i]
                h1 = (-np.conj(self.sn[k]) * self.H[k, i]
                      + self.cs[k] * self.H[k + 1, i])
                self.H[k, i], self.H[k + 1, i] = h0, h1
            cs, sn, d = _complex_givens(self.H[i, i], self.H[i + 1, i])
            self.cs[i], self.sn[i] = cs, sn
            self.H[i, i] = d
            self.H[i + 1, i] = 0.0
            g0 = cs * self.g[i]
            g1 = -np.conj(sn) * self.g[i]
            self.g[i], self.g[i + 1] = g0, g1
            return abs(g1)
        for k in range(i):
            h0 = self.cs[k] * self.H[k, i] + self.sn[k] * self.H[k + 1, i]
            h1 = -self.sn[k] * self.H[k, i] + self.cs[k] * self.H[k + 1, i]
            self.H[k, i], self.H[k + 1, i] = h0, h1
        # new rotation
        d = math.hypot(self.H[i, i].real, self.H[i + 1, i].real)
        if d == 0.0:
            self.cs[i], self.sn[i] = 1.0, 0.0
        else:
            self.cs[i] = self.H[i, i].real / d
            self.sn[i] = self.H[i + 1, i].real / d
        self.H[i, i] = d
        self.H[i + 1, i] = 0.0
        g0 = self.cs[i] * self.g[i]
        g1 = -self.sn[i] * self.g[i]
        self.g[i], self.g[i + 1] = g0, g1
        return abs(g1)

    def _update(self, x, P, i, left):
        b = self.backend
        y = np.zeros(i + 1, dtype=self.H.dtype)
        for k in range(i, -1, -1):
            s = self.g[k] - self.H[k, k + 1 : i + 1] @ y[k + 1 : i + 1]
            y[k] = s / self.H[k, k]
        # accumulate z = sum y_k v_k, then x += P z (right) or x += z (left)
        b.clear(self.w)
        for k in range(i + 1):
            b.axpby(y[k], self.v[k], 1.0, self.w)
        if left:
            b.axpby(1.0, self.w, 1.0, x)
        else:
            P.apply(self.w, self.tmp)
            b.axpby(1.0, self.tmp, 1.0, x)

    def solve(self, A, P, rhs, x):
        prm = self.prm
        left = prm["pside"] == "left"
        norm_rhs = self.norm(rhs)
        if norm_rhs == 0.0:
            if not prm["ns_search"]:
                self.backend.clear(x)
                return 0, 0.0
            norm_rhs = 1.0
        eps = max(prm["tol"] * norm_rhs, prm["abstol"])

        iters = 0
        res = 2 * eps
        while iters < prm["maxiter"]:
            beta = self._restart(A, P, rhs, x, left)
            res = beta
            if res <= eps:
                break
            i = -1
            while i + 1 < self.m and iters < prm["maxiter"]:
                i += 1
                res = self._iteration(A, P, i, left)
                iters += 1
                if res <= eps:
                    break
            self._update(x, P, i, left)
            if res <= eps:
                break
        return iters, res / norm_rhs


def _complex_givens(h0, h1):
    """Complex Givens rotation: returns (cs real, sn complex, d) with
    cs*h0 + sn*h1 = d and -conj(sn)*h0 + cs*h1 = 0."""
    import numpy as _np

    a0, a1 = abs(h0), abs(h1)
    r = _np.hypot(a0, a1)
    if r == 0.0:
        return 1.0, 0.0 + 0.0j, 0.0
    if a0 == 0.0:
        return 0.0, _np.conj(h1) / a1, a1
    cs = a0 / r
    sn = (h0 / a0) * _np.conj(h1) / r
    return cs, sn, (h0 / a0) * r
