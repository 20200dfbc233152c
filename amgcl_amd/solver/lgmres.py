"""LGMRES: 'loose' GMRES augmented with error-approximation vectors from
previous restart cycles.

Parity: amgcl/solver/lgmres.hpp:94 (M=30 inner iterations, K=3 augmented
vectors, always_reset semantics simplified: the augmentation pool persists
across restarts).
"""
import math

import numpy as np

from .base import SolverBase


class LGMRES(SolverBase):
    @staticmethod
    def defaults():
        return {"M": 30, "K": 3, "store_Av": True}

    def __init__(self, n, prm=None, backend=None, inner_product=None):
        self._init_common(n, prm, backend, inner_product)
        b = self.backend
        m = int(self.prm["M"])
        if m < 1:
            raise ValueError("gmres restart M must be >= 1")
        k = int(self.prm["K"])
        self.m, self.k = m, k
        self.r = b.vector(n)
        self.w = b.vector(n)
        self.v = [b.vector(n) for _ in range(m + k + 1)]
        self.z = [b.vector(n) for _ in range(m + k)]
        self.aug = []  # list of (z_err,) previous outer corrections
        self.n = n

    def solve(self, A, P, rhs, x):
        b = self.backend
        prm = self.prm
        mk = self.m
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
            naug = min(len(self.aug), self.k)
            msize = self.m + naug
            dt = getattr(self.backend, "dtype", np.float64)
            dt = (np.complex128 if "complex" in str(dt)
                  else (np.float32 if "float32" in str(dt) else np.float64))
            cplx = np.issubdtype(np.dtype(dt), np.complexfloating)
            H = np.zeros((msize + 1, msize), dtype=dt)
            cs = np.zeros(msize + 1)
            sn = np.zeros(msize + 1, dtype=dt)
            g = np.zeros(msize + 1, dtype=dt)
            g[0] = beta
            b.axpby(1.0 / beta, self.r, 0.0, self.v[0])
            dx_dirs = []
            i = -1
            while i + 1 < msize and iters < prm["maxiter"]:
                i += 1
                if i < self.m:
                    P.apply(self.v[i], self.z[i])
                    zdir = self.z[i]
                else:
                    zdir = self.aug[i - self.m]  # augmented direction
                dx_dirs.append(zdir)
                b.spmv(1.0, A, zdir, 0.0, self.w)
                for kk in range(i + 1):
                    h = self.inner(self.w, self.v[kk])
                    H[kk, i] = h
                    b.axpby(-h, self.v[kk], 1.0, self.w)
                hn = self.norm(self.w)
                H[i + 1, i] = hn
                if hn != 0.0:
                    b.axpby(1.0 / hn, self.w, 0.0, self.v[i + 1])
                if cplx:
                    for kk in range(i):
                        h0 = cs[kk] * H[kk, i] + sn[kk] * H[kk + 1, i]
                        h1 = -np.conj(sn[kk]) * H[kk, i] + cs[kk] * H[kk + 1, i]
                        H[kk, i], H[kk + 1, i] = h0, h1
                    cs_, sn_, d = _complex_givens(H[i, i], H[i + 1, i])
                    if d == 0.0:
                        cs_, sn_ = 1.0, 0.0
                    cs[i], sn[i] = cs_, sn_
                    H[i, i] = d
                    H[i + 1, i] = 0.0
                    g0, g1 = cs_ * g[i], -np.conj(sn_) * g[i]
                else:
                    for kk in range(i):
                        h0 = cs[kk] * H[kk, i] + sn[kk] * H[kk + 1, i]
                        h1 = -sn[kk] * H[kk, i] + cs[kk] * H[kk + 1, i]
                        H[kk, i], H[kk + 1, i] = h0, h1
                    d = math.hypot(H[i, i].real, H[i + 1, i].real)
                    cs[i] = H[i, i].real / d if d else 1.0
                    sn[i] = H[i + 1, i].real / d if d else 0.0
                    H[i, i] = d
                    H[i + 1, i] = 0.0
                    g0, g1 = cs[i] * g[i], -sn[i] * g[i]
                g[i], g[i + 1] = g0, g1
                res = abs(g1)
                iters += 1
                if res <= eps:
                    break
            y = np.zeros(i + 1, dtype=H.dtype)
            for kk in range(i, -1, -1):
                y[kk] = (g[kk] - H[kk, kk + 1 : i + 1] @ y[kk + 1 : i + 1]) / H[kk, kk]
            # outer correction dx = sum y_k dir_k (store as next aug vector)
            dx = b.vector(self.n)
            for kk in range(i + 1):
                b.axpby(y[kk], dx_dirs[kk], 1.0, dx)
            b.axpby(1.0, dx, 1.0, x)
            nrm = self.norm(dx)
            if nrm > 0:
                b.axpby(1.0 / nrm - 1.0, dx, 1.0, dx)  # normalize in place
                self.aug.insert(0, dx)
                del self.aug[self.k :]
            if res <= eps:
                break
        return iters, res / norm_rhs


from .gmres import _complex_givens  # noqa: E402
