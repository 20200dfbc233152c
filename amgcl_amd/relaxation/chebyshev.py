"""Chebyshev polynomial smoother.

Parity: amgcl/relaxation/chebyshev.hpp:54 — degree-5 polynomial on
[rho*lower, rho] with rho from Gershgorin (power_iters=0) or power iteration;
apply is the standard Chebyshev iteration built on residual/spmv/axpby, so it
runs entirely on the backend (GPU-capable).
"""
import numpy as np

from ..params import merge_params


class Chebyshev:
    gpu_supported = True

    @staticmethod
    def defaults():
        return {"degree": 5, "lower": 1.0 / 30.0, "power_iters": 0, "scale": False}

    def __init__(self, A, prm, backend):
        from ..matrix import CSR

        p = merge_params(self.defaults(), prm)
        self.degree = int(p["degree"])
        if self.degree < 1:
            raise ValueError("chebyshev degree must be >= 1")
        self.backend = backend
        self.scale = bool(p["scale"])

        if not isinstance(A, CSR):
            # device-resident level: Gershgorin + diagonal on the GPU
            from ..backend import hip_setup

            if self.scale:
                d32 = hip_setup.diagonal(A)
                self.Dinv = 1.0 / d32
            else:
                self.Dinv = None
            rho = hip_setup.gershgorin(A, self.scale)
        else:
            d = np.asarray(A.diagonal())
            if self.scale:
                self.Dinv = backend.from_host(1.0 / d)
            else:
                self.Dinv = None
            rho = self._spectral_radius(A, d, int(p["power_iters"]))
        hi = rho
        lo = rho * float(p["lower"])
        self.theta = 0.5 * (hi + lo)
        self.delta = 0.5 * (hi - lo)
        self.sigma1 = self.theta / self.delta if self.delta > 0 else 1.0

        n = A.nrows
        self._d = backend.vector(n)
        self._r = backend.vector(n)

    def _spectral_radius(self, A, diag, power_iters):
        if power_iters > 0:
            from ..coarsening.smoothed_aggregation import spectral_radius_dinv_a

            if self.scale:
                return spectral_radius_dinv_a(A, power_iters)
            # power iteration on A itself
            rng = np.random.default_rng(1234)
            b0 = rng.random(A.nrows)
            b0 /= np.linalg.norm(b0)
            rho = 1.0
            b1 = np.empty_like(b0)
            for _ in range(power_iters):
                A.spmv(1.0, b0, 0.0, b1)
                rho = float(np.dot(b0, b1))
                nrm = np.linalg.norm(b1)
                if nrm == 0:
                    break
                b0 = b1 / nrm
            return abs(rho)
        # Gershgorin bound (amgcl/backend/builtin.hpp:781, power_iters=0)
        rowsum = np.zeros(A.nrows)
        np.add.at(rowsum, np.repeat(np.arange(A.nrows), np.diff(A.ptr)), np.abs(A.val))
        if self.scale:
            rowsum /= np.abs(diag)
        return float(rowsum.max())

    def _polynomial(self, A, rhs, x, tmp):
        """Chebyshev iteration (Saad, Alg 12.1): smooths x in place."""
        b = self.backend
        d, r = self._d, self._r
        b.residual(rhs, A, x, r)
        if self.Dinv is not None:
            b.vmul(1.0, self.Dinv, r, 0.0, r)
        rho = 1.0 / self.sigma1
        b.axpby(1.0 / self.theta, r, 0.0, d)
        for _ in range(self.degree):
            b.axpby(1.0, d, 1.0, x)          # x += d
            b.residual(rhs, A, x, r)          # r = rhs - A x
            if self.Dinv is not None:
                b.vmul(1.0, self.Dinv, r, 0.0, r)
            rho_next = 1.0 / (2.0 * self.sigma1 - rho)
            b.axpby(2.0 * rho_next / self.delta, r, rho_next * rho, d)
            rho = rho_next

    def apply_pre(self, A, rhs, x, tmp):
        self._polynomial(A, rhs, x, tmp)

    def apply_post(self, A, rhs, x, tmp):
        self._polynomial(A, rhs, x, tmp)

    def apply(self, A, rhs, x, tmp=None):
        self.backend.clear(x)
        if tmp is None:
            tmp = self.backend.vector(len_of(x))
        self._polynomial(A, rhs, x, tmp)


def len_of(x):
    try:
        return x.shape[0]
    except AttributeError:
        return len(x)
