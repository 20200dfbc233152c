"""Smoothed aggregation coarsening (the default).

Parity: amgcl/coarsening/smoothed_aggregation.hpp:56-242 —
P = (I - omega * D^-1 * A_F) * P_tent with the filtered matrix A_F folding
weak connections into the diagonal; omega = relax*2/3 (or relax*4/3/rho with
spectral-radius estimation); R = P^T; Galerkin coarse operator; eps_strong
halved after each level.
"""
import numpy as np

from .. import _core
from ..matrix import CSR, galerkin
from ..params import merge_params


class SmoothedAggregation:
    @staticmethod
    def defaults():
        return {
            "eps_strong": 0.08,
            "relax": 1.0,
            "estimate_spectral_radius": False,
            "power_iters": 5,
        }

    def __init__(self, prm=None):
        self.prm = merge_params(self.defaults(), prm)
        self.eps_strong = float(self.prm["eps_strong"])

    def transfer_operators(self, A):
        from ..profiler import prof

        if not isinstance(A, CSR):
            return self._transfer_operators_device(A)
        with prof.scope("aggregates"):
            # large levels: deterministic parallel MIS(2) aggregation
            # (mirrors the reference's own distributed pmis design);
            # small levels: the exact greedy reference pass.
            agg = _core.aggregates_parallel if A.nrows > 100_000 else _core.aggregates
            naggr, aggr_id, strong = agg(A.nrows, A.ptr, A.col, A.val, self.eps_strong)
        self.eps_strong *= 0.5  # reference halves eps per level (s_a.hpp:140)

        omega = float(self.prm["relax"])
        if self.prm["estimate_spectral_radius"]:
            omega *= (4.0 / 3.0) / spectral_radius_dinv_a(A, int(self.prm["power_iters"]))
        else:
            omega *= 2.0 / 3.0

        with prof.scope("smooth_P"):
            pp, pc, pv = _core.smoothed_prolongation(
                A.nrows, A.ptr, A.col, A.val, strong, aggr_id, naggr, omega
            )
            P = CSR(A.nrows, naggr, pp, pc, pv)
        with prof.scope("transpose_R"):
            R = P.transpose()
        return P, R

    def _transfer_operators_device(self, A):
        """Device twin (backend/hip_setup.py): same algorithm, same keys."""
        from ..backend import hip_setup
        from ..profiler import prof

        with prof.scope("aggregates(dev)"):
            naggr, ids, strong = hip_setup.aggregates(A, self.eps_strong)
        self.eps_strong *= 0.5
        omega = float(self.prm["relax"]) * (2.0 / 3.0)
        with prof.scope("smooth_P(dev)"):
            P = hip_setup.smoothed_prolongation(A, strong, ids, naggr, omega)
        with prof.scope("transpose_R(dev)"):
            R = hip_setup.transpose(P)
        return P, R

    def coarse_operator(self, A, P, R):
        return galerkin(R, A, P)


def spectral_radius_dinv_a(A: CSR, iters):
    """Power iteration estimate of rho(D^-1 A)
    (parity: amgcl/backend/builtin.hpp:781 spectral_radius<true>)."""
    rng = np.random.default_rng(12345)
    d = 1.0 / A.diagonal()
    b0 = rng.random(A.nrows)
    b0 /= np.linalg.norm(b0)
    rho = 2.0
    b1 = np.empty_like(b0)
    for _ in range(max(1, iters)):
        A.spmv(1.0, b0, 0.0, b1)
        b1 *= d
        rho = float(np.dot(b0, b1))
        nrm = np.linalg.norm(b1)
        if nrm == 0:
            break
        b0, b1 = b1 / nrm, b0
    return abs(rho)
