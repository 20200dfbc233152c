"""Non-smoothed aggregation.

Parity: amgcl/coarsening/aggregation.hpp:71 — P is the piecewise-constant
tentative prolongation; coarse operator is the scaled Galerkin product with
an over-correction factor (amgcl/coarsening/detail/scaled_galerkin.hpp:36).
"""
from .. import _core
from ..matrix import CSR, galerkin
from ..params import merge_params


class Aggregation:
    @staticmethod
    def defaults():
        return {"eps_strong": 0.08, "over_interp": 1.5}

    def __init__(self, prm=None):
        self.prm = merge_params(self.defaults(), prm)
        self.eps_strong = float(self.prm["eps_strong"])

    def transfer_operators(self, A):
        if not isinstance(A, CSR):
            from ..backend import hip_setup

            naggr, ids, _strong = hip_setup.aggregates(A, self.eps_strong)
            self.eps_strong *= 0.5
            P = hip_setup.tentative_prolongation(A, ids, naggr)
            return P, hip_setup.transpose(P)
        naggr, aggr_id, _strong = _core.aggregates(
            A.nrows, A.ptr, A.col, A.val, self.eps_strong
        )
        self.eps_strong *= 0.5
        pp, pc, pv = _core.tentative_prolongation(A.nrows, aggr_id, naggr)
        P = CSR(A.nrows, naggr, pp, pc, pv)
        return P, P.transpose()

    def coarse_operator(self, A, P, R):
        Ac = galerkin(R, A, P)
        scale = 1.0 / float(self.prm["over_interp"])
        Ac.val *= scale
        return Ac
