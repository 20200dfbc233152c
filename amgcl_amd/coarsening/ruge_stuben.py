"""Classic Ruge-Stuben coarsening.

Parity: amgcl/coarsening/ruge_stuben.hpp:54 — strong negative couplings
(eps_strong = 0.25), lambda-bucket C/F splitting, direct interpolation with
truncation (eps_trunc = 0.2) and rescaling. Host-only setup (the C/F split is
inherently sequential); on the HIP backend the device matrix is downloaded
first (AMG's overflow fallback hook).
"""
from .. import _core
from ..matrix import CSR, galerkin
from ..params import merge_params


class RugeStuben:
    @staticmethod
    def defaults():
        return {"eps_strong": 0.25, "do_trunc": True, "eps_trunc": 0.2}

    def __init__(self, prm=None):
        self.prm = merge_params(self.defaults(), prm)

    def transfer_operators(self, A):
        if getattr(A, "is_complex", False):
            raise ValueError("ruge_stuben is real-valued; complex systems use "
                             "smoothed_aggregation (or the 2x2-real adapter)")
        if not isinstance(A, CSR):
            # host-only algorithm: trigger make_solver/AMG's host fallback
            raise OverflowError("ruge_stuben runs on the host")
        pp, pc, pv, nc = _core.ruge_stuben(
            A.nrows, A.ptr, A.col, A.val,
            float(self.prm["eps_strong"]), bool(self.prm["do_trunc"]),
            float(self.prm["eps_trunc"]),
        )
        P = CSR(A.nrows, nc, pp, pc, pv)
        return P, P.transpose()

    def coarse_operator(self, A, P, R):
        return galerkin(R, A, P)
