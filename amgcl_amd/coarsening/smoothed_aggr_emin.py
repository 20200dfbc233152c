"""Energy-minimizing smoothed aggregation (nonsymmetric-friendly).

Parity: amgcl/coarsening/smoothed_aggr_emin.hpp:55-363 —
filtered matrix Af (strong entries, weak folded into the diagonal), and
per-coarse-column energy-minimizing weights

    omega_j = (AP, ADAP)_j / (ADAP, ADAP)_j,   ADAP = Af D^-1 AP

with P = P_tent - D^-1 (Af P_tent) Omega and R smoothed with Af^T
(so P and R differ for nonsymmetric operators). Host-only setup.
"""
import numpy as np

from .. import _core
from ..matrix import CSR, galerkin
from ..params import merge_params


class SmoothedAggrEMin:
    @staticmethod
    def defaults():
        return {"eps_strong": 0.08, "block_size": 1}

    def __init__(self, prm=None):
        self.prm = merge_params(self.defaults(), prm)
        self.eps_strong = float(self.prm["eps_strong"])

    def _filtered(self, A: CSR, strong):
        import scipy.sparse as sp

        row_of = np.repeat(np.arange(A.nrows), np.diff(A.ptr))
        S = np.asarray(strong).astype(bool)
        diag_mask = A.col == row_of
        weak = (~S) & (~diag_mask)
        dia = np.zeros(A.nrows)
        np.add.at(dia, row_of[diag_mask], A.val[diag_mask])
        np.add.at(dia, row_of[weak], A.val[weak])
        keep = S | diag_mask
        vals = A.val.copy()
        vals[diag_mask] = dia[row_of[diag_mask]]
        Af = sp.csr_matrix((vals[keep], (row_of[keep], A.col[keep])), shape=A.shape)
        Af.sort_indices()
        return Af, dia

    def transfer_operators(self, A):
        if getattr(A, "is_complex", False):
            raise ValueError("smoothed_aggr_emin is real-valued; complex "
                             "systems use smoothed_aggregation")
        if not isinstance(A, CSR):
            raise OverflowError("smoothed_aggr_emin runs on the host")
        import scipy.sparse as sp

        bsize = int(self.prm["block_size"])
        if bsize > 1 and A.nrows % bsize == 0:
            pp, pc, pv = _core.pointwise_matrix(A.nrows, A.ptr, A.col, A.val, bsize)
            np_ = A.nrows // bsize
            agg = _core.aggregates_parallel if np_ > 100_000 else _core.aggregates
            naggr, id_p, strong_p = agg(np_, pp, pc, pv, self.eps_strong)
            aggr_id = np.where(np.repeat(np.asarray(id_p), bsize) >= 0,
                               np.repeat(np.asarray(id_p), bsize), -2).astype(np.int32)
            strong = _core.expand_strong(A.nrows, A.ptr, A.col, bsize, pp, pc, strong_p)
        else:
            agg = _core.aggregates_parallel if A.nrows > 100_000 else _core.aggregates
            naggr, aggr_id, strong = agg(A.nrows, A.ptr, A.col, A.val, self.eps_strong)
        self.eps_strong *= 0.5

        tp, tc, tv = _core.tentative_prolongation(A.nrows, aggr_id, naggr)
        Pt = CSR(A.nrows, naggr, tp, tc, tv).to_scipy()

        Af, dia = self._filtered(A, strong)
        dinv = sp.diags(1.0 / dia)

        AP = (Af @ Pt).tocsr()
        ADAP = (Af @ (dinv @ AP)).tocsr()
        num = np.asarray(AP.multiply(ADAP).sum(axis=0)).ravel()
        den = np.asarray(ADAP.multiply(ADAP).sum(axis=0)).ravel()
        omega = np.where(den > 0, num / np.maximum(den, 1e-300), 0.0)
        omega = np.maximum(omega, 0.0)  # negative energy weights are clipped
        Om = sp.diags(omega)

        P = (Pt - dinv @ AP @ Om).tocsr()
        # restriction smoothed with Af^T (nonsymmetric support)
        ATP = (Af.T.tocsr() @ Pt).tocsr()
        R = (Pt - dinv @ ATP @ Om).T.tocsr()
        P.sort_indices()
        R.sort_indices()
        return CSR.from_scipy(P), CSR.from_scipy(R)

    def coarse_operator(self, A, P, R):
        return galerkin(R, A, P)
