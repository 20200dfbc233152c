"""Fully-coupled distributed CPR (constrained pressure residual).

Parity: amgcl/mpi/cpr.hpp — the quasi-IMPES pressure matrix App is a
distributed operator (one hierarchy over all ranks via the nested
distributed pressure solver) and the global smoother runs on the full
distributed system. With cell-interleaved unknowns and strips aligned to
cell boundaries, the restriction W^T and prolongation U are block-diagonal
over ranks, so App's strip assembles from the local strip alone:

    App[i, j] = sum_k w[i*b+k] * K(i*b+k, j*b)   (pressure column of cell j)

and only the pressure SOLVE communicates (halo exchange + reductions inside
the nested distributed solver), exactly like the reference's distributed
first stage.
"""
import numpy as np

from ..matrix import CSR
from ..params import merge_params
from .dist_matrix import DistMatrix


class DistCPR:
    @staticmethod
    def defaults():
        return {
            "block_size": 2,
            "drs": False,
            "drs_eps_ps": 0.02,
            "psolver": {"precond": {"class": "amg"},
                        "solver": {"type": "preonly"}},
            "sprecond": {"type": "spai0"},  # global smoother (diagonal family)
        }

    def __init__(self, strip: CSR, prm, backend, group=None):
        import scipy.sparse as sp

        from .solver import DistSolver

        p = merge_params(self.defaults(), prm, opaque=("psolver", "sprecond"))
        self.backend = backend  # DistBackend
        base = backend.base if hasattr(backend, "base") else backend
        self.base = base
        b = int(p["block_size"])
        self.bsize = b
        n_loc = strip.nrows
        if n_loc % b:
            raise ValueError("strip size not divisible by block_size "
                             "(partition strips on cell boundaries)")
        ncells = n_loc // b

        self.A = DistMatrix(strip, base, group)
        if self.A.row_beg % b:
            raise ValueError("strip offset not aligned to cell boundaries")

        # quasi-IMPES weights from the FULL strip rows
        m = strip.to_scipy()  # rows local, cols global
        if p["drs"]:
            rs = np.abs(m).sum(axis=1)
            rs = np.asarray(rs).ravel()
            w = 1.0 / np.maximum(rs, float(p["drs_eps_ps"]))
        else:
            w = np.ones(n_loc)
        rows = np.arange(n_loc) // b
        Wt = sp.csr_matrix((w, (rows, np.arange(n_loc))), shape=(ncells, n_loc))
        # App strip: sum the weighted cell equations, keep pressure columns
        # (global column c contributes to global cell c // b iff c % b == 0)
        Ared = (Wt @ m).tocsr()
        keep = (Ared.indices % b) == 0
        rr = np.repeat(np.arange(ncells, dtype=np.int64), np.diff(Ared.indptr))[keep]
        cc = (Ared.indices[keep] // b).astype(np.int64)
        App = sp.coo_matrix((Ared.data[keep], (rr, cc)),
                            shape=(ncells, self.A.n_global // b)).tocsr()
        App.sort_indices()

        self.psolve = DistSolver(CSR.from_scipy(App), dict(p["psolver"]),
                                 backend=base, group=group)

        # global smoother weights from the full strip rows (diagonal family)
        sprm = dict(p["sprecond"] or {})
        kind = sprm.pop("type", "spai0")
        dia = np.zeros(n_loc)
        loc = self.A.A_loc_host
        rl = np.repeat(np.arange(n_loc), np.diff(loc.ptr))
        dmask = np.asarray(loc.col) == np.arange(n_loc)[rl]
        np.add.at(dia, rl[dmask], np.asarray(loc.val)[dmask])
        if kind == "spai0":
            den = np.asarray(m.multiply(m).sum(axis=1)).ravel()
            ms = np.divide(dia, den, out=np.zeros_like(dia), where=den != 0)
        elif kind == "damped_jacobi":
            damping = float(sprm.pop("damping", 0.72))
            ms = np.divide(damping, dia, out=np.zeros_like(dia), where=dia != 0)
        else:
            raise ValueError(f"dist cpr smoother '{kind}' not supported")
        self.M = base.from_host(ms)

        self.Wt = base.matrix(CSR.from_scipy(Wt))
        U = sp.csr_matrix((np.ones(ncells),
                           (np.arange(ncells) * b, np.arange(ncells))),
                          shape=(n_loc, ncells))
        self.U = base.matrix(CSR.from_scipy(U))
        self.rp = base.vector(ncells)
        self.tmp = base.vector(n_loc)
        self.r2 = base.vector(n_loc)

    def system_matrix(self):
        return self.A

    def apply(self, rhs, x):
        bk = self.backend
        base = self.base
        base.spmv(1.0, self.Wt, rhs, 0.0, self.rp)       # local restriction
        pc, _, _ = self.psolve(self.rp)                   # distributed stage 1
        base.spmv(1.0, self.U, pc, 0.0, x)                # local prolongation
        bk.residual(rhs, self.A, x, self.r2)              # halo-exchanged
        base.vmul(1.0, self.M, self.r2, 0.0, self.tmp)    # global smoother
        base.axpby(1.0, self.tmp, 1.0, x)
