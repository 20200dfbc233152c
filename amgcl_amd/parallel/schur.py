"""Fully-coupled distributed Schur pressure correction.

Parity: amgcl/mpi/schur_pressure_correction.hpp:62-570 — the global system
is split by the pressure mask into distributed field blocks

    K = [[Kuu, Kup], [Kpu, Kpp]]   (each block a distributed matrix),

the Schur complement is approximated explicitly as
S = Kpp - Kpu diag(Kuu)^-1 Kup (SIMPLE/SIMPLEC) with the cross-rank terms of
the product included (the diag(Kuu)^-1 Kup rows of ghost u-columns are
fetched from their owners over Kpu's comm pattern), and the nested velocity
and pressure solvers are themselves distributed solvers. One application is
the usual correction sweep; every SpMV in it is a halo-exchanged distributed
SpMV over the rectangular blocks.
"""
import numpy as np

from ..matrix import CSR
from ..params import merge_params
from .dist_matrix import DistMatrix


class DistSchurPressureCorrection:
    @staticmethod
    def defaults():
        return {
            "usolver": {"precond": {"class": "relaxation", "type": "spai0"},
                        "solver": {"type": "preonly"}},
            "psolver": {"precond": {"class": "amg"},
                        "solver": {"type": "preonly"}},
            "pmask_raw": None,      # boolean mask over MY rows
            "pmask_pattern": None,  # e.g. "%4"
            "simplec": True,
        }

    def __init__(self, strip: CSR, prm, backend, group=None):
        import torch.distributed as dist

        from .solver import DistSolver

        p = merge_params(self.defaults(), prm, opaque=("usolver", "psolver"))
        self.backend = backend  # DistBackend
        base = backend.base if hasattr(backend, "base") else backend
        self.base = base
        self.group = group
        self.dist = dist

        n_loc = strip.nrows
        if p["pmask_raw"] is not None:
            pm = np.asarray(p["pmask_raw"], dtype=bool)
        elif p["pmask_pattern"]:
            pat = str(p["pmask_pattern"])
            b = int(pat[1:])
            # the pattern applies to GLOBAL indices, so ranks agree
            sizes = [None] * dist.get_world_size(group)
            dist.all_gather_object(sizes, n_loc, group=group)
            beg = int(np.sum(sizes[: dist.get_rank(group)]))
            pm = (np.arange(beg, beg + n_loc) % b) == (b - 1)
        else:
            raise ValueError("schur needs pmask_raw or pmask_pattern")
        self.pmask = pm
        self.uidx = np.where(~pm)[0]
        self.pidx = np.where(pm)[0]
        nu_loc, np_loc = len(self.uidx), len(self.pidx)

        # the full operator (also the outer Krylov matrix)
        self.A = DistMatrix(strip, base, group)
        A = self.A
        world, rank = A.world, A.rank

        # global typed numbering: u and p are each packed by rank
        u_sizes = [None] * world
        p_sizes = [None] * world
        dist.all_gather_object(u_sizes, nu_loc, group=group)
        dist.all_gather_object(p_sizes, np_loc, group=group)
        u_beg = int(np.sum(u_sizes[:rank]))
        p_beg = int(np.sum(p_sizes[:rank]))
        typed_loc = np.empty(n_loc, dtype=np.int64)
        typed_loc[self.uidx] = u_beg + np.arange(nu_loc)
        typed_loc[self.pidx] = p_beg + np.arange(np_loc)
        self.n_u_glob = int(np.sum(u_sizes))
        self.n_p_glob = int(np.sum(p_sizes))

        # fetch (is_p, typed id) of my ghost columns from their owners over
        # the operator's comm pattern (reference exchanges the same boundary
        # metadata in schur_pressure_correction.hpp init)
        packets = {}
        for r, idx in zip(A.send_ranks, A.send_idx):
            rows = np.asarray(idx.cpu() if hasattr(idx, "cpu") else idx,
                              dtype=np.int64)
            packets[r] = (pm[rows], typed_loc[rows])
        gathered = [None] * world
        dist.all_gather_object(gathered, packets, group=group)
        pm_ghost = np.zeros(A.n_ghost, dtype=bool)
        typed_ghost = np.zeros(A.n_ghost, dtype=np.int64)
        off = 0
        for r, cnt in zip(A.recv_ranks, A.recv_counts):
            pmr, tyr = gathered[r][rank]
            pm_ghost[off : off + cnt] = pmr
            typed_ghost[off : off + cnt] = tyr
            off += cnt

        # split the strip into the 4 field blocks with global typed columns
        import scipy.sparse as sp

        def entries(csr_part, colmap_pm, colmap_ty):
            if csr_part is None:
                return (np.empty(0, np.int64), np.empty(0, bool),
                        np.empty(0, np.int64), np.empty(0))
            rows = np.repeat(np.arange(csr_part.nrows, dtype=np.int64),
                             np.diff(csr_part.ptr))
            cols = np.asarray(csr_part.col)
            return rows, colmap_pm[cols], colmap_ty[cols], np.asarray(csr_part.val)

        lr, lcp, lct, lv = entries(A.A_loc_host, pm, typed_loc)
        rr, rcp, rct, rv = entries(A.A_rem_host, pm_ghost, typed_ghost)
        rows = np.concatenate([lr, rr])
        cp = np.concatenate([lcp, rcp])
        ct = np.concatenate([lct, rct])
        vals = np.concatenate([lv, rv])
        row_is_p = pm[rows]
        urow_of = np.full(n_loc, -1, dtype=np.int64)
        urow_of[self.uidx] = np.arange(nu_loc)
        prow_of = np.full(n_loc, -1, dtype=np.int64)
        prow_of[self.pidx] = np.arange(np_loc)

        def block(rmask, cmask, rmap, ncols, nrows):
            sel = rmask & cmask
            M = sp.coo_matrix(
                (vals[sel], (rmap[rows[sel]], ct[sel])), shape=(nrows, ncols)
            ).tocsr()
            M.sort_indices()
            return CSR.from_scipy(M)

        Kuu = block(~row_is_p, ~cp, urow_of, self.n_u_glob, nu_loc)
        Kup = block(~row_is_p, cp, urow_of, self.n_p_glob, nu_loc)
        Kpu = block(row_is_p, ~cp, prow_of, self.n_u_glob, np_loc)
        Kpp = block(row_is_p, cp, prow_of, self.n_p_glob, np_loc)

        # SIMPLE(C) diagonal of Kuu over the FULL row (local + remote parts)
        if p["simplec"]:
            ru = np.repeat(np.arange(nu_loc), np.diff(Kuu.ptr))
            dsum = np.zeros(nu_loc)
            np.add.at(dsum, ru, np.abs(np.asarray(Kuu.val)))
        else:
            dsum = np.zeros(nu_loc)
            ru = np.repeat(np.arange(nu_loc), np.diff(Kuu.ptr))
            dia_mask = np.asarray(Kuu.col) == (u_beg + np.arange(nu_loc))[ru]
            np.add.at(dsum, ru[dia_mask], np.asarray(Kuu.val)[dia_mask])
        self.dinv_host = np.divide(1.0, dsum, out=np.zeros_like(dsum),
                                   where=dsum != 0)

        # rectangular distributed blocks for the correction sweeps
        self.Kup_d = DistMatrix(Kup, base, group, col_sizes=p_sizes)
        self.Kpu_d = DistMatrix(Kpu, base, group, col_sizes=u_sizes)

        # explicit distributed S = Kpp - Kpu D^-1 Kup: the D^-1 Kup rows of
        # ghost u-columns come from their owners over Kpu_d's comm pattern
        B = Kup.to_scipy().tocsr()
        B = sp.diags(self.dinv_host) @ B
        packets = {}
        for r, idx in zip(self.Kpu_d.send_ranks, self.Kpu_d.send_idx):
            rws = np.asarray(idx.cpu() if hasattr(idx, "cpu") else idx,
                             dtype=np.int64)
            sub = B[rws].tocsr()
            packets[r] = (sub.indptr, sub.indices, sub.data)
        gathered = [None] * world
        dist.all_gather_object(gathered, packets, group=group)
        mid = self.Kpu_d.A_loc_host.to_scipy() @ B
        if self.Kpu_d.n_ghost:
            blocks = []
            for r in self.Kpu_d.recv_ranks:
                ptr, col, val = gathered[r][rank]
                blocks.append(sp.csr_matrix((val, col, ptr),
                                            shape=(len(ptr) - 1, self.n_p_glob)))
            B_ghost = sp.vstack(blocks, format="csr")
            mid = mid + self.Kpu_d.A_rem_host.to_scipy() @ B_ghost
        S = (Kpp.to_scipy() - mid).tocsr()
        S.sort_indices()
        S_strip = CSR.from_scipy(S)

        self.usolve = DistSolver(Kuu, dict(p["usolver"]), backend=base,
                                 group=group)
        self.psolve = DistSolver(S_strip, dict(p["psolver"]), backend=base,
                                 group=group)

        self.dinv = base.from_host(self.dinv_host)
        self.rhs_u = base.vector(nu_loc)
        self.rhs_p = base.vector(np_loc)
        self.tmp_u = base.vector(nu_loc)
        if getattr(base, "name", "") == "hip":
            import torch

            self.uidx_d = torch.from_numpy(self.uidx.astype(np.int32)).to(base.device)
            self.pidx_d = torch.from_numpy(self.pidx.astype(np.int32)).to(base.device)
        else:
            self.uidx_d, self.pidx_d = self.uidx, self.pidx

    def system_matrix(self):
        return self.A

    def apply(self, rhs, x):
        bk = self.backend
        base = self.base
        base.gather(rhs, self.uidx_d, self.rhs_u)
        base.gather(rhs, self.pidx_d, self.rhs_p)
        u, _, _ = self.usolve(self.rhs_u)
        bk.spmv(-1.0, self.Kpu_d, u, 1.0, self.rhs_p)
        pvec, _, _ = self.psolve(self.rhs_p)
        bk.spmv(1.0, self.Kup_d, pvec, 0.0, self.tmp_u)
        base.vmul(-1.0, self.dinv, self.tmp_u, 1.0, u)
        base.scatter(u, self.uidx_d, x)
        base.scatter(pvec, self.pidx_d, x)
