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
            "nullspace_raw": None,  # (n x k) near-nullspace block (e.g. RBM)
            "block_size": 1,        # dofs per grid point (pointwise aggregation)
        }

    def __init__(self, prm=None):
        self.prm = merge_params(self.defaults(), prm)
        self.eps_strong = float(self.prm["eps_strong"])
        ns = self.prm["nullspace_raw"]
        self.B = None if ns is None else np.ascontiguousarray(ns, dtype=np.float64)

    def transfer_operators(self, A):
        from ..profiler import prof

        if not isinstance(A, CSR):
            return self._transfer_operators_device(A)
        if A.is_complex:
            return self._transfer_operators_complex(A)
        with prof.scope("aggregates"):
            # large levels: deterministic parallel MIS(2) aggregation
            # (mirrors the reference's own distributed pmis design);
            # small levels: the exact greedy reference pass.
            bsize = int(self.prm["block_size"])
            if bsize > 1 and A.nrows % bsize == 0:
                # pointwise (block) aggregation: condense to points, aggregate,
                # expand (parity: coarsening/pointwise_aggregates.hpp:85)
                pp, pc, pv = _core.pointwise_matrix(A.nrows, A.ptr, A.col, A.val, bsize)
                np_ = A.nrows // bsize
                agg = _core.aggregates_parallel if np_ > 100_000 else _core.aggregates
                naggr, id_p, strong_p = agg(np_, pp, pc, pv, self.eps_strong)
                aggr_id = np.where(
                    np.repeat(np.asarray(id_p), bsize) >= 0,
                    np.repeat(np.asarray(id_p), bsize), -2
                ).astype(np.int32)
                strong = _core.expand_strong(A.nrows, A.ptr, A.col, bsize,
                                             pp, pc, strong_p)
            else:
                agg = _core.aggregates_parallel if A.nrows > 100_000 else _core.aggregates
                naggr, aggr_id, strong = agg(A.nrows, A.ptr, A.col, A.val, self.eps_strong)
        self.eps_strong *= 0.5  # reference halves eps per level (s_a.hpp:140)

        omega = float(self.prm["relax"])
        if self.prm["estimate_spectral_radius"]:
            omega *= (4.0 / 3.0) / spectral_radius_dinv_a(A, int(self.prm["power_iters"]))
        else:
            omega *= 2.0 / 3.0

        with prof.scope("smooth_P"):
            if self.B is None:
                pp, pc, pv = _core.smoothed_prolongation(
                    A.nrows, A.ptr, A.col, A.val, strong, aggr_id, naggr, omega
                )
                P = CSR(A.nrows, naggr, pp, pc, pv)
            else:
                # nullspace path: per-aggregate QR tentative + explicit
                # filtered-Jacobi smoother matrix (reference: tentative_
                # prolongation.hpp:134-207 + smoothed_aggregation smoothing)
                k = self.B.shape[1]
                tp, tc, tv, Bnew = _core.tentative_nullspace(
                    A.nrows, aggr_id, naggr, self.B, k
                )
                P_tent = CSR(A.nrows, naggr * k, tp, tc, tv)
                sp_, sc_, sv_ = _core.filtered_smoother_matrix(
                    A.nrows, A.ptr, A.col, A.val, strong, omega
                )
                S_F = CSR(A.nrows, A.nrows, sp_, sc_, sv_)
                P = S_F @ P_tent
                self.B = np.asarray(Bnew).reshape(naggr * k, k)
                self.prm["block_size"] = 1  # coarse levels: k dofs per aggregate
                if k > 1:
                    self.prm["block_size"] = k  # aggregate coarse points (k dofs each)
        with prof.scope("transpose_R"):
            R = P.transpose()
        return P, R

    def _transfer_operators_complex(self, A):
        """Native complex-valued coarsening (parity:
        amgcl/value_type/complex.hpp — the reference instantiates the same
        templates over std::complex).  Strength uses |a_ij|^2 >
        eps^2 |a_ii||a_jj| (the complex specialization of the criterion);
        aggregation runs on the magnitude matrix; P is Jacobi-smoothed with
        complex arithmetic; R = P^H (adjoint)."""
        import scipy.sparse as sp

        if int(self.prm["block_size"]) > 1 or self.B is not None:
            raise ValueError("complex path supports scalar SA without nullspace")
        n = A.nrows
        mag = np.abs(A.val)
        agg = _core.aggregates_parallel if n > 100_000 else _core.aggregates
        naggr, aggr_id, strong = agg(n, A.ptr, A.col, mag, self.eps_strong)
        self.eps_strong *= 0.5
        aggr_id = np.asarray(aggr_id)
        strong = np.asarray(strong).astype(bool)
        omega = float(self.prm["relax"]) * (2.0 / 3.0)

        m = A.to_scipy().tocsr()
        row_of = np.repeat(np.arange(n), np.diff(m.indptr))
        offdiag = m.indices != row_of
        # filtered matrix: weak off-diagonal entries folded into the diagonal
        weak = offdiag & ~strong
        dia = np.asarray(m.diagonal()).copy()
        np.add.at(dia, row_of[weak], m.data[weak])
        keep = strong.copy()
        Af = sp.csr_matrix((m.data[keep], m.indices[keep],
                            np.concatenate([[0], np.cumsum(
                                np.bincount(row_of[keep], minlength=n))])),
                           shape=(n, n))
        # P_tent: one 1 per row at its aggregate (removed rows empty)
        ok = aggr_id >= 0
        Pt = sp.csr_matrix((np.ones(ok.sum()), (np.flatnonzero(ok), aggr_id[ok])),
                           shape=(n, naggr))
        dinv = np.where(dia != 0, 1.0 / np.where(dia == 0, 1, dia), 0)
        S = sp.identity(n, format="csr") - sp.diags(omega * dinv) @ Af
        P = (S @ Pt).tocsr()
        P.sort_indices()
        Pc = CSR.from_scipy(P)
        return Pc, Pc.transpose()  # transpose() is the adjoint for complex

    def _transfer_operators_device(self, A):
        """Device twin (backend/hip_setup.py): same algorithm, same keys.
        Block (pointwise) aggregation and the nullspace-QR tentative run on
        the device too (torch-orchestrated over the device aggregation /
        spgemm engines — host twins: pointwise_matrix, expand_strong,
        tentative_nullspace, filtered_smoother_matrix in core.cpp);
        AMGCL_HOST_NULLSPACE=1 restores the host fallback."""
        import os

        from ..backend import hip_setup
        from ..profiler import prof

        bsize = int(self.prm["block_size"])
        if (bsize > 1 or self.B is not None) and os.environ.get(
                "AMGCL_HOST_NULLSPACE"):
            raise OverflowError("block/nullspace coarsening forced to the host")

        if bsize > 1 and A.nrows % bsize == 0:
            with prof.scope("aggregates(dev)"):
                naggr, ids, strong = self._pointwise_aggregates_device(A, bsize)
        else:
            with prof.scope("aggregates(dev)"):
                naggr, ids, strong = hip_setup.aggregates(A, self.eps_strong)
        self.eps_strong *= 0.5
        omega = float(self.prm["relax"])
        if self.prm["estimate_spectral_radius"]:
            # power iteration on D^-1 A entirely on the device
            omega *= (4.0 / 3.0) / _spectral_radius_device(
                A, int(self.prm["power_iters"])
            )
        else:
            omega *= 2.0 / 3.0
        with prof.scope("smooth_P(dev)"):
            if self.B is None:
                P = hip_setup.smoothed_prolongation(A, strong, ids, naggr, omega)
            else:
                P = self._nullspace_prolongation_device(A, strong, ids, naggr,
                                                        omega)
        with prof.scope("transpose_R(dev)"):
            R = hip_setup.transpose(P)
        return P, R

    def _pointwise_aggregates_device(self, A, bsize):
        """Pointwise (block) aggregation on the device: condense the block
        rows to a scalar matrix of Frobenius norms (torch sparse coalesce),
        run the device MIS on it, expand ids and the strong mask back to
        scalar unknowns (host twins: pointwise_matrix + expand_strong)."""
        import torch

        from ..backend import hip_setup
        from ..backend.hip import DeviceCSR

        from ..profiler import prof

        n = A.nrows
        dev = A.val.device
        t64 = torch.int64
        with prof.scope("pt_build"):
            lens = (A.ptr[1:] - A.ptr[:-1]).to(t64)
            row_of = torch.repeat_interleave(
                torch.arange(n, device=dev, dtype=t64), lens)
            col64 = A.col.to(t64)
            np_ = n // bsize
            rpt = row_of // bsize
            cpt = col64 // bsize
            # one sort + segment-sum (sparse coalesce measured ~4x slower),
            # and the inverse map doubles as the scalar->pointwise entry
            # lookup for the strong-mask expansion below
            key = rpt * np_ + cpt
            sk, perm = torch.sort(key)
            uk, inv = torch.unique_consecutive(sk, return_inverse=True)
            pv = torch.zeros(uk.numel(), dtype=torch.float64, device=dev)
            pv.scatter_add_(0, inv, (A.val * A.val)[perm])
            prow = uk // np_
            pcol = uk % np_
            pval = pv.sqrt()
            pos = torch.empty(key.numel(), dtype=t64, device=dev)
            pos[perm] = inv
        pptr = torch.zeros(np_ + 1, dtype=t64, device=dev)
        torch.cumsum(torch.bincount(prow, minlength=np_), 0, out=pptr[1:])
        Apt = DeviceCSR.from_tensors(
            np_, np_, pptr.to(torch.int32).contiguous(),
            pcol.to(torch.int32).contiguous(), pval.contiguous())
        with prof.scope("pt_mis"):
            naggr, ids_pt, S_pt = hip_setup.aggregates(Apt, self.eps_strong)
        with prof.scope("pt_expand"):
            ids = ids_pt.repeat_interleave(bsize).contiguous()
            # strong mask per scalar entry = the pointwise entry's flag
            # (diagonal-point entries are never strong); pos computed above
            S = S_pt[pos]
            S = torch.where(rpt == cpt, torch.zeros_like(S), S).contiguous()
        self._keep_pt = (Apt,)  # keep alive until setup finishes
        return naggr, ids, S

    def _nullspace_prolongation_device(self, A, strong, ids, naggr, omega):
        """Nullspace tentative prolongation (per-aggregate batched QR) +
        filtered-Jacobi smoothing, device-resident (host twins:
        tentative_nullspace + filtered_smoother_matrix).  Assumes every row
        carries a diagonal entry (true for the supported generators)."""
        import torch

        from ..backend import hip_setup
        from ..backend.hip import DeviceCSR

        n = A.nrows
        dev = A.val.device
        t64 = torch.int64
        k = self.B.shape[1]
        B_d = torch.from_numpy(np.ascontiguousarray(self.B)).to(dev)

        from ..profiler import prof

        ids64 = ids.to(t64)
        assigned = ids64 >= 0
        rows_assigned = torch.nonzero(assigned).ravel()
        agg_of = ids64[rows_assigned]
        order = torch.argsort(agg_of, stable=True)
        members = rows_assigned[order]            # grouped by aggregate
        magg = agg_of[order]
        counts = torch.bincount(magg, minlength=naggr)
        maxd = int(counts.max().item())
        offs = torch.zeros(naggr + 1, dtype=t64, device=dev)
        torch.cumsum(counts, 0, out=offs[1:])
        slot = (torch.arange(members.numel(), device=dev, dtype=t64)
                - offs[:-1].repeat_interleave(counts))
        prof.tic("ns_qr")
        batch = torch.zeros(naggr, maxd, k, dtype=torch.float64, device=dev)
        batch[magg, slot] = B_d[members]
        # vectorized MGS with one re-orthogonalization (the host engine's
        # exact algorithm, batched over aggregates; the padded zero rows
        # contribute nothing to the inner products).  Measured faster than
        # MAGMA's batched small-matrix QR and keeps host parity.
        Q = batch
        Rf = torch.zeros(naggr, k, k, dtype=torch.float64, device=dev)
        for c in range(k):
            qc = Q[:, :, c]
            for _pass in range(2):
                for p in range(c):
                    qp = Q[:, :, p]
                    h = (qp * qc).sum(dim=1)
                    qc -= h.unsqueeze(1) * qp
                    Rf[:, p, c] += h
            nrm = qc.square().sum(dim=1).sqrt()
            Rf[:, c, c] = nrm
            safe = nrm > 1e-300
            qc /= torch.where(safe, nrm, torch.ones_like(nrm)).unsqueeze(1)
        # coarse-level nullspace = per-aggregate R blocks (naggr*k x k)
        self.B = Rf.reshape(naggr * k, k).cpu().numpy()
        self.prm["block_size"] = k if k > 1 else 1

        prof.toc("ns_qr")
        # P_tent rows in row order: row i gets Q[id[i], slot_of(i), :]
        slot_of = torch.zeros(n, dtype=t64, device=dev)
        slot_of[members] = slot
        vals = Q[ids64[rows_assigned], slot_of[rows_assigned]]  # (nass, k)
        pptr = torch.zeros(n + 1, dtype=t64, device=dev)
        torch.cumsum(assigned.to(t64) * k, 0, out=pptr[1:])
        pcols = (ids64[rows_assigned].unsqueeze(1) * k
                 + torch.arange(k, device=dev, dtype=t64)).reshape(-1)
        P_tent = DeviceCSR.from_tensors(
            n, naggr * k, pptr.to(torch.int32).contiguous(),
            pcols.to(torch.int32).contiguous(), vals.reshape(-1).contiguous())

        # S_F = I - omega Df^-1 Af (weak off-diagonals folded into Df)
        lens = (A.ptr[1:] - A.ptr[:-1]).to(t64)
        row_of = torch.repeat_interleave(
            torch.arange(n, device=dev, dtype=t64), lens)
        col64 = A.col.to(t64)
        dia_mask = row_of == col64
        Sb = strong.to(torch.bool)
        contrib = torch.where(dia_mask | ~Sb, A.val, torch.zeros_like(A.val))
        dia = torch.zeros(n, dtype=torch.float64, device=dev)
        dia.scatter_add_(0, row_of, contrib)
        w = torch.where(dia != 0, -omega / dia,
                        torch.zeros_like(dia))
        keep = dia_mask | Sb
        kr = row_of[keep]
        kv = torch.where(dia_mask[keep],
                         torch.full_like(A.val[keep], 1.0 - omega),
                         w[kr] * A.val[keep])
        sptr = torch.zeros(n + 1, dtype=t64, device=dev)
        torch.cumsum(torch.bincount(kr, minlength=n), 0, out=sptr[1:])
        S_F = DeviceCSR.from_tensors(
            n, n, sptr.to(torch.int32).contiguous(),
            A.col[keep].contiguous(), kv.contiguous())
        with prof.scope("ns_spgemm"):
            return hip_setup.spgemm(S_F, P_tent, sort=True)

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


def _spectral_radius_device(A, iters):
    """Power-iteration estimate of rho(D^-1 A) on a device matrix."""
    import torch

    from ..backend import hip_setup
    from ..backend.hip import HipBackend

    hip = HipBackend.__new__(HipBackend)  # op-only use; no state needed
    d = hip_setup.diagonal(A)
    dinv = 1.0 / d
    g = torch.Generator(device=d.device).manual_seed(12345)
    b0 = torch.rand(A.nrows, dtype=torch.float64, device=d.device, generator=g)
    b0 /= b0.norm()
    b1 = torch.empty_like(b0)
    rho = 2.0
    for _ in range(max(1, iters)):
        hip.spmv(1.0, A, b0, 0.0, b1)
        b1 *= dinv
        rho = float(b0.dot(b1))
        nrm = float(b1.norm())
        if nrm == 0:
            break
        b0 = b1 / nrm
        b1 = torch.empty_like(b0)
    return abs(rho)
