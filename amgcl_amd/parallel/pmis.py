"""Cross-rank pmis aggregation + distributed transfer operators.

Parity: amgcl/mpi/coarsening/pmis.hpp:50-712 — distributed maximal
independent set (distance 2) aggregation with deterministic tie-breaking,
where aggregates DO cross rank boundaries: root selection, 1-ring claims and
provisional adoption all negotiate over the halo. The algorithm is the same
round-synchronous MIS(2) as the host/device engines (csrc/core/core.cpp
aggregates_parallel, csrc/hip/setup.hip) with the identical deterministic
hash keys, so on the same global matrix the distributed run reproduces the
single-process aggregation EXACTLY (tested).

Transfer operators: the tentative/smoothed prolongation strips reference
remote coarse columns (rectangular DistMatrix over the coarse partition), the
restriction strips are assembled by a transpose exchange, and the Galerkin
triple product sums per-rank P^T (A P) contributions shipped to their coarse
row owners (the reference's distributed_matrix product machinery).
"""
import numpy as np

from ..matrix import CSR
from .dist_matrix import DistMatrix

UNDEF = np.int64(-1)
REMOVED = np.int64(-2)


def agg_key(ids):
    """Deterministic MIS key, identical to core.cpp:agg_key (hash<<32 | id).
    The low 32 bits carry the id, so a segment-max over keys also yields the
    argmax id."""
    x = np.asarray(ids, dtype=np.int64).astype(np.uint32)
    x = x ^ (x >> np.uint32(16))
    x = (x * np.uint32(0x7FEB352D)) & np.uint32(0xFFFFFFFF)
    x = x ^ (x >> np.uint32(15))
    x = (x * np.uint32(0x846CA68B)) & np.uint32(0xFFFFFFFF)
    x = x ^ (x >> np.uint32(16))
    return (x.astype(np.uint64) << np.uint64(32)) | np.asarray(
        ids, dtype=np.int64).astype(np.uint32).astype(np.uint64)


def _seg_max(vals, ptr, n):
    """Per-row max of vals (uint64) over CSR segments; 0 for empty rows."""
    out = np.zeros(n, dtype=np.uint64)
    nonempty = np.flatnonzero(np.diff(ptr) > 0)
    if len(nonempty):
        red = np.maximum.reduceat(vals, ptr[:-1][nonempty])
        out[nonempty] = red
    return out


class _Halo:
    """Setup-time neighbor exchange of per-boundary-row arrays over a
    DistMatrix's comm pattern (the generic `exchange` of the reference's
    comm_pattern, mpi/distributed_matrix.hpp:276)."""

    def __init__(self, A: DistMatrix, dist, group):
        self.A = A
        self.dist = dist
        self.group = group
        self.send_rows = [
            np.asarray(ix.cpu() if hasattr(ix, "cpu") else ix, dtype=np.int64)
            for ix in A.send_idx
        ]

    def __call__(self, *arrays):
        """For each local array, returns the ghost-aligned mirror."""
        A = self.A
        packets = {}
        for r, rows in zip(A.send_ranks, self.send_rows):
            packets[r] = tuple(a[rows] for a in arrays)
        gathered = [None] * A.world
        self.dist.all_gather_object(gathered, packets, group=self.group)
        outs = []
        for k, a in enumerate(arrays):
            g = np.zeros(A.n_ghost, dtype=a.dtype)
            off = 0
            for r, cnt in zip(A.recv_ranks, A.recv_counts):
                g[off : off + cnt] = gathered[r][A.rank][k]
                off += cnt
            outs.append(g)
        return outs[0] if len(outs) == 1 else outs


def strong_masks(A: DistMatrix, eps, halo):
    """Strong-connection masks for the local and remote strip parts, using
    the FULL row (the decoupled path only sees A_loc)."""
    loc, rem = A.A_loc_host, A.A_rem_host
    d = np.zeros(A.n_loc)
    rl = np.repeat(np.arange(A.n_loc), np.diff(loc.ptr))
    dm = np.asarray(loc.col) == np.arange(A.n_loc)[rl]
    np.add.at(d, rl[dm], np.asarray(loc.val)[dm])
    d_ghost = halo(d) if A.n_ghost else np.zeros(0)
    eps2 = eps * eps
    vl = np.asarray(loc.val)
    S_loc = (~dm) & (eps2 * d[rl] * d[np.asarray(loc.col)] < vl * vl)
    if rem is not None:
        rr = np.repeat(np.arange(A.n_loc), np.diff(rem.ptr))
        vr = np.asarray(rem.val)
        S_rem = eps2 * d[rr] * d_ghost[np.asarray(rem.col)] < vr * vr
    else:
        S_rem = np.zeros(0, dtype=bool)
    return S_loc, S_rem, d


def pmis_aggregates(A: DistMatrix, eps, dist, group):
    """Distributed MIS(2) aggregation. Returns id[n_loc] holding GLOBAL root
    ids (-2 for isolated rows) plus the strong masks."""
    halo = _Halo(A, dist, group)
    S_loc, S_rem, _ = strong_masks(A, eps, halo)
    loc, rem = A.A_loc_host, A.A_rem_host
    n = A.n_loc
    gid = np.arange(A.row_beg, A.row_end, dtype=np.int64)
    key = agg_key(gid)
    g_gid = np.asarray(A.ghost_global, dtype=np.int64)

    # strong edge lists (row -> local col) and (row -> ghost col), compressed
    lr = np.repeat(np.arange(n, dtype=np.int64), np.diff(loc.ptr))[S_loc]
    lc = np.asarray(loc.col)[S_loc]
    lptr = np.concatenate([[0], np.cumsum(np.bincount(lr, minlength=n))])
    order = np.argsort(lr, kind="stable")
    lc = lc[order]
    if rem is not None:
        rr = np.repeat(np.arange(n, dtype=np.int64), np.diff(rem.ptr))[S_rem]
        rc = np.asarray(rem.col)[S_rem]
        rptr = np.concatenate([[0], np.cumsum(np.bincount(rr, minlength=n))])
        ro = np.argsort(rr, kind="stable")
        rc = rc[ro]
    else:
        rptr = np.zeros(n + 1, dtype=np.int64)
        rc = np.zeros(0, dtype=np.int64)

    has_strong = (np.diff(lptr) + np.diff(rptr)) > 0
    ids = np.where(has_strong, UNDEF, REMOVED)
    prov = np.zeros(n, dtype=bool)
    g_key = agg_key(g_gid)

    for _round in range(64):
        g_ids, g_prov = (halo(ids, prov) if A.n_ghost
                         else (np.zeros(0, np.int64), np.zeros(0, bool)))
        undef = ids == UNDEF
        g_undef = g_ids == UNDEF

        # pass 1: m1 = max UNDEF key within distance 1 (incl. ghosts)
        m1 = np.where(undef, key, np.uint64(0))
        lv = np.where(undef[lc], key[lc], np.uint64(0))
        m1 = np.maximum(m1, _seg_max(lv, lptr, n))
        if len(rc):
            rv = np.where(g_undef[rc], g_key[rc], np.uint64(0))
            m1 = np.maximum(m1, _seg_max(rv, rptr, n))
        g_m1 = halo(m1) if A.n_ghost else np.zeros(0, np.uint64)

        # pass 2: u is a root iff its key is the distance-2 max
        m2 = np.maximum(m1, _seg_max(m1[lc], lptr, n))
        if len(rc):
            m2 = np.maximum(m2, _seg_max(g_m1[rc], rptr, n))
        new_root = undef & (m2 == key)
        ids[new_root] = gid[new_root]
        prov[new_root] = False

        # pass 3: new roots claim their strong 1-ring (UNDEF or provisional),
        # local targets directly, remote targets via their owners
        er = np.repeat(np.arange(n, dtype=np.int64), np.diff(lptr))
        sel = new_root[er]
        tgt = lc[sel]
        ok = (ids[tgt] == UNDEF) | prov[tgt]
        ids[tgt[ok]] = gid[er[sel]][ok]
        prov[tgt[ok]] = False
        packets = {}
        if len(rc):
            er2 = np.repeat(np.arange(n, dtype=np.int64), np.diff(rptr))
            sel2 = new_root[er2]
            cg = g_gid[rc[sel2]]
            rootg = gid[er2[sel2]]
            owner = np.searchsorted(A.col_begs, cg, side="right") - 1
            for r in A.recv_ranks:
                m = owner == r
                if m.any():
                    packets[r] = (cg[m], rootg[m])
        gathered = [None] * A.world
        dist.all_gather_object(gathered, packets, group=group)
        for r in range(A.world):
            pkt = gathered[r].get(A.rank) if gathered[r] else None
            if pkt is None:
                continue
            cg, rootg = pkt
            li = cg - A.row_beg
            ok = (ids[li] == UNDEF) | prov[li]
            ids[li[ok]] = rootg[ok]
            prov[li[ok]] = False

        # pass 4: adoption reads the POST-claim state (fresh ghosts), exactly
        # like the serial two-phase mark/commit
        g_ids, g_prov = (halo(ids, prov) if A.n_ghost
                         else (np.zeros(0, np.int64), np.zeros(0, bool)))
        undef = ids == UNDEF
        firm_l = (ids[lc] >= 0) & ~prov[lc]
        cand_l = np.where(firm_l, agg_key(np.where(firm_l, ids[lc], 0)),
                          np.uint64(0))
        best = _seg_max(cand_l, lptr, n)
        if len(rc):
            firm_r = (g_ids[rc] >= 0) & ~g_prov[rc]
            cand_r = np.where(firm_r, agg_key(np.where(firm_r, g_ids[rc], 0)),
                              np.uint64(0))
            best = np.maximum(best, _seg_max(cand_r, rptr, n))
        adopt = undef & (best > 0)
        ids[adopt] = (best[adopt] & np.uint64(0xFFFFFFFF)).astype(np.int64)
        prov[adopt] = True

        remaining = int((ids == UNDEF).sum())
        tot = [None] * A.world
        dist.all_gather_object(tot, remaining, group=group)
        if sum(tot) == 0:
            break
    else:
        raise RuntimeError("distributed pmis did not converge")
    return ids, S_loc, S_rem


def renumber(A: DistMatrix, ids, dist, group):
    """Compact global coarse numbering (rank-major by node order — identical
    to the serial renumber for ordered 1-D strips). Returns (coarse of each
    local row or -1, naggr sizes per rank, root->coarse resolver for ghosts)."""
    gid = np.arange(A.row_beg, A.row_end, dtype=np.int64)
    is_root = ids == gid
    naggr_loc = int(is_root.sum())
    sizes = [None] * A.world
    dist.all_gather_object(sizes, naggr_loc, group=group)
    coarse_begs = np.concatenate([[0], np.cumsum(sizes)]).astype(np.int64)
    my_beg = int(coarse_begs[A.rank])
    root_to_coarse = np.full(A.n_loc, -1, dtype=np.int64)
    root_to_coarse[is_root] = my_beg + np.arange(naggr_loc)

    # resolve roots of my rows that live on other ranks
    roots = np.unique(ids[ids >= 0])
    owner = np.searchsorted(A.row_begs, roots, side="right") - 1
    queries = {}
    for r in range(A.world):
        if r == A.rank:
            continue
        sel = roots[owner == r]
        if len(sel):
            queries[r] = sel
    gathered = [None] * A.world
    dist.all_gather_object(gathered, queries, group=group)
    answers = {}
    for r in range(A.world):
        q = gathered[r].get(A.rank) if gathered[r] else None
        if q is not None:
            answers[r] = root_to_coarse[q - A.row_beg]
    gathered2 = [None] * A.world
    dist.all_gather_object(gathered2, answers, group=group)
    lookup = {}
    for root, c in zip(roots[owner == A.rank],
                       root_to_coarse[(roots[owner == A.rank]) - A.row_beg]):
        lookup[int(root)] = int(c)
    for r in range(A.world):
        a = gathered2[r].get(A.rank) if gathered2[r] else None
        if a is not None:
            for root, c in zip(queries[r], a):
                lookup[int(root)] = int(c)

    coarse_of = np.full(A.n_loc, -1, dtype=np.int64)
    m = ids >= 0
    coarse_of[m] = np.array([lookup[int(x)] for x in ids[m]], dtype=np.int64)
    return coarse_of, sizes, coarse_begs


def smoothed_p_strip(A: DistMatrix, coarse_of, S_loc, S_rem, n_c_glob, omega,
                     dist, group):
    """Smoothed prolongation strip (rows = my fine rows, cols = GLOBAL coarse
    ids, which may live on other ranks): P = (I - omega Df^-1 Af) P_tent."""
    import scipy.sparse as sp

    halo = _Halo(A, dist, group)
    g_coarse = halo(coarse_of) if A.n_ghost else np.zeros(0, np.int64)
    loc, rem = A.A_loc_host, A.A_rem_host
    n = A.n_loc

    # filtered diagonal over the full row
    rl = np.repeat(np.arange(n, dtype=np.int64), np.diff(loc.ptr))
    dm = np.asarray(loc.col) == np.arange(n)[rl]
    dia = np.zeros(n)
    np.add.at(dia, rl[dm], np.asarray(loc.val)[dm])
    weak_l = (~S_loc) & (~dm)
    np.add.at(dia, rl[weak_l], np.asarray(loc.val)[weak_l])
    if rem is not None:
        rr = np.repeat(np.arange(n, dtype=np.int64), np.diff(rem.ptr))
        np.add.at(dia, rr[~S_rem], np.asarray(rem.val)[~S_rem])
    w = np.divide(-omega, dia, out=np.zeros_like(dia), where=dia != 0)

    rows, cols, vals = [], [], []
    own = coarse_of >= 0
    rows.append(np.flatnonzero(own))
    cols.append(coarse_of[own])
    vals.append(np.full(int(own.sum()), 1.0 - omega))
    sel = S_loc & (coarse_of[np.asarray(loc.col)] >= 0)
    rows.append(rl[sel])
    cols.append(coarse_of[np.asarray(loc.col)[sel]])
    vals.append(w[rl[sel]] * np.asarray(loc.val)[sel])
    if rem is not None:
        sel = S_rem & (g_coarse[np.asarray(rem.col)] >= 0)
        rows.append(rr[sel])
        cols.append(g_coarse[np.asarray(rem.col)[sel]])
        vals.append(w[rr[sel]] * np.asarray(rem.val)[sel])
    P = sp.coo_matrix(
        (np.concatenate(vals), (np.concatenate(rows), np.concatenate(cols))),
        shape=(n, n_c_glob),
    ).tocsr()
    P.sort_indices()
    return P


def transpose_exchange(A: DistMatrix, P, coarse_begs, dist, group):
    """R strip = rows owned in the COARSE partition, global FINE columns:
    ship each P entry to its coarse row owner (the distributed transpose of
    mpi/distributed_matrix.hpp)."""
    import scipy.sparse as sp

    n_glob_fine = A.n_global
    rank, world = A.rank, A.world
    Pc = P.tocoo()
    fine_g = Pc.row + A.row_beg
    owner = np.searchsorted(coarse_begs, Pc.col, side="right") - 1
    packets = {}
    for r in range(world):
        m = owner == r
        if r != rank and m.any():
            packets[r] = (Pc.col[m], fine_g[m], Pc.data[m])
    gathered = [None] * world
    dist.all_gather_object(gathered, packets, group=group)
    nc_loc = int(coarse_begs[rank + 1] - coarse_begs[rank])
    rows = [Pc.col[owner == rank] - coarse_begs[rank]]
    cols = [fine_g[owner == rank]]
    vals = [Pc.data[owner == rank]]
    for r in range(world):
        pkt = gathered[r].get(rank) if gathered[r] else None
        if pkt is not None:
            c, f, v = pkt
            rows.append(c - coarse_begs[rank])
            cols.append(f)
            vals.append(v)
    R = sp.coo_matrix(
        (np.concatenate(vals), (np.concatenate(rows), np.concatenate(cols))),
        shape=(nc_loc, n_glob_fine),
    ).tocsr()
    R.sort_indices()
    return R


def galerkin_strip(A: DistMatrix, P, coarse_begs, dist, group):
    """Distributed Ac = P^T A P: each rank computes its fine-row contribution
    W = P_strip^T (A P)_strip and ships W's rows to their coarse owners."""
    import scipy.sparse as sp

    halo_rows = _exchange_rows(A, P, dist, group)
    mid = A.A_loc_host.to_scipy() @ P
    if A.A_rem_host is not None:
        mid = mid + A.A_rem_host.to_scipy() @ halo_rows
    W = (P.T @ mid).tocoo()
    rank, world = A.rank, A.world
    owner = np.searchsorted(coarse_begs, W.row, side="right") - 1
    packets = {}
    for r in range(world):
        m = owner == r
        if r != rank and m.any():
            packets[r] = (W.row[m], W.col[m], W.data[m])
    gathered = [None] * world
    dist.all_gather_object(gathered, packets, group=group)
    nc_loc = int(coarse_begs[rank + 1] - coarse_begs[rank])
    mine = owner == rank
    rows = [W.row[mine] - coarse_begs[rank]]
    cols = [W.col[mine]]
    vals = [W.data[mine]]
    for r in range(world):
        pkt = gathered[r].get(rank) if gathered[r] else None
        if pkt is not None:
            rr, cc, vv = pkt
            rows.append(rr - coarse_begs[rank])
            cols.append(cc)
            vals.append(vv)
    Ac = sp.coo_matrix(
        (np.concatenate(vals), (np.concatenate(rows), np.concatenate(cols))),
        shape=(nc_loc, int(coarse_begs[-1])),
    ).tocsr()
    Ac.sum_duplicates()
    Ac.sort_indices()
    return Ac


def _exchange_rows(A: DistMatrix, P, dist, group):
    """Rows of P for A's ghost fine columns (P already has global columns)."""
    import scipy.sparse as sp

    packets = {}
    for r, idx in zip(A.send_ranks, A.send_idx):
        rows = np.asarray(idx.cpu() if hasattr(idx, "cpu") else idx,
                          dtype=np.int64)
        sub = P[rows].tocsr()
        packets[r] = (sub.indptr, sub.indices, sub.data)
    gathered = [None] * A.world
    dist.all_gather_object(gathered, packets, group=group)
    if A.n_ghost == 0:
        return sp.csr_matrix((0, P.shape[1]))
    blocks = []
    for r in A.recv_ranks:
        ptr, col, val = gathered[r][A.rank]
        blocks.append(sp.csr_matrix((val, col, ptr),
                                    shape=(len(ptr) - 1, P.shape[1])))
    return sp.vstack(blocks, format="csr")
