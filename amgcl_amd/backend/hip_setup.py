"""Device-side AMG setup orchestration.

Python driver for the gfx950 setup kernels (csrc/hip/setup.hip): the whole
smoothed-aggregation hierarchy — strong connections, parallel aggregation,
prolongation smoothing, R = P^T, and the Galerkin triple product — is built
in device memory, eliminating the reference's host-assembly + per-level H2D
upload design (amgcl/amg.hpp:467-512, backend/hip.hpp:382-409), which is the
dominant cost on CPU-quota'd GPU nodes.

The algorithms are the exact device twins of the host engine
(csrc/core/core.cpp), with identical deterministic aggregation keys, so the
device-built hierarchy matches the host-built one level for level.
"""
import numpy as np

from ..matrix import CSR
from ._hiplib import check, lib


def _stream():
    import torch

    return torch.cuda.current_stream().cuda_stream


def _torch():
    import torch

    return torch


def device_csr(nrows, ncols, ptr, col, val, subw=0):
    from .hip import DeviceCSR

    return DeviceCSR.from_tensors(nrows, ncols, ptr, col, val, subw)


_SCRATCH = {}


def _scratch_i32(n, dev):
    """Process-lifetime int32 scratch (grow-only, keyed by device).  The
    aggregation frontier lists and the SpGEMM big-row worklists are pure
    scratch, but allocating them per call (up to 1.6 GB at 512^3) can miss
    the torch allocator cache after interleaved allocations — a cold
    hipMalloc costs ~28 ms/GB (profiles/README r02)."""
    t = _torch()
    key = str(dev)
    buf = _SCRATCH.get(key)
    if buf is None or buf.numel() < n:
        _SCRATCH[key] = buf = t.empty(int(n * 1.25), dtype=t.int32, device=dev)
    return buf[:n]


def _new_ptr(n, device):
    t = _torch()
    return t.zeros(n + 1, dtype=t.int32, device=device)


def _scan_ptr(ptr):
    """Inclusive scan of counts stored in ptr[1:]; ptr[0] stays 0."""
    n = ptr.numel() - 1
    check(lib().amg_scan_i32(ptr[1:].data_ptr(), n, _stream()), "scan")
    return ptr


def diagonal(A):
    t = _torch()
    d = t.empty(A.nrows, dtype=t.float64, device=A.val.device)
    check(lib().amg_setup_diag(A.nrows, A.ptr.data_ptr(), A.col.data_ptr(),
                               A.val.data_ptr(), d.data_ptr(), _stream()), "diag")
    return d


def spai0(A):
    t = _torch()
    m = t.empty(A.nrows, dtype=t.float64, device=A.val.device)
    check(lib().amg_setup_spai0(A.nrows, A.ptr.data_ptr(), A.col.data_ptr(),
                                A.val.data_ptr(), m.data_ptr(), _stream()), "spai0")
    return m


def gershgorin(A, scale=False):
    """Gershgorin spectral-radius bound on the device."""
    t = _torch()
    out = t.zeros(1, dtype=t.float64, device=A.val.device)
    check(lib().amg_gershgorin(A.nrows, A.ptr.data_ptr(), A.col.data_ptr(),
                               A.val.data_ptr(), 1 if scale else 0, out.data_ptr(),
                               _stream()), "gershgorin")
    return float(out.item())


def aggregates(A, eps_strong):
    """Device twin of _core.aggregates_parallel: returns (naggr, id, strong)."""
    t = _torch()
    dev = A.val.device
    n = A.nrows
    d = diagonal(A)
    S = t.empty(A.nnz, dtype=t.uint8, device=dev)
    check(lib().amg_setup_strong(n, A.ptr.data_ptr(), A.col.data_ptr(), A.val.data_ptr(),
                                 d.data_ptr(), float(eps_strong) ** 2, S.data_ptr(),
                                 _stream()), "strong")
    ids = t.empty(n, dtype=t.int32, device=dev)
    check(lib().amg_agg_init(n, A.ptr.data_ptr(), S.data_ptr(), ids.data_ptr(),
                             _stream()), "agg_init")
    prov = t.zeros(n, dtype=t.uint8, device=dev)
    m1 = t.empty(n, dtype=t.int64, device=dev)
    newroot = t.empty(n, dtype=t.uint8, device=dev)
    near = t.empty(n, dtype=t.uint8, device=dev)
    remaining = t.zeros(1, dtype=t.int32, device=dev)
    # the whole round loop runs inside the library (one ctypes call):
    # per-round host gaps (launch latency + torch dispatch + .item() syncs)
    # cost ~11 ms/round at 134M rows when driven from Python
    import ctypes

    rounds = ctypes.c_int(0)
    # frontier-compaction scratch: two ping-pong active lists + counter
    lists = _scratch_i32(3 * n, dev)
    rc = lib().amg_agg_run(n, A.ptr.data_ptr(), A.col.data_ptr(), S.data_ptr(),
                           ids.data_ptr(), prov.data_ptr(), m1.data_ptr(),
                           newroot.data_ptr(), near.data_ptr(), remaining.data_ptr(),
                           2, 64, ctypes.byref(rounds), lists.data_ptr(), _stream())
    if rc == 9999:
        raise RuntimeError("device aggregation did not converge")
    check(rc, "agg_run")
    mark = t.empty(n, dtype=t.int32, device=dev)
    check(lib().amg_agg_renumber(n, ids.data_ptr(), mark.data_ptr(), _stream()),
          "agg_renumber")
    naggr = int(mark[-1].item())
    if naggr == 0:
        raise RuntimeError("empty level in aggregation")
    return naggr, ids, S


def smoothed_prolongation(A, S, ids, naggr, omega):
    t = _torch()
    dev = A.val.device
    n = A.nrows
    pptr = _new_ptr(n, dev)
    overflow = t.zeros(1, dtype=t.int32, device=dev)
    check(lib().amg_psmooth_count(n, A.ptr.data_ptr(), A.col.data_ptr(), S.data_ptr(),
                                  ids.data_ptr(), pptr[1:].data_ptr(),
                                  overflow.data_ptr(), _stream()), "psmooth_count")
    _scan_ptr(pptr)
    ov, nnz = (int(v) for v in t.cat([overflow, pptr[-1:]]).cpu())
    if ov != 0:
        raise OverflowError("P row exceeds device buffer; use host setup")
    pcol = t.empty(nnz, dtype=t.int32, device=dev)
    pval = t.empty(nnz, dtype=t.float64, device=dev)
    check(lib().amg_psmooth_fill(n, A.ptr.data_ptr(), A.col.data_ptr(), A.val.data_ptr(),
                                 S.data_ptr(), ids.data_ptr(), float(omega),
                                 pptr[1:].data_ptr(), pcol.data_ptr(), pval.data_ptr(),
                                 _stream()), "psmooth_fill")
    return device_csr(n, naggr, pptr, pcol, pval)


def tentative_prolongation(A, ids, naggr):
    t = _torch()
    dev = A.val.device
    n = A.nrows
    pptr = _new_ptr(n, dev)
    check(lib().amg_ptent_count(n, ids.data_ptr(), pptr[1:].data_ptr(), _stream()),
          "ptent_count")
    _scan_ptr(pptr)
    nnz = int(pptr[-1].item())
    pcol = t.empty(nnz, dtype=t.int32, device=dev)
    pval = t.empty(nnz, dtype=t.float64, device=dev)
    check(lib().amg_ptent_fill(n, ids.data_ptr(), pptr[1:].data_ptr(), pcol.data_ptr(),
                               pval.data_ptr(), _stream()), "ptent_fill")
    return device_csr(n, naggr, pptr, pcol, pval)


def transpose(A):
    t = _torch()
    dev = A.val.device
    tptr = _new_ptr(A.ncols, dev)
    check(lib().amg_transpose_count(A.nnz, A.col.data_ptr(), tptr[1:].data_ptr(),
                                    _stream()), "tcount")
    _scan_ptr(tptr)
    cursor = tptr[:-1].clone()
    tcol = t.empty(A.nnz, dtype=t.int32, device=dev)
    tval = t.empty(A.nnz, dtype=t.float64, device=dev)
    check(lib().amg_transpose_scatter(A.nrows, A.ptr.data_ptr(), A.col.data_ptr(),
                                      A.val.data_ptr(), cursor.data_ptr(),
                                      tcol.data_ptr(), tval.data_ptr(), _stream()),
          "tscatter")
    # rows are left unsorted: no solve/setup kernel depends on intra-row
    # order (gather bandwidth depends on the address SET, not order), and
    # sorting 40+-entry rows in global memory costs more than everything else
    return device_csr(A.ncols, A.nrows, tptr, tcol, tval)


def spgemm(A, B, sort=True):
    t = _torch()
    dev = A.val.device
    cptr = _new_ptr(A.nrows, dev)
    ub = t.empty(A.nrows, dtype=t.int32, device=dev)
    overflow = t.zeros(1, dtype=t.int32, device=dev)
    # flags/scan + big-row worklist + device-side count (no host sync)
    bigscratch = _scratch_i32(2 * A.nrows + 1, dev)
    check(lib().amg_spgemm_count(A.nrows, A.ptr.data_ptr(), A.col.data_ptr(),
                                 B.ptr.data_ptr(), B.col.data_ptr(), ub.data_ptr(),
                                 cptr[1:].data_ptr(), overflow.data_ptr(),
                                 bigscratch.data_ptr(), _stream()),
          "spgemm_count")
    _scan_ptr(cptr)
    ov, nnz = (int(v) for v in t.cat([overflow, cptr[-1:]]).cpu())
    if ov != 0:
        raise OverflowError("spgemm row exceeds LDS hash; use host setup")
    ccol = t.empty(nnz, dtype=t.int32, device=dev)
    cval = t.empty(nnz, dtype=t.float64, device=dev)
    check(lib().amg_spgemm_fill(A.nrows, A.ptr.data_ptr(), A.col.data_ptr(),
                                A.val.data_ptr(), B.ptr.data_ptr(), B.col.data_ptr(),
                                B.val.data_ptr(), ub.data_ptr(), cptr[1:].data_ptr(),
                                ccol.data_ptr(), cval.data_ptr(), 1 if sort else 0,
                                bigscratch.data_ptr(), _stream()),
          "spgemm_fill")
    return device_csr(A.nrows, B.ncols, cptr, ccol, cval)


def download(A):
    """Device CSR -> host CSR (sorted rows)."""
    import scipy.sparse as sp

    m = sp.csr_matrix(
        (A.val.cpu().numpy(), A.col.cpu().numpy(), A.ptr.cpu().numpy()),
        shape=(A.nrows, A.ncols),
    )
    m.sort_indices()
    return CSR(A.nrows, A.ncols, m.indptr, m.indices, m.data)


def poisson3d_device(n, device="cuda"):
    if 7 * n**3 >= 2**31:
        raise ValueError(
            f"n={n}: {7*n**3/1e9:.2f}G nonzeros exceed the framework's int32 "
            "index design (2^31); the single-GPU ceiling is ~673^3 (305M "
            "unknowns, 2.1G nnz) — shard larger problems across ranks "
            "(bench.py --gpus N)")
    """7-point Poisson fixture generated directly on the GPU."""
    return poisson3d_device_strip(n, 0, n**3, device)


def poisson3d_device_strip(n, row_beg, row_end, device="cuda", nz=None):
    """Row strip [row_beg, row_end) of the 7-point Poisson operator on an
    n x n x nz box (nz defaults to n: the cube) with GLOBAL columns,
    generated directly in device memory — the per-rank distributed fixture
    (reference: examples/mpi/mpi_solver.cpp assemble_poisson3d).  Global
    column ids are int32, so the GLOBAL problem may reach 2^31 unknowns
    (~1290^3) across ranks."""
    nz = n if nz is None else nz
    ntot = n * n * nz
    if ntot > 2**31:
        raise ValueError(f"global column ids exceed int32 ({ntot/1e9:.2f}G unknowns)")
    nloc = row_end - row_beg
    if 7 * nloc >= 2**31:
        raise ValueError("strip nonzeros exceed int32; use more ranks")
    t = _torch()
    ptr = _new_ptr(nloc, device)
    check(lib().amg_poisson_cnt(n, nz, row_beg, row_end, ptr[1:].data_ptr(),
                                _stream()), "poisson_cnt")
    _scan_ptr(ptr)
    nnz = int(ptr[-1].item())
    col = t.empty(nnz, dtype=t.int32, device=device)
    val = t.empty(nnz, dtype=t.float64, device=device)
    check(lib().amg_poisson_fill(n, nz, row_beg, row_end, ptr[1:].data_ptr(),
                                 col.data_ptr(), val.data_ptr(), _stream()),
          "poisson_fill")
    return device_csr(nloc, ntot, ptr, col, val)


def split_strip_torch(ptr, col, val, col_beg, col_end):
    """Local/remote split of a row strip with global columns, in torch ops
    (device-agnostic: runs on the GPU for device strips; tested on CPU
    tensors against the C++ _core.split_strip).  Returns
    (lp, lc, lv, rp, rc, rv, ghost_global): local part with columns
    renumbered to [0, col_end-col_beg), remote part with columns renumbered
    into the sorted ghost_global id list (parity:
    amgcl/mpi/distributed_matrix.hpp:370-430)."""
    t = _torch()
    nloc = ptr.numel() - 1
    dev = col.device
    lens = (ptr[1:] - ptr[:-1]).to(t.int64)
    row_ids = t.repeat_interleave(t.arange(nloc, device=dev, dtype=t.int64), lens)
    mask = (col >= col_beg) & (col < col_end)
    lloc = t.bincount(row_ids[mask], minlength=nloc)
    lp = t.zeros(nloc + 1, dtype=t.int64, device=dev)
    t.cumsum(lloc, 0, out=lp[1:])
    lc = (col[mask] - col_beg).contiguous()
    lv = val[mask].contiguous()
    maskr = ~mask
    rcg = col[maskr]
    rows_r = row_ids[maskr]
    lrem = t.bincount(rows_r, minlength=nloc) if rcg.numel() else t.zeros(
        nloc, dtype=t.int64, device=dev)
    rp = t.zeros(nloc + 1, dtype=t.int64, device=dev)
    t.cumsum(lrem, 0, out=rp[1:])
    ghost_global = t.unique(rcg.to(t.int64))
    rc = t.searchsorted(ghost_global, rcg.to(t.int64)).to(t.int32).contiguous()
    rv = val[maskr].contiguous()
    return (lp.to(t.int32).contiguous(), lc, lv,
            rp.to(t.int32).contiguous(), rc, rv, ghost_global)
