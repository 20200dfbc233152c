"""The MI355X HIP backend.

The single GPU compute backend (north star: no multi-backend dispatch).
Vectors are fp64 torch HIP tensors (memory managed by torch's caching
allocator, interoperable with torch.distributed/RCCL); matrices are
device-resident CSR triples; every solve-phase op launches a hand-written
gfx950 kernel from libamghip.so on the current torch stream.

Replaces the reference HIP backend's hipSPARSE SpMV + rocThrust vector ops
(amgcl/backend/hip.hpp:219-708) with fused custom kernels, and its host
skyline-LU coarse solve (hip.hpp:73-96, D2H/H2D per cycle) with a
device-resident dense-inverse GEMV.
"""
import numpy as np

from ..matrix import CSR
from . import register
from ._hiplib import check, lib


def _stream():
    import torch

    return torch.cuda.current_stream().cuda_stream


class DeviceCSR:
    __slots__ = ("nrows", "ncols", "ptr", "col", "val", "subw",
                 "nslice", "soff", "scol", "sval", "srows")

    def __init__(self, csr: CSR, device, subw=0):
        import torch

        self.nrows = csr.nrows
        self.ncols = csr.ncols
        self.ptr = torch.from_numpy(np.asarray(csr.ptr)).to(device)
        self.col = torch.from_numpy(np.asarray(csr.col)).to(device)
        self.val = torch.from_numpy(np.asarray(csr.val)).to(device)
        self.subw = subw  # 0 = auto by mean row length
        self.nslice = 0
        self.soff = self.scol = self.sval = self.srows = None

    @classmethod
    def from_tensors(cls, nrows, ncols, ptr, col, val, subw=0):
        self = cls.__new__(cls)
        self.nrows = int(nrows)
        self.ncols = int(ncols)
        self.ptr, self.col, self.val = ptr, col, val
        self.subw = subw
        self.nslice = 0
        self.soff = self.scol = self.sval = self.srows = None
        return self

    @property
    def nnz(self):
        return self.col.numel()

    def bytes(self):
        b = (self.ptr.numel() + self.col.numel()) * 4 + self.val.numel() * 8
        if self.nslice:
            b += self.soff.numel() * 8 + self.scol.numel() * 4
            b += self.sval.numel() * self.sval.element_size()
        return b

    def build_sell(self, sigma=0):
        """Build the SELL-64 image of this matrix (kernels.hip rationale:
        wave-native layout; lane = row, slices column-major).  The CSR arrays
        are kept — transfers and non-SELL paths still use them.

        sigma > 0 sorts rows by length inside sigma-sized windows to cut
        slice padding.  MEASURED (r02): a NET LOSS on the 512^3 hierarchy
        (solve 0.43 -> 0.64 s) — the permutation turns the perfectly
        coalesced per-wave x/rhs/y accesses into window-local gathers,
        which costs far more than the ~24% padding it saves.  Off by
        default; kept as an explicit option (and for matrices so ragged
        that padding dominates)."""
        if self.nslice:
            return self
        import os
        import time

        import torch

        dbg = os.environ.get("AMGCL_DEBUG_SELL")
        if dbg:
            torch.cuda.synchronize()
            _t0 = time.perf_counter()
        n = self.nrows
        dev = self.val.device
        nslice = (n + 63) // 64
        lens = (self.ptr[1:] - self.ptr[:-1]).to(torch.int64)
        pad = nslice * 64 - n
        if pad:
            lens = torch.cat([lens, torch.zeros(pad, dtype=torch.int64, device=dev)])
        w = lens.view(nslice, 64).max(dim=1).values
        soff = torch.zeros(nslice + 1, dtype=torch.int64, device=dev)
        torch.cumsum(w * 64, dim=0, out=soff[1:])
        if dbg:
            _t1 = time.perf_counter()  # torch ops enqueued
        total = int(soff[-1].item())
        srows = None
        nnz = self.col.numel()
        if sigma and total > 1.08 * nnz and n > (1 << 16):
            # sigma-sorted slices: order rows by length inside sigma-sized
            # windows to cut the slice padding of ragged (coarse SA) levels;
            # the permutation stays window-local, so the scattered x/rhs/y
            # accesses remain L2-local
            npad = ((n + sigma - 1) // sigma) * sigma
            lens2 = (self.ptr[1:] - self.ptr[:-1]).to(torch.int64)
            if npad > n:
                lens2 = torch.cat([lens2, torch.zeros(npad - n, dtype=torch.int64,
                                                      device=dev)])
            lw = lens2.view(-1, sigma)
            slens, order = torch.sort(lw, dim=1, descending=True, stable=True)
            base = (torch.arange(lw.shape[0], device=dev, dtype=torch.int64)
                    * sigma).unsqueeze(1)
            rowids = (order + base).reshape(-1)
            srows = torch.where(rowids < n, rowids,
                                torch.full_like(rowids, -1)).to(torch.int32)
            srows = srows.contiguous()
            nslice = npad // 64
            w = slens.reshape(-1).view(nslice, 64).max(dim=1).values
            soff = torch.zeros(nslice + 1, dtype=torch.int64, device=dev)
            torch.cumsum(w * 64, dim=0, out=soff[1:])
            total = int(soff[-1].item())
        if dbg:
            _t2 = time.perf_counter()  # sync + D2H of the total
        # ONE allocation for col+val (empty, not zeros: the fill kernel
        # writes the padding itself).  Measured: separate multi-GB
        # allocations stalled the GPU ~11 ms each at 512^3 (allocator
        # growth), see profiles/README.md round 2.
        esize = self.val.element_size()
        blob = torch.empty(total * (4 + esize), dtype=torch.uint8, device=dev)
        scol = blob[: total * 4].view(torch.int32)
        sval = blob[total * 4 :].view(self.val.dtype)
        if dbg:
            _t3 = time.perf_counter()  # allocation
        fn = (lib().amg_sell_fill_f32 if self.val.dtype == torch.float32
              else lib().amg_sell_fill_f64)
        check(fn(n, nslice, self.ptr.data_ptr(), self.col.data_ptr(),
                 self.val.data_ptr(), soff.data_ptr(),
                 srows.data_ptr() if srows is not None else 0,
                 scol.data_ptr(), sval.data_ptr(), _stream()), "sell_fill")
        self.nslice, self.soff, self.scol, self.sval = nslice, soff, scol, sval
        self.srows = srows
        if dbg:
            torch.cuda.synchronize()
            _t4 = time.perf_counter()
            print(f"[sell] n={n:>10} total={total:>11} enq={(_t1-_t0)*1e3:7.1f} "
                  f"item={(_t2-_t1)*1e3:7.1f} alloc={(_t3-_t2)*1e3:7.1f} "
                  f"fill={(_t4-_t3)*1e3:7.1f} ms", flush=True)
        return self


class DeviceBSR:
    """Block CSR (row-major blocks): the solve-phase container for
    block-valued matrices (reference: builtin_hybrid / vexcl_static_matrix).
    BSR SpMV is bandwidth-bound at fp64 (0.25 flop/byte), so the kernels are
    unrolled block loads, not MFMA — see csrc/hip/block.hip header note."""

    __slots__ = ("nbrows", "bsize", "ptr", "col", "val", "nrows", "ncols")

    def __init__(self, csr: CSR, bsize, device):
        import torch

        from .. import _core

        bp, bc, bv = _core.csr_to_bsr(csr.nrows, csr.ptr, csr.col, csr.val, int(bsize))
        self.nbrows = csr.nrows // int(bsize)
        self.bsize = int(bsize)
        self.nrows = csr.nrows
        self.ncols = csr.ncols
        self.ptr = torch.from_numpy(np.asarray(bp)).to(device)
        self.col = torch.from_numpy(np.asarray(bc)).to(device)
        self.val = torch.from_numpy(np.asarray(bv)).to(device)

    @property
    def nnz(self):
        return self.val.numel()

    @classmethod
    def from_device(cls, dcsr, bsize, device):
        """Convert a device CSR to BSR entirely on the GPU (torch ops): the
        block pattern is the coalesced pointwise pattern, values scatter by
        (block index, r, c).  Host twin: _core.csr_to_bsr."""
        import torch

        b = int(bsize)
        n = dcsr.nrows
        nb = n // b
        dev = dcsr.val.device
        t64 = torch.int64
        lens = (dcsr.ptr[1:] - dcsr.ptr[:-1]).to(t64)
        row_of = torch.repeat_interleave(torch.arange(n, device=dev, dtype=t64),
                                         lens)
        col64 = dcsr.col.to(t64)
        rpt = row_of // b
        cpt = col64 // b
        key = rpt * nb + cpt
        ukey = torch.unique(key, sorted=True)  # block pattern, row-major
        blk = torch.searchsorted(ukey, key)    # block index per scalar entry
        nblocks = ukey.numel()
        bptr = torch.zeros(nb + 1, dtype=t64, device=dev)
        torch.cumsum(torch.bincount(ukey // nb, minlength=nb), 0, out=bptr[1:])
        bcol = (ukey % nb).to(torch.int32).contiguous()
        bval = torch.zeros(nblocks * b * b, dtype=dcsr.val.dtype, device=dev)
        dest = blk * (b * b) + (row_of % b) * b + (col64 % b)
        bval[dest] = dcsr.val
        self = cls.__new__(cls)
        self.nbrows = nb
        self.bsize = b
        self.nrows = n
        self.ncols = dcsr.ncols
        self.ptr = bptr.to(torch.int32).contiguous()
        self.col = bcol
        self.val = bval
        return self


@register("hip")
class HipBackend:
    name = "hip"

    def __init__(self, device=None, dtype=None):
        import torch

        if not torch.cuda.is_available():
            raise RuntimeError("hip backend requires a GPU (torch.cuda unavailable)")
        self.torch = torch
        self.device = torch.device(device or "cuda")
        # default value type for vectors; complex128 selects the native
        # complex kernels (parity: amgcl/value_type/complex.hpp on the
        # device backends)
        self.dtype = dtype or torch.float64
        if self.dtype not in (torch.float64, torch.complex128):
            raise ValueError("hip backend supports float64 and complex128")
        lib()  # fail loudly now if the kernel library is missing
        self._dotbuf = torch.zeros(2, dtype=torch.float64, device=self.device)
        self._dothost = torch.zeros(2, dtype=torch.float64, pin_memory=True)

    # --- containers -------------------------------------------------------
    def matrix(self, csr):
        if isinstance(csr, DeviceCSR):
            return csr
        return DeviceCSR(csr, self.device)

    def vector(self, n, dtype=None):
        return self.torch.zeros(n, dtype=dtype or self.dtype, device=self.device)

    def from_host(self, a):
        dt = np.complex128 if (np.iscomplexobj(a)
                               or self.dtype == self.torch.complex128) else np.float64
        t = self.torch.from_numpy(np.ascontiguousarray(a, dtype=dt))
        return t.to(self.device)

    def to_host(self, v):
        return v.cpu().numpy()

    # --- primitives --------------------------------------------------------
    @staticmethod
    def _fn(name, t):
        import torch

        sfx = {torch.float32: "_f32", torch.float64: "_f64",
               torch.complex128: "_c128"}[t.dtype]
        return getattr(lib(), "amg_" + name + sfx)

    def spmv(self, alpha, A, x, beta, y):
        if isinstance(A, DeviceBSR):
            check(lib().amg_bsr_spmv_f64(A.nbrows, A.bsize, A.ptr.data_ptr(),
                                         A.col.data_ptr(), A.val.data_ptr(),
                                         x.data_ptr(), alpha, beta, y.data_ptr(),
                                         _stream()), "bsr_spmv")
            return
        if A.val.is_complex():
            a, b = complex(alpha), complex(beta)
            check(lib().amg_spmv_c128(A.nrows, A.nnz, A.ptr.data_ptr(),
                                      A.col.data_ptr(), A.val.data_ptr(),
                                      x.data_ptr(), a.real, a.imag, b.real, b.imag,
                                      y.data_ptr(), A.subw, _stream()), "cspmv")
            return
        if getattr(A, "nslice", 0):
            check(self._fn("sell_spmv", A.sval)(A.nrows, A.nslice, A.soff.data_ptr(),
                                                A.scol.data_ptr(), A.sval.data_ptr(),
                                                A.srows.data_ptr() if A.srows is not None else 0,
                                                x.data_ptr(), alpha, beta,
                                                y.data_ptr(), _stream()), "sell_spmv")
            return
        check(self._fn("spmv", A.val)(A.nrows, A.nnz, A.ptr.data_ptr(), A.col.data_ptr(),
                                      A.val.data_ptr(), x.data_ptr(), alpha, beta,
                                      y.data_ptr(), A.subw, _stream()), "spmv")

    def residual(self, b, A, x, r):
        if isinstance(A, DeviceBSR):
            check(lib().amg_bsr_residual_f64(A.nbrows, A.bsize, A.ptr.data_ptr(),
                                             A.col.data_ptr(), A.val.data_ptr(),
                                             b.data_ptr(), x.data_ptr(), r.data_ptr(),
                                             _stream()), "bsr_residual")
            return
        if A.val.is_complex():
            check(lib().amg_residual_c128(A.nrows, A.nnz, A.ptr.data_ptr(),
                                          A.col.data_ptr(), A.val.data_ptr(),
                                          b.data_ptr(), x.data_ptr(), r.data_ptr(),
                                          A.subw, _stream()), "cresidual")
            return
        if getattr(A, "nslice", 0):
            check(self._fn("sell_residual", A.sval)(A.nrows, A.nslice,
                                                    A.soff.data_ptr(), A.scol.data_ptr(),
                                                    A.sval.data_ptr(),
                                                    A.srows.data_ptr() if A.srows is not None else 0,
                                                    b.data_ptr(),
                                                    x.data_ptr(), r.data_ptr(),
                                                    _stream()), "sell_residual")
            return
        check(self._fn("residual", A.val)(A.nrows, A.nnz, A.ptr.data_ptr(),
                                          A.col.data_ptr(), A.val.data_ptr(),
                                          b.data_ptr(), x.data_ptr(), r.data_ptr(),
                                          A.subw, _stream()), "residual")

    def relax_diag(self, A, M, rhs, x, t):
        """t = M∘(rhs - A x); x += t (fused single pass over A + axpby)."""
        if isinstance(A, DeviceBSR):
            check(lib().amg_bsr_relax_f64(A.nbrows, A.bsize, A.ptr.data_ptr(),
                                          A.col.data_ptr(), A.val.data_ptr(),
                                          M.data_ptr(), rhs.data_ptr(), x.data_ptr(),
                                          t.data_ptr(), _stream()), "bsr_relax")
            self.axpby(1.0, t, 1.0, x)
            return
        if A.val.is_complex():
            check(lib().amg_relax_diag_c128(A.nrows, A.nnz, A.ptr.data_ptr(),
                                            A.col.data_ptr(), A.val.data_ptr(),
                                            M.data_ptr(), rhs.data_ptr(),
                                            x.data_ptr(), t.data_ptr(), A.subw,
                                            _stream()), "crelax")
            self.axpby(1.0, t, 1.0, x)
            return
        if getattr(A, "nslice", 0):
            # SELL relax writes x_new = x + M(rhs - Ax) into t, then copy back
            check(self._fn("sell_relax", A.sval)(A.nrows, A.nslice, A.soff.data_ptr(),
                                                 A.scol.data_ptr(), A.sval.data_ptr(),
                                                 A.srows.data_ptr() if A.srows is not None else 0,
                                                 M.data_ptr(), rhs.data_ptr(),
                                                 x.data_ptr(), t.data_ptr(),
                                                 _stream()), "sell_relax")
            x.copy_(t)
            return
        check(self._fn("relax_diag", A.val)(A.nrows, A.nnz, A.ptr.data_ptr(),
                                            A.col.data_ptr(), A.val.data_ptr(),
                                            M.data_ptr(), rhs.data_ptr(), x.data_ptr(),
                                            t.data_ptr(), A.subw, _stream()),
              "relax_diag")
        self.axpby(1.0, t, 1.0, x)

    def clear(self, x):
        if x.is_complex():
            check(lib().amg_fill_c128(x.numel(), 0.0, 0.0, x.data_ptr(),
                                      _stream()), "cfill")
            return
        check(self._fn("fill", x)(x.numel(), 0.0, x.data_ptr(), _stream()), "fill")

    def copy(self, x, y):
        y.copy_(x)

    def axpby(self, a, x, b, y):
        if x.is_complex():
            a, b = complex(a), complex(b)
            check(lib().amg_axpby_c128(x.numel(), a.real, a.imag, x.data_ptr(),
                                       b.real, b.imag, y.data_ptr(), _stream()),
                  "caxpby")
            return
        check(self._fn("axpby", x)(x.numel(), a, x.data_ptr(), b, y.data_ptr(),
                                   _stream()), "axpby")

    def axpbypcz(self, a, x, b, y, c, z):
        if x.is_complex():
            a, b, c = complex(a), complex(b), complex(c)
            check(lib().amg_axpbypcz_c128(x.numel(), a.real, a.imag, x.data_ptr(),
                                          b.real, b.imag, y.data_ptr(), c.real,
                                          c.imag, z.data_ptr(), _stream()),
                  "caxpbypcz")
            return
        check(self._fn("axpbypcz", x)(x.numel(), a, x.data_ptr(), b, y.data_ptr(), c,
                                      z.data_ptr(), _stream()), "axpbypcz")

    def vmul(self, a, m, x, b, z):
        if x.is_complex():
            a, b = complex(a), complex(b)
            check(lib().amg_vmul_c128(x.numel(), a.real, a.imag, m.data_ptr(),
                                      x.data_ptr(), b.real, b.imag, z.data_ptr(),
                                      _stream()), "cvmul")
            return
        check(self._fn("vmul", x)(x.numel(), a, m.data_ptr(), x.data_ptr(), b,
                                  z.data_ptr(), _stream()), "vmul")

    def cast(self, src, dst):
        import torch

        if src.dtype == dst.dtype:
            dst.copy_(src)
        elif src.dtype == torch.float64:
            check(lib().amg_cast_d2s(src.numel(), src.data_ptr(), dst.data_ptr(),
                                     _stream()), "cast")
        else:
            check(lib().amg_cast_s2d(src.numel(), src.data_ptr(), dst.data_ptr(),
                                     _stream()), "cast")

    def dot(self, x, y):
        if x.is_complex():
            self._dotbuf.zero_()
            check(lib().amg_dot_c128(x.numel(), x.data_ptr(), y.data_ptr(),
                                     self._dotbuf.data_ptr(), _stream()), "cdot")
            self._dothost.copy_(self._dotbuf, non_blocking=False)
            return complex(float(self._dothost[0]), float(self._dothost[1]))
        check(self._fn("dot", x)(x.numel(), x.data_ptr(), y.data_ptr(),
                                 self._dotbuf.data_ptr(), _stream()), "dot")
        return float(self._dotbuf[0].item())

    def dot2(self, x1, y1, x2, y2):
        if x1.is_complex():
            return self.dot(x1, y1), self.dot(x2, y2)
        check(lib().amg_dot2_f64(x1.numel(), x1.data_ptr(), y1.data_ptr(),
                                 x2.data_ptr(), y2.data_ptr(),
                                 self._dotbuf.data_ptr(), _stream()), "dot2")
        self._dothost.copy_(self._dotbuf, non_blocking=False)
        return float(self._dothost[0]), float(self._dothost[1])

    def blkdiag_vmul(self, bsize, M, x, y):
        """y_i = M_i x_i with M a flat (nblocks*B*B) row-major block-diagonal."""
        nb = x.numel() // bsize
        check(lib().amg_blkdiag_vmul_f64(nb, bsize, M.data_ptr(), x.data_ptr(),
                                         y.data_ptr(), _stream()), "blkdiag_vmul")

    def gather(self, x, idx, buf):
        if x.is_complex():
            self.torch.index_select(x, 0, idx.to(self.torch.int64), out=buf)
            return
        check(self._fn("gather", x)(idx.numel(), x.data_ptr(), idx.data_ptr(),
                                    buf.data_ptr(), _stream()), "gather")

    def scatter(self, buf, idx, x):
        if x.is_complex():
            x[idx.to(self.torch.int64)] = buf
            return
        check(self._fn("scatter", x)(idx.numel(), buf.data_ptr(), idx.data_ptr(),
                                     x.data_ptr(), _stream()), "scatter")

    # --- coarse direct solver ---------------------------------------------
    def coarse_solver(self, csr, kind="dense"):
        if kind == "skyline":
            # host profile-LU with a round-trip per cycle (reference's coarse
            # shape, hip.hpp:73-96); O(profile) memory vs dense O(n^2)
            return HostSkylineSolver(csr, self)
        if kind == "splu":
            # host SuperLU with a D2H/H2D round-trip per cycle — the
            # reference's coarse-solve shape (hip.hpp:73-96); only worth it
            # for very large, very sparse coarse levels
            return HostSpluSolver(csr, self)
        if kind != "dense":
            raise ValueError(f"unknown direct_solver '{kind}' "
                             "(dense, skyline, splu)")
        if isinstance(csr, DeviceCSR):
            return DeviceDenseSolver.from_device(csr, self)
        return DeviceDenseSolver(csr, self)

    def synchronize(self):
        self.torch.cuda.synchronize()


class HostSkylineSolver:
    """Host skyline (profile) LU coarse solve with a device round-trip per
    application (parity: amgcl/solver/skyline_lu.hpp via backend/hip.hpp
    hip_skyline_lu:73)."""

    def __init__(self, csr, backend):
        from .cpu import SkylineCoarseSolver

        if isinstance(csr, DeviceCSR):
            from . import hip_setup

            csr = hip_setup.download(csr)
        self._s = SkylineCoarseSolver(csr)
        self.n = csr.nrows

    def __call__(self, f, u):
        import torch

        out = np.empty(self.n)
        self._s(f.cpu().numpy().astype(np.float64), out)
        u.copy_(torch.from_numpy(out).to(u.device, dtype=u.dtype))


class HostSpluSolver:
    """Host SuperLU coarse solve with a device round-trip per application
    (the reference's coarse-solve shape, backend/hip.hpp:73-96; opt-in via
    direct_solver='splu')."""

    def __init__(self, csr, backend):
        import scipy.sparse.linalg as spla

        if isinstance(csr, DeviceCSR):
            from . import hip_setup

            csr = hip_setup.download(csr)
        self._lu = spla.splu(csr.to_scipy().tocsc())
        self.n = csr.nrows

    def __call__(self, f, u):
        import torch

        sol = self._lu.solve(f.cpu().numpy().astype(np.float64))
        u.copy_(torch.from_numpy(sol).to(u.device, dtype=u.dtype))


class DeviceDenseSolver:
    """Coarsest-level solve as device-resident dense GEMV with a precomputed
    inverse (design note in backend/cpu.py:DenseCoarseSolver)."""

    def __init__(self, csr: CSR, backend):
        # densify on the host (cheap), invert on the DEVICE: hipSOLVER
        # takes ~25 ms at n=2900 where single-threaded LAPACK takes ~300+
        import torch

        a = csr.to_scipy().toarray()
        self.n = csr.nrows
        dense = torch.from_numpy(np.ascontiguousarray(a)).to(backend.device)
        self.inv = torch.linalg.inv(dense).contiguous().ravel()

    @classmethod
    def from_device(cls, dcsr, backend):
        """Densify + invert entirely on the GPU (no download round-trip; the
        host LAPACK route measured 3.7x slower at n=846)."""
        import torch

        self = cls.__new__(cls)
        n = dcsr.nrows
        dense = torch.zeros(n, n, dtype=dcsr.val.dtype, device=dcsr.val.device)
        lens = dcsr.ptr[1:] - dcsr.ptr[:-1]
        rows = torch.repeat_interleave(
            torch.arange(n, device=dcsr.val.device, dtype=torch.int64),
            lens.to(torch.int64))
        dense[rows, dcsr.col.to(torch.int64)] = dcsr.val
        self.n = n
        # MEASURED (r02 scope profile): the host-numpy inverse costs ~88 ms
        # warm at n=846 (download + single-threaded LAPACK + upload) vs
        # ~24 ms for the device torch.linalg.inv — keep the device path.
        self.inv = torch.linalg.inv(dense).contiguous().ravel()
        return self

    def __call__(self, f, u):
        import torch

        if self.inv.is_complex():
            check(lib().amg_gemv_c128(self.n, self.inv.data_ptr(), f.data_ptr(),
                                      u.data_ptr(), _stream()), "cgemv")
            return
        fn = lib().amg_gemv_f32 if self.inv.dtype == torch.float32 else lib().amg_gemv_f64
        check(fn(self.n, self.inv.data_ptr(), f.data_ptr(), u.data_ptr(), _stream()),
              "gemv")
