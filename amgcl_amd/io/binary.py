"""Raw binary matrix dump/restore with size headers; also serves as the
checkpoint format, incl. per-rank strip reads.

Two on-disk layouts are supported for reading:

* native: 8-byte ``AMGCLAMD`` magic + int64 {nrows, ncols, nnz} header,
  int64 ptr, int32 col, float64 val;
* the reference layout (amgcl/io/binary.hpp:70-173, examples/mm2bin.cpp):
  a single ``size_t rows`` followed by int64 ptr / int64 col / float64 val,
  no magic — detected by the absence of the magic, so files produced by the
  upstream ``mm2bin`` tooling load directly (square matrices).

Writing uses the native layout.
"""
import numpy as np

from ..matrix import CSR

_MAGIC = b"AMGCLAMD"


def write_crs(path, m: CSR):
    with open(path, "wb") as f:
        f.write(_MAGIC)
        np.array([m.nrows, m.ncols, m.nnz], dtype=np.int64).tofile(f)
        m.ptr.astype(np.int64).tofile(f)
        m.col.tofile(f)
        m.val.tofile(f)


def _read_crs_reference(f, row_beg, row_end):
    """Reference amgcl binary layout: size_t n, ptrdiff_t ptr[n+1],
    ptrdiff_t col[nnz], double val[nnz] (amgcl/io/binary.hpp:108-121)."""
    f.seek(0)
    (nrows,) = np.fromfile(f, dtype=np.int64, count=1)
    nrows = int(nrows)
    ptr_base = 8
    if row_beg is None:
        row_beg, row_end = 0, nrows
    f.seek(ptr_base + 8 * row_beg)
    ptr = np.fromfile(f, dtype=np.int64, count=row_end - row_beg + 1)
    f.seek(ptr_base + 8 * nrows)
    (nnz,) = np.fromfile(f, dtype=np.int64, count=1)
    lo, hi = int(ptr[0]), int(ptr[-1])
    col_base = ptr_base + 8 * (nrows + 1)
    f.seek(col_base + 8 * lo)
    col = np.fromfile(f, dtype=np.int64, count=hi - lo).astype(np.int32)
    f.seek(col_base + 8 * int(nnz) + 8 * lo)
    val = np.fromfile(f, dtype=np.float64, count=hi - lo)
    return CSR(row_end - row_beg, nrows, ptr - lo, col, val)


def read_crs(path, row_beg=None, row_end=None):
    with open(path, "rb") as f:
        if f.read(8) != _MAGIC:
            return _read_crs_reference(f, row_beg, row_end)
        nrows, ncols, nnz = np.fromfile(f, dtype=np.int64, count=3)
        ptr = np.fromfile(f, dtype=np.int64, count=nrows + 1)
        if len(ptr) != nrows + 1:
            raise ValueError(f"truncated binary matrix file '{path}'")
        if row_beg is None:
            col = np.fromfile(f, dtype=np.int32, count=nnz)
            val = np.fromfile(f, dtype=np.float64, count=nnz)
            if len(col) != nnz or len(val) != nnz:
                raise ValueError(f"truncated binary matrix file '{path}'")
            return CSR(nrows, ncols, ptr, col, val)
        # strip read
        lo, hi = int(ptr[row_beg]), int(ptr[row_end])
        base = f.tell()
        f.seek(base + 4 * lo)
        col = np.fromfile(f, dtype=np.int32, count=hi - lo)
        f.seek(base + 4 * nnz + 8 * lo)
        val = np.fromfile(f, dtype=np.float64, count=hi - lo)
        return CSR(row_end - row_beg, ncols, ptr[row_beg : row_end + 1] - lo, col, val)


def write_dense(path, a):
    a = np.asarray(a, dtype=np.float64)
    with open(path, "wb") as f:
        f.write(_MAGIC)
        shape = a.shape if a.ndim == 2 else (a.shape[0], 1)
        np.array(shape, dtype=np.int64).tofile(f)
        a.tofile(f)


def read_dense(path):
    with open(path, "rb") as f:
        head = f.read(8)
        if head != _MAGIC:
            # reference layout: size_t n, size_t m, double v[n*m]
            f.seek(0)
            nrows, ncols = np.fromfile(f, dtype=np.int64, count=2)
        else:
            nrows, ncols = np.fromfile(f, dtype=np.int64, count=2)
        a = np.fromfile(f, dtype=np.float64, count=nrows * ncols)
        return a.reshape(nrows, ncols) if ncols > 1 else a
