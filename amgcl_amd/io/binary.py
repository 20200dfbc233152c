"""Raw binary matrix dump/restore with size headers
(parity: amgcl/io/binary.hpp:70-173; also serves as the checkpoint format,
incl. per-rank strips — SURVEY §5.4)."""
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


def read_crs(path, row_beg=None, row_end=None):
    with open(path, "rb") as f:
        if f.read(8) != _MAGIC:
            raise ValueError("bad binary matrix file")
        nrows, ncols, nnz = np.fromfile(f, dtype=np.int64, count=3)
        ptr = np.fromfile(f, dtype=np.int64, count=nrows + 1)
        if row_beg is None:
            col = np.fromfile(f, dtype=np.int32, count=nnz)
            val = np.fromfile(f, dtype=np.float64, count=nnz)
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
        if f.read(8) != _MAGIC:
            raise ValueError("bad binary dense file")
        nrows, ncols = np.fromfile(f, dtype=np.int64, count=2)
        a = np.fromfile(f, dtype=np.float64, count=nrows * ncols)
        return a.reshape(nrows, ncols) if ncols > 1 else a
