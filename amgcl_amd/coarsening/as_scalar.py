"""as_scalar coarsening wrapper + BSR (block-valued) input adapter.

Parity: amgcl/coarsening/as_scalar.hpp:46 — the reference wraps any base
coarsening so a *block-valued* system is unblocked to its scalar expansion,
the base coarsening runs on scalars, and the transfer operators are
re-blocked.  In this framework the setup always runs on scalar CSR (the
builtin_hybrid design: scalar setup quality + BSR solve storage via
AMG(block_value=B)), so the wrapper's job splits into:

* `unblock_bsr(...)`: accept a block-valued (BSR) *input* matrix and expand
  it to the scalar CSR the setup engine consumes (the reference's unblock
  step, as_scalar.hpp:69);
* `AsScalar`: a registry-visible coarsening wrapper delegating to its base
  (the re-block step is AMG's existing `block_value` storage conversion).
"""
import numpy as np

from ..matrix import CSR
from . import make_coarsening


def unblock_bsr(nbrows, bsize, ptr, col, val, ncols=None):
    """Expand a BSR matrix (row-major B x B blocks in CSR block order) to
    scalar CSR.  `val` is (nblocks, B, B) or flat nblocks*B*B."""
    b = int(bsize)
    ptr = np.asarray(ptr, dtype=np.int64)
    col = np.asarray(col, dtype=np.int64)
    val = np.asarray(val, dtype=np.float64).reshape(-1, b, b)
    nblocks = len(col)
    n = int(nbrows) * b
    blk_per_row = np.diff(ptr)
    # scalar row i*b+r has blk_per_row[i]*b entries
    sptr = np.zeros(n + 1, dtype=np.int64)
    sptr[1:] = np.repeat(blk_per_row * b, b)
    np.cumsum(sptr, out=sptr)
    # destination order: for each block row i, scalar row r: all blocks j,
    # then scalar cols c
    brow_of = np.repeat(np.arange(nbrows, dtype=np.int64), blk_per_row)
    scol = (col[:, None] * b + np.arange(b)[None, :])  # nblocks x b
    scol_rows = np.broadcast_to(scol[:, None, :], (nblocks, b, b))
    # gather per scalar row: order blocks by (brow, r, j, c)
    order = np.argsort(brow_of, kind="stable")  # already sorted; identity
    sval = val.transpose(0, 1, 2)  # (j, r, c)
    # build arrays grouped by (brow, r): concatenate each block's r-th row
    out_col = np.empty(nblocks * b * b, dtype=np.int32)
    out_val = np.empty(nblocks * b * b, dtype=np.float64)
    # position of block j's (r, :) slice inside the scalar CSR:
    # base(brow) + (pos within row)*b columns... compute via per-block offset
    pos_in_row = np.arange(nblocks, dtype=np.int64) - ptr[brow_of]
    for r in range(b):
        dst = sptr[brow_of * b + r] + pos_in_row * b
        idx = (dst[:, None] + np.arange(b)[None, :]).ravel()
        out_col[idx] = scol_rows[:, r, :].ravel().astype(np.int32)
        out_val[idx] = sval[:, r, :].ravel()
    m = int(ncols) * b if ncols is not None else n
    return CSR(n, m, sptr, out_col, out_val)


class AsScalar:
    """Registry wrapper: {"type": "as_scalar", "base": {...}} — delegates to
    the base coarsening (setup is scalar by design; see module docstring)."""

    @staticmethod
    def defaults():
        return {"base": {"type": "smoothed_aggregation"}}

    def __init__(self, prm=None):
        prm = dict(prm or {})
        base = prm.pop("base", None) or {"type": "smoothed_aggregation"}
        if prm:
            raise ValueError(f"unknown as_scalar options: {sorted(prm)}")
        self.base = make_coarsening(dict(base))

    def transfer_operators(self, A):
        return self.base.transfer_operators(A)

    def coarse_operator(self, A, P, R):
        return self.base.coarse_operator(A, P, R)
