#!/usr/bin/env python3
"""MFMA vs unrolled BSR SpMV head-to-head on the elasticity config
(BASELINE config #3 shape: B=3/4 block matrices).  VERDICT r01 next-step #4:
settle the MFMA question with data, not prose."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

import amgcl_amd as am
from amgcl_amd.backend import make_backend
from amgcl_amd.backend._hiplib import check, lib
from amgcl_amd.backend.hip import DeviceBSR, _stream
from amgcl_amd.matrix import CSR


def timeit(fn, iters=50):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 96
    B = 4
    hip = make_backend("hip")
    # elasticity-shaped block system: 7-pt stencil blocks of BxB
    Ap, _ = am.poisson3d(n)
    import scipy.sparse as sp

    rng = np.random.default_rng(1)
    base = Ap.to_scipy().tocsr()
    blocks = rng.standard_normal((base.nnz, B, B)) * 0.1
    # make it block-diagonally dominant
    row_of = np.repeat(np.arange(base.shape[0]), np.diff(base.indptr))
    diag_mask = row_of == base.indices
    blocks[diag_mask] += np.eye(B) * 8.0
    m = sp.bsr_matrix((blocks, base.indices, base.indptr),
                      shape=(base.shape[0] * B, base.shape[0] * B)).tocsr()
    m.sort_indices()
    A = CSR(m.shape[0], m.shape[1], m.indptr, m.indices, m.data)
    Ad = DeviceBSR(A, B, hip.device)
    nb = Ad.nbrows
    x = torch.rand(m.shape[0], dtype=torch.float64, device=hip.device)
    y = torch.zeros_like(x)
    gb = (Ad.val.numel() * 8 + Ad.col.numel() * 4 + (nb + 1) * 4
          + x.numel() * 8 * 2) / 1e9

    dt_unrolled = timeit(lambda: hip.spmv(1.0, Ad, x, 0.0, y))
    y_ref = y.cpu().numpy().copy()
    dt_mfma = timeit(lambda: check(
        lib().amg_bsr_spmv_mfma4_f64(nb, Ad.ptr.data_ptr(), Ad.col.data_ptr(),
                                     Ad.val.data_ptr(), x.data_ptr(), 1.0, 0.0,
                                     y.data_ptr(), _stream()), "mfma4"))
    err = np.abs(y.cpu().numpy() - y_ref).max() / np.abs(y_ref).max()
    print(f"BSR B={B} n={m.shape[0]} ({nb} block rows, {Ad.val.numel()//16} blocks)")
    print(f"unrolled: {dt_unrolled*1e6:8.1f} us  {gb/dt_unrolled:7.0f} GB/s")
    print(f"mfma    : {dt_mfma*1e6:8.1f} us  {gb/dt_mfma:7.0f} GB/s")
    print(f"relative diff vs unrolled: {err:.2e}")
    print("winner:", "mfma" if dt_mfma < dt_unrolled else "unrolled")


if __name__ == "__main__":
    main()
