#!/usr/bin/env python3
"""Per-kernel bandwidth microbenchmark on the real AMG hierarchy (GPU).

Builds the 3D Poisson hierarchy on-device, then times each solve-phase kernel
per level and per SUBW variant, reporting effective GB/s (nominal bytes:
nnz*(12B val+col) + row ptr + in/out vectors; x-gather counted once).
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import amgcl_amd as am
from amgcl_amd.backend import make_backend
from amgcl_amd.backend.hip_setup import poisson3d_device


def timeit(fn, iters=20):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--size", type=int, default=384)
    ap.add_argument("--sort", action="store_true", help="sort level rows first")
    args = ap.parse_args()

    hip = make_backend("hip")
    A = poisson3d_device(args.size)
    solve = am.make_solver(A, {"solver": {"type": "cg", "tol": 1e-6}}, backend=hip)
    amg = solve.P
    print(amg)

    if args.sort:
        from amgcl_amd.backend._hiplib import check, lib
        from amgcl_amd.backend.hip_setup import _stream

        for lvl in amg.levels:
            Ad = lvl.A
            check(lib().amg_sort_rows(Ad.nrows, Ad.ptr[1:].data_ptr(),
                                      Ad.col.data_ptr(), Ad.val.data_ptr(),
                                      _stream()), "sort")
        torch.cuda.synchronize()
        print("(rows sorted)")

    for li, lvl in enumerate(amg.levels):
        Ad = lvl.A
        n, nnz = Ad.nrows, int(Ad.nnz)
        if n < 2000:
            continue
        x = torch.rand(Ad.ncols, dtype=torch.float64, device=hip.device)
        y = torch.zeros(n, dtype=torch.float64, device=hip.device)
        b = torch.rand(n, dtype=torch.float64, device=hip.device)
        m = torch.rand(n, dtype=torch.float64, device=hip.device)
        gb_spmv = (nnz * 12 + (n + 1) * 4 + min(nnz, Ad.ncols) * 8 + n * 8) / 1e9
        row = f"L{li} n={n:>9} nnz={nnz:>10} mean={nnz/n:5.1f} | "
        best = (None, 0)
        sell_state = (Ad.nslice, Ad.soff, Ad.scol, Ad.sval)
        Ad.nslice = 0  # force the CSR kernels for the subw sweep
        for subw in (1, 2, 4, 8, 16, 32, 64):
            Ad.subw = subw
            dt = timeit(lambda: hip.spmv(1.0, Ad, x, 0.0, y))
            bw = gb_spmv / dt
            row += f"s{subw}:{bw:5.0f} "
            if bw > best[1]:
                best = (subw, bw)
        Ad.subw = 0
        dt = timeit(lambda: hip.spmv(1.0, Ad, x, 0.0, y))
        row += f"| auto:{gb_spmv/dt:5.0f} GB/s (best s{best[0]})"
        print(row)

        Ad.subw = best[0]
        t = torch.zeros(n, dtype=torch.float64, device=hip.device)
        dt = timeit(lambda: hip.relax_diag(Ad, m, b, y, t))
        gb = gb_spmv + n * 8 * 4 / 1e9
        print(f"   relax_diag(best): {gb/dt:5.0f} GB/s   ", end="")
        dt = timeit(lambda: hip.residual(b, Ad, x, y))
        print(f"residual: {(gb_spmv + n*8/1e9)/dt:5.0f} GB/s")
        Ad.subw = 0
        # SELL-64 image (same nominal-bytes denominator as CSR for a fair
        # apples-to-apples effective number; padding inflates actual bytes)
        Ad.nslice, Ad.soff, Ad.scol, Ad.sval = sell_state
        if not Ad.nslice:
            try:
                Ad.build_sell()
            except Exception as e:
                print(f"   (sell build failed: {e})")
        if Ad.nslice:
            pad = Ad.sval.numel() / max(nnz, 1)
            dt = timeit(lambda: hip.spmv(1.0, Ad, x, 0.0, y))
            s_spmv = gb_spmv / dt
            dt = timeit(lambda: hip.relax_diag(Ad, m, b, y, t))
            s_rel = gb / dt
            dt = timeit(lambda: hip.residual(b, Ad, x, y))
            s_res = (gb_spmv + n * 8 / 1e9) / dt
            print(f"   SELL (pad x{pad:4.2f}): spmv {s_spmv:5.0f}  "
                  f"relax {s_rel:5.0f}  residual {s_res:5.0f} GB/s")

    n = amg.levels[0].rows
    x = torch.rand(n, dtype=torch.float64, device=hip.device)
    y = torch.rand(n, dtype=torch.float64, device=hip.device)
    dt = timeit(lambda: hip.dot(x, y))
    print(f"dot(n={n}): {n*16/1e9/dt:5.0f} GB/s  {dt*1e6:.0f} us")
    dt = timeit(lambda: hip.axpby(1.1, x, 0.9, y))
    print(f"axpby: {n*24/1e9/dt:5.0f} GB/s")


if __name__ == "__main__":
    main()
