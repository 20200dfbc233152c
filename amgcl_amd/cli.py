"""CLI solver driver (parity: examples/solver.cpp — read a MatrixMarket or
binary system, configure the runtime solver from key=value args / JSON,
solve, print the hierarchy and profile).

Usage:
  python -m amgcl_amd.cli -A matrix.mtx [-f rhs.mtx] [-p key=value ...]
  python -m amgcl_amd.cli --poisson 64 --backend hip -p solver.type=bicgstab
"""
import argparse
import json
import sys
import time

import numpy as np


def main(argv=None):
    ap = argparse.ArgumentParser(description="amgcl_amd solver CLI")
    ap.add_argument("-A", "--matrix", help="system matrix (.mtx or .bin)")
    ap.add_argument("-f", "--rhs", help="right-hand side (.mtx or .bin)")
    ap.add_argument("--poisson", type=int, help="generate n^3 Poisson instead")
    ap.add_argument("-p", "--prm", action="append", default=[],
                    help="key=value solver parameter (e.g. solver.type=cg)")
    ap.add_argument("-P", "--prm-file", help="JSON parameter file")
    ap.add_argument("--backend", default="cpu", choices=["cpu", "hip"])
    ap.add_argument("--reorder", action="store_true", help="Cuthill-McKee reorder")
    ap.add_argument("--scale", action="store_true", help="symmetric diagonal scaling")
    ap.add_argument("-o", "--out", help="write solution (binary)")
    ap.add_argument("--convert", nargs=2, metavar=("SRC", "DST"),
                    help="convert a matrix between .mtx and .bin and exit")
    args = ap.parse_args(argv)

    import amgcl_amd as am
    from amgcl_amd.params import set_kv

    if args.convert:
        # mm2bin / bin2mm converters (reference examples/{mm2bin,bin2mm}.cpp)
        from amgcl_amd import io

        src_p, dst = args.convert
        m = io.read_crs(src_p) if src_p.endswith(".bin") else io.mm_read(src_p)
        if dst.endswith(".bin"):
            io.write_crs(dst, m)
        else:
            io.mm_write(dst, m)
        print(f"converted {src_p} -> {dst} ({m.nrows}x{m.ncols}, {m.nnz} nnz)",
              file=sys.stderr)
        return 0

    if args.poisson:
        A, b = am.poisson3d(args.poisson, rhs="random")
    elif args.matrix:
        from amgcl_amd import io

        A = (io.read_crs(args.matrix) if args.matrix.endswith(".bin")
             else io.mm_read(args.matrix))
        if args.rhs:
            b = (io.read_dense(args.rhs) if args.rhs.endswith(".bin")
                 else io.mm_read(args.rhs))
            b = np.asarray(b).ravel()
        else:
            b = np.ones(A.nrows)
    else:
        ap.error("need -A or --poisson")

    prm = json.load(open(args.prm_file)) if args.prm_file else {}
    for kv in args.prm:
        key, _, value = kv.partition("=")
        set_kv(prm, key, value)

    reord = scale = None
    if args.reorder:
        from amgcl_amd.adapter import Reordered

        reord = Reordered(A)
        A, b = reord.A, reord.forward(b)
    if args.scale:
        from amgcl_amd.adapter import ScaledProblem

        scale = ScaledProblem(A)
        A, b = scale.A, scale.scale_rhs(b)

    t0 = time.perf_counter()
    solve = am.make_solver(A, prm, backend=args.backend)
    t1 = time.perf_counter()
    print(solve, file=sys.stderr)
    x, iters, resid = solve(b)
    t2 = time.perf_counter()

    xh = solve.backend.to_host(x)
    if scale is not None:
        xh = scale.unscale_x(xh)
    if reord is not None:
        xh = reord.inverse(xh)

    print(f"iters:  {iters}")
    print(f"error:  {resid:.3e}")
    print(f"setup:  {t1 - t0:.4f} s")
    print(f"solve:  {t2 - t1:.4f} s")
    if args.out:
        from amgcl_amd.io import write_dense

        write_dense(args.out, xh)
    return 0


if __name__ == "__main__":
    sys.exit(main())
