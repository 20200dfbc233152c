"""Mixed-precision AMG: fp32 hierarchy under a fp64 Krylov loop.

Reference analogue: examples/mixed_precision.cpp. On the HIP backend the
native driver runs the whole fp32 V-cycle + fp64 CG in one C++ call per
solve; the fp32 hierarchy roughly halves the preconditioner's memory traffic.
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import amgcl_amd as am


def main(n=128):
    A, b = am.poisson3d(n, rhs="random")
    for precision in ("fp64", "mixed"):
        prm = {"solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}}
        if precision == "mixed":
            prm["precond"] = {"class": "amg", "precision": "mixed"}
        backend = "hip" if _has_gpu() else "cpu"
        if precision == "mixed" and backend != "hip":
            print("mixed precision needs the hip backend; skipping")
            continue
        solve = am.make_solver(A, prm, backend=backend)
        t0 = time.perf_counter()
        x, iters, resid = solve(b)
        print(f"{precision}: {iters} iters, resid {resid:.2e}, "
              f"{time.perf_counter() - t0:.3f} s")


def _has_gpu():
    try:
        import torch

        return torch.cuda.is_available()
    except Exception:
        return False


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 128)
