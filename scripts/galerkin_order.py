"""A/B the Galerkin association on the fine level: R*(A*P) vs (R*A)*P.

Builds the 512^3 level-0 transfer operators with the production path, then
times each product chain warm (device-synced)."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))


def main(n=512):
    import torch

    from amgcl_amd.backend import hip_setup
    from amgcl_amd.backend.hip_setup import poisson3d_device

    A = poisson3d_device(n)
    naggr, ids, S = hip_setup.aggregates(A, 0.08)
    P = hip_setup.smoothed_prolongation(A, S, ids, naggr, 0.666667)
    R = hip_setup.transpose(P)
    print(f"A {A.nrows}x{A.ncols} nnz={A.nnz}  P nnz={P.nnz}")
    torch.cuda.synchronize()

    def timed(fn, label):
        for _ in range(2):
            C = fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        C = fn()
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        print(f"{label:10s} {dt*1000:8.1f} ms   Ac nnz={C.nnz}")
        return C

    AP = timed(lambda: hip_setup.spgemm(A, P, sort=False), "A*P")
    timed(lambda: hip_setup.spgemm(R, AP), "R*(AP)")
    try:
        RA = timed(lambda: hip_setup.spgemm(R, A, sort=False), "R*A")
        print(f"  RA nnz={RA.nnz}")
        timed(lambda: hip_setup.spgemm(RA, P), "(RA)*P")
    except OverflowError as e:
        print("R*A overflow:", e)


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 512)
