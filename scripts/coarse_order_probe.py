"""Upper-bound probe: does a locality-blocked (Morton) numbering of the
LEVEL-1 coarse unknowns raise the SELL SpMV bandwidth?

The production coarse numbering is first-fine-row order (x-fastest
lexicographic).  Any 1-D numbering of a 3-D grid leaves two far directions;
Morton order bounds the index distance of all three.  This probe cheats by
using the known 512^3 grid geometry to compute aggregate centroids — if even
this ideal ordering does not beat lexicographic, no algebraic reordering
will, and the idea dies here (see profiles/README.md).
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))


def morton3(ix, iy, iz, bits=10):
    import torch

    out = torch.zeros_like(ix)
    for b in range(bits):
        out |= ((ix >> b) & 1) << (3 * b + 0)
        out |= ((iy >> b) & 1) << (3 * b + 1)
        out |= ((iz >> b) & 1) << (3 * b + 2)
    return out


def bench_spmv(A, label, reps=50):
    import torch

    from amgcl_amd.backend.hip import HipBackend

    be = HipBackend()
    x = torch.randn(A.ncols, dtype=torch.float64, device="cuda")
    y = torch.zeros(A.nrows, dtype=torch.float64, device="cuda")
    for _ in range(5):
        be.spmv(1.0, A, x, 0.0, y)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        be.spmv(1.0, A, x, 0.0, y)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / reps
    gb = (A.nnz * 12 + A.nrows * 8 + A.ncols * 8) / 1e9
    print(f"{label:28s} {dt*1e3:7.3f} ms  {gb/dt:7.0f} GB/s  nnz={A.nnz}")


def main(n=512):
    import torch

    from amgcl_amd.backend import hip_setup
    from amgcl_amd.backend.hip import DeviceCSR
    from amgcl_amd.backend.hip_setup import poisson3d_device

    A = poisson3d_device(n)
    naggr, ids, S = hip_setup.aggregates(A, 0.08)
    P = hip_setup.smoothed_prolongation(A, S, ids, naggr, 0.666667)
    R = hip_setup.transpose(P)
    A1 = hip_setup.spgemm(R, hip_setup.spgemm(A, P, sort=False))
    print(f"A1 {A1.nrows} rows nnz={A1.nnz} ({A1.nnz/A1.nrows:.1f}/row)")

    A1.build_sell()
    bench_spmv(A1, "lex (production)")

    # aggregate centroids from the known grid geometry
    idx = torch.arange(n**3, device="cuda", dtype=torch.int64)
    ix, iy, iz = idx % n, (idx // n) % n, idx // (n * n)
    ids64 = ids.to(torch.int64)
    cnt = torch.bincount(ids64, minlength=naggr).double().clamp(min=1)
    cx = torch.bincount(ids64, weights=ix.double(), minlength=naggr) / cnt
    cy = torch.bincount(ids64, weights=iy.double(), minlength=naggr) / cnt
    cz = torch.bincount(ids64, weights=iz.double(), minlength=naggr) / cnt
    code = morton3(cx.long(), cy.long(), cz.long())
    perm = torch.argsort(code)  # new order: old aggregate perm[k] -> slot k
    inv = torch.empty_like(perm)
    inv[perm] = torch.arange(naggr, device="cuda")

    # symmetric permutation of A1 via scipy (exactness only; setup-time cost
    # is irrelevant for this probe)
    import numpy as np
    import scipy.sparse as sp

    m = sp.csr_matrix(
        (A1.val.cpu().numpy(), A1.col.cpu().numpy(), A1.ptr.cpu().numpy()),
        shape=(A1.nrows, A1.ncols))
    pm = perm.cpu().numpy()
    m2 = m[pm][:, pm].tocsr()
    m2.sort_indices()
    B = DeviceCSR.from_tensors(
        A1.nrows, A1.ncols,
        torch.tensor(m2.indptr, dtype=torch.int32, device="cuda"),
        torch.tensor(m2.indices, dtype=torch.int32, device="cuda"),
        torch.tensor(m2.data, dtype=torch.float64, device="cuda"))
    B.build_sell()
    bench_spmv(B, "morton (upper bound)")

    # algebraic (coordinate-free) candidates: group level-1 nodes by their
    # OWN next-level aggregates — single level (clusters ~30) and a two-level
    # chain (clusters ~900), the hierarchy's built-in locality blocking
    naggr2, ids2, S2 = hip_setup.aggregates(A1, 0.04)
    key1 = ids2.long()
    pm1 = torch.argsort(key1, stable=True).cpu().numpy()
    m4 = m[pm1][:, pm1].tocsr()
    D = DeviceCSR.from_tensors(
        A1.nrows, A1.ncols,
        torch.tensor(m4.indptr, dtype=torch.int32, device="cuda"),
        torch.tensor(m4.indices, dtype=torch.int32, device="cuda"),
        torch.tensor(m4.data, dtype=torch.float64, device="cuda"))
    D.build_sell()
    bench_spmv(D, "agg-grouped (1 level)")

    P1 = hip_setup.smoothed_prolongation(A1, S2, ids2, naggr2, 0.666667)
    R1 = hip_setup.transpose(P1)
    A2 = hip_setup.spgemm(R1, hip_setup.spgemm(A1, P1, sort=False))
    naggr3, ids3, _ = hip_setup.aggregates(A2, 0.02)
    key2 = ids3.long()[ids2.long()] * (naggr2 + 1) + ids2.long()
    pm2 = torch.argsort(key2, stable=True).cpu().numpy()
    m5 = m[pm2][:, pm2].tocsr()
    E = DeviceCSR.from_tensors(
        A1.nrows, A1.ncols,
        torch.tensor(m5.indptr, dtype=torch.int32, device="cuda"),
        torch.tensor(m5.indices, dtype=torch.int32, device="cuda"),
        torch.tensor(m5.data, dtype=torch.float64, device="cuda"))
    E.build_sell()
    bench_spmv(E, "agg-grouped (2 levels)")

    # also: pure random order = the locality floor
    rp = torch.randperm(naggr).cpu().numpy()
    m3 = m[rp][:, rp].tocsr()
    m3.sort_indices()
    C = DeviceCSR.from_tensors(
        A1.nrows, A1.ncols,
        torch.tensor(m3.indptr, dtype=torch.int32, device="cuda"),
        torch.tensor(m3.indices, dtype=torch.int32, device="cuda"),
        torch.tensor(m3.data, dtype=torch.float64, device="cuda"))
    C.build_sell()
    bench_spmv(C, "random (floor)")


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 512)
