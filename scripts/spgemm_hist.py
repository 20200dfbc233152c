"""Histogram of upper-bound and exact output row lengths for the two
level-0 Galerkin products (decides SpGEMM tier engineering)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

import amgcl_amd as am
from amgcl_amd.backend import make_backend
from amgcl_amd.backend.hip_setup import poisson3d_device

hip = make_backend("hip")
A = poisson3d_device(int(sys.argv[1]) if len(sys.argv) > 1 else 384)
s = am.make_solver(A, {"solver": {"type": "cg", "tol": 1e-6}}, backend=hip)
amg = s.P

def hist(name, lens):
    lens = lens.to(torch.int64)
    bins = torch.tensor([0, 16, 32, 48, 64, 96, 128, 192, 256, 512, 1 << 30],
                        device=lens.device)
    h = torch.bucketize(lens, bins[1:], right=False)
    cnt = torch.bincount(h, minlength=len(bins) - 1).cpu()
    tot = int(lens.sum())
    print(f"{name}: n={lens.numel()} mean={tot/max(lens.numel(),1):.1f}")
    labels = ["<=16", "17-32", "33-48", "49-64", "65-96", "97-128", "129-192",
              "193-256", "257-512", ">512"]
    for lab, c in zip(labels, cnt.tolist()):
        if c:
            print(f"   {lab:>8}: {c:>10}  ({100*c/lens.numel():5.1f}%)")

for li in range(min(2, len(amg.levels) - 1)):
    L = amg.levels[li]
    Ad, P, R = L.A, L.P, L.R
    lenP = (P.ptr[1:] - P.ptr[:-1]).to(torch.float64)
    # ub per A-row: sum of len(P_col) over the row = A_pattern @ lenP
    ones = torch.ones_like(Ad.val)
    ub = torch.zeros(Ad.nrows, dtype=torch.float64, device=lenP.device)
    from amgcl_amd.backend.hip import DeviceCSR
    pat = DeviceCSR.from_tensors(Ad.nrows, Ad.ncols, Ad.ptr, Ad.col, ones)
    hip.spmv(1.0, pat, lenP, 0.0, ub)
    hist(f"L{li} A*P ub", ub)
    # exact lens of AP and of Ac
    from amgcl_amd.backend import hip_setup
    AP = hip_setup.spgemm(Ad, P, sort=False)
    hist(f"L{li} A*P exact", AP.ptr[1:] - AP.ptr[:-1])
    lenAP = (AP.ptr[1:] - AP.ptr[:-1]).to(torch.float64)
    ubR = torch.zeros(R.nrows, dtype=torch.float64, device=lenP.device)
    patR = DeviceCSR.from_tensors(R.nrows, R.ncols, R.ptr, R.col,
                                  torch.ones_like(R.val))
    hip.spmv(1.0, patR, lenAP, 0.0, ubR)
    hist(f"L{li} R*(AP) ub", ubR)
    Ac = hip_setup.spgemm(R, AP, sort=False)
    hist(f"L{li} R*(AP) exact", Ac.ptr[1:] - Ac.ptr[:-1])
