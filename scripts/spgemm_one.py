"""Profile exactly ONE Galerkin product (A0*P0 or R0*AP) in isolation."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import amgcl_amd as am
from amgcl_amd.backend import make_backend
from amgcl_amd.backend.hip_setup import poisson3d_device, aggregates, smoothed_prolongation, transpose, spgemm

which = sys.argv[1] if len(sys.argv) > 1 else "ap"
hip = make_backend("hip")
A = poisson3d_device(512)
naggr, ids, S = aggregates(A, 0.08)
P = smoothed_prolongation(A, S, ids, naggr, 2.0/3.0)
torch.cuda.synchronize()
import time
if which == "ap":
    t0=time.perf_counter(); AP = spgemm(A, P, sort=False); torch.cuda.synchronize()
    print("A*P wall", (time.perf_counter()-t0)*1e3, "ms nnz", AP.nnz)
else:
    AP = spgemm(A, P, sort=False)
    R = transpose(P)
    torch.cuda.synchronize()
    t0=time.perf_counter(); Ac = spgemm(R, AP); torch.cuda.synchronize()
    print("R*AP wall", (time.perf_counter()-t0)*1e3, "ms nnz", Ac.nnz)
