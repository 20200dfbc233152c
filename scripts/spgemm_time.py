import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import amgcl_amd as am
from amgcl_amd.backend import make_backend
from amgcl_amd.backend.hip_setup import poisson3d_device, spgemm

hip = make_backend("hip")
A = poisson3d_device(int(sys.argv[1]) if len(sys.argv) > 1 else 512)
s = am.make_solver(A, {"solver": {"type": "cg", "tol": 1e-6}}, backend=hip)
amg = s.P

def t(fn):
    torch.cuda.synchronize(); t0 = time.perf_counter()
    r = fn(); torch.cuda.synchronize()
    return r, (time.perf_counter() - t0) * 1e3

for li in range(min(2, len(amg.levels) - 1)):
    L = amg.levels[li]
    AP, dt1 = t(lambda: spgemm(L.A, L.P, sort=False))
    _, dt1b = t(lambda: spgemm(L.A, L.P, sort=False))
    Ac, dt2 = t(lambda: spgemm(L.R, AP))
    _, dt2b = t(lambda: spgemm(L.R, AP))
    print(f"L{li}: A*P {dt1:7.1f} ms (warm {dt1b:6.1f})   R*(AP) {dt2:7.1f} ms (warm {dt2b:6.1f})"
          f"   nnz(AP)={AP.nnz} nnz(Ac)={Ac.nnz}", flush=True)
