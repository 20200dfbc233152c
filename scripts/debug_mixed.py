import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
import amgcl_amd as am
from amgcl_amd.backend import make_backend
from amgcl_amd.backend.hip import DeviceCSR

hip = make_backend("hip")
A, b = am.poisson3d(16, rhs="random")

# 1) f32 kernels vs f64 reference
Ad64 = DeviceCSR(A, hip.device)
Ad32 = DeviceCSR.from_tensors(Ad64.nrows, Ad64.ncols, Ad64.ptr, Ad64.col,
                              Ad64.val.to(torch.float32), Ad64.subw)
x = torch.rand(A.nrows, dtype=torch.float64, device=hip.device)
y64 = hip.vector(A.nrows)
hip.spmv(1.0, Ad64, x, 0.0, y64)
x32 = x.to(torch.float32); y32 = hip.vector(A.nrows, torch.float32)
hip.spmv(1.0, Ad32, x32, 0.0, y32)
print("spmv f32 vs f64 max err:", (y32.double()-y64).abs().max().item())
m = torch.rand(A.nrows, dtype=torch.float32, device=hip.device)+0.5
t32 = hip.vector(A.nrows, torch.float32)
b32 = torch.rand(A.nrows, dtype=torch.float32, device=hip.device)
xx = x32.clone()
hip.relax_diag(Ad32, m, b32, xx, t32)
ref = xx.double()  # just check finite
print("relax_diag f32 finite:", torch.isfinite(t32).all().item(), torch.isfinite(xx).all().item())
r32 = hip.vector(A.nrows, torch.float32)
hip.residual(b32, Ad32, x32, r32)
print("residual f32 max err:", (r32.double()-(b32.double()- (Ad64 and 0) - 0)).abs().max().item() if False else "skip")
hip.axpby(1.5, b32, -0.5, r32); print("axpby finite:", torch.isfinite(r32).all().item())
hip.clear(r32); print("fill f32 ok:", (r32==0).all().item())
z64 = hip.vector(A.nrows); hip.cast(x32, z64); print("cast s2d err:", (z64-x32.double()).abs().max().item())
z32 = hip.vector(A.nrows, torch.float32); hip.cast(x, z32); print("cast d2s err:", (z32.double()-x).abs().max().item())
print("dot f32:", hip.dot(x32, x32), "vs", float(x32.double().dot(x32.double())))

# 2) mixed AMG apply, step by step
prm = {"precond": {"class": "amg", "precision": "mixed", "coarse_enough": 500},
       "solver": {"type": "cg", "tol": 1e-8, "maxiter": 50}}
s = am.make_solver(A, prm, backend=hip)
P = s.P
rd = hip.from_host(b)
sd = hip.vector(A.nrows)
P.apply(rd, sd)
print("apply out finite:", torch.isfinite(sd).all().item(), "norm", sd.norm().item())
print("r32 norm", P._r32.norm().item(), "x32 norm", P._x32.norm().item())
for i, lvl in enumerate(P.levels):
    print(f"L{i} A dtype {lvl.A.val.dtype} t {None if lvl.t is None else lvl.t.dtype}",
          "M", None if lvl.relax is None else lvl.relax.M.dtype,
          "finite t", None if lvl.t is None else torch.isfinite(lvl.t).all().item())
x1, it, res = s(b)
print("mixed solve:", it, res)
