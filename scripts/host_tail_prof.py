import cProfile, pstats, io, sys, time
sys.path.insert(0, ".")
import torch
import amgcl_amd as am
from amgcl_amd.backend.hip_setup import poisson3d_device
A = poisson3d_device(512)
prm = {"solver": {"type": "cg", "tol": 1e-6, "maxiter": 100}}
s = am.make_solver(A, prm, backend="hip"); del s
torch.cuda.synchronize()
pr = cProfile.Profile()
t0 = time.perf_counter()
pr.enable()
s = am.make_solver(A, prm, backend="hip")
torch.cuda.synchronize()
pr.disable()
print(f"setup wall {time.perf_counter()-t0:.3f}")
out = io.StringIO()
ps = pstats.Stats(pr, stream=out).sort_stats("cumulative")
ps.print_stats(35)
txt = out.getvalue()
print("\n".join(l for l in txt.splitlines() if " 0.0" not in l[:40] or "cumtime" in l)[:4000])
