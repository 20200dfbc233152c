"""Leak soak: repeated full setup+solve cycles; device memory must plateau
(the native driver owns hipMalloc'd buffers, graphs, events — __del__ must
release them; torch tensors recycle through the caching allocator)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

import amgcl_amd as am
from amgcl_amd.backend import make_backend
from amgcl_amd.backend.hip_setup import poisson3d_device

hip = make_backend("hip")
n = int(sys.argv[1]) if len(sys.argv) > 1 else 96
iters = int(sys.argv[2]) if len(sys.argv) > 2 else 40
peaks = []
for i in range(iters):
    A = poisson3d_device(n)
    g = torch.Generator(device="cuda").manual_seed(i)
    b = torch.randn(n**3, dtype=torch.float64, device="cuda", generator=g)
    s = am.make_solver(A, {"solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}},
                       backend=hip)
    x, it, r = s(b)
    assert r < 1e-8
    del s, A, b, x
    torch.cuda.synchronize()
    alloc = torch.cuda.memory_allocated() / 1e6
    free, total = torch.cuda.mem_get_info()
    peaks.append((alloc, (total - free) / 1e6))
    if i % 10 == 0 or i == iters - 1:
        print(f"cycle {i:3d}: torch_alloc {alloc:9.1f} MB  device_used "
              f"{(total-free)/1e6:9.1f} MB", flush=True)
# plateau check: last-10 device_used growth < 50 MB
grow = peaks[-1][1] - peaks[-11][1]
print(f"device_used growth over last 10 cycles: {grow:.1f} MB")
assert abs(grow) < 50, "device memory is growing -> leak"
print("SOAK_OK")
