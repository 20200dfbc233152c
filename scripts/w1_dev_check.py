"""Single-rank RCCL check of the device-strip DistMatrix branch (no ghosts):
device strip generation -> torch split -> local device hierarchy ->
deflated CG, verified against the host operator."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29640")
os.environ["RANK"] = "0"
os.environ["WORLD_SIZE"] = "1"
import numpy as np
import torch
import torch.distributed as dist

torch.cuda.set_device(0)
dist.init_process_group("nccl", rank=0, world_size=1)
import amgcl_amd as am
from amgcl_amd.backend.hip_setup import poisson3d_device_strip
from amgcl_amd.parallel import make_dist_solver

n = 64
strip = poisson3d_device_strip(n, 0, n**3)
rng = np.random.default_rng(42)
bh = rng.standard_normal(n**3)
b = torch.from_numpy(bh).cuda()
idx = np.arange(n**3)
coords = np.stack([idx % n, (idx // n) % n, idx // (n * n)], 1).astype(float)
s = make_dist_solver(strip, {"precond": {"class": "amg"},
                             "solver": {"type": "cg", "tol": 1e-8, "maxiter": 200},
                             "deflation": {"type": "linear", "coords_raw": coords}},
                     backend="hip")
x, it, r = s(b)
A, _ = am.poisson3d(n)
xh = s.backend.to_host(x)
tr = np.linalg.norm(bh - A @ xh) / np.linalg.norm(bh)
print("W1_DEV_OK", it, r, tr)
assert r < 1e-8 and tr < 1e-7
dist.destroy_process_group()
