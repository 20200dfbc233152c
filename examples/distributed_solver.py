#!/usr/bin/env python3
"""Distributed solve with subdomain deflation (reference analogue:
examples/mpi/mpi_solver.cpp, runtime_sdd.cpp). Run:

  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 examples/distributed_solver.py [n]

Uses RCCL on GPUs (one rank per GPU), gloo on CPU-only hosts.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import os
import sys

import numpy as np
import torch
import torch.distributed as dist

import amgcl_amd as am
from amgcl_amd.parallel import make_dist_solver


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 32
    if "RANK" not in os.environ:
        print("launch under torch.distributed, e.g.:\n"
              "  torchrun --nnodes=1 --nproc-per-node 2 --master-addr 127.0.0.1 "
              "examples/distributed_solver.py")
        return
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if torch.cuda.is_available():
        torch.cuda.set_device(rank % torch.cuda.device_count())
        dist.init_process_group("nccl")
        backend = "hip"
    else:
        dist.init_process_group("gloo")
        backend = "cpu"

    strip, b, row_beg, row_end = am.poisson3d_strip(n, rank, world, rhs="ones")
    idx = np.arange(row_beg, row_end)
    coords = np.stack([idx % n, (idx // n) % n, idx // (n * n)], axis=1)
    solve = make_dist_solver(
        strip,
        {"precond": {"class": "amg"},
         "solver": {"type": "cg", "tol": 1e-8, "maxiter": 300},
         "deflation": {"type": "linear", "coords_raw": coords}},
        backend=backend,
    )
    x, iters, resid = solve(b)
    if rank == 0:
        print(f"world={world}  iters: {iters}  resid: {resid:.3e}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
