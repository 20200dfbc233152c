"""Dump the device-built AMG level matrices to raw binaries for the
torch-free PMC harness (same gpurun call consumes them)."""
import os
import struct
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))

import numpy as np

import amgcl_amd as am
from amgcl_amd.backend import make_backend
from amgcl_amd.backend.hip_setup import download, poisson3d_device


def main(n, outdir):
    hip = make_backend("hip")
    A = poisson3d_device(n)
    solve = am.make_solver(A, {"solver": {"type": "cg", "tol": 1e-6}}, backend=hip)
    os.makedirs(outdir, exist_ok=True)
    for k, lvl in enumerate(solve.P.levels[:3]):
        m = lvl.A
        h = download(m) if not hasattr(m, "to_scipy") else m
        ptr = np.asarray(h.ptr, dtype=np.int32)
        col = np.asarray(h.col, dtype=np.int32)
        val = np.asarray(h.val, dtype=np.float64)
        with open(os.path.join(outdir, f"lv{k}.bin"), "wb") as f:
            f.write(struct.pack("<qq", h.nrows, len(col)))
            f.write(ptr.tobytes())
            f.write(col.tobytes())
            f.write(val.tobytes())
        print(f"lv{k}: n={h.nrows} nnz={len(col)}")


if __name__ == "__main__":
    main(int(sys.argv[1]), sys.argv[2])
