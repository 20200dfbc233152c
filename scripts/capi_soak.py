"""Leak soak for the torch-free GPU C API device setup: repeated
create/solve/destroy at 128^3 while watching free HBM via hipMemGetInfo.
A leak in the pool/blob bookkeeping shows as monotonically shrinking
free memory."""
import ctypes
import os
import sys

import numpy as np
import scipy.sparse as sp

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def main(m=128, cycles=12):
    lib = ctypes.CDLL(os.path.join(ROOT, "amgcl_amd", "_hip", "libamghip.so"))
    hip = ctypes.CDLL("libamdhip64.so")
    hip.hipMemGetInfo.argtypes = [ctypes.POINTER(ctypes.c_size_t),
                                  ctypes.POINTER(ctypes.c_size_t)]
    lib.amgcl_amd_gpu_solver_create.restype = ctypes.c_void_p
    lib.amgcl_amd_gpu_solver_create.argtypes = (
        [ctypes.c_int] + [ctypes.c_void_p] * 3 + [ctypes.c_char_p])
    lib.amgcl_amd_gpu_solver_solve.restype = ctypes.c_int
    lib.amgcl_amd_gpu_solver_solve.argtypes = (
        [ctypes.c_void_p] * 3
        + [ctypes.POINTER(ctypes.c_int), ctypes.POINTER(ctypes.c_double)])
    lib.amgcl_amd_gpu_solver_destroy.argtypes = [ctypes.c_void_p]

    T = sp.diags([-1.0, 2.0, -1.0], [-1, 0, 1], shape=(m, m), format="csr")
    I = sp.identity(m, format="csr")
    A = (sp.kron(sp.kron(T, I), I) + sp.kron(sp.kron(I, T), I)
         + sp.kron(sp.kron(I, I), T)).tocsr()
    A.sort_indices()
    n = A.shape[0]
    ptr = A.indptr.astype(np.int32)
    col = A.indices.astype(np.int32)
    val = A.data
    b = np.ones(n)
    x = np.zeros(n)
    cfg = b"solver.type=cg;solver.tol=1e-6;precond.setup=device"

    frees = []
    for c in range(cycles):
        x[:] = 0.0  # fresh start: otherwise cycle c>0 converges at iter 0
        h = lib.amgcl_amd_gpu_solver_create(
            n, ptr.ctypes.data, col.ctypes.data, val.ctypes.data, cfg)
        assert h, "create failed"
        it = ctypes.c_int(0)
        res = ctypes.c_double(0.0)
        rc = lib.amgcl_amd_gpu_solver_solve(h, b.ctypes.data, x.ctypes.data,
                                            ctypes.byref(it), ctypes.byref(res))
        assert rc == 0 and res.value < 1e-6, (rc, res.value)
        lib.amgcl_amd_gpu_solver_destroy(h)
        f, t = ctypes.c_size_t(0), ctypes.c_size_t(0)
        hip.hipMemGetInfo(ctypes.byref(f), ctypes.byref(t))
        frees.append(f.value)
        print(f"cycle {c}: free {f.value/2**30:.2f} GiB iters={it.value}")
    # allow the pool to retain a working set; fail on monotone decline
    drop = (frees[2] - frees[-1]) / 2**30
    assert drop < 0.5, f"leak: free memory dropped {drop:.2f} GiB after warmup"
    print("SOAK_OK")


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 128)
