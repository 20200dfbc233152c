"""Torch-free GPU C API benchmark: build a 3D Poisson system with scipy,
then setup+solve entirely through libamghip.so via ctypes — torch is never
imported.  Times amgcl_amd_gpu_solver_create (upload + hierarchy) and
_solve separately, for both precond.setup=device and =host."""
import ctypes
import os
import sys
import time

import numpy as np
import scipy.sparse as sp

assert "torch" not in sys.modules
ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def poisson3d_scipy(m):
    T = sp.diags([-1.0, 2.0, -1.0], [-1, 0, 1], shape=(m, m), format="csr")
    I = sp.identity(m, format="csr")
    A = (sp.kron(sp.kron(T, I), I) + sp.kron(sp.kron(I, T), I)
         + sp.kron(sp.kron(I, I), T)).tocsr()
    A.sort_indices()
    return A


def main(m=256):
    lib = ctypes.CDLL(os.path.join(ROOT, "amgcl_amd", "_hip", "libamghip.so"))
    lib.amgcl_amd_gpu_solver_create.restype = ctypes.c_void_p
    lib.amgcl_amd_gpu_solver_create.argtypes = (
        [ctypes.c_int] + [ctypes.c_void_p] * 3 + [ctypes.c_char_p])
    lib.amgcl_amd_gpu_solver_solve.restype = ctypes.c_int
    lib.amgcl_amd_gpu_solver_solve.argtypes = (
        [ctypes.c_void_p] * 3
        + [ctypes.POINTER(ctypes.c_int), ctypes.POINTER(ctypes.c_double)])
    lib.amgcl_amd_gpu_solver_destroy.argtypes = [ctypes.c_void_p]

    t0 = time.perf_counter()
    A = poisson3d_scipy(m)
    n = A.shape[0]
    ptr = A.indptr.astype(np.int32)
    col = A.indices.astype(np.int32)
    val = A.data
    print(f"scipy gen {m}^3: {time.perf_counter()-t0:.1f} s, nnz={A.nnz}")
    rng = np.random.default_rng(42)
    b = rng.standard_normal(n)
    x = np.zeros(n)
    x[:] = 1.0  # pre-fault the pages: a cold 134 MB pageable H2D otherwise
    # charges ~100 ms of page faults + staging to whichever mode runs first

    only = os.environ.get("CAPI_BENCH_MODE")
    for mode in ((only,) if only else ("device", "host")):
        cfg = (f"solver.type=cg;solver.tol=1e-6;precond.coarse_enough=1000;"
               f"precond.setup={mode}").encode()
        for rep in range(2):  # first = cold allocator, second = warm
            t0 = time.perf_counter()
            h = lib.amgcl_amd_gpu_solver_create(
                n, ptr.ctypes.data, col.ctypes.data, val.ctypes.data, cfg)
            t1 = time.perf_counter()
            assert h, "create failed"
            it = ctypes.c_int(0)
            res = ctypes.c_double(0.0)
            solves = []
            for s in range(3):  # production pattern: many solves per handle
                x[:] = 0.0
                ts = time.perf_counter()
                rc = lib.amgcl_amd_gpu_solver_solve(
                    h, b.ctypes.data, x.ctypes.data, ctypes.byref(it),
                    ctypes.byref(res))
                solves.append(time.perf_counter() - ts)
                assert rc == 0 and res.value < 1e-6, (rc, res.value)
            lib.amgcl_amd_gpu_solver_destroy(h)
            sv = "/".join(f"{s:.3f}" for s in solves)
            print(f"setup={mode:6s} rep={rep} create={t1-t0:7.3f} s "
                  f"solve={sv} s iters={it.value} resid={res.value:.2e}")
    assert "torch" not in sys.modules
    print("TORCHFREE_BENCH_OK")


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 256)
