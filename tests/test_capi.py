"""C API tests: the standalone C-callable library (no Python dependency
inside) driven through ctypes.

Parity: the reference ships lib/ as a C shared library over the host
backend with 1-based (Fortran) entry points; these tests exercise the same
surface on amgcl_amd's plain-C++/OpenMP shim.
"""
import ctypes
import os

import numpy as np
import pytest

import amgcl_amd as am


@pytest.fixture(scope="module")
def capi():
    from amgcl_amd.build import build_capi_lib

    path = build_capi_lib(verbose=False)
    lib = ctypes.CDLL(path)
    h = ctypes.c_void_p
    lib.amgcl_amd_params_create.restype = h
    lib.amgcl_amd_params_seti.argtypes = [h, ctypes.c_char_p, ctypes.c_int]
    lib.amgcl_amd_params_setf.argtypes = [h, ctypes.c_char_p, ctypes.c_double]
    lib.amgcl_amd_params_sets.argtypes = [h, ctypes.c_char_p, ctypes.c_char_p]
    lib.amgcl_amd_params_destroy.argtypes = [h]
    for fn in ("amgcl_amd_solver_create", "amgcl_amd_solver_create_f",
               "amgcl_amd_precond_create", "amgcl_amd_precond_create_f"):
        getattr(lib, fn).restype = h
        getattr(lib, fn).argtypes = [ctypes.c_int, ctypes.c_void_p,
                                     ctypes.c_void_p, ctypes.c_void_p, h]
    lib.amgcl_amd_solver_solve.restype = ctypes.c_int
    lib.amgcl_amd_solver_solve.argtypes = [h, ctypes.c_void_p, ctypes.c_void_p,
                                           ctypes.POINTER(ctypes.c_int),
                                           ctypes.POINTER(ctypes.c_double)]
    lib.amgcl_amd_solver_solve_mtx.restype = ctypes.c_int
    lib.amgcl_amd_solver_solve_mtx.argtypes = [h] + [ctypes.c_void_p] * 5 + [
        ctypes.POINTER(ctypes.c_int), ctypes.POINTER(ctypes.c_double)]
    lib.amgcl_amd_precond_apply.argtypes = [h, ctypes.c_void_p, ctypes.c_void_p]
    lib.amgcl_amd_precond_report.restype = ctypes.c_int
    lib.amgcl_amd_precond_report.argtypes = [h, ctypes.c_char_p, ctypes.c_int]
    lib.amgcl_amd_solver_report.restype = ctypes.c_int
    lib.amgcl_amd_solver_report.argtypes = [h, ctypes.c_char_p, ctypes.c_int]
    lib.amgcl_amd_solver_destroy.argtypes = [h]
    lib.amgcl_amd_precond_destroy.argtypes = [h]
    return lib


def _arrays(A):
    ptr = np.ascontiguousarray(A.ptr, dtype=np.int32)
    col = np.ascontiguousarray(A.col, dtype=np.int32)
    val = np.ascontiguousarray(A.val, dtype=np.float64)
    return ptr, col, val


def _ptr(a):
    return a.ctypes.data_as(ctypes.c_void_p)


@pytest.mark.parametrize("solver", ["cg", "bicgstab"])
def test_capi_solve(capi, solver):
    A, b = am.poisson3d(20, rhs="random")
    ptr, col, val = _arrays(A)
    prm = capi.amgcl_amd_params_create()
    capi.amgcl_amd_params_sets(prm, b"solver.type", solver.encode())
    capi.amgcl_amd_params_setf(prm, b"solver.tol", 1e-8)
    capi.amgcl_amd_params_seti(prm, b"precond.coarse_enough", 500)
    s = capi.amgcl_amd_solver_create(A.nrows, _ptr(ptr), _ptr(col), _ptr(val), prm)
    capi.amgcl_amd_params_destroy(prm)
    x = np.zeros(A.nrows)
    it = ctypes.c_int(0)
    res = ctypes.c_double(0.0)
    rc = capi.amgcl_amd_solver_solve(s, _ptr(b), _ptr(x), ctypes.byref(it),
                                     ctypes.byref(res))
    assert rc == 0
    assert res.value < 1e-8
    assert 0 < it.value < 60
    assert np.linalg.norm(b - A @ x) / np.linalg.norm(b) < 1e-7

    # lagged-preconditioner variant: same hierarchy, scaled matrix
    val2 = np.ascontiguousarray(2.0 * val)
    x2 = np.zeros(A.nrows)
    rc = capi.amgcl_amd_solver_solve_mtx(s, _ptr(ptr), _ptr(col), _ptr(val2),
                                         _ptr(b), _ptr(x2), ctypes.byref(it),
                                         ctypes.byref(res))
    assert rc == 0
    np.testing.assert_allclose(x2 * 2.0, x, rtol=1e-5, atol=1e-8)

    buf = ctypes.create_string_buffer(4096)
    n = capi.amgcl_amd_solver_report(s, buf, 4096)
    assert 0 < n < 4096
    assert b"operator complexity" in buf.value
    capi.amgcl_amd_solver_destroy(s)


def test_capi_precond_and_fortran_indexing(capi):
    A, b = am.poisson3d(12, rhs="random")
    ptr, col, val = _arrays(A)
    p0 = capi.amgcl_amd_precond_create(A.nrows, _ptr(ptr), _ptr(col), _ptr(val),
                                       None)
    x = np.zeros(A.nrows)
    capi.amgcl_amd_precond_apply(p0, _ptr(b), _ptr(x))
    # one V-cycle must reduce the residual
    assert np.linalg.norm(b - A @ x) < 0.5 * np.linalg.norm(b)

    # 1-based (Fortran) arrays produce the identical preconditioner
    ptr1 = np.ascontiguousarray(ptr + 1)
    col1 = np.ascontiguousarray(col + 1)
    p1 = capi.amgcl_amd_precond_create_f(A.nrows, _ptr(ptr1), _ptr(col1),
                                         _ptr(val), None)
    x1 = np.zeros(A.nrows)
    capi.amgcl_amd_precond_apply(p1, _ptr(b), _ptr(x1))
    np.testing.assert_allclose(x1, x, rtol=1e-13, atol=1e-15)
    capi.amgcl_amd_precond_destroy(p0)
    capi.amgcl_amd_precond_destroy(p1)


def test_capi_iteration_parity_with_python(capi):
    """The C API hierarchy must land in the same iteration class as the
    Python host engine on the same problem."""
    A, b = am.poisson3d(16, rhs="random")
    s_py = am.make_solver(A, {"precond": {"class": "amg", "coarse_enough": 500},
                              "solver": {"type": "cg", "tol": 1e-8,
                                         "maxiter": 100}})
    _, it_py, _ = s_py(b)

    ptr, col, val = _arrays(A)
    prm = capi.amgcl_amd_params_create()
    capi.amgcl_amd_params_setf(prm, b"solver.tol", 1e-8)
    capi.amgcl_amd_params_seti(prm, b"precond.coarse_enough", 500)
    s = capi.amgcl_amd_solver_create(A.nrows, _ptr(ptr), _ptr(col), _ptr(val), prm)
    capi.amgcl_amd_params_destroy(prm)
    x = np.zeros(A.nrows)
    it = ctypes.c_int(0)
    res = ctypes.c_double(0.0)
    capi.amgcl_amd_solver_solve(s, _ptr(b), _ptr(x), ctypes.byref(it),
                                ctypes.byref(res))
    capi.amgcl_amd_solver_destroy(s)
    assert res.value < 1e-8
    assert abs(it.value - it_py) <= 5


def test_capi_cycle_and_relax_params(capi):
    """Parameter plumbing: W-cycle, extra sweeps, damped-Jacobi relaxation
    through the params handle."""
    A, b = am.poisson3d(14, rhs="random")
    ptr, col, val = _arrays(A)
    prm = capi.amgcl_amd_params_create()
    capi.amgcl_amd_params_seti(prm, b"precond.npre", 2)
    capi.amgcl_amd_params_seti(prm, b"precond.npost", 2)
    capi.amgcl_amd_params_seti(prm, b"precond.ncycle", 2)
    capi.amgcl_amd_params_sets(prm, b"precond.relax.type", b"damped_jacobi")
    capi.amgcl_amd_params_setf(prm, b"precond.relax.damping", 0.72)
    capi.amgcl_amd_params_seti(prm, b"precond.coarse_enough", 300)
    capi.amgcl_amd_params_setf(prm, b"solver.tol", 1e-8)
    s = capi.amgcl_amd_solver_create(A.nrows, _ptr(ptr), _ptr(col), _ptr(val), prm)
    capi.amgcl_amd_params_destroy(prm)
    x = np.zeros(A.nrows)
    it = ctypes.c_int(0)
    res = ctypes.c_double(0.0)
    rc = capi.amgcl_amd_solver_solve(s, _ptr(b), _ptr(x), ctypes.byref(it),
                                     ctypes.byref(res))
    capi.amgcl_amd_solver_destroy(s)
    assert rc == 0 and res.value < 1e-8
    assert np.linalg.norm(b - A @ x) / np.linalg.norm(b) < 1e-7


def test_fortran_module_compiles_and_solves(tmp_path):
    """Compile the Fortran interface module + the poisson_f example with
    ROCm's amdflang and run a full solve through the 1-based C entry points
    (parity: reference fortran/poisson.f90).  Skips if no Fortran compiler
    is present."""
    import os
    import shutil
    import subprocess

    fc = shutil.which("amdflang") or shutil.which("flang") or shutil.which("gfortran")
    if fc is None:
        pytest.skip("no Fortran compiler in image")
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    mod = os.path.join(root, "amgcl_amd", "csrc", "capi", "amgcl_amd.f90")
    ex = os.path.join(root, "examples", "poisson_f.f90")
    libdir = os.path.join(root, "amgcl_amd", "_capi")
    exe = str(tmp_path / "poisson_f")
    subprocess.check_call(
        [fc, "-O1", "-o", exe, mod, ex, f"-L{libdir}", "-lamgclamd_c",
         f"-Wl,-rpath,{libdir}"],
        cwd=str(tmp_path))
    out = subprocess.check_output([exe], text=True)
    assert "FORTRAN_OK" in out


@pytest.mark.gpu
def test_torch_free_gpu_solve():
    """A full AMG+CG GPU solve with torch ABSENT from the process
    (VERDICT r01 #3 'Done' criterion): a subprocess using only
    ctypes+numpy drives the torch-free GPU C API (capi_gpu.hip ->
    libamghip.so), which builds the hierarchy with the host C++ engine,
    uploads it with raw hipMalloc and solves through the native driver."""
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    prog = r"""
import ctypes, sys
import numpy as np
assert "torch" not in sys.modules
lib = ctypes.CDLL(r"%s/amgcl_amd/_hip/libamghip.so")
lib.amgcl_amd_gpu_solver_create.restype = ctypes.c_void_p
lib.amgcl_amd_gpu_solver_create.argtypes = [ctypes.c_int] + [ctypes.c_void_p]*3 + [ctypes.c_char_p]
lib.amgcl_amd_gpu_solver_solve.restype = ctypes.c_int
lib.amgcl_amd_gpu_solver_solve.argtypes = [ctypes.c_void_p]*3 + [ctypes.POINTER(ctypes.c_int), ctypes.POINTER(ctypes.c_double)]
m = 32
n = m**3
import itertools
ptr = [0]; col = []; val = []
for k in range(m):
    for j in range(m):
        for i in range(m):
            r = (k*m + j)*m + i
            for (dk,dj,di,v) in ((-1,0,0,-1.),(0,-1,0,-1.),(0,0,-1,-1.),(0,0,0,6.),(0,0,1,-1.),(0,1,0,-1.),(1,0,0,-1.)):
                kk,jj,ii = k+dk, j+dj, i+di
                if 0 <= kk < m and 0 <= jj < m and 0 <= ii < m:
                    col.append((kk*m + jj)*m + ii); val.append(v)
            ptr.append(len(col))
ptr = np.asarray(ptr, dtype=np.int32); col = np.asarray(col, dtype=np.int32)
val = np.asarray(val, dtype=np.float64)
b = np.ones(n); x = np.zeros(n)
h = lib.amgcl_amd_gpu_solver_create(n, ptr.ctypes.data, col.ctypes.data, val.ctypes.data,
                                    b"solver.type=cg;solver.tol=1e-8;precond.coarse_enough=1000")
assert h, "create failed"
it = ctypes.c_int(0); res = ctypes.c_double(0.0)
rc = lib.amgcl_amd_gpu_solver_solve(h, b.ctypes.data, x.ctypes.data,
                                    ctypes.byref(it), ctypes.byref(res))
assert rc == 0, rc
assert res.value < 1e-8 and it.value < 40, (it.value, res.value)
assert "torch" not in sys.modules
print("TORCHFREE_OK", it.value, res.value)
""" % root
    out = subprocess.check_output([sys.executable, "-c", prog], text=True,
                                  stderr=subprocess.STDOUT, timeout=300)
    assert "TORCHFREE_OK" in out


@pytest.mark.gpu
def test_gpu_capi_from_c(tmp_path):
    """Compile examples/gpu_capi.c with hipcc and run it — a complete GPU
    solve from plain C with no Python/torch in the process."""
    import shutil
    import subprocess

    hipcc = shutil.which("hipcc") or "/opt/rocm/bin/hipcc"
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    libdir = os.path.join(root, "amgcl_amd", "_hip")
    exe = str(tmp_path / "gpu_capi")
    subprocess.check_call(
        [hipcc, os.path.join(root, "examples", "gpu_capi.c"), "-o", exe,
         f"-L{libdir}", "-lamghip", f"-Wl,-rpath,{libdir}"])
    out = subprocess.check_output([exe], text=True, timeout=240)
    assert "rc=0" in out


@pytest.mark.gpu
def test_torch_free_gpu_device_setup(tmp_path):
    """Torch-free DEVICE-resident setup through the GPU C API
    (precond.setup=device): the SA hierarchy — strong connections, MIS
    aggregation, smoothed P, R=P^T, Galerkin, SELL images — is built by the
    setup.hip kernels orchestrated from C++ (capi_gpu.hip
    build_device_levels), with the host engine finishing only the tail
    below precond.device_handoff.  n=80^3 (512k rows) also exercises the
    C-side SELL-64 image build of the fine level."""
    import subprocess
    import sys

    import amgcl_amd as am

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    A, _ = am.poisson3d(80)
    import numpy as np

    npz = str(tmp_path / "a80.npz")
    np.savez(npz, ptr=np.asarray(A.ptr, dtype=np.int32),
             col=np.asarray(A.col, dtype=np.int32),
             val=np.asarray(A.val, dtype=np.float64))
    prog = r"""
import ctypes, sys
import numpy as np
assert "torch" not in sys.modules
d = np.load(r"%s")
ptr, col, val = d["ptr"], d["col"], d["val"]
n = len(ptr) - 1
lib = ctypes.CDLL(r"%s/amgcl_amd/_hip/libamghip.so")
lib.amgcl_amd_gpu_solver_create.restype = ctypes.c_void_p
lib.amgcl_amd_gpu_solver_create.argtypes = [ctypes.c_int] + [ctypes.c_void_p]*3 + [ctypes.c_char_p]
lib.amgcl_amd_gpu_solver_solve.restype = ctypes.c_int
lib.amgcl_amd_gpu_solver_solve.argtypes = [ctypes.c_void_p]*3 + [ctypes.POINTER(ctypes.c_int), ctypes.POINTER(ctypes.c_double)]
b = np.ones(n); x = np.zeros(n)
h = lib.amgcl_amd_gpu_solver_create(n, ptr.ctypes.data, col.ctypes.data, val.ctypes.data,
    b"solver.type=cg;solver.tol=1e-8;precond.coarse_enough=1000;precond.setup=device")
assert h, "create failed"
it = ctypes.c_int(0); res = ctypes.c_double(0.0)
rc = lib.amgcl_amd_gpu_solver_solve(h, b.ctypes.data, x.ctypes.data,
                                    ctypes.byref(it), ctypes.byref(res))
assert rc == 0, rc
assert res.value < 1e-8 and it.value < 40, (it.value, res.value)
# host-setup reference on the same system: same convergence class
h2 = lib.amgcl_amd_gpu_solver_create(n, ptr.ctypes.data, col.ctypes.data, val.ctypes.data,
    b"solver.type=cg;solver.tol=1e-8;precond.coarse_enough=1000;precond.setup=host")
x2 = np.zeros(n); it2 = ctypes.c_int(0); res2 = ctypes.c_double(0.0)
rc = lib.amgcl_amd_gpu_solver_solve(h2, b.ctypes.data, x2.ctypes.data,
                                    ctypes.byref(it2), ctypes.byref(res2))
assert rc == 0, rc
assert res2.value < 1e-8
assert abs(it.value - it2.value) <= 3, (it.value, it2.value)
err = np.abs(x - x2).max() / np.abs(x2).max()
assert err < 1e-6, err
lib.amgcl_amd_gpu_solver_destroy.argtypes = [ctypes.c_void_p]
lib.amgcl_amd_gpu_solver_destroy(h)
lib.amgcl_amd_gpu_solver_destroy(h2)
assert "torch" not in sys.modules
print("DEVICE_SETUP_OK", it.value, it2.value, res.value)
""" % (npz, root)
    out = subprocess.check_output([sys.executable, "-c", prog], text=True,
                                  stderr=subprocess.STDOUT, timeout=600)
    assert "DEVICE_SETUP_OK" in out


def test_capi_rejects_unknown_solver(capi):
    """The compiled engine carries CG/BiCGStab; asking for anything else
    must fail at create (NULL handle), never silently substitute."""
    A, b = am.poisson3d(8)
    ptr, col, val = _arrays(A)
    prm = capi.amgcl_amd_params_create()
    capi.amgcl_amd_params_sets(prm, b"solver.type", b"gmres")
    s = capi.amgcl_amd_solver_create(A.nrows, _ptr(ptr), _ptr(col), _ptr(val), prm)
    capi.amgcl_amd_params_destroy(prm)
    assert not s
