"""Native solve driver bindings.

Hands the whole AMG-preconditioned CG/BiCGStab solve to the C++ driver
(csrc/hip/driver.hip) — one ctypes call per solve, no Python in the hot
path. Engaged automatically by make_solver for HIP-backend AMG with diagonal
smoothers (SPAI0/damped Jacobi), dense-inverse coarse solve, and cg/bicgstab;
anything else uses the generic Python orchestration.
"""
import ctypes

from ._hiplib import lib as _lib


class LevelDescC(ctypes.Structure):
    _fields_ = [
        ("nrows", ctypes.c_int64),
        ("nnz", ctypes.c_int64),
        ("ptr", ctypes.c_void_p),
        ("col", ctypes.c_void_p),
        ("val", ctypes.c_void_p),
        ("subw", ctypes.c_int),
        ("pnnz", ctypes.c_int64),
        ("pptr", ctypes.c_void_p),
        ("pcol", ctypes.c_void_p),
        ("pval", ctypes.c_void_p),
        ("psubw", ctypes.c_int),
        ("rnnz", ctypes.c_int64),
        ("rptr", ctypes.c_void_p),
        ("rcol", ctypes.c_void_p),
        ("rval", ctypes.c_void_p),
        ("rsubw", ctypes.c_int),
        ("M", ctypes.c_void_p),
        ("f", ctypes.c_void_p),
        ("u", ctypes.c_void_p),
        ("t", ctypes.c_void_p),
        ("nslice", ctypes.c_int64),
        ("soff", ctypes.c_void_p),
        ("scol", ctypes.c_void_p),
        ("sval", ctypes.c_void_p),
        ("pnslice", ctypes.c_int64),
        ("psoff", ctypes.c_void_p),
        ("pscol", ctypes.c_void_p),
        ("psval", ctypes.c_void_p),
        ("rnslice", ctypes.c_int64),
        ("rsoff", ctypes.c_void_p),
        ("rscol", ctypes.c_void_p),
        ("rsval", ctypes.c_void_p),
        ("srows", ctypes.c_void_p),
        ("psrows", ctypes.c_void_p),
        ("rsrows", ctypes.c_void_p),
        ("cheb_degree", ctypes.c_int),
        ("cheb_theta", ctypes.c_double),
        ("cheb_delta", ctypes.c_double),
        ("cheb_sigma1", ctypes.c_double),
        ("cheb_d", ctypes.c_void_p),
        ("bsize", ctypes.c_int),
        ("nbrows", ctypes.c_int64),
        ("bptr", ctypes.c_void_p),
        ("bcol", ctypes.c_void_p),
        ("bval", ctypes.c_void_p),
        ("ilu_iters", ctypes.c_int),
        ("ilu_damping", ctypes.c_double),
        ("ilu_jdamping", ctypes.c_double),
        ("lptr", ctypes.c_void_p),
        ("lcol", ctypes.c_void_p),
        ("lval", ctypes.c_void_p),
        ("uptr", ctypes.c_void_p),
        ("ucol", ctypes.c_void_p),
        ("uval", ctypes.c_void_p),
        ("ilu_dinv", ctypes.c_void_p),
        ("ilu_y", ctypes.c_void_p),
        ("ilu_s", ctypes.c_void_p),
        ("ilu_b", ctypes.c_void_p),
    ]


_driver_bound = False


def _bind():
    global _driver_bound
    L = _lib()
    if not _driver_bound:
        L.amg_driver_create.argtypes = [
            ctypes.POINTER(LevelDescC), ctypes.c_int, ctypes.c_void_p, ctypes.c_int64,
            ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int,
            ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
            ctypes.c_int, ctypes.c_void_p,
        ]
        L.amg_driver_create.restype = ctypes.c_void_p
        L.amg_driver_destroy.argtypes = [ctypes.c_void_p]
        L.amg_driver_destroy.restype = None
        L.amg_driver_cg.argtypes = (
            [ctypes.c_void_p] + [ctypes.c_void_p] * 7
            + [ctypes.c_double, ctypes.c_double, ctypes.c_int,
               ctypes.POINTER(ctypes.c_int64), ctypes.POINTER(ctypes.c_double)]
        )
        L.amg_driver_cg.restype = ctypes.c_int
        L.amg_driver_bicgstab.argtypes = (
            [ctypes.c_void_p] + [ctypes.c_void_p] * 10
            + [ctypes.c_double, ctypes.c_double, ctypes.c_int,
               ctypes.POINTER(ctypes.c_int64), ctypes.POINTER(ctypes.c_double)]
        )
        L.amg_driver_bicgstab.restype = ctypes.c_int
        L.amg_driver_precond.argtypes = [ctypes.c_void_p] * 4
        L.amg_driver_precond.restype = ctypes.c_int
        _driver_bound = True
    return L


def _ptr(t):
    return ctypes.c_void_p(0 if t is None else t.data_ptr())


class NativeDriver:
    """Owns the C driver handle plus references to every device tensor it
    points at (torch would otherwise free them under the caching allocator)."""

    def __init__(self, amg, backend, solver_kind="cg", solver_prm=None):
        import torch

        from ..relaxation.chebyshev import Chebyshev
        from ..relaxation.ilu0 import ILU0
        from ..relaxation.spai0 import DiagonalSmootherBase
        from .hip import DeviceBSR, DeviceCSR, DeviceDenseSolver

        self.backend = backend
        self.solver_kind = solver_kind
        solver_prm = solver_prm or {"tol": 1e-8, "abstol": 0.0, "maxiter": 100}
        self.tol = float(solver_prm["tol"])
        self.abstol = float(solver_prm["abstol"])
        self.maxiter = int(solver_prm["maxiter"])

        levels = amg.levels
        if not all(isinstance(l.A, (DeviceCSR, DeviceBSR)) for l in levels):
            raise TypeError("native driver needs device-resident levels")
        self._mixed = bool(getattr(amg, "_mixed", False))
        def relax_ok(r):
            if isinstance(r, (DiagonalSmootherBase, Chebyshev)):
                return True
            # ILU0 with the iterated-Jacobi solve (graph-capturable); the
            # exact cooperative sptrsv stays on the generic path
            return (isinstance(r, ILU0) and not getattr(r, "_serial", True)
                    and not getattr(r, "_exact", False))

        for l in levels[:-1]:
            if not relax_ok(l.relax):
                raise TypeError(
                    "native driver supports diagonal/Chebyshev/ILU0(jacobi)")
        if amg.coarse_solve is not None and not isinstance(
            amg.coarse_solve, DeviceDenseSolver
        ):
            raise TypeError("native driver needs the dense coarse solver")
        if amg.coarse_solve is None and not relax_ok(levels[-1].relax):
            raise TypeError("native coarsest smoother unsupported")

        self._keep = []  # tensor refs
        descs = (LevelDescC * len(levels))()
        n0 = levels[0].A.nrows
        for i, l in enumerate(levels):
            d = descs[i]
            A = l.A
            if isinstance(A, DeviceBSR):
                d.nrows, d.nnz = A.nrows, int(A.nnz)
                d.bsize, d.nbrows = A.bsize, A.nbrows
                d.bptr, d.bcol, d.bval = _ptr(A.ptr), _ptr(A.col), _ptr(A.val)
            else:
                d.nrows, d.nnz = A.nrows, int(A.nnz)
                d.ptr, d.col, d.val = _ptr(A.ptr), _ptr(A.col), _ptr(A.val)
                d.subw = A.subw or _auto_subw(A)
            if l.P is not None:
                P, R = l.P, l.R
                d.pnnz = int(P.nnz)
                d.pptr, d.pcol, d.pval = _ptr(P.ptr), _ptr(P.col), _ptr(P.val)
                d.psubw = P.subw or _auto_subw(P)
                d.rnnz = int(R.nnz)
                d.rptr, d.rcol, d.rval = _ptr(R.ptr), _ptr(R.col), _ptr(R.val)
                d.rsubw = R.subw or _auto_subw(R)
            if getattr(A, "nslice", 0):
                d.nslice = A.nslice
                d.soff, d.scol, d.sval = _ptr(A.soff), _ptr(A.scol), _ptr(A.sval)
                d.srows = _ptr(A.srows)
                self._keep.extend([A.soff, A.scol, A.sval, A.srows])
            if l.P is not None and getattr(l.P, "nslice", 0):
                d.pnslice = l.P.nslice
                d.psoff, d.pscol, d.psval = (_ptr(l.P.soff), _ptr(l.P.scol),
                                             _ptr(l.P.sval))
                d.psrows = _ptr(l.P.srows)
                self._keep.extend([l.P.soff, l.P.scol, l.P.sval, l.P.srows])
            if l.R is not None and getattr(l.R, "nslice", 0):
                d.rnslice = l.R.nslice
                d.rsoff, d.rscol, d.rsval = (_ptr(l.R.soff), _ptr(l.R.scol),
                                             _ptr(l.R.sval))
                d.rsrows = _ptr(l.R.srows)
                self._keep.extend([l.R.soff, l.R.scol, l.R.sval, l.R.srows])
            relax = l.relax
            if isinstance(relax, ILU0):
                d.ilu_iters = relax.solve_iters
                d.ilu_damping = relax.damping
                d.ilu_jdamping = relax.solve_damping
                d.lptr, d.lcol, d.lval = (_ptr(relax.L.ptr), _ptr(relax.L.col),
                                          _ptr(relax.L.val))
                d.uptr, d.ucol, d.uval = (_ptr(relax.U.ptr), _ptr(relax.U.col),
                                          _ptr(relax.U.val))
                d.ilu_dinv = _ptr(relax.Dinv)
                work = [torch.empty(A.nrows, dtype=torch.float64,
                                    device=backend.device) for _ in range(3)]
                d.ilu_y, d.ilu_s, d.ilu_b = (_ptr(work[0]), _ptr(work[1]),
                                             _ptr(work[2]))
                self._keep.extend([relax.L.ptr, relax.L.col, relax.L.val,
                                   relax.U.ptr, relax.U.col, relax.U.val,
                                   relax.Dinv] + work)
            elif isinstance(relax, Chebyshev):
                d.cheb_degree = relax.degree
                d.cheb_theta = relax.theta
                d.cheb_delta = relax.delta
                d.cheb_sigma1 = relax.sigma1
                cheb_d = torch.empty(A.nrows, dtype=torch.float64,
                                     device=backend.device)
                d.cheb_d = _ptr(cheb_d)
                self._keep.append(cheb_d)
                d.M = _ptr(relax.Dinv)  # optional D^-1 scaling (may be null)
                if relax.Dinv is not None:
                    self._keep.append(relax.Dinv)
            else:
                d.M = _ptr(relax.M if relax is not None else None)
            d.f = _ptr(l.f)
            d.u = _ptr(l.u)
            d.t = _ptr(l.t)
            self._keep.extend([A.ptr, A.col, A.val, l.f, l.u, l.t])
            if l.P is not None:
                self._keep.extend([l.P.ptr, l.P.col, l.P.val, l.R.ptr, l.R.col, l.R.val])
            if relax is not None and not isinstance(relax, (Chebyshev, ILU0)):
                self._keep.append(relax.M)

        inv = amg.coarse_solve.inv if amg.coarse_solve is not None else None
        if inv is not None:
            self._keep.append(inv)
        prm = amg.prm
        L = _bind()
        stream = torch.cuda.current_stream().cuda_stream
        if self._mixed:
            # the Krylov loop keeps iterating with the fp64 fine operator;
            # the LevelDescs above point at the fp32 hierarchy for the cycle
            A64 = amg._A64
            self._keep.extend([A64.ptr, A64.col, A64.val])
            a64 = (int(A64.nnz), _ptr(A64.ptr), _ptr(A64.col), _ptr(A64.val),
                   A64.subw or _auto_subw(A64))
        else:
            a64 = (0, None, None, None, 0)
        self.handle = L.amg_driver_create(
            descs, len(levels), _ptr(inv),
            amg.coarse_solve.n if amg.coarse_solve is not None else 0,
            int(prm["npre"]), int(prm["npost"]), int(prm["ncycle"]),
            int(prm["pre_cycles"]), int(self._mixed), a64[0], a64[1], a64[2],
            a64[3], a64[4], ctypes.c_void_p(stream),
        )
        if not self.handle:
            raise RuntimeError("amg_driver_create failed")
        nwork = 5 if solver_kind == "cg" else 8
        self._work = [torch.empty(n0, dtype=torch.float64, device=backend.device)
                      for _ in range(nwork)]
        self._descs = descs

    def solve(self, rhs, x):
        L = _bind()
        iters = ctypes.c_int64(0)
        resid = ctypes.c_double(0.0)
        w = [_ptr(t) for t in self._work]
        if self.solver_kind == "cg":
            rc = L.amg_driver_cg(self.handle, _ptr(rhs), _ptr(x), *w, self.tol,
                                 self.abstol, self.maxiter,
                                 ctypes.byref(iters), ctypes.byref(resid))
        else:
            rc = L.amg_driver_bicgstab(self.handle, _ptr(rhs), _ptr(x), *w, self.tol,
                                       self.abstol, self.maxiter,
                                       ctypes.byref(iters), ctypes.byref(resid))
        if rc != 0:
            raise RuntimeError(f"native solve failed rc={rc}")
        return int(iters.value), float(resid.value)

    def precond_apply(self, rhs, x):
        """x = M^-1 rhs (one native V-cycle stack application)."""
        L = _bind()
        rc = L.amg_driver_precond(self.handle, _ptr(rhs), _ptr(x), _ptr(self._work[0]))
        if rc != 0:
            raise RuntimeError(f"native precond failed rc={rc}")

    def __del__(self):
        try:
            _bind().amg_driver_destroy(self.handle)
        except Exception:
            pass


def _auto_subw(A):
    m = A.nnz / max(A.nrows, 1)
    for lim, sw in ((4, 2), (10, 4), (24, 8), (128, 16)):
        if m <= lim:
            return sw
    return 32


def try_native(make_solver_obj):
    """Attach a native driver to a MakeSolver when the configuration allows;
    returns None otherwise."""
    from ..precond.amg import AMG
    from ..solver.bicgstab import BiCGStab
    from ..solver.cg import CG

    P, S, backend = make_solver_obj.P, make_solver_obj.S, make_solver_obj.backend
    if getattr(backend, "name", "") != "hip":
        return None
    if "complex" in str(getattr(backend, "dtype", "")):
        return None  # complex solves use the generic per-kernel path
    if not isinstance(P, AMG):
        return None
    if isinstance(S, CG):
        kind = "cg"
    elif isinstance(S, BiCGStab) and S.prm["pside"] == "right" and not S.prm["check_after"]:
        kind = "bicgstab"
    else:
        return None
    if S.prm.get("verbose") or S.prm.get("ns_search"):
        return None
    try:
        return NativeDriver(P, backend, kind, S.prm)
    except TypeError:
        return None
