"""make_solver: bundle a preconditioner and a Krylov solver.

Parity: amgcl/make_solver.hpp:45-231 — constructed from the host matrix and a
nested params dict {"precond": {...}, "solver": {...}}; callable as
solve(rhs, x) -> (iters, relative residual); nestable; allows solving with a
matrix different from the setup matrix (lagged preconditioner).
"""
import numpy as np

from .matrix import CSR
from .params import merge_params
from .precond import make_preconditioner
from .solver import make_solver_component


class MakeSolver:
    @staticmethod
    def defaults():
        return {
            "precond": {"class": "amg"},
            "solver": {"type": "bicgstab"},
        }

    def __init__(self, A, prm=None, backend=None, inner_product=None):
        if backend is None:
            from .backend import make_backend

            backend = make_backend("cpu")
        self.backend = backend
        # precond/solver subtrees have open-ended keys; validate only the top
        prm = dict(prm or {})
        for key in prm:
            if key not in ("precond", "solver"):
                raise ValueError(f"unknown parameter '{key}'")
        nr = getattr(A, "nrows", None)
        nc = getattr(A, "ncols", None)
        if nr is not None and nc is not None and nr != nc:
            raise ValueError(f"system matrix must be square, got {nr}x{nc}")
        self.A_host = A
        self.P = make_preconditioner(A, prm.get("precond"), backend)
        self.S = make_solver_component(
            A.nrows, prm.get("solver"), backend, inner_product
        )
        # native C++ solve driver (HIP + AMG + diagonal smoothers + cg/bicgstab)
        self._native = None
        if inner_product is None and getattr(backend, "name", "") == "hip":
            from .backend.native import try_native

            self._native = try_native(self)

    def __call__(self, rhs, x=None, A=None):
        """Solve A x = rhs. Returns (x, iters, relative residual)."""
        b = self.backend
        if len(rhs) != self.A_host.nrows:
            raise ValueError(f"rhs has {len(rhs)} entries, system has "
                             f"{self.A_host.nrows} rows")
        rhs_dev = rhs if not isinstance(rhs, np.ndarray) or b.name == "cpu" else b.from_host(rhs)
        if x is None:
            x_dev = b.vector(self.A_host.nrows)
        else:
            x_dev = x if not isinstance(x, np.ndarray) or b.name == "cpu" else b.from_host(x)
        if self._native is not None and A is None:
            iters, resid = self._native.solve(rhs_dev, x_dev)
        else:
            iters, resid = self.S(self.P, rhs_dev, x_dev, A=A)
        return x_dev, iters, resid

    def solve(self, rhs, x=None, A=None):
        return self(rhs, x, A)

    def rebuild(self, A_new):
        """Rebuild the preconditioner for a matrix with changed coefficients
        (reuses the transfer operators; parity: amgcl/amg.hpp rebuild) and
        refresh the native driver, which holds raw pointers into the old
        level tensors."""
        self.P.rebuild(A_new)
        self.A_host = A_new
        if self._native is not None:
            from .backend.native import try_native

            self._native = try_native(self)

    def system_matrix(self):
        return self.P.system_matrix()

    def __str__(self):
        s = [f"Solver: {type(self.S).__name__}  (n={self.A_host.nrows})"]
        if hasattr(self.P, "levels"):
            s.append(str(self.P))
        return "\n".join(s)


def make_solver(A, prm=None, backend="cpu", **backend_kwargs):
    """Convenience front door.

    A may be a CSR, a scipy sparse matrix, or a (ptr, col, val) tuple
    (the crs_tuple adapter, cf. amgcl/adapter/crs_tuple.hpp:83).
    """
    if type(A).__name__ == "DeviceCSR":
        pass  # device-resident input: setup runs on the GPU
    elif not isinstance(A, CSR):
        if isinstance(A, tuple) and len(A) in (3, 4):
            if len(A) == 4:
                n, ptr, col, val = A
            else:
                ptr, col, val = A
                n = len(ptr) - 1
            A = CSR(n, n, ptr, col, val)
        else:
            import numpy as _np

            if isinstance(A, _np.ndarray):
                import scipy.sparse as _sp

                A = CSR.from_scipy(_sp.csr_matrix(_np.atleast_2d(A)))
            else:
                A = CSR.from_scipy(A)
    if isinstance(backend, str):
        from .backend import make_backend

        if (backend in ("cpu", "hip") and isinstance(A, CSR) and A.is_complex
                and "dtype" not in backend_kwargs):
            # native complex solve on BOTH backends (parity:
            # amgcl/value_type/complex.hpp — the reference instantiates the
            # same templates over std::complex; HIP runs hand-written
            # double2 kernels).  adapter.complex_to_real remains as the
            # 2x2-real alternative route.
            if backend == "hip":
                import torch

                backend_kwargs["dtype"] = torch.complex128
            else:
                backend_kwargs["dtype"] = np.complex128
        backend = make_backend(backend, **backend_kwargs)
    return MakeSolver(A, prm, backend)
