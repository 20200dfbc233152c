"""Iterative (Krylov) solvers.

Solver concept (parity: amgcl/solver/cg.hpp:127-218): constructed with the
problem size + params + backend (+ replaceable inner product — the only hook
needed to make a solver distributed, cf. amgcl/solver/detail/
default_inner_product.hpp:61 and amgcl/mpi/inner_product.hpp:44);
call solve(A, P, rhs, x) -> (iters, relative residual).
"""
from .base import SolverBase  # noqa: F401
from .cg import CG
from .bicgstab import BiCGStab
from .gmres import GMRES
from .richardson import Richardson
from .preonly import PreOnly

REGISTRY = {
    "cg": CG,
    "bicgstab": BiCGStab,
    "gmres": GMRES,
    "richardson": Richardson,
    "preonly": PreOnly,
}


def make_solver_component(n, prm=None, backend=None, inner_product=None):
    prm = dict(prm or {})
    kind = prm.pop("type", "bicgstab")
    for name, mod in (("bicgstabl", "bicgstabl"), ("fgmres", "fgmres"),
                      ("lgmres", "lgmres"), ("idrs", "idrs")):
        if kind == name and name not in REGISTRY:
            import importlib

            m = importlib.import_module(f".{mod}", __package__)
            REGISTRY[name] = getattr(m, name.upper() if name != "bicgstabl" else "BiCGStabL")
    if kind not in REGISTRY:
        raise ValueError(f"unknown solver '{kind}'")
    return REGISTRY[kind](n, prm, backend, inner_product)
