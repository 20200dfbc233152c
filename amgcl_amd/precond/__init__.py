"""Preconditioners.

Preconditioner concept (parity: amgcl/amg.hpp:289-306): constructed from the
host system matrix + params + backend; .apply(rhs, x) approximately solves
A x = rhs; .system_matrix() returns the backend matrix the Krylov solver
iterates with.
"""
from .amg import AMG
from .dummy import Dummy
from .relaxation_precond import RelaxationPrecond

REGISTRY = {
    "amg": AMG,
    "relaxation": RelaxationPrecond,
    "dummy": Dummy,
}


def _lazy(kind):
    if kind == "schur_pressure_correction" and kind not in REGISTRY:
        from .schur import SchurPressureCorrection

        REGISTRY[kind] = SchurPressureCorrection
    if kind == "deflation" and kind not in REGISTRY:
        from .deflated import DeflatedPrecond

        REGISTRY[kind] = DeflatedPrecond
    if kind in ("cpr", "cpr_drs") and kind not in REGISTRY:
        from .cpr import CPR

        REGISTRY["cpr"] = CPR

        class CPRDRS(CPR):
            @staticmethod
            def defaults():
                d = CPR.defaults()
                d["drs"] = True
                return d

        REGISTRY["cpr_drs"] = CPRDRS


def make_preconditioner(A, prm=None, backend=None):
    prm = dict(prm or {})
    kind = prm.pop("class", "amg")
    _lazy(kind)
    if kind not in REGISTRY:
        raise ValueError(f"unknown preconditioner class '{kind}'")
    return REGISTRY[kind](A, prm, backend)
