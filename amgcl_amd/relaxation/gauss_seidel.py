"""Gauss-Seidel smoother (CPU backend only, like the reference).

Parity: amgcl/relaxation/gauss_seidel.hpp:58 — forward sweep in apply_pre,
backward sweep in apply_post. The reference restricts GS to the builtin
backend via relaxation_is_supported; the HIP backend likewise rejects it
(a multicolor variant is the planned GPU path).
"""
from .. import _core
from ..params import merge_params


class GaussSeidel:
    gpu_supported = False

    @staticmethod
    def defaults():
        return {"serial": True}

    def __init__(self, A, prm, backend):
        merge_params(self.defaults(), prm)
        if backend.name != "cpu":
            raise ValueError("gauss_seidel is CPU-only (reference parity)")
        self.A_host = A
        self.backend = backend

    def apply_pre(self, A, rhs, x, tmp):
        _core.gauss_seidel(self.A_host.nrows, self.A_host.ptr, self.A_host.col,
                           self.A_host.val, rhs, x, True)

    def apply_post(self, A, rhs, x, tmp):
        _core.gauss_seidel(self.A_host.nrows, self.A_host.ptr, self.A_host.col,
                           self.A_host.val, rhs, x, False)

    def apply(self, A, rhs, x, tmp=None):
        self.backend.clear(x)
        self.apply_pre(A, rhs, x, tmp)
        self.apply_post(A, rhs, x, tmp)
