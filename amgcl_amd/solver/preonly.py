"""Single preconditioner application (for solver nesting).

Parity: amgcl/solver/preonly.hpp:54.
"""
from .base import SolverBase


class PreOnly(SolverBase):
    @staticmethod
    def defaults():
        return {}

    def __init__(self, n, prm=None, backend=None, inner_product=None):
        self._init_common(n, prm, backend, inner_product)
        self.r = self.backend.vector(n)

    def solve(self, A, P, rhs, x):
        P.apply(rhs, x)
        self.backend.residual(rhs, A, x, self.r)
        nr = self.norm(rhs)
        return 1, self.norm(self.r) / nr if nr else 0.0
