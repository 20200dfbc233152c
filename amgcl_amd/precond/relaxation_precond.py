"""Use any smoother as a single-level preconditioner.

Parity: amgcl/relaxation/as_preconditioner.hpp:43.
"""
from ..relaxation import make_relaxation_factory


class RelaxationPrecond:
    @staticmethod
    def defaults():
        return {"type": "spai0"}

    def __init__(self, A, prm=None, backend=None):
        if backend is None:
            from ..backend import make_backend

            backend = make_backend("cpu")
        self.backend = backend
        factory = make_relaxation_factory(prm)
        self.relax = factory(A, backend)
        self._A = backend.matrix(A)
        self._tmp = backend.vector(A.nrows)

    def system_matrix(self):
        return self._A

    def apply(self, rhs, x):
        self.relax.apply(self._A, rhs, x, self._tmp)
