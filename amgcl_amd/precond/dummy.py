"""Identity preconditioner (parity: amgcl/preconditioner/dummy.hpp:42)."""
from ..params import merge_params


class Dummy:
    @staticmethod
    def defaults():
        return {}

    def __init__(self, A, prm=None, backend=None):
        if backend is None:
            from ..backend import make_backend

            backend = make_backend("cpu")
        merge_params(self.defaults(), prm)
        self.backend = backend
        self._A = backend.matrix(A)

    def system_matrix(self):
        return self._A

    def apply(self, rhs, x):
        self.backend.copy(rhs, x)
