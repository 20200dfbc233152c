"""SPAI-0 smoother (the reference's default/benchmark smoother).

Parity: amgcl/relaxation/spai0.hpp:50-116 — diagonal sparse approximate
inverse with weights m_i = a_ii / sum_j a_ij^2; each application is
x += M ∘ (rhs - A x).
"""
from .. import _core
from ..params import merge_params


class DiagonalSmootherBase:
    """Shared apply logic for smoothers of the form x += M ∘ (rhs - A x).

    Uses the backend's fused relax_diag op when available (single pass over
    A on the HIP backend), otherwise residual + vmul.
    """

    def _setup_m(self, m_host, backend):
        self.backend = backend
        import numpy as np

        self.M = backend.from_host(m_host) if isinstance(m_host, np.ndarray) else m_host
        self._fused = hasattr(backend, "relax_diag")

    def _step(self, A, rhs, x, tmp):
        b = self.backend
        if self._fused:
            b.relax_diag(A, self.M, rhs, x, tmp)  # tmp = M∘(rhs-Ax); x += tmp
        else:
            b.residual(rhs, A, x, tmp)
            b.vmul(1.0, self.M, tmp, 1.0, x)

    def apply_pre(self, A, rhs, x, tmp):
        self._step(A, rhs, x, tmp)

    def apply_post(self, A, rhs, x, tmp):
        self._step(A, rhs, x, tmp)

    def apply(self, A, rhs, x, tmp=None):
        # single application to zero initial guess: x = M ∘ rhs
        self.backend.vmul(1.0, self.M, rhs, 0.0, x)


class Spai0(DiagonalSmootherBase):
    gpu_supported = True

    @staticmethod
    def defaults():
        return {}

    def __init__(self, A, prm, backend):
        from ..matrix import CSR

        merge_params(self.defaults(), prm)
        if isinstance(A, CSR):
            if A.is_complex:
                # complex SPAI-0: m_i = conj(a_ii) / sum_j |a_ij|^2
                # (minimizes ||I - M A||_F with diagonal M over C)
                import numpy as np

                row_of = np.repeat(np.arange(A.nrows), np.diff(A.ptr))
                s = np.zeros(A.nrows)
                np.add.at(s, row_of, np.abs(A.val) ** 2)
                m = np.conj(np.asarray(A.diagonal())) / np.where(s == 0, 1, s)
            else:
                m = _core.spai0(A.nrows, A.ptr, A.col, A.val)
        else:  # device-resident level matrix: compute weights on device
            from ..backend import hip_setup

            m = hip_setup.spai0(A)
        self._setup_m(m, backend)


class Spai1:
    """SPAI-1 smoother: approximate inverse with A's sparsity
    (parity: amgcl/relaxation/spai1.hpp:54). Apply: x += M (rhs - A x)."""

    gpu_supported = True

    @staticmethod
    def defaults():
        return {}

    def __init__(self, A, prm, backend):
        from ..matrix import CSR

        merge_params(self.defaults(), prm)
        if not isinstance(A, CSR):
            from ..backend import hip_setup

            A = hip_setup.download(A)
        mp, mc, mv = _core.spai1(A.nrows, A.ptr, A.col, A.val)
        self.backend = backend
        self.M = backend.matrix(CSR(A.nrows, A.ncols, mp, mc, mv))
        self.n = A.nrows

    def _step(self, A, rhs, x, tmp):
        b = self.backend
        b.residual(rhs, A, x, tmp)
        t2 = getattr(self, "_t2", None)
        if t2 is None:
            t2 = self._t2 = b.vector(self.n)
        b.spmv(1.0, self.M, tmp, 0.0, t2)
        b.axpby(1.0, t2, 1.0, x)

    apply_pre = _step
    apply_post = _step

    def apply(self, A, rhs, x, tmp=None):
        self.backend.spmv(1.0, self.M, rhs, 0.0, x)
