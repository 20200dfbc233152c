"""Damped Jacobi smoother.

Parity: amgcl/relaxation/damped_jacobi.hpp:54 — x += omega * D^-1 (rhs - A x).
"""
import numpy as np

from ..params import merge_params
from .spai0 import DiagonalSmootherBase


class DampedJacobi(DiagonalSmootherBase):
    gpu_supported = True

    @staticmethod
    def defaults():
        return {"damping": 0.72}

    def __init__(self, A, prm, backend):
        from ..matrix import CSR

        p = merge_params(self.defaults(), prm)
        if isinstance(A, CSR):
            m = float(p["damping"]) / np.asarray(A.diagonal())
        else:
            from ..backend import hip_setup

            m = float(p["damping"]) / hip_setup.diagonal(A)
        self._setup_m(m, backend)
