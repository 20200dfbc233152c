"""Shared-memory deflated solver (parity: amgcl/deflated_solver.hpp:45).

User-supplied deflation vectors Z (columns); E = Z^T A Z is factorized once
and each preconditioner application is augmented with the coarse-space
projection:  apply(r) = P_inner(r) + Z E^-1 Z^T r  (additive coarse-space
correction), with the Krylov solver iterating on the projected operator
through make_solver as usual. This is the single-process counterpart of
parallel/deflation.py.
"""
import numpy as np

from ..matrix import CSR
from ..params import merge_params


class DeflatedPrecond:
    @staticmethod
    def defaults():
        return {
            "Z_raw": None,          # (n x k) deflation vectors; None -> constant
            "inner": {"class": "amg"},
        }

    def __init__(self, A, prm=None, backend=None):
        from . import make_preconditioner

        if backend is None:
            from ..backend import make_backend

            backend = make_backend("cpu")
        self.backend = backend
        p = merge_params(self.defaults(), prm, opaque=("inner",))

        host = A
        if not isinstance(A, CSR):
            from ..backend import hip_setup

            host = hip_setup.download(A)
        n = host.nrows
        Z = p["Z_raw"]
        Z = np.ones((n, 1)) if Z is None else np.asarray(Z, dtype=np.float64)
        if Z.ndim == 1:
            Z = Z[:, None]
        self.k = Z.shape[1]

        AZ = np.stack([host @ np.ascontiguousarray(Z[:, j]) for j in range(self.k)],
                      axis=1)
        E = Z.T @ AZ
        self.Einv = np.linalg.pinv(E)

        self.inner = make_preconditioner(A, dict(p["inner"]), backend)
        self._A = self.inner.system_matrix()
        is_hip = backend.name == "hip"
        if is_hip:
            import torch

            self.Z = torch.from_numpy(np.ascontiguousarray(Z)).to(backend.device)
            self.Einv_d = torch.from_numpy(self.Einv).to(backend.device)
        else:
            self.Z = Z
            self.Einv_d = self.Einv
        self._tmp = backend.vector(n)

    def system_matrix(self):
        return self._A

    def apply(self, rhs, x):
        self.inner.apply(rhs, x)
        # coarse correction: x += Z E^-1 Z^T rhs
        lam = self.Einv_d @ (self.Z.T @ rhs)
        corr = self.Z @ lam
        self.backend.axpby(1.0, corr, 1.0, x)
