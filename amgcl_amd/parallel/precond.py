"""Distributed preconditioners.

LocalBlockPrecond: additive-Schwarz-style block preconditioner — a full
local preconditioner (AMG by default) on the rank's diagonal block.
Parity: amgcl/mpi/block_preconditioner.hpp:49 (and the local-AMG part of
mpi/subdomain_deflation.hpp). On the HIP backend the local hierarchy is
built by the device setup engine and applied by the native driver.
"""
from ..precond import make_preconditioner


class LocalBlockPrecond:
    def __init__(self, dist_A, prm=None, backend=None):
        base = backend.base if hasattr(backend, "base") else backend
        # on HIP, A_loc is already device-resident: the local hierarchy is
        # built by the device setup engine with no extra upload
        A_loc = dist_A.A_loc if getattr(base, "name", "") == "hip" else dist_A.A_loc_host
        self.local = make_preconditioner(A_loc, dict(prm or {}), base)
        self.A = dist_A
        self._native = None
        if getattr(base, "name", "") == "hip" and hasattr(self.local, "levels"):
            from ..backend.native import NativeDriver

            try:
                self._native = NativeDriver(self.local, base)
            except TypeError:
                self._native = None

    def system_matrix(self):
        return self.A

    def apply(self, rhs, x):
        if self._native is not None:
            self._native.precond_apply(rhs, x)
        else:
            self.local.apply(rhs, x)
