"""Distributed preconditioners.

LocalBlockPrecond: additive-Schwarz-style block preconditioner — a full
local preconditioner (AMG by default) on the rank's diagonal block.
Parity: amgcl/mpi/block_preconditioner.hpp:49 (and the local-AMG part of
mpi/subdomain_deflation.hpp).
"""
from ..precond import make_preconditioner


class LocalBlockPrecond:
    def __init__(self, dist_A, prm=None, backend=None):
        # build the local preconditioner on the diagonal block
        base = backend.base if hasattr(backend, "base") else backend
        self.local = make_preconditioner(dist_A.A_loc_host, dict(prm or {}), base)
        self.A = dist_A

    def system_matrix(self):
        return self.A

    def apply(self, rhs, x):
        self.local.apply(rhs, x)
