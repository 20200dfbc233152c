"""Distributed backend: wraps a base (cpu/hip) backend so the *unchanged*
Krylov solvers run distributed — spmv/residual on a DistMatrix overlap the
halo exchange with the local part (parity: mpi/distributed_matrix.hpp:520-547),
and inner products all-reduce (parity: mpi/inner_product.hpp:44).
"""
import numpy as np

from .dist_matrix import DistMatrix


class DistInnerProduct:
    """Replaceable inner product hook for the solvers (the reference's only
    change to make a solver distributed — solver/detail/default_inner_product
    vs mpi/inner_product)."""

    def __init__(self, backend, group=None):
        import torch
        import torch.distributed as dist

        self.backend = backend.base if hasattr(backend, "base") else backend
        self.dist = dist
        self.group = group
        dev = getattr(self.backend, "device", "cpu")
        self._buf = torch.zeros(2, dtype=torch.float64,
                                device=dev if self.backend.name == "hip" else "cpu")

    def __call__(self, x, y):
        self._buf[0] = self.backend.dot(x, y)
        self.dist.all_reduce(self._buf[:1], group=self.group)
        return float(self._buf[0].item())

    def dot2(self, x1, y1, x2, y2):
        d1, d2 = self.backend.dot2(x1, y1, x2, y2)
        self._buf[0], self._buf[1] = d1, d2
        self.dist.all_reduce(self._buf, group=self.group)
        return float(self._buf[0].item()), float(self._buf[1].item())


class DistBackend:
    """Same primitive set as a single-device backend; matrices may be
    DistMatrix (halo-exchanged) or plain local matrices."""

    def __init__(self, base, group=None):
        self.base = base
        self.group = group
        self.name = "dist+" + base.name
        self.device = getattr(base, "device", "cpu")

    # containers
    def matrix(self, strip):
        return DistMatrix(strip, self.base, self.group)

    def local_matrix(self, csr):
        return self.base.matrix(csr)

    def vector(self, n):
        return self.base.vector(n)

    def from_host(self, a):
        return self.base.from_host(a)

    def to_host(self, v):
        return self.base.to_host(v)

    # primitives
    def spmv(self, alpha, A, x, beta, y):
        from .deflation import ProjectedDistMatrix

        if isinstance(A, ProjectedDistMatrix):
            if beta != 0.0:
                raise ValueError("projected spmv supports beta=0 only")
            self.spmv(alpha, A.inner, x, 0.0, y)
            A.defl.project(y)
        elif isinstance(A, DistMatrix):
            works = A.start_exchange(x)
            self.base.spmv(alpha, A.A_loc, x, beta, y)  # overlapped with halo
            A.finish_exchange(works)
            if A.A_rem is not None:
                self.base.spmv(alpha, A.A_rem, A.x_rem, 1.0, y)
        else:
            self.base.spmv(alpha, A, x, beta, y)

    def residual(self, b, A, x, r):
        from .deflation import ProjectedDistMatrix

        if isinstance(A, ProjectedDistMatrix):
            self.spmv(1.0, A, x, 0.0, r)  # r = P A x
            self.base.axpby(1.0, b, -1.0, r)  # r = b - P A x
        elif isinstance(A, DistMatrix):
            works = A.start_exchange(x)
            self.base.residual(b, A.A_loc, x, r)
            A.finish_exchange(works)
            if A.A_rem is not None:
                self.base.spmv(-1.0, A.A_rem, A.x_rem, 1.0, r)
        else:
            self.base.residual(b, A, x, r)

    def clear(self, x):
        self.base.clear(x)

    def copy(self, x, y):
        self.base.copy(x, y)

    def axpby(self, a, x, b, y):
        self.base.axpby(a, x, b, y)

    def axpbypcz(self, a, x, b, y, c, z):
        self.base.axpbypcz(a, x, b, y, c, z)

    def vmul(self, a, m, x, b, z):
        self.base.vmul(a, m, x, b, z)

    def gather(self, x, idx, buf):
        self.base.gather(x, idx, buf)

    def scatter(self, buf, idx, x):
        self.base.scatter(buf, idx, x)

    def relax_diag(self, A, M, rhs, x, t):
        # fused diagonal relaxation is only used on local matrices
        if isinstance(A, DistMatrix):
            raise TypeError("relax_diag on a DistMatrix is not supported")
        return self.base.relax_diag(A, M, rhs, x, t)

    # global reductions (delegated to DistInnerProduct by solvers; these are
    # here for components that call the backend directly)
    def dot(self, x, y):
        return DistInnerProduct(self.base, self.group)(x, y)

    def dot2(self, x1, y1, x2, y2):
        return DistInnerProduct(self.base, self.group).dot2(x1, y1, x2, y2)

    def coarse_solver(self, csr):
        return self.base.coarse_solver(csr)

    def synchronize(self):
        self.base.synchronize()
