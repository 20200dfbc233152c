"""Solver base class (params, replaceable inner product)."""


class SolverBase:
    """Common machinery: params, norm via replaceable inner product."""

    common_defaults = {
        "tol": 1e-8,
        "abstol": 0.0,
        "maxiter": 100,
        "verbose": False,
        "ns_search": False,
    }

    def _init_common(self, n, prm, backend, inner_product):
        from ..params import merge_params

        if backend is None:
            from ..backend import make_backend

            backend = make_backend("cpu")
        self.backend = backend
        defaults = dict(self.common_defaults)
        defaults.update(self.defaults())
        self.prm = merge_params(defaults, prm)
        self.n = int(n)
        self.inner = inner_product or backend.dot
        self.inner2 = (
            inner_product.dot2
            if (inner_product is not None and hasattr(inner_product, "dot2"))
            else (backend.dot2 if inner_product is None else None)
        )

    def norm(self, x):
        import math

        return math.sqrt(abs(self.inner(x, x)))

    def __call__(self, P, rhs, x, A=None):
        if A is None:
            A = P.system_matrix()
        return self.solve(A, P, rhs, x)
