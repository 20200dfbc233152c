"""Compute backends.

The backend concept mirrors the reference's backend layer
(amgcl/backend/interface.hpp:191-443): a small set of parallel primitives
(spmv, residual, axpby, axpbypcz, vmul, dot, clear, copy, gather/scatter)
plus matrix/vector containers and a coarse direct solver factory. Unlike the
reference there is exactly ONE GPU backend — hand-written gfx950 HIP kernels
— plus the OpenMP CPU backend used for setup-phase numerics and CPU testing.
"""

_REGISTRY = {}


def register(name):
    def deco(cls):
        _REGISTRY[name] = cls
        return cls
    return deco


def make_backend(name="cpu", **kwargs):
    if name not in _REGISTRY:
        # lazy imports so the HIP backend is only touched when requested
        if name == "cpu":
            from . import cpu  # noqa: F401
        elif name == "hip":
            from . import hip  # noqa: F401
    if name not in _REGISTRY:
        raise ValueError(f"unknown backend '{name}'")
    return _REGISTRY[name](**kwargs)


from . import cpu  # noqa: E402,F401  (cpu backend is always available)
