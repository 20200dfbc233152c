"""Relaxation (smoother) components.

Smoother concept (parity: amgcl/relaxation/spai0.hpp:50-116): constructed from
the host CSR level matrix + params + backend; apply_pre(A, rhs, x, tmp) and
apply_post(A, rhs, x, tmp) smooth in place through backend primitives;
apply(A, rhs, x, tmp) is the single-shot form used when the smoother acts as
a standalone preconditioner.
"""
from .spai0 import Spai0, Spai1
from .damped_jacobi import DampedJacobi
from .chebyshev import Chebyshev
from .gauss_seidel import GaussSeidel
from .ilu0 import ILU0, ILU0ChowPatel, ILUK, ILUP, ILUT
from .block_ilu0 import BlockILU0
from .as_block import AsBlock

REGISTRY = {
    "spai0": Spai0,
    "spai1": Spai1,
    "damped_jacobi": DampedJacobi,
    "chebyshev": Chebyshev,
    "gauss_seidel": GaussSeidel,
    "ilu0": ILU0,
    "block_ilu0": BlockILU0,
    "iluk": ILUK,
    "ilup": ILUP,
    "ilut": ILUT,
    "ilu0_chow_patel": ILU0ChowPatel,
    "as_block": AsBlock,
}


def make_relaxation_factory(prm=None):
    """Returns a factory f(A_host, backend) -> smoother."""
    prm = dict(prm or {})
    kind = prm.pop("type", "spai0")
    if kind not in REGISTRY:
        raise ValueError(f"unknown relaxation '{kind}'")
    cls = REGISTRY[kind]

    def factory(A, backend):
        return cls(A, prm, backend)

    factory.kind = kind
    factory.cls = cls
    return factory
