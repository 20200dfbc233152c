"""amgcl_amd — MI355X-native algebraic multigrid solver framework.

A from-scratch redesign of ddemidov/amgcl for AMD Instinct MI355X (gfx950):
the AMG hierarchy (smoothed aggregation / Ruge-Stuben, Galerkin product) is
assembled on the host by an OpenMP C++ engine, and the solve phase runs
through a single hand-written CDNA4 HIP backend (CSR SpMV, fused smoothers,
Krylov vector primitives), scaling to the 8 GPUs of one node over RCCL/xGMI.

Public API mirrors the reference's composition style:
    solve = amgcl_amd.make_solver(A, {"precond": {...}, "solver": {...}},
                                  backend="hip")
    x, iters, resid = solve(rhs)
"""

import os as _os

# GNU OpenMP's default active spin-wait makes the many small parallel regions
# of the solve phase ~40x slower when interleaved with Python/numpy work;
# passive waiting is the right default for this call pattern.
_os.environ.setdefault("OMP_WAIT_POLICY", "PASSIVE")


def _effective_cpus():
    """nproc capped by the cgroup CPU quota (GPU boxes expose 256 CPUs but
    cap the container at a much smaller cpu.max; oversubscribed OpenMP
    threads then thrash against CFS throttling)."""
    n = _os.cpu_count() or 1
    try:
        quota, period = open("/sys/fs/cgroup/cpu.max").read().split()
        if quota != "max":
            n = min(n, max(1, int(int(quota) / int(period))))
    except (OSError, ValueError):
        pass
    return n


_os.environ.setdefault("OMP_NUM_THREADS", str(_effective_cpus()))

from . import build as _build_mod

try:
    from . import _core  # noqa: F401
except ImportError:  # first use: build in-tree
    _build_mod.build_core_ext()
    from . import _core  # noqa: F401

from .matrix import CSR, galerkin  # noqa: F401,E402
from .generators import poisson3d, poisson3d_strip  # noqa: F401,E402
from .make_solver import make_solver, MakeSolver  # noqa: F401,E402
from .precond.amg import AMG  # noqa: F401,E402
from .backend import make_backend  # noqa: F401,E402
from .profiler import prof  # noqa: F401,E402

__version__ = "0.1.0"
