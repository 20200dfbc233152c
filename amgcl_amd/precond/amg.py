"""The AMG hierarchy preconditioner.

Parity: amgcl/amg.hpp:68-602 — levels built on the host (coarsening +
Galerkin) until coarse_enough, each level's operators moved to the backend;
V/W-cycle recursion with npre/npost smoothing, residual restriction, coarse
correction prolongation, and a direct solve at the coarsest level.

MI355X-native deltas from the reference:
  - coarsest level solved by a precomputed dense inverse applied as a
    device-resident GEMV (no D<->H round-trip; cf. backend/hip.hpp:73-96);
  - the whole cycle is expressible as a fixed kernel sequence, enabling
    hipGraph capture on the HIP backend (small-level launch overhead).
"""
import numpy as np

from ..matrix import CSR
from ..params import merge_params
from ..coarsening import make_coarsening
from ..relaxation import make_relaxation_factory


def _host_handoff():
    """Row threshold below which coarse levels leave the device engine for
    the host (tiny kernels are launch-bound; the host engine also supports
    every smoother/solver).  Env-tunable for A/B measurements."""
    import os

    return int(os.environ.get("AMGCL_HOST_HANDOFF", "20000"))


class Level:
    __slots__ = ("A", "P", "R", "P_build", "R_build", "f", "u", "t", "relax",
                 "rows", "nnz", "A_host")

    def __init__(self):
        self.A = self.P = self.R = None
        self.P_build = self.R_build = None
        self.f = self.u = self.t = None
        self.relax = None
        self.rows = self.nnz = 0
        self.A_host = None


class AMG:
    @staticmethod
    def defaults():
        return {
            "coarsening": {"type": "smoothed_aggregation"},
            "relax": {"type": "spai0"},
            "coarse_enough": 3000,
            "max_levels": 100,
            "npre": 1,
            "npost": 1,
            "ncycle": 1,      # 1 = V-cycle, 2 = W-cycle
            "pre_cycles": 1,
            "direct_coarse": True,
            "direct_solver": "dense",  # or "splu" (scipy SuperLU)
            "keep_host_matrices": False,
            "precision": "fp64",  # "mixed" = fp32 hierarchy under fp64 Krylov
            "block_value": 0,      # >1: store level operators as BSR blocks
            # SELL-64 images of level operators (wave-native layout;
            # kernels.hip rationale).  "auto" converts HIP-resident operators
            # with >= sell_min_rows rows: measured +25% on the fine level and
            # +17% on level 1, but a LOSS below ~8k slices (parallelism
            # starvation), hence the row threshold.  sell_min_mean kept as an
            # extra gate (0 = off).
            "sell": "auto",
            "sell_min_rows": 500000,
            "sell_min_mean": 0.0,
            "sell_transfers": True,  # also convert P/R transfer operators
        }

    def __init__(self, A, prm=None, backend=None):
        if backend is None:
            from ..backend import make_backend

            backend = make_backend("cpu")
        self.backend = backend
        self.prm = merge_params(self.defaults(), prm, opaque=("coarsening", "relax"))
        p = self.prm
        if p["precision"] not in ("fp64", "mixed"):
            raise ValueError(f"precision must be fp64|mixed, got '{p['precision']}'")
        if int(p["block_value"]) < 0:
            raise ValueError("block_value must be >= 0")
        if int(p["ncycle"]) < 1 or int(p["pre_cycles"]) < 1:
            raise ValueError("ncycle and pre_cycles must be >= 1")
        if int(p["npre"]) < 0 or int(p["npost"]) < 0:
            raise ValueError("npre/npost must be >= 0")
        if int(p["max_levels"]) < 1:
            raise ValueError("max_levels must be >= 1")
        self.levels = []
        self.coarse_solve = None
        self._coarsening = None
        self._mixed = False
        self._build(A)
        if self.prm["precision"] == "mixed":
            self._to_mixed()
        if int(self.prm["block_value"]) > 1:
            self._to_block(int(self.prm["block_value"]))
        self._build_sell()

    def _build_sell(self):
        """Attach SELL-64 images to coarse-level operators (measured: CSR
        sub-wave SpMV reaches only ~2 TB/s on SA coarse levels because the
        x-gather scatters across the sub-wave; see csrc/hip/kernels.hip)."""
        import os

        prm = self.prm
        if prm["sell"] in (False, "off") or os.environ.get("AMGCL_NO_SELL"):
            return
        if self.backend.name != "hip":
            return
        from ..backend.hip import DeviceCSR

        min_rows = int(prm["sell_min_rows"])
        min_mean = float(prm["sell_min_mean"])

        def eligible(M):
            if not isinstance(M, DeviceCSR) or M.nrows < min_rows:
                return False
            if M.val.is_complex():
                return False  # SELL kernels are real-valued (f64/f32)
            return M.nnz / max(M.nrows, 1) >= min_mean

        for lvl in self.levels:
            if eligible(lvl.A):
                lvl.A.build_sell()
            if prm["sell_transfers"]:
                for M in (lvl.P, lvl.R):
                    if M is not None and eligible(M):
                        M.build_sell()

    def rebuild(self, A_new):
        """Reuse the transfer operators for a matrix with changed coefficients
        (time-dependent problems; parity: amgcl/amg.hpp:250-269 rebuild)."""
        backend = self.backend
        relax_factory = make_relaxation_factory(self.prm["relax"])
        A = A_new
        for lvl in self.levels[:-1]:
            # mirror the device->host handoff of the original build: if this
            # level's transfers were built on the host, bring A back down
            if isinstance(lvl.P_build, CSR) and not isinstance(A, CSR):
                from ..backend import hip_setup

                A = hip_setup.download(A)
            lvl.rows, lvl.nnz = A.nrows, A.nnz
            if lvl.A_host is not None:
                lvl.A_host = A
            lvl.A = backend.matrix(A)
            lvl.relax = relax_factory(A, backend)
            A = self._coarsening.coarse_operator(A, lvl.P_build, lvl.R_build)
        last = self.levels[-1]
        if not isinstance(A, CSR) and A.nrows <= _host_handoff():
            from ..backend import hip_setup

            A = hip_setup.download(A)
        last.rows, last.nnz = A.nrows, A.nnz
        if last.A_host is not None:
            last.A_host = A
        last.A = backend.matrix(A)
        if self.prm["direct_coarse"]:
            self.coarse_solve = backend.coarse_solver(
                A, kind=self.prm["direct_solver"])
        else:
            last.relax = relax_factory(A, backend)
        self._build_sell()

    # --- setup (host) ------------------------------------------------------
    def _build(self, A: CSR):
        prm = self.prm
        coarsening = make_coarsening(prm["coarsening"])
        self._coarsening = coarsening
        relax_factory = make_relaxation_factory(prm["relax"])
        backend = self.backend

        coarse_enough = int(prm["coarse_enough"])
        A_host = A
        from ..profiler import prof

        while True:
            lvl = Level()
            lvl.rows, lvl.nnz = A_host.nrows, A_host.nnz
            if prm["keep_host_matrices"] or backend.name == "cpu":
                lvl.A_host = A_host
            with prof.scope("move_to_backend"):
                lvl.A = backend.matrix(A_host)
            last = (
                A_host.nrows <= coarse_enough
                or len(self.levels) + 1 >= int(prm["max_levels"])
            )
            if not last:
                with prof.scope("relax_setup"):
                    lvl.relax = relax_factory(A_host, backend)
                lvl.f = backend.vector(A_host.nrows) if self.levels else None
                lvl.u = backend.vector(A_host.nrows) if self.levels else None
                lvl.t = backend.vector(A_host.nrows)
                try:
                    with prof.scope("transfer_operators"):
                        P, R = coarsening.transfer_operators(A_host)
                except OverflowError:
                    # device setup hit a row denser than its LDS buffers:
                    # fall back to the host engine for this and deeper levels
                    from ..backend import hip_setup

                    A_host = hip_setup.download(A_host)
                    lvl.A_host = A_host if (prm["keep_host_matrices"] or backend.name == "cpu") else None
                    with prof.scope("transfer_operators"):
                        P, R = coarsening.transfer_operators(A_host)
                except RuntimeError:
                    # empty level (all nodes removed): stop coarsening here
                    last = True
                if not last:
                    if P.ncols == 0:
                        last = True
            if last:
                # coarsest level
                lvl.f = backend.vector(A_host.nrows) if self.levels else None
                lvl.u = backend.vector(A_host.nrows) if self.levels else None
                lvl.t = backend.vector(A_host.nrows)
                if prm["direct_coarse"]:
                    with prof.scope("coarse_solver"):
                        self.coarse_solve = backend.coarse_solver(
                            A_host, kind=prm["direct_solver"])
                else:
                    lvl.relax = relax_factory(A_host, backend)
                self.levels.append(lvl)
                break

            with prof.scope("move_to_backend"):
                lvl.P = backend.matrix(P)
                lvl.R = backend.matrix(R)
            lvl.P_build, lvl.R_build = P, R
            self.levels.append(lvl)
            with prof.scope("galerkin"):
                A_host = coarsening.coarse_operator(A_host, P, R)
            # device-setup path: hand small coarse levels back to the host
            # engine (tiny kernels are launch-bound; the host path also
            # supports every smoother/coarse solver)
            if not isinstance(A_host, CSR) and A_host.nrows <= _host_handoff():
                from ..backend import hip_setup

                A_host = hip_setup.download(A_host)

    def _to_mixed(self):
        """Convert the hierarchy to fp32 storage (mixed precision: fp32
        preconditioner under a fp64 Krylov loop; parity: SURVEY §5.9,
        reference examples/mixed_precision.cpp, backend/detail/mixing.hpp)."""
        if self.backend.name != "hip":
            raise ValueError("precision='mixed' requires the hip backend")
        import torch

        from ..backend.hip import DeviceCSR
        from ..relaxation.spai0 import DiagonalSmootherBase

        f32 = torch.float32

        def conv(m):
            if m is None:
                return None
            return DeviceCSR.from_tensors(m.nrows, m.ncols, m.ptr, m.col,
                                          m.val.to(f32), m.subw)

        # the outer Krylov loop keeps iterating with the fp64 fine operator
        # (backend/detail/mixing.hpp semantics: solver backend fp64, precond
        # backend fp32); only the preconditioner's internals go fp32
        self._A64 = self.levels[0].A
        for lvl in self.levels:
            lvl.A = conv(lvl.A)
            lvl.P = conv(lvl.P)
            lvl.R = conv(lvl.R)
            for name in ("f", "u", "t"):
                v = getattr(lvl, name)
                if v is not None:
                    setattr(lvl, name, v.to(f32))
            if lvl.relax is not None:
                if not isinstance(lvl.relax, DiagonalSmootherBase):
                    raise ValueError("mixed precision supports diagonal smoothers")
                lvl.relax.M = lvl.relax.M.to(f32)
        if self.coarse_solve is not None and hasattr(self.coarse_solve, "inv"):
            # HostSpluSolver (direct_solver="splu") has no dense inverse to
            # convert; its factorized solve simply stays fp64.
            self.coarse_solve.inv = self.coarse_solve.inv.to(f32)
        n0 = self.levels[0].rows
        self._r32 = self.backend.vector(n0, f32)
        self._x32 = self.backend.vector(n0, f32)
        self._mixed = True

    def _to_block(self, bsize):
        """Store level operators as BSR for block-valued solve kernels
        (parity: amgcl/backend/builtin_hybrid.hpp:43 — scalar setup quality,
        block solve speed). Only levels whose size is divisible by the block
        size are converted; transfers stay scalar CSR."""
        if self.backend.name != "hip":
            raise ValueError("block_value storage requires the hip backend")
        if self._mixed:
            raise ValueError("block_value is fp64-only for now")
        from ..backend import hip_setup
        from ..backend.hip import DeviceBSR, DeviceCSR

        for lvl in self.levels:
            A = lvl.A
            if A.nrows % bsize:
                continue
            if isinstance(A, DeviceCSR):
                # device-resident conversion, no D2H round-trip
                lvl.A = DeviceBSR.from_device(A, bsize, self.backend.device)
                continue
            host = lvl.A_host
            if host is None or not isinstance(host, CSR):
                continue
            lvl.A = DeviceBSR(host, bsize, self.backend.device)

    # --- solve-phase -------------------------------------------------------
    def system_matrix(self):
        if self._mixed:
            return self._A64
        return self.levels[0].A

    def cycle(self, i, f, u):
        """One multigrid cycle at level i with rhs f and iterate u
        (parity: amgcl/amg.hpp:514-553)."""
        b = self.backend
        lvl = self.levels[i]
        prm = self.prm

        if i + 1 == len(self.levels):
            if self.coarse_solve is not None:
                self.coarse_solve(f, u)
            else:
                for _ in range(int(prm["npre"])):
                    lvl.relax.apply_pre(lvl.A, f, u, lvl.t)
                for _ in range(int(prm["npost"])):
                    lvl.relax.apply_post(lvl.A, f, u, lvl.t)
            return

        nxt = self.levels[i + 1]
        for _ in range(int(prm["npre"])):
            lvl.relax.apply_pre(lvl.A, f, u, lvl.t)
        b.residual(f, lvl.A, u, lvl.t)
        b.spmv(1.0, lvl.R, lvl.t, 0.0, nxt.f)
        b.clear(nxt.u)
        for _ in range(int(prm["ncycle"])):
            self.cycle(i + 1, nxt.f, nxt.u)
        b.spmv(1.0, lvl.P, nxt.u, 1.0, u)
        for _ in range(int(prm["npost"])):
            lvl.relax.apply_post(lvl.A, f, u, lvl.t)

    def apply(self, rhs, x):
        """x = M^-1 rhs via pre_cycles cycles from a zero initial guess
        (parity: amgcl/amg.hpp:289-297)."""
        b = self.backend
        if self._mixed:
            b.cast(rhs, self._r32)
            b.clear(self._x32)
            for _ in range(int(self.prm["pre_cycles"])):
                self.cycle(0, self._r32, self._x32)
            b.cast(self._x32, x)
            return
        b.clear(x)
        for _ in range(int(self.prm["pre_cycles"])):
            self.cycle(0, rhs, x)

    # --- observability (parity: amgcl/amg.hpp:561-598) ---------------------
    def __str__(self):
        lines = ["level     unknowns       nonzeros"]
        total_rows = sum(l.rows for l in self.levels)
        total_nnz = sum(l.nnz for l in self.levels)
        for i, l in enumerate(self.levels):
            pct = 100.0 * l.nnz / total_nnz if total_nnz else 0
            lines.append(f"{i:5d} {l.rows:12d} {l.nnz:14d} ({pct:5.2f}%)")
        oc = total_nnz / self.levels[0].nnz if self.levels[0].nnz else 0
        gc = total_rows / self.levels[0].rows if self.levels[0].rows else 0
        lines.append(f"operator complexity: {oc:.2f}")
        lines.append(f"grid complexity:     {gc:.2f}")
        return "\n".join(lines)

    def bytes(self):
        tot = 0
        for l in self.levels:
            if l.A_host is not None:
                tot += l.A_host.bytes()
        return tot
