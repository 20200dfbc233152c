"""True per-phase setup wall time: synchronize the GPU at every profiler
scope boundary so async attribution cannot blur phases (diagnostic only —
the sync itself adds a little time)."""
import contextlib
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import amgcl_amd as am
from amgcl_amd import profiler


class SyncProf(profiler.Profiler):
    @contextlib.contextmanager
    def scope(self, name):
        import torch

        torch.cuda.synchronize()
        t0 = time.perf_counter()
        with super().scope(name):
            yield
        torch.cuda.synchronize()
        self.extra = getattr(self, "extra", {})
        self.extra[name] = self.extra.get(name, 0.0) + time.perf_counter() - t0


def main(n=512):
    import torch

    profiler.prof.__class__ = SyncProf
    from amgcl_amd.backend.hip_setup import poisson3d_device
    A = poisson3d_device(n)
    b = torch.randn(A.nrows, dtype=torch.float64, device="cuda")
    torch.cuda.synchronize()
    for run in range(2):
        profiler.prof.extra = {}
        t0 = time.perf_counter()
        s = am.make_solver(A, {"solver": {"type": "cg", "tol": 1e-6,
                                          "maxiter": 100}}, backend="hip")
        torch.cuda.synchronize()
        wall = time.perf_counter() - t0
        if run == 1:
            print(f"setup wall {wall:.3f} s (with per-scope syncs)")
            for k, v in sorted(profiler.prof.extra.items(), key=lambda kv: -kv[1]):
                print(f"  {k:24s} {v*1000:8.1f} ms")
        del s


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 512)
