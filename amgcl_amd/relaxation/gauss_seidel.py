"""Gauss-Seidel smoother.

Parity: amgcl/relaxation/gauss_seidel.hpp:58 — forward sweep in apply_pre,
backward sweep in apply_post; serial (or level-scheduled) sweeps on the CPU
backend. On the HIP backend this framework EXCEEDS the reference (which
blocks GS on GPU backends via relaxation_is_supported): rows are colored by
a deterministic parallel Jones-Plassmann pass and each color sweeps as one
race-free GPU kernel (multicolor Gauss-Seidel).
"""
import numpy as np

from .. import _core
from ..params import merge_params


class GaussSeidel:
    gpu_supported = True  # multicolor on the HIP backend

    @staticmethod
    def defaults():
        return {"serial": True}

    def __init__(self, A, prm, backend):
        from ..matrix import CSR

        merge_params(self.defaults(), prm)
        self.backend = backend
        if backend.name == "cpu":
            self.A_host = A
            self._gpu = False
            return
        if not isinstance(A, CSR):
            from ..backend import hip_setup

            A = hip_setup.download(A)
        self._gpu = True
        colors, ncolors = _core.color_graph(A.nrows, A.ptr, A.col)
        colors = np.asarray(colors)
        order = np.argsort(colors, kind="stable")
        counts = np.bincount(colors, minlength=ncolors)
        import torch

        from ..backend._hiplib import check, lib

        self._lib = lib
        self._check = check
        self.rowlists = []
        off = 0
        dev = backend.device
        for c in range(ncolors):
            cnt = int(counts[c])
            self.rowlists.append(
                torch.from_numpy(order[off : off + cnt].astype(np.int32)).to(dev)
            )
            off += cnt

    def _sweep(self, A, rhs, x, forward):
        from ..backend.hip import _stream

        lists = self.rowlists if forward else list(reversed(self.rowlists))
        for rl in lists:
            if rl.numel() == 0:
                continue
            self._check(self._lib().amg_gs_color_f64(
                rl.numel(), rl.data_ptr(), A.ptr.data_ptr(), A.col.data_ptr(),
                A.val.data_ptr(), rhs.data_ptr(), x.data_ptr(), _stream()), "gs_color")

    def apply_pre(self, A, rhs, x, tmp):
        if self._gpu:
            self._sweep(A, rhs, x, True)
        else:
            _core.gauss_seidel(self.A_host.nrows, self.A_host.ptr, self.A_host.col,
                               self.A_host.val, rhs, x, True)

    def apply_post(self, A, rhs, x, tmp):
        if self._gpu:
            self._sweep(A, rhs, x, False)
        else:
            _core.gauss_seidel(self.A_host.nrows, self.A_host.ptr, self.A_host.col,
                               self.A_host.val, rhs, x, False)

    def apply(self, A, rhs, x, tmp=None):
        self.backend.clear(x)
        self.apply_pre(A, rhs, x, tmp)
        self.apply_post(A, rhs, x, tmp)
