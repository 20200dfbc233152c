"""Gauss-Seidel smoother.

Parity: amgcl/relaxation/gauss_seidel.hpp:58 — forward sweep in apply_pre,
backward sweep in apply_post; serial (or level-scheduled) sweeps on the CPU
backend. On the HIP backend this framework EXCEEDS the reference (which
blocks GS on GPU backends via relaxation_is_supported): rows are colored by
a deterministic parallel Jones-Plassmann pass and each color sweeps as one
race-free GPU kernel (multicolor Gauss-Seidel).
"""
import os

import numpy as np

from .. import _core
from ..params import merge_params


class GaussSeidel:
    gpu_supported = True  # multicolor on the HIP backend

    @staticmethod
    def defaults():
        # serial=None picks the reference's heuristic (parallel when the
        # OpenMP pool is wide enough, gauss_seidel.hpp:83); True forces the
        # serial lexicographic sweep; False forces the multicolor sweep
        return {"serial": None}

    def __init__(self, A, prm, backend):
        from ..matrix import CSR

        p = merge_params(self.defaults(), prm)
        self.backend = backend
        if backend.name == "cpu":
            self.A_host = A
            self._gpu = False
            serial = p["serial"]
            if serial is None:
                serial = os.cpu_count() < 4 or A.nrows < 10_000
            self._serial = bool(serial)
            if not self._serial:
                # deterministic CPU-parallel sweep: multicolor ordering (the
                # reference's level-scheduled parallel GS races on same-level
                # upper reads; colors are race-free AND deterministic)
                colors, ncolors = _core.color_graph(A.nrows, A.ptr, A.col)
                colors = np.asarray(colors)
                self._order = np.argsort(colors, kind="stable").astype(np.int32)
                counts = np.bincount(colors, minlength=ncolors)
                self._cptr = np.concatenate([[0], np.cumsum(counts)]).astype(np.int32)
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

    def _cpu_sweep(self, rhs, x, forward):
        H = self.A_host
        if self._serial:
            _core.gauss_seidel(H.nrows, H.ptr, H.col, H.val, rhs, x, forward)
        else:
            _core.gauss_seidel_colored(H.nrows, H.ptr, H.col, H.val, rhs, x,
                                       self._order, self._cptr, forward)

    def apply_pre(self, A, rhs, x, tmp):
        if self._gpu:
            self._sweep(A, rhs, x, True)
        else:
            self._cpu_sweep(rhs, x, True)

    def apply_post(self, A, rhs, x, tmp):
        if self._gpu:
            self._sweep(A, rhs, x, False)
        else:
            self._cpu_sweep(rhs, x, False)

    def apply(self, A, rhs, x, tmp=None):
        self.backend.clear(x)
        self.apply_pre(A, rhs, x, tmp)
        self.apply_post(A, rhs, x, tmp)
