"""ctypes bindings to the hand-written gfx950 kernel library (libamghip.so).

The library is pure HIP (no torch dependency); tensors cross the boundary as
raw device pointers + the current torch HIP stream. Fails loudly if the
library is missing on a GPU machine — GPU ops must never fall back silently
to eager PyTorch.
"""
import ctypes
import os

_LIB = None

_SIGS = {
    "amg_spmv_f64": [ctypes.c_int64, ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p,
                     ctypes.c_void_p, ctypes.c_void_p, ctypes.c_double, ctypes.c_double,
                     ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p],
    "amg_residual_f64": [ctypes.c_int64, ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p,
                         ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                         ctypes.c_int, ctypes.c_void_p],
    "amg_relax_diag_f64": [ctypes.c_int64, ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p,
                           ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                           ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p],
    "amg_axpby_f64": [ctypes.c_int64, ctypes.c_double, ctypes.c_void_p, ctypes.c_double,
                      ctypes.c_void_p, ctypes.c_void_p],
    "amg_axpbypcz_f64": [ctypes.c_int64, ctypes.c_double, ctypes.c_void_p, ctypes.c_double,
                         ctypes.c_void_p, ctypes.c_double, ctypes.c_void_p, ctypes.c_void_p],
    "amg_vmul_f64": [ctypes.c_int64, ctypes.c_double, ctypes.c_void_p, ctypes.c_void_p,
                     ctypes.c_double, ctypes.c_void_p, ctypes.c_void_p],
    "amg_fill_f64": [ctypes.c_int64, ctypes.c_double, ctypes.c_void_p, ctypes.c_void_p],
    "amg_dot_f64": [ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                    ctypes.c_void_p],
    "amg_dot2_f64": [ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                     ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p],
    "amg_gather_f64": [ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                       ctypes.c_void_p],
    "amg_scatter_f64": [ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                        ctypes.c_void_p],
    "amg_gemv_f64": [ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                     ctypes.c_void_p],
    # --- fp32 variants (mixed precision) + casts ---
    "amg_spmv_f32": [ctypes.c_int64, ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p,
                     ctypes.c_void_p, ctypes.c_void_p, ctypes.c_double, ctypes.c_double,
                     ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p],
    "amg_residual_f32": [ctypes.c_int64, ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p,
                         ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                         ctypes.c_int, ctypes.c_void_p],
    "amg_relax_diag_f32": [ctypes.c_int64, ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p,
                           ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                           ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p],
    "amg_axpby_f32": [ctypes.c_int64, ctypes.c_double, ctypes.c_void_p, ctypes.c_double,
                      ctypes.c_void_p, ctypes.c_void_p],
    "amg_axpbypcz_f32": [ctypes.c_int64, ctypes.c_double, ctypes.c_void_p, ctypes.c_double,
                         ctypes.c_void_p, ctypes.c_double, ctypes.c_void_p, ctypes.c_void_p],
    "amg_vmul_f32": [ctypes.c_int64, ctypes.c_double, ctypes.c_void_p, ctypes.c_void_p,
                     ctypes.c_double, ctypes.c_void_p, ctypes.c_void_p],
    "amg_fill_f32": [ctypes.c_int64, ctypes.c_double, ctypes.c_void_p, ctypes.c_void_p],
    "amg_dot_f32": [ctypes.c_int64] + [ctypes.c_void_p] * 4,
    "amg_gather_f32": [ctypes.c_int64] + [ctypes.c_void_p] * 4,
    "amg_scatter_f32": [ctypes.c_int64] + [ctypes.c_void_p] * 4,
    "amg_gemv_f32": [ctypes.c_int64] + [ctypes.c_void_p] * 4,
    "amg_cast_d2s": [ctypes.c_int64] + [ctypes.c_void_p] * 3,
    "amg_cast_s2d": [ctypes.c_int64] + [ctypes.c_void_p] * 3,
    "amg_gs_color_f64": [ctypes.c_int64] + [ctypes.c_void_p] * 7,
    # exact level-scheduled triangular solve (cooperative kernel)
    "amg_sptrsv_f64": [ctypes.c_int64] + [ctypes.c_void_p] * 7
                      + [ctypes.c_int, ctypes.c_void_p],
    "amg_coop_supported": [],
    # --- complex128 solve kernels (host complex setup + device solve) ---
    "amg_spmv_c128": [ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 4
                     + [ctypes.c_double] * 4 + [ctypes.c_void_p, ctypes.c_int,
                                                ctypes.c_void_p],
    "amg_residual_c128": [ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 6
                         + [ctypes.c_int, ctypes.c_void_p],
    "amg_relax_diag_c128": [ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 7
                           + [ctypes.c_int, ctypes.c_void_p],
    "amg_axpby_c128": [ctypes.c_int64] + [ctypes.c_double] * 2 + [ctypes.c_void_p]
                      + [ctypes.c_double] * 2 + [ctypes.c_void_p] * 2,
    "amg_axpbypcz_c128": [ctypes.c_int64] + [ctypes.c_double] * 2
                         + [ctypes.c_void_p] + [ctypes.c_double] * 2
                         + [ctypes.c_void_p] + [ctypes.c_double] * 2
                         + [ctypes.c_void_p] * 2,
    "amg_vmul_c128": [ctypes.c_int64] + [ctypes.c_double] * 2
                     + [ctypes.c_void_p] * 2 + [ctypes.c_double] * 2
                     + [ctypes.c_void_p] * 2,
    "amg_dot_c128": [ctypes.c_int64] + [ctypes.c_void_p] * 4,
    "amg_gemv_c128": [ctypes.c_int64] + [ctypes.c_void_p] * 4,
    "amg_fill_c128": [ctypes.c_int64] + [ctypes.c_double] * 2
                     + [ctypes.c_void_p] * 2,
    # --- SELL-64 (wave-native sliced-ELL) solve kernels ---
    "amg_sell_spmv_f64": [ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 5
                         + [ctypes.c_double, ctypes.c_double]
                         + [ctypes.c_void_p] * 2,
    "amg_sell_residual_f64": [ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 8,
    "amg_sell_relax_f64": [ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 9,
    "amg_sell_fill_f64": [ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 8,
    "amg_sell_spmv_f32": [ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 5
                         + [ctypes.c_double, ctypes.c_double]
                         + [ctypes.c_void_p] * 2,
    "amg_sell_residual_f32": [ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 8,
    "amg_sell_relax_f32": [ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 9,
    "amg_sell_fill_f32": [ctypes.c_int64, ctypes.c_int64] + [ctypes.c_void_p] * 8,
    # --- block (BSR) solve kernels ---
    "amg_bsr_spmv_f64": [ctypes.c_int64, ctypes.c_int, ctypes.c_void_p, ctypes.c_void_p,
                         ctypes.c_void_p, ctypes.c_void_p, ctypes.c_double,
                         ctypes.c_double, ctypes.c_void_p, ctypes.c_void_p],
    "amg_bsr_residual_f64": [ctypes.c_int64, ctypes.c_int] + [ctypes.c_void_p] * 7,
    "amg_bsr_relax_f64": [ctypes.c_int64, ctypes.c_int] + [ctypes.c_void_p] * 8,
    "amg_blkdiag_vmul_f64": [ctypes.c_int64, ctypes.c_int] + [ctypes.c_void_p] * 4,
    "amg_bsr_spmv_mfma4_f64": [ctypes.c_int64] + [ctypes.c_void_p] * 4
                              + [ctypes.c_double, ctypes.c_double]
                              + [ctypes.c_void_p] * 2,
    # --- device-side setup engine (setup.hip) ---
    "amg_setup_diag": [ctypes.c_int64] + [ctypes.c_void_p] * 5,
    "amg_setup_strong": [ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                         ctypes.c_void_p, ctypes.c_double, ctypes.c_void_p, ctypes.c_void_p],
    "amg_setup_spai0": [ctypes.c_int64] + [ctypes.c_void_p] * 5,
    "amg_scan_i32": [ctypes.c_void_p, ctypes.c_int64, ctypes.c_void_p],
    "amg_agg_init": [ctypes.c_int64] + [ctypes.c_void_p] * 4,
    "amg_agg_round": [ctypes.c_int64] + [ctypes.c_void_p] * 10,
    "amg_agg_run": [ctypes.c_int64] + [ctypes.c_void_p] * 9 + [
        ctypes.c_int, ctypes.c_int, ctypes.c_void_p, ctypes.c_void_p,
        ctypes.c_void_p],
    "amg_agg_renumber": [ctypes.c_int64] + [ctypes.c_void_p] * 3,
    "amg_psmooth_count": [ctypes.c_int64] + [ctypes.c_void_p] * 7,
    "amg_psmooth_fill": [ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                         ctypes.c_void_p, ctypes.c_void_p, ctypes.c_double, ctypes.c_void_p,
                         ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p],
    "amg_ptent_count": [ctypes.c_int64] + [ctypes.c_void_p] * 3,
    "amg_ptent_fill": [ctypes.c_int64] + [ctypes.c_void_p] * 4,
    "amg_transpose_count": [ctypes.c_int64] + [ctypes.c_void_p] * 3,
    "amg_transpose_scatter": [ctypes.c_int64] + [ctypes.c_void_p] * 7,
    "amg_sort_rows": [ctypes.c_int64] + [ctypes.c_void_p] * 4,
    "amg_spgemm_count": [ctypes.c_int64] + [ctypes.c_void_p] * 9,
    "amg_spgemm_fill": [ctypes.c_int64] + [ctypes.c_void_p] * 10 + [ctypes.c_int]
                       + [ctypes.c_void_p] * 2,
    "amg_gershgorin": [ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p, ctypes.c_void_p],
    "amg_poisson_cnt": [ctypes.c_int64] * 4 + [ctypes.c_void_p] * 2,
    "amg_poisson_fill": [ctypes.c_int64] * 4 + [ctypes.c_void_p] * 4,
}


def lib():
    global _LIB
    if _LIB is None:
        path = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                            "_hip", "libamghip.so")
        if not os.path.exists(path):
            # try to build (works even without a GPU: hipcc cross-compiles)
            from ..build import build_hip_lib

            build_hip_lib()
        if not os.path.exists(path):
            raise RuntimeError(
                f"libamghip.so not found at {path}; run `python setup.py build_ext "
                "--inplace` — GPU ops do not fall back to eager torch"
            )
        _LIB = ctypes.CDLL(path)
        for name, argtypes in _SIGS.items():
            fn = getattr(_LIB, name)
            fn.argtypes = argtypes
            fn.restype = ctypes.c_int
    return _LIB


def check(rc, what):
    if rc != 0:
        raise RuntimeError(f"HIP kernel '{what}' failed with hipError_t={rc}")
