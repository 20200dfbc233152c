// amgcl_amd — shared device helpers for the gfx950 kernels.
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

#define WAVE 64

// XCD-aware logical block id (guide cdna_hip_programming.md §1): the command
// processor round-robins workgroups across the 8 XCDs, so blocks with
// blockIdx%8==k run on XCD k.  Remapping gives each XCD one contiguous slice
// of the logical index space: the x-gather bands of neighboring CSR rows
// then land in a single XCD's L2 instead of being replicated in all eight.
// Affects speed only, never correctness.
// MEASURED (kbw384, r02): the swizzle LOST bandwidth on every level
// (L0 4107->3337 GB/s): the natural round-robin dispatch already places
// same-XCD blocks ~8 blocks (~512 rows) apart, inside the stencil's +-n
// x-band, so L2 locality is better WITHOUT the remap.  Off by default,
// kept for A/B (-DAMGCL_SWIZZLE).
__device__ static inline int64_t amg_logical_block() {
#ifdef AMGCL_SWIZZLE
    unsigned g = gridDim.x;
    if ((g & 7u) == 0u)
        return (int64_t)(blockIdx.x & 7u) * (int64_t)(g >> 3) + (blockIdx.x >> 3);
#endif
    return blockIdx.x;
}

// MEASURED (kbw384, r02): nontemporal val/col loads LOST bandwidth badly on
// cache-resident coarse levels (L2-level 3653->2088 GB/s: a 15 MB matrix is
// L2/LLC-resident across V-cycle passes, and the evict-first hint destroys
// that residency) and did not help the streaming fine level either.  Off by
// default, kept for A/B (-DAMGCL_NT).
#ifdef AMGCL_NT
#define AMG_STREAM_LD(p) __builtin_nontemporal_load(p)
#else
#define AMG_STREAM_LD(p) (*(p))
#endif

// Launch geometry for memory-bound grid-stride kernels (guide §6 G11): cap
// at ~8 blocks/CU and grid-stride the rest; rounded to a multiple of 8 so
// the XCD swizzle divides evenly.
static inline int amg_nblocks(int64_t work, int block = 256, int cap = 2048) {
    int64_t b = (work + block - 1) / block;
    if (b < 1) b = 1;
    if (b > cap) b = cap;
    if (b > 8) b = (b + 7) & ~(int64_t)7;
    return (int)b;
}

// Per-level operator descriptor consumed by the native solve driver
// (driver.hip amg_driver_create); filled either from Python (backend/
// native.py, torch tensors) or from the torch-free GPU C API
// (capi_gpu.hip, raw hipMalloc buffers).
struct LevelDesc {
    int64_t nrows, nnz;
    const int *ptr;
    const int *col;
    const double *val;
    int subw;
    int64_t pnnz;  // P: nrows x next->nrows
    const int *pptr;
    const int *pcol;
    const double *pval;
    int psubw;
    int64_t rnnz;  // R: next->nrows x nrows
    const int *rptr;
    const int *rcol;
    const double *rval;
    int rsubw;
    const double *M;  // diagonal smoother weights
    double *f;
    double *u;
    double *t;  // workspace (f/u unused at level 0)
    // optional SELL-64 images of A / P / R (see kernels.hip)
    int64_t nslice;  // 0 = no SELL
    const int64_t *soff;
    const int *scol;
    const void *sval;
    int64_t pnslice;
    const int64_t *psoff;
    const int *pscol;
    const void *psval;
    int64_t rnslice;
    const int64_t *rsoff;
    const int *rscol;
    const void *rsval;
    // optional sigma-sort permutations (slot -> row id, < 0 = padding)
    const int *srows;
    const int *psrows;
    const int *rsrows;
    // Chebyshev smoothing (cheb_degree > 0 replaces the diagonal smoother;
    // M then holds the optional D^-1 scaling or null)
    int cheb_degree;
    double cheb_theta;
    double cheb_delta;
    double cheb_sigma1;
    void *cheb_d;  // extra per-level work vector for the recurrence
    // BSR storage of A (bsize > 0; ptr/col/val above are then unused for A)
    int bsize;
    int64_t nbrows;
    const int *bptr;
    const int *bcol;
    const double *bval;
    // ILU(0) smoothing via damped-Jacobi iterated triangular solves
    // (ilu_iters > 0; the reference's GPU-native ilu_solve.hpp route).
    // L strictly lower (unit diag implied), U strictly upper; ilu_dinv is
    // the inverted diagonal of U.
    int ilu_iters;
    double ilu_damping;       // outer relaxation damping
    double ilu_jdamping;      // inner Jacobi damping (0.72)
    const int *lptr;
    const int *lcol;
    const double *lval;
    const int *uptr;
    const int *ucol;
    const double *uval;
    const double *ilu_dinv;
    double *ilu_y;            // work vectors (level-sized)
    double *ilu_s;
    double *ilu_b;
};
