// amgcl_amd — block-valued (BSR) solve-phase kernels for gfx950.
//
// Reference analogue: the VexCL static_matrix backend's block-unrolled SpMV
// (amgcl/backend/vexcl_static_matrix.hpp:450) and builtin_hybrid's
// "scalar setup, block storage" design (amgcl/backend/builtin_hybrid.hpp:43).
//
// Design note (MI355X): fp64 BSR SpMV is memory-bandwidth-bound — a BxB
// block contributes 2*B*B flops for 8*B*B value bytes, i.e. arithmetic
// intensity 0.25 flop/byte, ~1e-3 of the fp64 compute:bandwidth balance
// point.  MEASURED (round 2, profiles/mfma_vs_unrolled_bsr_r02.log): a
// correct v_mfma_f64_16x16x4_f64 variant (below) reaches 2586 GB/s vs
// 5680 GB/s for these unrolled kernels on the elasticity config — the
// unrolled form IS the speed-of-light design (val rows contiguous, one
// scalar row per lane within the block row, x-block reuse for free).
//
// Layout: block values row-major within each block, blocks in CSR order.
// B lanes cooperate on one block row (lane = scalar row within the block).

#include <hip/hip_runtime.h>

#include <cstdint>

static inline int nblk_b(int64_t work, int block = 256, int cap = 4096) {
    int64_t b = (work + block - 1) / block;
    return (int)(b < 1 ? 1 : (b > cap ? cap : b));
}

// y = alpha * A x + beta * y ; one lane per scalar row, B lanes per block row
template <int B, bool BETA0>
__global__ void bsr_spmv_k(int64_t nbrows, const int *__restrict__ ptr,
                           const int *__restrict__ col, const double *__restrict__ val,
                           const double *__restrict__ x, double alpha, double beta,
                           double *__restrict__ y) {
    int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int lr = (int)(tid % B);
    int64_t brow = tid / B;
    int64_t stride = ((int64_t)gridDim.x * blockDim.x) / B;
    for (; brow < nbrows; brow += stride) {
        double s = 0.0;
        for (int j = ptr[brow]; j < ptr[brow + 1]; ++j) {
            const double *blk = val + (int64_t)j * B * B + lr * B;
            const double *xb = x + (int64_t)col[j] * B;
#pragma unroll
            for (int c = 0; c < B; ++c) s += blk[c] * xb[c];
        }
        int64_t row = brow * B + lr;
        y[row] = BETA0 ? alpha * s : alpha * s + beta * y[row];
    }
}

template <int B>
__global__ void bsr_residual_k(int64_t nbrows, const int *__restrict__ ptr,
                               const int *__restrict__ col, const double *__restrict__ val,
                               const double *__restrict__ rhs, const double *__restrict__ x,
                               double *__restrict__ r) {
    int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int lr = (int)(tid % B);
    int64_t brow = tid / B;
    int64_t stride = ((int64_t)gridDim.x * blockDim.x) / B;
    for (; brow < nbrows; brow += stride) {
        double s = 0.0;
        for (int j = ptr[brow]; j < ptr[brow + 1]; ++j) {
            const double *blk = val + (int64_t)j * B * B + lr * B;
            const double *xb = x + (int64_t)col[j] * B;
#pragma unroll
            for (int c = 0; c < B; ++c) s += blk[c] * xb[c];
        }
        int64_t row = brow * B + lr;
        r[row] = rhs[row] - s;
    }
}

// fused diagonal relaxation step: t = M ∘ (rhs - A x)
template <int B>
__global__ void bsr_relax_k(int64_t nbrows, const int *__restrict__ ptr,
                            const int *__restrict__ col, const double *__restrict__ val,
                            const double *__restrict__ M, const double *__restrict__ rhs,
                            const double *__restrict__ x, double *__restrict__ t) {
    int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int lr = (int)(tid % B);
    int64_t brow = tid / B;
    int64_t stride = ((int64_t)gridDim.x * blockDim.x) / B;
    for (; brow < nbrows; brow += stride) {
        double s = 0.0;
        for (int j = ptr[brow]; j < ptr[brow + 1]; ++j) {
            const double *blk = val + (int64_t)j * B * B + lr * B;
            const double *xb = x + (int64_t)col[j] * B;
#pragma unroll
            for (int c = 0; c < B; ++c) s += blk[c] * xb[c];
        }
        int64_t row = brow * B + lr;
        t[row] = M[row] * (rhs[row] - s);
    }
}

// block-diagonal matvec y_i = M_i x_i (M: nbrows row-major BxB blocks).
// Used by the as_block smoother wrapper (reference: relaxation/as_block.hpp,
// where the block-typed base smoother's M is a block per point).
template <int B>
__global__ void blkdiag_vmul_k(int64_t nbrows, const double *__restrict__ M,
                               const double *__restrict__ x, double *__restrict__ y) {
    int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int lr = (int)(tid % B);
    int64_t brow = tid / B;
    int64_t stride = ((int64_t)gridDim.x * blockDim.x) / B;
    for (; brow < nbrows; brow += stride) {
        const double *m = M + brow * B * B + lr * B;
        const double *xb = x + brow * B;
        double s = 0.0;
#pragma unroll
        for (int c = 0; c < B; ++c) s += m[c] * xb[c];
        y[brow * B + lr] = s;
    }
}

#define BSR_LAUNCH(kname, ...)                                              \
    switch (bsize) {                                                        \
        case 2: kname<2><<<grid, 256, 0, s>>>(__VA_ARGS__); break;          \
        case 3: kname<3><<<grid, 256, 0, s>>>(__VA_ARGS__); break;          \
        case 4: kname<4><<<grid, 256, 0, s>>>(__VA_ARGS__); break;          \
        default: return (int)hipErrorInvalidValue;                          \
    }

extern "C" int amg_bsr_spmv_f64(int64_t nbrows, int bsize, const int *ptr, const int *col,
                                const double *val, const double *x, double alpha,
                                double beta, double *y, hipStream_t s) {
    int grid = nblk_b(nbrows * bsize);
    if (beta == 0.0) {
        switch (bsize) {
            case 2: bsr_spmv_k<2, true><<<grid, 256, 0, s>>>(nbrows, ptr, col, val, x, alpha, beta, y); break;
            case 3: bsr_spmv_k<3, true><<<grid, 256, 0, s>>>(nbrows, ptr, col, val, x, alpha, beta, y); break;
            case 4: bsr_spmv_k<4, true><<<grid, 256, 0, s>>>(nbrows, ptr, col, val, x, alpha, beta, y); break;
            default: return (int)hipErrorInvalidValue;
        }
    } else {
        switch (bsize) {
            case 2: bsr_spmv_k<2, false><<<grid, 256, 0, s>>>(nbrows, ptr, col, val, x, alpha, beta, y); break;
            case 3: bsr_spmv_k<3, false><<<grid, 256, 0, s>>>(nbrows, ptr, col, val, x, alpha, beta, y); break;
            case 4: bsr_spmv_k<4, false><<<grid, 256, 0, s>>>(nbrows, ptr, col, val, x, alpha, beta, y); break;
            default: return (int)hipErrorInvalidValue;
        }
    }
    return (int)hipGetLastError();
}

extern "C" int amg_bsr_residual_f64(int64_t nbrows, int bsize, const int *ptr,
                                    const int *col, const double *val, const double *rhs,
                                    const double *x, double *r, hipStream_t s) {
    int grid = nblk_b(nbrows * bsize);
    BSR_LAUNCH(bsr_residual_k, nbrows, ptr, col, val, rhs, x, r)
    return (int)hipGetLastError();
}

extern "C" int amg_bsr_relax_f64(int64_t nbrows, int bsize, const int *ptr, const int *col,
                                 const double *val, const double *M, const double *rhs,
                                 const double *x, double *t, hipStream_t s) {
    int grid = nblk_b(nbrows * bsize);
    BSR_LAUNCH(bsr_relax_k, nbrows, ptr, col, val, M, rhs, x, t)
    return (int)hipGetLastError();
}

extern "C" int amg_blkdiag_vmul_f64(int64_t nbrows, int bsize, const double *M,
                                    const double *x, double *y, hipStream_t s) {
    int grid = nblk_b(nbrows * bsize);
    BSR_LAUNCH(blkdiag_vmul_k, nbrows, M, x, y)
    return (int)hipGetLastError();
}

// ---------------------------------------------------------------------------
// MFMA variant of the B=4 BSR SpMV (the VERDICT-mandated head-to-head;
// reference analogue of the question: vexcl_static_matrix block kernels).
//
// Mapping: one wave handles FOUR 4x4 block rows per step of
// v_mfma_f64_16x16x4_f64 (D[16x16] += A[16x4]*B[4x16]).  The M dimension
// carries the 16 scalar rows of the 4 block rows, K the 4 scalar columns of
// each block-row's current BSR block, and the B operand replicates each
// block row's gathered x block across its 4 N-columns, so the needed
// results are the diagonal D[m][m].  15/16 of the MFMA lanes compute
// unused dot products — the point of the experiment: flops are free at
// AI 0.25 flop/byte, so IF the kernel is purely bandwidth-bound the MFMA
// version should tie the unrolled one.  Measured result is recorded in
// profiles/README.md (round 2).
// Fragment layout for v_mfma_f64_16x16x4_f64, decoded EMPIRICALLY on
// gfx950 (scripts/mfma_probe, gpurun_out/mfma_probe.log):
//   A: lane l holds A[4*(l&3) + ((l>>2)&3)][l>>4]   (4x4-transposed sub-index)
//   B: lane l holds B[l>>4][l&15]
//   D: lane l item i holds D[4*(l>>4)+i][l&15]
// ---------------------------------------------------------------------------
typedef double d4_t __attribute__((ext_vector_type(4)));

template <bool BETA0>
__global__ void bsr_spmv_mfma4_k(int64_t nbrows, const int *__restrict__ ptr,
                                 const int *__restrict__ col,
                                 const double *__restrict__ val,
                                 const double *__restrict__ x, double alpha,
                                 double beta, double *__restrict__ y) {
    const int wpb = blockDim.x / 64;
    const int wid = threadIdx.x / 64, lane = threadIdx.x & 63;
    const int sub = lane & 15;
    const int k = lane >> 4;               // scalar column within a block
    const int ma = 4 * (sub & 3) + (sub >> 2);  // A row this lane feeds
    const int ga = ma >> 2;                // A-side block row within group
    const int ra = ma & 3;                 // scalar row within that block
    const int gb = sub >> 2;               // B-side: col n = sub -> group n/4
    int64_t grp = (int64_t)blockIdx.x * wpb + wid;
    const int64_t ngrp = (nbrows + 3) >> 2;
    const int64_t gstride = (int64_t)gridDim.x * wpb;
    for (; grp < ngrp; grp += gstride) {
        const int64_t brow0 = grp << 2;
        const int64_t browA = brow0 + ga;
        const int64_t browB = brow0 + gb;
        int begA = 0, lenA = 0, begB = 0, lenB = 0;
        if (browA < nbrows) {
            begA = ptr[browA];
            lenA = ptr[browA + 1] - begA;
        }
        if (browB < nbrows) {
            begB = ptr[browB];
            lenB = ptr[browB + 1] - begB;
        }
        int maxlen = lenA;
#pragma unroll
        for (int off = 32; off > 0; off >>= 1) {
            int o = __shfl_down(maxlen, off);
            if (o > maxlen) maxlen = o;
        }
        maxlen = __shfl(maxlen, 0);
        d4_t acc = {0.0, 0.0, 0.0, 0.0};
        for (int t = 0; t < maxlen; ++t) {
            double a = 0.0, b = 0.0;
            if (t < lenA) a = val[(int64_t)(begA + t) * 16 + ra * 4 + k];
            if (t < lenB) b = x[(int64_t)col[begB + t] * 4 + k];
            acc = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc, 0, 0, 0);
        }
        // extract the diagonal D[m][m]: lane l item i holds D[4*(l>>4)+i][l&15]
        const int n = lane & 15;
        const int i = n - 4 * (lane >> 4);
        if (i >= 0 && i < 4) {
            const int64_t row = brow0 * 4 + n;
            if (row < nbrows * 4)
                y[row] = BETA0 ? alpha * acc[i] : alpha * acc[i] + beta * y[row];
        }
    }
}

extern "C" int amg_bsr_spmv_mfma4_f64(int64_t nbrows, const int *ptr, const int *col,
                                      const double *val, const double *x, double alpha,
                                      double beta, double *y, hipStream_t s) {
    int64_t ngrp = (nbrows + 3) >> 2;
    int grid = nblk_b(ngrp * 64);
    if (beta == 0.0)
        bsr_spmv_mfma4_k<true><<<grid, 256, 0, s>>>(nbrows, ptr, col, val, x, alpha,
                                                    beta, y);
    else
        bsr_spmv_mfma4_k<false><<<grid, 256, 0, s>>>(nbrows, ptr, col, val, x, alpha,
                                                     beta, y);
    return (int)hipGetLastError();
}
