// amgcl_amd — block-valued (BSR) solve-phase kernels for gfx950.
//
// Reference analogue: the VexCL static_matrix backend's block-unrolled SpMV
// (amgcl/backend/vexcl_static_matrix.hpp:450) and builtin_hybrid's
// "scalar setup, block storage" design (amgcl/backend/builtin_hybrid.hpp:43).
//
// Design note (MI355X): fp64 BSR SpMV is memory-bandwidth-bound — a BxB
// block contributes 2*B*B flops for 8*B*B value bytes, i.e. arithmetic
// intensity 0.25 flop/byte, ~1e-3 of the fp64 compute:bandwidth balance
// point. Matrix cores (MFMA) therefore cannot speed this kernel up; the
// speed-of-light design is maximal-bandwidth block loads (val rows
// contiguous, one scalar row per lane within the block row) plus the x-reuse
// that BSR gives for free (each x block is loaded once per block row).
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
