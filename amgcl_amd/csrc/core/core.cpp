// amgcl_amd host setup engine (CPU, OpenMP).
//
// Algebraic-multigrid hierarchy construction primitives: CSR utilities
// (transpose / SpGEMM / diagonal), greedy aggregation, smoothed-aggregation
// prolongation, SPAI-0 smoother setup, and OpenMP solve-phase reference
// primitives (SpMV / residual) used by the CPU backend and as the numerics
// reference for the HIP kernels.
//
// Behavioral parity references (ddemidov/amgcl, studied not copied):
//   - aggregation semantics:       amgcl/coarsening/plain_aggregates.hpp:114-207
//   - SA prolongation smoothing:   amgcl/coarsening/smoothed_aggregation.hpp:130-242
//   - Galerkin triple product:     amgcl/coarsening/detail/galerkin.hpp:42
//   - SpGEMM (Saad marker):        amgcl/detail/spgemm.hpp:62
//   - SPAI-0 weights:              amgcl/relaxation/spai0.hpp:66-77
//   - Poisson fixture:             tests/sample_problem.hpp:11
//
// All arrays are numpy: ptr/col int32, values float64 (fp32 instantiations
// where noted). nnz must stay < 2^31 per matrix (asserted).

#include <pybind11/pybind11.h>
#include <sys/mman.h>
#include <pybind11/numpy.h>
#include <omp.h>

#include <algorithm>
#include <atomic>
#include <cmath>
#include <cstdint>
#include <cstring>
#include <memory>
#include <stdexcept>
#include <vector>

namespace py = pybind11;

using i32 = int32_t;
using i64 = int64_t;

template <typename T>
using arr = py::array_t<T, py::array::c_style | py::array::forcecast>;

struct CsrView {
    i64 nrows, ncols;
    const i32 *ptr, *col;
    const double *val;
};


static void scan_i32(i32 *a, i64 n);

// THP hint + parallel write pre-touch for large fresh arrays. Fault cost on
// 4 KiB pages (3.3M faults for a 13 GB spgemm output) dominated setup on the
// 2-socket EPYC host; 2 MiB pages + parallel faulting removes it and gives
// NUMA-interleaved first touch.
static void pretouch(void *p, i64 bytes) {
    if (bytes < (i64(64) << 20)) return;
    uintptr_t s = ((uintptr_t)p + 4095) & ~(uintptr_t)4095;
    uintptr_t e = (uintptr_t)p + bytes;
    if (e > s) madvise((void *)s, e - s, MADV_HUGEPAGE);
    char *c = (char *)p;
#pragma omp parallel for schedule(static)
    for (i64 off = 0; off < bytes; off += 4096) c[off] = 0;
}

template <typename T>
static arr<T> big_arr(i64 n) {
    arr<T> a(n);
    pretouch(a.mutable_data(), n * (i64)sizeof(T));
    return a;
}


static CsrView view(i64 nrows, i64 ncols, const arr<i32> &ptr, const arr<i32> &col,
                    const arr<double> &val) {
    if (ptr.size() != nrows + 1) throw std::runtime_error("bad ptr size");
    if (col.size() != val.size()) throw std::runtime_error("col/val size mismatch");
    return CsrView{nrows, ncols, ptr.data(), col.data(), val.data()};
}

// ---------------------------------------------------------------------------
// Poisson 7-point fixture (parity: tests/sample_problem.hpp:11-80).
// Unit cube, Dirichlet handled by truncated stencils; diag = 2/hx^2+2/hy^2+2/hz^2,
// off-diagonals -1/h^2; hy = hx*aniso, hz = hy*aniso.
// ---------------------------------------------------------------------------
static py::tuple poisson3d(i64 n, double anisotropy) {
    const i64 n3 = n * n * n;
    const double hx = 1.0, hy = hx * anisotropy, hz = hy * anisotropy;
    const double wx = 1.0 / (hx * hx), wy = 1.0 / (hy * hy), wz = 1.0 / (hz * hz);
    const double dia = 2 * wx + 2 * wy + 2 * wz;

    arr<i32> ptr(n3 + 1);
    i32 *P = ptr.mutable_data();
    P[0] = 0;
#pragma omp parallel for schedule(static)
    for (i64 idx = 0; idx < n3; ++idx) {
        i64 i = idx % n, j = (idx / n) % n, k = idx / (n * n);
        int cnt = 1 + (k > 0) + (j > 0) + (i > 0) + (i + 1 < n) + (j + 1 < n) + (k + 1 < n);
        P[idx + 1] = cnt;
    }
    scan_i32(P + 1, n3);
    const i64 nnz = P[n3];
    if (nnz >= (i64(1) << 31)) throw std::runtime_error("nnz exceeds int32");

    arr<i32> col = big_arr<i32>(nnz);
    arr<double> val = big_arr<double>(nnz);
    i32 *C = col.mutable_data();
    double *V = val.mutable_data();
#pragma omp parallel for schedule(static)
    for (i64 idx = 0; idx < n3; ++idx) {
        i64 i = idx % n, j = (idx / n) % n, k = idx / (n * n);
        i64 h = P[idx];
        if (k > 0)     { C[h] = (i32)(idx - n * n); V[h] = -wz; ++h; }
        if (j > 0)     { C[h] = (i32)(idx - n);     V[h] = -wy; ++h; }
        if (i > 0)     { C[h] = (i32)(idx - 1);     V[h] = -wx; ++h; }
        C[h] = (i32)idx; V[h] = dia; ++h;
        if (i + 1 < n) { C[h] = (i32)(idx + 1);     V[h] = -wx; ++h; }
        if (j + 1 < n) { C[h] = (i32)(idx + n);     V[h] = -wy; ++h; }
        if (k + 1 < n) { C[h] = (i32)(idx + n * n); V[h] = -wz; ++h; }
    }
    return py::make_tuple(ptr, col, val);
}

// Row strip [row_beg, row_end) of the n^3 Poisson matrix with GLOBAL column
// indices (for the distributed tests/benchmarks; parity:
// examples/mpi/mpi_solver.cpp:47 assembles the local strip per rank).
static py::tuple poisson3d_strip(i64 n, i64 row_beg, i64 row_end) {
    const i64 nloc = row_end - row_beg;
    const double dia = 6.0;
    arr<i32> ptr(nloc + 1);
    i32 *P = ptr.mutable_data();
    P[0] = 0;
#pragma omp parallel for schedule(static)
    for (i64 r = 0; r < nloc; ++r) {
        i64 idx = row_beg + r;
        i64 i = idx % n, j = (idx / n) % n, k = idx / (n * n);
        P[r + 1] = 1 + (k > 0) + (j > 0) + (i > 0) + (i + 1 < n) + (j + 1 < n) + (k + 1 < n);
    }
    scan_i32(P + 1, nloc);
    const i64 nnz = P[nloc];
    arr<i32> col = big_arr<i32>(nnz);
    arr<double> val = big_arr<double>(nnz);
    i32 *C = col.mutable_data();
    double *V = val.mutable_data();
#pragma omp parallel for schedule(static)
    for (i64 r = 0; r < nloc; ++r) {
        i64 idx = row_beg + r;
        i64 i = idx % n, j = (idx / n) % n, k = idx / (n * n);
        i64 h = P[r];
        if (k > 0)     { C[h] = (i32)(idx - n * n); V[h] = -1.0; ++h; }
        if (j > 0)     { C[h] = (i32)(idx - n);     V[h] = -1.0; ++h; }
        if (i > 0)     { C[h] = (i32)(idx - 1);     V[h] = -1.0; ++h; }
        C[h] = (i32)idx; V[h] = dia; ++h;
        if (i + 1 < n) { C[h] = (i32)(idx + 1);     V[h] = -1.0; ++h; }
        if (j + 1 < n) { C[h] = (i32)(idx + n);     V[h] = -1.0; ++h; }
        if (k + 1 < n) { C[h] = (i32)(idx + n * n); V[h] = -1.0; ++h; }
    }
    return py::make_tuple(ptr, col, val);
}

// Box-grid variant (nx x ny x nz) of the strip generator: the weak-scaling
// fixture (per-rank cube, domain elongated along z as ranks are added —
// reference benchmarks.rst weak-scaling setup).
static py::tuple poisson3d_box_strip(i64 nx, i64 ny, i64 nz, i64 row_beg, i64 row_end) {
    const i64 nloc = row_end - row_beg;
    const double dia = 6.0;
    arr<i32> ptr(nloc + 1);
    i32 *P = ptr.mutable_data();
    P[0] = 0;
#pragma omp parallel for schedule(static)
    for (i64 r = 0; r < nloc; ++r) {
        i64 idx = row_beg + r;
        i64 i = idx % nx, j = (idx / nx) % ny, k = idx / (nx * ny);
        P[r + 1] = 1 + (k > 0) + (j > 0) + (i > 0) + (i + 1 < nx) + (j + 1 < ny) +
                   (k + 1 < nz);
    }
    scan_i32(P + 1, nloc);
    const i64 nnz = P[nloc];
    arr<i32> col = big_arr<i32>(nnz);
    arr<double> val = big_arr<double>(nnz);
    i32 *C = col.mutable_data();
    double *V = val.mutable_data();
#pragma omp parallel for schedule(static)
    for (i64 r = 0; r < nloc; ++r) {
        i64 idx = row_beg + r;
        i64 i = idx % nx, j = (idx / nx) % ny, k = idx / (nx * ny);
        i64 h = P[r];
        if (k > 0)      { C[h] = (i32)(idx - nx * ny); V[h] = -1.0; ++h; }
        if (j > 0)      { C[h] = (i32)(idx - nx);      V[h] = -1.0; ++h; }
        if (i > 0)      { C[h] = (i32)(idx - 1);       V[h] = -1.0; ++h; }
        C[h] = (i32)idx; V[h] = dia; ++h;
        if (i + 1 < nx) { C[h] = (i32)(idx + 1);       V[h] = -1.0; ++h; }
        if (j + 1 < ny) { C[h] = (i32)(idx + nx);      V[h] = -1.0; ++h; }
        if (k + 1 < nz) { C[h] = (i32)(idx + nx * ny); V[h] = -1.0; ++h; }
    }
    return py::make_tuple(ptr, col, val);
}

// Split a distributed row strip (global columns) into the square local part
// (columns renumbered to local) and the remote part over deduplicated ghost
// columns (parity: amgcl/mpi/distributed_matrix.hpp:370-430).
// Returns (lptr, lcol, lval, rptr, rcol, rval, ghost_global[i64]).
static py::tuple split_strip(i64 nloc, i64 row_beg, i64 row_end, arr<i32> ptr_a,
                             arr<i32> col_a, arr<double> val_a) {
    // [row_beg, row_end) is the LOCAL COLUMN window: equal to the row range
    // for square operators, but wider/narrower for rectangular field blocks
    // (distributed Schur) and transfer strips (cross-rank pmis).
    const i32 *ptr = ptr_a.data();
    const i32 *col = col_a.data();
    const double *val = val_a.data();

    arr<i32> lptr(nloc + 1), rptr(nloc + 1);
    i32 *LP = lptr.mutable_data();
    i32 *RP = rptr.mutable_data();
    LP[0] = RP[0] = 0;
#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < nloc; ++i) {
        i32 lc = 0, rc = 0;
        for (i32 j = ptr[i]; j < ptr[i + 1]; ++j) {
            i64 c = col[j];
            if (c >= row_beg && c < row_end) ++lc;
            else ++rc;
        }
        LP[i + 1] = lc;
        RP[i + 1] = rc;
    }
    scan_i32(LP + 1, nloc);
    scan_i32(RP + 1, nloc);
    const i64 lnnz = LP[nloc], rnnz = RP[nloc];
    arr<i32> lcol = big_arr<i32>(lnnz), rcol = big_arr<i32>(rnnz);
    arr<double> lval = big_arr<double>(lnnz), rval = big_arr<double>(rnnz);
    i32 *LC = lcol.mutable_data();
    i32 *RC = rcol.mutable_data();
    double *LV = lval.mutable_data();
    double *RV = rval.mutable_data();
    // remote columns keep GLOBAL ids for now (renumbered after dedup)
#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < nloc; ++i) {
        i32 lh = LP[i], rh = RP[i];
        for (i32 j = ptr[i]; j < ptr[i + 1]; ++j) {
            i64 c = col[j];
            if (c >= row_beg && c < row_end) {
                LC[lh] = (i32)(c - row_beg);
                LV[lh] = val[j];
                ++lh;
            } else {
                RC[rh] = col[j];
                RV[rh] = val[j];
                ++rh;
            }
        }
    }
    // dedup + sort remote global columns
    std::vector<i32> ghosts(RC, RC + rnnz);
    std::sort(ghosts.begin(), ghosts.end());
    ghosts.erase(std::unique(ghosts.begin(), ghosts.end()), ghosts.end());
    arr<int64_t> ghost_out(ghosts.size());
    int64_t *G = ghost_out.mutable_data();
    for (size_t g = 0; g < ghosts.size(); ++g) G[g] = ghosts[g];
    // renumber remote columns to ghost indices
#pragma omp parallel for schedule(static)
    for (i64 j = 0; j < rnnz; ++j) {
        RC[j] = (i32)(std::lower_bound(ghosts.begin(), ghosts.end(), RC[j]) -
                      ghosts.begin());
    }
    return py::make_tuple(lptr, lcol, lval, rptr, rcol, rval, ghost_out);
}

// ---------------------------------------------------------------------------
// diagonal
// ---------------------------------------------------------------------------
static arr<double> diagonal(i64 nrows, arr<i32> ptr, arr<i32> col, arr<double> val) {
    auto A = view(nrows, nrows, ptr, col, val);
    arr<double> d(nrows);
    double *D = d.mutable_data();
#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < nrows; ++i) {
        double v = 0;
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j)
            if (A.col[j] == i) { v = A.val[j]; break; }
        D[i] = v;
    }
    return d;
}

// ---------------------------------------------------------------------------
// transpose (parallel count via atomics, scan, scatter).
// Rows of the result come out sorted by construction when input rows are
// iterated in order and scattered with per-column cursors.
// ---------------------------------------------------------------------------
static py::tuple transpose(i64 nrows, i64 ncols, arr<i32> ptr, arr<i32> col, arr<double> val) {
    auto A = view(nrows, ncols, ptr, col, val);
    const i64 nnz = (i64)col.size();

    arr<i32> tptr(ncols + 1);
    i32 *TP = tptr.mutable_data();
    std::memset(TP, 0, sizeof(i32) * (ncols + 1));
#pragma omp parallel for schedule(static)
    for (i64 j = 0; j < nnz; ++j) {
#pragma omp atomic
        TP[A.col[j] + 1]++;
    }
    scan_i32(TP + 1, ncols);

    arr<i32> tcol = big_arr<i32>(nnz);
    arr<double> tval = big_arr<double>(nnz);
    i32 *TC = tcol.mutable_data();
    double *TV = tval.mutable_data();
    std::vector<std::atomic<i32>> cur(ncols);
#pragma omp parallel for schedule(static)
    for (i64 c = 0; c < ncols; ++c) cur[c].store(TP[c], std::memory_order_relaxed);

#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < nrows; ++i) {
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
            i32 c = A.col[j];
            i32 h = cur[c].fetch_add(1, std::memory_order_relaxed);
            TC[h] = (i32)i;
            TV[h] = A.val[j];
        }
    }
    // rows may be unsorted (threads race across i); sort each row (rows are short)
#pragma omp parallel for schedule(dynamic, 1024)
    for (i64 c = 0; c < ncols; ++c) {
        i32 b = TP[c], e = TP[c + 1];
        // insertion sort by column index
        for (i32 k = b + 1; k < e; ++k) {
            i32 ck = TC[k]; double vk = TV[k];
            i32 m = k;
            while (m > b && TC[m - 1] > ck) { TC[m] = TC[m - 1]; TV[m] = TV[m - 1]; --m; }
            TC[m] = ck; TV[m] = vk;
        }
    }
    return py::make_tuple(tptr, tcol, tval);
}

// ---------------------------------------------------------------------------
// SpGEMM, Saad row-marker algorithm (parity: amgcl/detail/spgemm.hpp:62).
// Two passes: symbolic row sizes, then numeric fill; per-thread marker array.
// Output rows sorted.
// ---------------------------------------------------------------------------
// Per-thread open-addressing accumulator for one sparse row at a time.
// Fixed KB-scale footprint (L2-resident) instead of the classic O(ncols)
// marker array — the marker costs 256 threads x ncols x 4B of first-touch
// page faults on a 2-socket EPYC and was the dominant setup cost at 512^3.
struct RowHash {
    std::vector<i32> keys;
    std::vector<double> vals;
    std::vector<i32> used;
    uint32_t mask;

    explicit RowHash(int cap_pow2 = 1 << 11)
        : keys(cap_pow2, -1), vals(cap_pow2, 0.0), mask(cap_pow2 - 1) {
        used.reserve(cap_pow2);
    }

    void ensure(i64 row_ub) {
        // keep load factor <= 1/2
        if ((i64)(mask + 1) < 2 * row_ub) {
            i64 cap = mask + 1;
            while (cap < 2 * row_ub) cap *= 2;
            keys.assign(cap, -1);
            vals.assign(cap, 0.0);
            mask = (uint32_t)cap - 1;
            used.clear();
        }
    }

    inline uint32_t slot(i32 key) const {
        uint32_t h = ((uint32_t)key * 2654435761u) & mask;
        while (keys[h] != key && keys[h] != -1) h = (h + 1) & mask;
        return h;
    }

    inline void add(i32 key, double v) {
        uint32_t h = slot(key);
        if (keys[h] == -1) {
            keys[h] = key;
            vals[h] = v;
            used.push_back((i32)h);
        } else {
            vals[h] += v;
        }
    }

    inline int count_add(i32 key) {  // symbolic: returns 1 if new
        uint32_t h = slot(key);
        if (keys[h] == -1) {
            keys[h] = key;
            used.push_back((i32)h);
            return 1;
        }
        return 0;
    }

    inline void reset() {
        for (i32 h : used) {
            keys[h] = -1;
            vals[h] = 0.0;
        }
        used.clear();
    }
};

static py::tuple spgemm(i64 an, i64 am, i64 bm,
                        arr<i32> aptr, arr<i32> acol, arr<double> aval,
                        arr<i32> bptr, arr<i32> bcol, arr<double> bval) {
    auto A = view(an, am, aptr, acol, aval);
    auto B = view(am, bm, bptr, bcol, bval);

    arr<i32> cptr(an + 1);
    i32 *CP = cptr.mutable_data();
    CP[0] = 0;

    // dense per-thread markers are faster but cost nthreads*bm*4B of
    // first-touch pages; switch to per-row hash accumulators when that
    // footprint would thrash (the 256-core EPYC regime).
    const bool use_marker =
        (i64)omp_get_max_threads() * bm * 4 < (i64(1) << 29);

#pragma omp parallel
    {
        RowHash hash;
        std::vector<i32> marker(use_marker ? bm : 0, -1);
#pragma omp for schedule(dynamic, 4096)
        for (i64 i = 0; i < an; ++i) {
            i32 cnt = 0;
            if (use_marker) {
                for (i32 ja = A.ptr[i]; ja < A.ptr[i + 1]; ++ja) {
                    i32 ca = A.col[ja];
                    for (i32 jb = B.ptr[ca]; jb < B.ptr[ca + 1]; ++jb) {
                        i32 cb = B.col[jb];
                        if (marker[cb] != (i32)i) { marker[cb] = (i32)i; ++cnt; }
                    }
                }
            } else {
                i64 ub = 0;
                for (i32 ja = A.ptr[i]; ja < A.ptr[i + 1]; ++ja) {
                    i32 ca = A.col[ja];
                    ub += B.ptr[ca + 1] - B.ptr[ca];
                }
                hash.ensure(ub);
                for (i32 ja = A.ptr[i]; ja < A.ptr[i + 1]; ++ja) {
                    i32 ca = A.col[ja];
                    for (i32 jb = B.ptr[ca]; jb < B.ptr[ca + 1]; ++jb)
                        cnt += hash.count_add(B.col[jb]);
                }
                hash.reset();
            }
            CP[i + 1] = cnt;
        }
    }
    {
        i64 chk = 0;
        for (i64 i = 0; i < an; ++i) chk += CP[i + 1];
        if (chk >= (i64(1) << 31)) throw std::runtime_error("spgemm: nnz exceeds int32");
    }
    scan_i32(CP + 1, an);
    const i64 total = CP[an];

    arr<i32> ccol = big_arr<i32>(total);
    arr<double> cval = big_arr<double>(total);
    i32 *CC = ccol.mutable_data();
    double *CV = cval.mutable_data();

#pragma omp parallel
    {
        RowHash hash;
        std::vector<i32> marker(use_marker ? bm : 0, -1);
        std::vector<i32> marker_row(use_marker ? bm : 0, -1);
#pragma omp for schedule(dynamic, 4096)
        for (i64 i = 0; i < an; ++i) {
            const i32 row_beg = CP[i];
            i32 row_end = row_beg;
            if (use_marker) {
                for (i32 ja = A.ptr[i]; ja < A.ptr[i + 1]; ++ja) {
                    i32 ca = A.col[ja];
                    double va = A.val[ja];
                    for (i32 jb = B.ptr[ca]; jb < B.ptr[ca + 1]; ++jb) {
                        i32 cb = B.col[jb];
                        double vb = B.val[jb];
                        if (marker_row[cb] != (i32)i) {
                            marker_row[cb] = (i32)i;
                            marker[cb] = row_end;
                            CC[row_end] = cb;
                            CV[row_end] = va * vb;
                            ++row_end;
                        } else {
                            CV[marker[cb]] += va * vb;
                        }
                    }
                }
            } else {
                i64 ub = 0;
                for (i32 ja = A.ptr[i]; ja < A.ptr[i + 1]; ++ja) {
                    i32 ca = A.col[ja];
                    ub += B.ptr[ca + 1] - B.ptr[ca];
                }
                hash.ensure(ub);
                for (i32 ja = A.ptr[i]; ja < A.ptr[i + 1]; ++ja) {
                    i32 ca = A.col[ja];
                    double va = A.val[ja];
                    for (i32 jb = B.ptr[ca]; jb < B.ptr[ca + 1]; ++jb)
                        hash.add(B.col[jb], va * B.val[jb]);
                }
                for (i32 h : hash.used) {
                    CC[row_end] = hash.keys[h];
                    CV[row_end] = hash.vals[h];
                    ++row_end;
                }
                hash.reset();
            }
            // sort the (short) row
            for (i32 k = row_beg + 1; k < row_end; ++k) {
                i32 ck = CC[k]; double vk = CV[k];
                i32 m = k;
                while (m > row_beg && CC[m - 1] > ck) { CC[m] = CC[m - 1]; CV[m] = CV[m - 1]; --m; }
                CC[m] = ck; CV[m] = vk;
            }
        }
    }
    return py::make_tuple(cptr, ccol, cval);
}

// parallel inclusive scan over an i32 array (blocked 3-phase)
static void scan_i32(i32 *a, i64 n) {
    if (n < (1 << 16)) {
        for (i64 i = 1; i < n; ++i) a[i] += a[i - 1];
        return;
    }
    int nt = omp_get_max_threads();
    std::vector<i64> sums(nt + 1, 0);
#pragma omp parallel num_threads(nt)
    {
        int t = omp_get_thread_num();
        i64 b = n * t / nt, e = n * (t + 1) / nt;
        i64 s = 0;
        for (i64 i = b; i < e; ++i) { s += a[i]; }
        sums[t + 1] = s;
#pragma omp barrier
#pragma omp single
        for (int k = 1; k <= nt; ++k) sums[k] += sums[k - 1];
        i64 off = sums[t];
        for (i64 i = b; i < e; ++i) { off += a[i]; a[i] = (i32)off; }
    }
}

// ---------------------------------------------------------------------------
// Parallel aggregation: distance-2 maximal-independent-set roots with 1-ring
// claiming and leftover adoption. Parallel counterpart of the reference's
// greedy pass, mirroring its own distributed PMIS design
// (amgcl/mpi/coarsening/pmis.hpp:50 — deterministic random tie-breaking).
// Deterministic regardless of thread count (hash-keyed competition).
// ---------------------------------------------------------------------------
static inline uint64_t agg_key(i32 i) {
    uint32_t x = (uint32_t)i;
    x ^= x >> 16; x *= 0x7feb352dU; x ^= x >> 15; x *= 0x846ca68bU; x ^= x >> 16;
    return ((uint64_t)x << 32) | (uint32_t)i;
}

static inline void atomic_max_u64(std::atomic<uint64_t> &a, uint64_t v) {
    uint64_t cur = a.load(std::memory_order_relaxed);
    while (cur < v && !a.compare_exchange_weak(cur, v, std::memory_order_relaxed)) {
    }
}

static py::tuple aggregates_parallel(i64 nrows, arr<i32> ptr, arr<i32> col,
                                     arr<double> val, double eps_strong) {
    auto A = view(nrows, nrows, ptr, col, val);
    const i64 nnz = (i64)col.size();
    const double eps2 = eps_strong * eps_strong;

    arr<double> dia_a = diagonal(nrows, ptr, col, val);
    const double *D = dia_a.data();

    arr<uint8_t> strong(nnz);
    uint8_t *S = strong.mutable_data();
#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < nrows; ++i) {
        double edi = eps2 * D[i];
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
            i32 c = A.col[j];
            double v = A.val[j];
            S[j] = (c != (i32)i) && (edi * D[c] < v * v);
        }
    }

    // States mirror the greedy reference: UNDEF competes to seed; FIRM nodes
    // are roots or 1-ring members; PROVISIONAL nodes (2nd ring, adjacent to
    // an aggregate) hold an adoptive id, never seed, and may be re-claimed
    // by a later root's 1-ring — exactly the greedy pass's semantics, made
    // parallel via distance-2 MIS root selection with hashed keys.
    constexpr i32 UNDEF = -1, REMOVED = -2;
    arr<i32> id_a(nrows);
    i32 *id = id_a.mutable_data();  // UNDEF / REMOVED / root-node index
    std::vector<uint8_t> prov(nrows, 0);
    std::unique_ptr<std::atomic<uint64_t>[]> m1_owner(new std::atomic<uint64_t>[nrows]);
    std::atomic<uint64_t> *m1 = m1_owner.get();
    pretouch((void *)m1, nrows * (i64)sizeof(uint64_t));

#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < nrows; ++i) {
        m1[i].store(0, std::memory_order_relaxed);
        i32 st = REMOVED;
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j)
            if (S[j]) { st = UNDEF; break; }
        id[i] = st;
    }

    std::vector<i32> active;
    active.reserve(nrows);
    for (i64 i = 0; i < nrows; ++i)
        if (id[i] == UNDEF) active.push_back((i32)i);

    std::vector<i32> touched, next_active;
    touched.reserve(active.size());

    bool m1_dirty_all = false;
    while (!active.empty()) {
        const i64 na = (i64)active.size();
        if (na > nrows / 16) {
            // pass 1, dense gather (no atomics): one sweep over all rows
#pragma omp parallel for schedule(static)
            for (i64 i = 0; i < nrows; ++i) {
                uint64_t m = (id[i] == UNDEF) ? agg_key((i32)i) : 0;
                for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
                    if (!S[j]) continue;
                    i32 c = A.col[j];
                    if (id[c] == UNDEF) {
                        uint64_t k = agg_key(c);
                        if (k > m) m = k;
                    }
                }
                m1[i].store(m, std::memory_order_relaxed);
            }
            m1_dirty_all = true;
            touched.clear();
        } else {
            // clear stale m1, then sparse scatter from the active set
            if (m1_dirty_all) {
#pragma omp parallel for schedule(static)
                for (i64 i = 0; i < nrows; ++i)
                    m1[i].store(0, std::memory_order_relaxed);
                m1_dirty_all = false;
                touched.clear();
            } else {
#pragma omp parallel for schedule(static)
                for (i64 t = 0; t < (i64)touched.size(); ++t)
                    m1[touched[t]].store(0, std::memory_order_relaxed);
                touched.clear();
            }
#pragma omp parallel
            {
                std::vector<i32> local_touched;
#pragma omp for schedule(static) nowait
                for (i64 t = 0; t < na; ++t) {
                    i32 u = active[t];
                    uint64_t k = agg_key(u);
                    atomic_max_u64(m1[u], k);
                    local_touched.push_back(u);
                    for (i32 j = A.ptr[u]; j < A.ptr[u + 1]; ++j) {
                        if (!S[j]) continue;
                        i32 c = A.col[j];
                        atomic_max_u64(m1[c], k);
                        local_touched.push_back(c);
                    }
                }
#pragma omp critical
                touched.insert(touched.end(), local_touched.begin(), local_touched.end());
            }
        }

        // pass 2: u is a root iff its key is the max within distance 2 of
        // the UNDEF subgraph (propagated through any intermediate node).
#pragma omp parallel for schedule(static)
        for (i64 t = 0; t < na; ++t) {
            i32 u = active[t];
            uint64_t key = agg_key(u);
            uint64_t m2 = m1[u].load(std::memory_order_relaxed);
            for (i32 j = A.ptr[u]; j < A.ptr[u + 1] && m2 <= key; ++j) {
                if (!S[j]) continue;
                uint64_t v = m1[A.col[j]].load(std::memory_order_relaxed);
                if (v > m2) m2 = v;
            }
            if (m2 == key) id[u] = u;  // new root (FIRM)
        }

        // pass 3: new roots claim their strong 1-ring FIRM, overwriting
        // provisional members (greedy: "later claimed by other aggregates").
        // With a *symmetric* strength mask two new roots are never within
        // distance 2 and claims cannot conflict; the mask is row-wise, so for
        // nonsymmetric matrices two roots CAN claim the same node.  Resolve
        // claims deterministically with an atomic max on the root key
        // (m1 is free after pass 2): clear -> max -> commit, so the winner is
        // independent of thread count / schedule.
#pragma omp parallel for schedule(static)
        for (i64 t = 0; t < na; ++t) {
            i32 u = active[t];
            if (id[u] != u) continue;
            for (i32 j = A.ptr[u]; j < A.ptr[u + 1]; ++j) {
                i32 c = A.col[j];
                if (!S[j]) continue;
                if (id[c] == UNDEF || prov[c]) m1[c].store(0, std::memory_order_relaxed);
            }
        }
#pragma omp parallel for schedule(static)
        for (i64 t = 0; t < na; ++t) {
            i32 u = active[t];
            if (id[u] != u) continue;
            uint64_t key = agg_key(u);
            for (i32 j = A.ptr[u]; j < A.ptr[u + 1]; ++j) {
                i32 c = A.col[j];
                if (!S[j]) continue;
                if (id[c] == UNDEF || prov[c]) atomic_max_u64(m1[c], key);
            }
        }
#pragma omp parallel for schedule(static)
        for (i64 t = 0; t < na; ++t) {
            i32 u = active[t];
            if (id[u] != u) continue;
            uint64_t key = agg_key(u);
            for (i32 j = A.ptr[u]; j < A.ptr[u + 1]; ++j) {
                i32 c = A.col[j];
                if (!S[j]) continue;
                if ((id[c] == UNDEF || prov[c]) &&
                    m1[c].load(std::memory_order_relaxed) == key) {
                    id[c] = u;
                    prov[c] = 0;
                }
            }
        }

        // pass 4 (two-phase): remaining UNDEF nodes adjacent to a FIRM member
        // become provisional members (greedy 2nd-ring claim). The mark phase
        // reads the stable post-claim state; the commit phase applies it —
        // a fused pass would race with neighbors turning provisional.
        std::vector<i32> choice(na);
#pragma omp parallel for schedule(static)
        for (i64 t = 0; t < na; ++t) {
            i32 u = active[t];
            i32 root = -1;
            if (id[u] == UNDEF) {
                uint64_t best = 0;
                for (i32 j = A.ptr[u]; j < A.ptr[u + 1]; ++j) {
                    i32 c = A.col[j];
                    if (!S[j] || c == u) continue;
                    if (id[c] >= 0 && !prov[c]) {
                        uint64_t k = agg_key(id[c]);
                        if (k > best) { best = k; root = id[c]; }
                    }
                }
            }
            choice[t] = root;
        }
        next_active.clear();
#pragma omp parallel
        {
            std::vector<i32> local_next;
#pragma omp for schedule(static) nowait
            for (i64 t = 0; t < na; ++t) {
                i32 u = active[t];
                if (id[u] != UNDEF) continue;
                if (choice[t] >= 0) {
                    id[u] = choice[t];
                    prov[u] = 1;
                } else {
                    local_next.push_back(u);
                }
            }
#pragma omp critical
            next_active.insert(next_active.end(), local_next.begin(), local_next.end());
        }
        std::swap(active, next_active);
    }

    // renumber root nodes to compact aggregate ids (deterministic: node order)
    std::vector<i32> mark(nrows, 0);
#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < nrows; ++i)
        if (id[i] == (i32)i) mark[i] = 1;
    scan_i32(mark.data(), nrows);  // inclusive
    i64 count = mark[nrows - 1];
    if (!count) throw std::runtime_error("empty level in aggregation");
#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < nrows; ++i)
        if (id[i] >= 0) id[i] = mark[id[i]] - 1;

    return py::make_tuple((i64)count, id_a, strong);
}

// ---------------------------------------------------------------------------
// Greedy aggregation (parity: amgcl/coarsening/plain_aggregates.hpp:114-207).
// Strong coupling: i!=j and eps^2*|a_ii*a_jj| < a_ij^2.
// Returns (naggr, id[i32] with -1 never present after renumber; removed = -2),
// plus the strong-connection mask aligned with A's nonzeros.
// ---------------------------------------------------------------------------
static py::tuple aggregates(i64 nrows, arr<i32> ptr, arr<i32> col, arr<double> val,
                            double eps_strong) {
    auto A = view(nrows, nrows, ptr, col, val);
    const i64 nnz = (i64)col.size();
    const double eps2 = eps_strong * eps_strong;

    arr<double> dia_a = diagonal(nrows, ptr, col, val);
    const double *D = dia_a.data();

    arr<uint8_t> strong(nnz);
    uint8_t *S = strong.mutable_data();
#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < nrows; ++i) {
        double edi = eps2 * D[i];
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
            i32 c = A.col[j];
            double v = A.val[j];
            S[j] = (c != (i32)i) && (edi * D[c] < v * v);
        }
    }

    constexpr i32 UNDEF = -1, REMOVED = -2;
    arr<i32> id_a(nrows);
    i32 *id = id_a.mutable_data();
#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < nrows; ++i) {
        i32 st = REMOVED;
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j)
            if (S[j]) { st = UNDEF; break; }
        id[i] = st;
    }

    // serial greedy pass with neighbor expansion
    i64 count = 0;
    std::vector<i32> neib;
    for (i64 i = 0; i < nrows; ++i) {
        if (id[i] != UNDEF) continue;
        i32 cur = (i32)count++;
        id[i] = cur;
        neib.clear();
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
            i32 c = A.col[j];
            if (S[j] && id[c] != REMOVED) { id[c] = cur; neib.push_back(c); }
        }
        // provisionally claim second-ring undefined points
        for (i32 c : neib)
            for (i32 j = A.ptr[c]; j < A.ptr[c + 1]; ++j) {
                i32 cc = A.col[j];
                if (S[j] && id[cc] == UNDEF) id[cc] = cur;
            }
    }
    if (!count) throw std::runtime_error("empty level in aggregation");

    // renumber: drop aggregates that vanished during expansion
    std::vector<i32> cnt(count, 0);
    for (i64 i = 0; i < nrows; ++i)
        if (id[i] >= 0) cnt[id[i]] = 1;
    for (i64 a = 1; a < (i64)count; ++a) cnt[a] += cnt[a - 1];
    if ((i64)count > cnt[count - 1]) {
        count = cnt[count - 1];
#pragma omp parallel for schedule(static)
        for (i64 i = 0; i < nrows; ++i)
            if (id[i] >= 0) id[i] = cnt[id[i]] - 1;
    }
    return py::make_tuple((i64)count, id_a, strong);
}

// ---------------------------------------------------------------------------
// Smoothed-aggregation prolongation, fused tentative+smoothing for the
// piecewise-constant (no nullspace) case
// (parity: amgcl/coarsening/smoothed_aggregation.hpp:157-232).
//   P = (I - omega * Df^-1 * Af) * P_tent
// where Af keeps strong off-diagonal entries and folds weak ones into the
// diagonal. Rows sorted.
// ---------------------------------------------------------------------------
static py::tuple smoothed_prolongation(i64 nrows, arr<i32> ptr, arr<i32> col, arr<double> val,
                                       arr<uint8_t> strong, arr<i32> id_a, i64 naggr,
                                       double omega) {
    auto A = view(nrows, nrows, ptr, col, val);
    const uint8_t *S = strong.data();
    const i32 *id = id_a.data();

    arr<i32> pptr(nrows + 1);
    i32 *PP = pptr.mutable_data();
    PP[0] = 0;

#pragma omp parallel
    {
        RowHash hash(256);
#pragma omp for schedule(static)
        for (i64 i = 0; i < nrows; ++i) {
            i64 row_len = A.ptr[i + 1] - A.ptr[i];
            hash.ensure(row_len);
            i32 cnt = 0;
            for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
                i32 c = A.col[j];
                if (c != (i32)i && !S[j]) continue;  // weak off-diagonal: skipped
                i32 a = id[c];
                if (a >= 0) cnt += hash.count_add(a);
            }
            hash.reset();
            PP[i + 1] = cnt;
        }
    }
    scan_i32(PP + 1, nrows);
    const i64 total = PP[nrows];

    arr<i32> pcol = big_arr<i32>(total);
    arr<double> pval = big_arr<double>(total);
    i32 *PC = pcol.mutable_data();
    double *PV = pval.mutable_data();

#pragma omp parallel
    {
        RowHash hash(256);
#pragma omp for schedule(static)
        for (i64 i = 0; i < nrows; ++i) {
            // filtered diagonal: original diagonal plus weak off-diagonal values
            double dia = 0;
            for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j)
                if (A.col[j] == (i32)i || !S[j]) dia += A.val[j];
            if (dia != 0.0) dia = -omega / dia;

            hash.ensure(A.ptr[i + 1] - A.ptr[i]);
            for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
                i32 c = A.col[j];
                if (c != (i32)i && !S[j]) continue;
                i32 a = id[c];
                if (a < 0) continue;
                double v = (c == (i32)i) ? (1.0 - omega) : dia * A.val[j];
                hash.add(a, v);
            }
            const i32 row_beg = PP[i];
            i32 row_end = row_beg;
            for (i32 h : hash.used) {
                PC[row_end] = hash.keys[h];
                PV[row_end] = hash.vals[h];
                ++row_end;
            }
            hash.reset();
            for (i32 k = row_beg + 1; k < row_end; ++k) {
                i32 ck = PC[k]; double vk = PV[k];
                i32 m = k;
                while (m > row_beg && PC[m - 1] > ck) { PC[m] = PC[m - 1]; PV[m] = PV[m - 1]; --m; }
                PC[m] = ck; PV[m] = vk;
            }
        }
    }
    return py::make_tuple(pptr, pcol, pval);
}

// Tentative (unsmoothed) prolongation, piecewise constant
// (parity: amgcl/coarsening/tentative_prolongation.hpp:208-228).
static py::tuple tentative_prolongation(i64 nrows, arr<i32> id_a, i64 naggr) {
    const i32 *id = id_a.data();
    arr<i32> pptr(nrows + 1);
    i32 *PP = pptr.mutable_data();
    PP[0] = 0;
    i64 total = 0;
    for (i64 i = 0; i < nrows; ++i) { total += (id[i] >= 0); PP[i + 1] = (i32)total; }
    arr<i32> pcol(total);
    arr<double> pval(total);
    i32 *PC = pcol.mutable_data();
    double *PV = pval.mutable_data();
#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < nrows; ++i)
        if (id[i] >= 0) { PC[PP[i]] = id[i]; PV[PP[i]] = 1.0; }
    (void)naggr;
    return py::make_tuple(pptr, pcol, pval);
}

// ---------------------------------------------------------------------------
// Pointwise (block) condensation for vector problems
// (parity: amgcl/backend/builtin.hpp:505 pointwise_matrix +
// amgcl/coarsening/pointwise_aggregates.hpp:85): condense the scalar matrix
// with node-major interleaved dofs (block_size unknowns per point) to a
// point matrix whose values are Frobenius norms of the blocks; aggregation
// then runs on points and ids/strength expand back to unknowns.
// ---------------------------------------------------------------------------
static py::tuple pointwise_matrix(i64 n, arr<i32> ptr_a, arr<i32> col_a,
                                  arr<double> val_a, i64 b) {
    auto A = view(n, n, ptr_a, col_a, val_a);
    if (n % b) throw std::runtime_error("pointwise: size not divisible by block");
    const i64 np_ = n / b;
    arr<i32> pptr(np_ + 1);
    i32 *PP = pptr.mutable_data();
    PP[0] = 0;
#pragma omp parallel
    {
        RowHash hash(256);
#pragma omp for schedule(static)
        for (i64 p = 0; p < np_; ++p) {
            i64 ub = 0;
            for (i64 r = p * b; r < (p + 1) * b; ++r) ub += A.ptr[r + 1] - A.ptr[r];
            hash.ensure(ub);
            i32 cnt = 0;
            for (i64 r = p * b; r < (p + 1) * b; ++r)
                for (i32 j = A.ptr[r]; j < A.ptr[r + 1]; ++j)
                    cnt += hash.count_add(A.col[j] / (i32)b);
            hash.reset();
            PP[p + 1] = cnt;
        }
    }
    scan_i32(PP + 1, np_);
    arr<i32> pcol = big_arr<i32>(PP[np_]);
    arr<double> pval = big_arr<double>(PP[np_]);
    i32 *PC = pcol.mutable_data();
    double *PV = pval.mutable_data();
#pragma omp parallel
    {
        RowHash hash(256);
#pragma omp for schedule(static)
        for (i64 p = 0; p < np_; ++p) {
            i64 ub = 0;
            for (i64 r = p * b; r < (p + 1) * b; ++r) ub += A.ptr[r + 1] - A.ptr[r];
            hash.ensure(ub);
            for (i64 r = p * b; r < (p + 1) * b; ++r)
                for (i32 j = A.ptr[r]; j < A.ptr[r + 1]; ++j)
                    hash.add(A.col[j] / (i32)b, A.val[j] * A.val[j]);
            i32 head = PP[p];
            for (i32 h : hash.used) {
                PC[head] = hash.keys[h];
                PV[head] = std::sqrt(hash.vals[h]);  // Frobenius norm of block
                ++head;
            }
            hash.reset();
            // sort row
            i32 beg = PP[p];
            for (i32 k2 = beg + 1; k2 < head; ++k2) {
                i32 ck = PC[k2];
                double vk = PV[k2];
                i32 m = k2;
                while (m > beg && PC[m - 1] > ck) {
                    PC[m] = PC[m - 1];
                    PV[m] = PV[m - 1];
                    --m;
                }
                PC[m] = ck;
                PV[m] = vk;
            }
        }
    }
    return py::make_tuple(pptr, pcol, pval);
}

// Greedy-parallel graph coloring (Jones-Plassmann MIS rounds with the same
// deterministic hash keys as the aggregation). Enables multicolor
// Gauss-Seidel sweeps on the GPU (rows of one color are independent).
static py::tuple color_graph(i64 n, arr<i32> ptr_a, arr<i32> col_a) {
    const i32 *ptr = ptr_a.data();
    const i32 *col = col_a.data();
    arr<i32> colors_a(n);
    i32 *colors = colors_a.mutable_data();
    std::fill(colors, colors + n, -1);
    std::vector<uint8_t> win(n);
    i64 remaining = n;
    i32 round = 0;
    while (remaining > 0) {
        if (round > 4096) throw std::runtime_error("coloring did not converge");
#pragma omp parallel for schedule(static)
        for (i64 i = 0; i < n; ++i) {
            win[i] = 0;
            if (colors[i] >= 0) continue;
            uint64_t k = agg_key((i32)i);
            bool best = true;
            for (i32 j = ptr[i]; j < ptr[i + 1]; ++j) {
                i32 c = col[j];
                if (c == (i64)i || colors[c] >= 0) continue;
                if (agg_key(c) > k) { best = false; break; }
            }
            win[i] = best;
        }
        i64 done = 0;
#pragma omp parallel for schedule(static) reduction(+ : done)
        for (i64 i = 0; i < n; ++i)
            if (win[i]) {
                colors[i] = round;
                ++done;
            }
        remaining -= done;
        ++round;
    }
    return py::make_tuple(colors_a, (i64)round);
}

// Scalar CSR -> BSR (block CSR) with zero fill
// (parity: amgcl/adapter/block_matrix.hpp:44 view + builtin_hybrid storage).
// Blocks row-major, block columns sorted.
static py::tuple csr_to_bsr(i64 n, arr<i32> ptr_a, arr<i32> col_a, arr<double> val_a,
                            i64 b) {
    auto A = view(n, n, ptr_a, col_a, val_a);
    if (n % b) throw std::runtime_error("csr_to_bsr: size not divisible by block");
    const i64 nb = n / b;
    arr<i32> bptr(nb + 1);
    i32 *BP = bptr.mutable_data();
    BP[0] = 0;
#pragma omp parallel
    {
        RowHash hash(256);
#pragma omp for schedule(static)
        for (i64 p = 0; p < nb; ++p) {
            i64 ub = 0;
            for (i64 r = p * b; r < (p + 1) * b; ++r) ub += A.ptr[r + 1] - A.ptr[r];
            hash.ensure(ub);
            i32 cnt = 0;
            for (i64 r = p * b; r < (p + 1) * b; ++r)
                for (i32 j = A.ptr[r]; j < A.ptr[r + 1]; ++j)
                    cnt += hash.count_add(A.col[j] / (i32)b);
            hash.reset();
            BP[p + 1] = cnt;
        }
    }
    scan_i32(BP + 1, nb);
    const i64 nnzb = BP[nb];
    arr<i32> bcol = big_arr<i32>(nnzb);
    arr<double> bval = big_arr<double>(nnzb * b * b);
    i32 *BC = bcol.mutable_data();
    double *BV = bval.mutable_data();
    pretouch(BV, nnzb * b * b * 8);
#pragma omp parallel
    {
        RowHash hash(256);
#pragma omp for schedule(static)
        for (i64 p = 0; p < nb; ++p) {
            // collect sorted distinct block columns
            i64 ub = 0;
            for (i64 r = p * b; r < (p + 1) * b; ++r) ub += A.ptr[r + 1] - A.ptr[r];
            hash.ensure(ub);
            i32 head = BP[p];
            i32 cnt = 0;
            for (i64 r = p * b; r < (p + 1) * b; ++r)
                for (i32 j = A.ptr[r]; j < A.ptr[r + 1]; ++j)
                    if (hash.count_add(A.col[j] / (i32)b)) BC[head + cnt++] = A.col[j] / (i32)b;
            hash.reset();
            std::sort(BC + head, BC + head + cnt);
            for (i64 k = (i64)head * b * b; k < (i64)(head + cnt) * b * b; ++k) BV[k] = 0.0;
            // scatter values: binary search the block column
            for (i64 r = p * b; r < (p + 1) * b; ++r) {
                int lr = (int)(r - p * b);
                for (i32 j = A.ptr[r]; j < A.ptr[r + 1]; ++j) {
                    i32 q = A.col[j] / (i32)b;
                    int lc = (int)(A.col[j] - (i64)q * b);
                    i32 lo = head, hi = head + cnt - 1, pos = -1;
                    while (lo <= hi) {
                        i32 mid = (lo + hi) / 2;
                        if (BC[mid] == q) { pos = mid; break; }
                        if (BC[mid] < q) lo = mid + 1;
                        else hi = mid - 1;
                    }
                    BV[(i64)pos * b * b + lr * b + lc] = A.val[j];
                }
            }
        }
    }
    return py::make_tuple(bptr, bcol, bval);
}

// Expand a point-level strong mask to the scalar entries: scalar entry (i,j)
// is strong iff the point pair (i/b, col/b) is strong in the point matrix.
static arr<uint8_t> expand_strong(i64 n, arr<i32> ptr_a, arr<i32> col_a, i64 b,
                                  arr<i32> pptr_a, arr<i32> pcol_a,
                                  arr<uint8_t> pstrong_a) {
    const i32 *ptr = ptr_a.data();
    const i32 *col = col_a.data();
    const i32 *pp = pptr_a.data();
    const i32 *pc = pcol_a.data();
    const uint8_t *ps = pstrong_a.data();
    const i64 nnz = (i64)col_a.size();
    arr<uint8_t> S(nnz);
    uint8_t *Sv = S.mutable_data();
#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < n; ++i) {
        i32 p = (i32)(i / b);
        for (i32 j = ptr[i]; j < ptr[i + 1]; ++j) {
            i32 q = col[j] / (i32)b;
            uint8_t st = 0;
            if (q != p) {
                // binary search q in point row p (sorted)
                i32 lo = pp[p], hi = pp[p + 1] - 1;
                while (lo <= hi) {
                    i32 mid = (lo + hi) / 2;
                    if (pc[mid] == q) { st = ps[mid]; break; }
                    if (pc[mid] < q) lo = mid + 1;
                    else hi = mid - 1;
                }
            }
            Sv[j] = st;
        }
    }
    return S;
}

// ---------------------------------------------------------------------------
// Nullspace-aware tentative prolongation
// (parity: amgcl/coarsening/tentative_prolongation.hpp:120-207 — per
// aggregate, a thin QR of the near-nullspace block B gives the P block (Q)
// and the coarse-level nullspace (R)). Modified Gram-Schmidt with one
// re-orthogonalization pass (k <= ~6 columns).
// Returns (Pptr, Pcol, Pval, Bnew) with P: n x (naggr*k).
// ---------------------------------------------------------------------------
static py::tuple tentative_nullspace(i64 n, arr<i32> id_a, i64 naggr,
                                     arr<double> B_a, i64 k) {
    const i32 *id = id_a.data();
    const double *B = B_a.data();  // n x k row-major

    // group rows by aggregate
    std::vector<i32> acnt(naggr + 1, 0);
    for (i64 i = 0; i < n; ++i)
        if (id[i] >= 0) ++acnt[id[i] + 1];
    for (i64 a = 0; a < naggr; ++a) acnt[a + 1] += acnt[a];
    std::vector<i32> members(acnt[naggr]);
    {
        std::vector<i32> cur(acnt.begin(), acnt.end() - 1);
        for (i64 i = 0; i < n; ++i)
            if (id[i] >= 0) members[cur[id[i]]++] = (i32)i;
    }

    arr<i32> pptr(n + 1);
    i32 *PP = pptr.mutable_data();
    PP[0] = 0;
#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < n; ++i) PP[i + 1] = (id[i] >= 0) ? (i32)k : 0;
    scan_i32(PP + 1, n);
    const i64 pnz = PP[n];
    arr<i32> pcol = big_arr<i32>(pnz);
    arr<double> pval = big_arr<double>(pnz);
    i32 *PC = pcol.mutable_data();
    double *PV = pval.mutable_data();
    arr<double> Bnew_a(naggr * k * k);
    double *Bnew = Bnew_a.mutable_data();

#pragma omp parallel
    {
        std::vector<double> Q;  // d x k column-major
#pragma omp for schedule(dynamic, 256)
        for (i64 a = 0; a < naggr; ++a) {
            i64 beg = acnt[a], end = acnt[a + 1];
            i64 d = end - beg;
            Q.assign(d * k, 0.0);
            for (i64 r = 0; r < d; ++r)
                for (i64 c = 0; c < (i64)k; ++c)
                    Q[c * d + r] = B[(i64)members[beg + r] * k + c];
            double *Rm = Bnew + a * k * k;  // k x k row-major
            std::fill(Rm, Rm + k * k, 0.0);
            // MGS with one re-orthogonalization
            for (i64 c = 0; c < (i64)k; ++c) {
                double *qc = &Q[c * d];
                for (int pass = 0; pass < 2; ++pass) {
                    for (i64 p = 0; p < c; ++p) {
                        const double *qp = &Q[p * d];
                        double h = 0;
                        for (i64 r = 0; r < d; ++r) h += qp[r] * qc[r];
                        for (i64 r = 0; r < d; ++r) qc[r] -= h * qp[r];
                        Rm[p * k + c] += h;
                    }
                }
                double nrm = 0;
                for (i64 r = 0; r < d; ++r) nrm += qc[r] * qc[r];
                nrm = std::sqrt(nrm);
                Rm[c * k + c] = nrm;
                if (nrm > 1e-300)
                    for (i64 r = 0; r < d; ++r) qc[r] /= nrm;
            }
            // write P rows
            for (i64 r = 0; r < d; ++r) {
                i32 row = members[beg + r];
                i32 head = PP[row];
                for (i64 c = 0; c < (i64)k; ++c) {
                    PC[head + c] = (i32)(a * k + c);
                    PV[head + c] = Q[c * d + r];
                }
            }
        }
    }
    return py::make_tuple(pptr, pcol, pval, Bnew_a);
}

// Filtered Jacobi smoother matrix S_F = I - omega * Df^-1 * Af as explicit
// CSR (strong off-diagonals kept, weak ones folded into the diagonal) —
// used to smooth a general (nullspace) tentative P via spgemm
// (parity: the implicit smoothing pass of smoothed_aggregation.hpp:157-232).
static py::tuple filtered_smoother_matrix(i64 n, arr<i32> ptr_a, arr<i32> col_a,
                                          arr<double> val_a, arr<uint8_t> strong_a,
                                          double omega) {
    auto A = view(n, n, ptr_a, col_a, val_a);
    const uint8_t *S = strong_a.data();
    arr<i32> sptr(n + 1);
    i32 *SP = sptr.mutable_data();
    SP[0] = 0;
#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < n; ++i) {
        i32 cnt = 0;
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j)
            if (A.col[j] == (i32)i || S[j]) ++cnt;
        // ensure a diagonal entry even if A has none
        bool has_dia = false;
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j)
            if (A.col[j] == (i32)i) { has_dia = true; break; }
        if (!has_dia) ++cnt;
        SP[i + 1] = cnt;
    }
    scan_i32(SP + 1, n);
    arr<i32> scol = big_arr<i32>(SP[n]);
    arr<double> sval = big_arr<double>(SP[n]);
    i32 *SC = scol.mutable_data();
    double *SV = sval.mutable_data();
#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < n; ++i) {
        double dia = 0;
        bool has_dia = false;
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
            if (A.col[j] == (i32)i) has_dia = true;
            if (A.col[j] == (i32)i || !S[j]) dia += A.val[j];
        }
        double w = (dia != 0.0) ? -omega / dia : 0.0;
        i32 head = SP[i];
        bool wrote_dia = false;
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
            i32 c = A.col[j];
            if (c == (i32)i) {
                SC[head] = c;
                SV[head] = 1.0 - omega;
                wrote_dia = true;
                ++head;
            } else if (S[j]) {
                SC[head] = c;
                SV[head] = w * A.val[j];
                ++head;
            }
        }
        if (!has_dia && !wrote_dia) {
            SC[head] = (i32)i;
            SV[head] = 1.0 - omega;
        }
    }
    return py::make_tuple(sptr, scol, sval);
}

// ---------------------------------------------------------------------------
// Ruge-Stuben (classic) coarsening
// (parity: amgcl/coarsening/ruge_stuben.hpp:54-458 — strong negative
// couplings -a_ij >= eps*max|a_ik^-|, standard lambda-bucket C/F splitting,
// direct interpolation with optional truncation+rescaling).
// Returns (Pptr, Pcol, Pval, nc).
// ---------------------------------------------------------------------------
static py::tuple ruge_stuben(i64 n, arr<i32> ptr_a, arr<i32> col_a, arr<double> val_a,
                             double eps_strong, bool do_trunc, double eps_trunc) {
    auto A = view(n, n, ptr_a, col_a, val_a);
    const i64 nnz = (i64)col_a.size();
    constexpr double tiny = 1e-300;

    // strong connections: S[j] = (col != i) && (a_ij < eps * min_k a_ik < 0)
    std::vector<uint8_t> S(nnz, 0);
    std::vector<char> cf(n, 'U');
#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < n; ++i) {
        double amin = 0.0;
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j)
            if (A.col[j] != (i32)i && A.val[j] < amin) amin = A.val[j];
        if (amin > -tiny) {
            cf[i] = 'F';  // no negative couplings
            continue;
        }
        amin *= eps_strong;
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j)
            S[j] = (A.col[j] != (i32)i) && (A.val[j] < amin);
    }

    // transpose of the strong graph (S^T: who strongly depends on i)
    std::vector<i32> tptr(n + 1, 0), tcol;
    for (i64 j = 0; j < nnz; ++j)
        if (S[j]) ++tptr[A.col[j] + 1];
    for (i64 i = 0; i < n; ++i) tptr[i + 1] += tptr[i];
    tcol.resize(tptr[n]);
    {
        std::vector<i32> cur(tptr.begin(), tptr.end() - 1);
        for (i64 i = 0; i < n; ++i)
            for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j)
                if (S[j]) tcol[cur[A.col[j]]++] = (i32)i;
    }

    // lambda-bucket C/F split (standard RS first pass)
    std::vector<i64> lambda(n);
    for (i64 i = 0; i < n; ++i) {
        i64 t = 0;
        for (i32 j = tptr[i]; j < tptr[i + 1]; ++j)
            t += (cf[tcol[j]] == 'U') ? 1 : 2;
        lambda[i] = t;
    }
    std::vector<i64> bptr(n + 2, 0), bcnt(n + 1, 0), i2n(n), n2i(n);
    for (i64 i = 0; i < n; ++i) ++bptr[lambda[i] + 1];
    for (i64 i = 0; i <= n; ++i) bptr[i + 1] += bptr[i];
    for (i64 i = 0; i < n; ++i) {
        i64 lam = lambda[i];
        i64 idx = bptr[lam] + bcnt[lam]++;
        i2n[idx] = i;
        n2i[i] = idx;
    }
    auto bucket_move = [&](i64 node, i64 from, i64 to_pos) {
        i64 old_pos = n2i[node];
        n2i[i2n[old_pos]] = to_pos;
        n2i[i2n[to_pos]] = old_pos;
        std::swap(i2n[old_pos], i2n[to_pos]);
        (void)from;
    };
    for (i64 top = n; top-- > 0;) {
        i64 i = i2n[top];
        i64 lam = lambda[i];
        if (lam == 0) {
            for (i64 k = 0; k < n; ++k)
                if (cf[k] == 'U') cf[k] = 'C';
            break;
        }
        --bcnt[lam];
        if (cf[i] == 'F') continue;
        cf[i] = 'C';
        for (i32 j = tptr[i]; j < tptr[i + 1]; ++j) {
            i32 c = tcol[j];
            if (cf[c] != 'U') continue;
            cf[c] = 'F';
            // bump lambdas of the new F's strong neighbours
            for (i32 aj = A.ptr[c]; aj < A.ptr[c + 1]; ++aj) {
                if (!S[aj]) continue;
                i32 ac = A.col[aj];
                i64 lam_a = lambda[ac];
                if (cf[ac] != 'U' || lam_a + 1 >= n) continue;
                bucket_move(ac, lam_a, bptr[lam_a] + bcnt[lam_a] - 1);
                --bcnt[lam_a];
                ++bcnt[lam_a + 1];
                bptr[lam_a + 1] = bptr[lam_a] + bcnt[lam_a];
                lambda[ac] = lam_a + 1;
            }
        }
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
            if (!S[j]) continue;
            i32 c = A.col[j];
            i64 lam_c = lambda[c];
            if (cf[c] != 'U' || lam_c == 0) continue;
            bucket_move(c, lam_c, bptr[lam_c]);
            --bcnt[lam_c];
            ++bcnt[lam_c - 1];
            ++bptr[lam_c];
            lambda[c] = lam_c - 1;
        }
    }

    // coarse index
    std::vector<i32> cidx(n, -1);
    i64 nc = 0;
    for (i64 i = 0; i < n; ++i)
        if (cf[i] == 'C') cidx[i] = (i32)nc++;
    if (!nc) throw std::runtime_error("empty level in ruge_stuben");

    // direct interpolation with optional truncation
    arr<i32> pptr(n + 1);
    i32 *PP = pptr.mutable_data();
    PP[0] = 0;
    std::vector<double> Amin(do_trunc ? n : 0), Amax(do_trunc ? n : 0);
#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < n; ++i) {
        i32 cnt = 0;
        if (cf[i] == 'C') {
            cnt = 1;
        } else if (do_trunc) {
            double amin = 0, amax = 0;
            for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
                if (!S[j] || cf[A.col[j]] != 'C') continue;
                amin = std::min(amin, A.val[j]);
                amax = std::max(amax, A.val[j]);
            }
            Amin[i] = amin * eps_trunc;
            Amax[i] = amax * eps_trunc;
            for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
                if (!S[j] || cf[A.col[j]] != 'C') continue;
                if (A.val[j] < Amin[i] || A.val[j] > Amax[i]) ++cnt;
            }
        } else {
            for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j)
                if (S[j] && cf[A.col[j]] == 'C') ++cnt;
        }
        PP[i + 1] = cnt;
    }
    scan_i32(PP + 1, n);
    const i64 pnz = PP[n];
    arr<i32> pcol = big_arr<i32>(pnz);
    arr<double> pval = big_arr<double>(pnz);
    i32 *PC = pcol.mutable_data();
    double *PV = pval.mutable_data();

#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < n; ++i) {
        i32 head = PP[i];
        if (cf[i] == 'C') {
            PC[head] = cidx[i];
            PV[head] = 1.0;
            continue;
        }
        double dia = 0, a_num = 0, a_den = 0, b_num = 0, b_den = 0, d_neg = 0, d_pos = 0;
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
            i32 c = A.col[j];
            double v = A.val[j];
            if (c == (i32)i) {
                dia = v;
                continue;
            }
            if (v < 0) {
                a_num += v;
                if (S[j] && cf[c] == 'C') {
                    a_den += v;
                    if (do_trunc && Amin[i] < v) d_neg += v;
                }
            } else {
                b_num += v;
                if (S[j] && cf[c] == 'C') {
                    b_den += v;
                    if (do_trunc && v < Amax[i]) d_pos += v;
                }
            }
        }
        double cf_neg = 1, cf_pos = 1;
        if (do_trunc) {
            if (std::abs(a_den - d_neg) > tiny) cf_neg = std::abs(a_den) / std::abs(a_den - d_neg);
            if (std::abs(b_den - d_pos) > tiny) cf_pos = std::abs(b_den) / std::abs(b_den - d_pos);
        }
        if (b_num > 0 && std::abs(b_den) < tiny) dia += b_num;
        double alpha = std::abs(a_den) > tiny
                           ? -cf_neg * std::abs(a_num) / (std::abs(dia) * std::abs(a_den))
                           : 0.0;
        double beta = std::abs(b_den) > tiny
                          ? -cf_pos * std::abs(b_num) / (std::abs(dia) * std::abs(b_den))
                          : 0.0;
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
            i32 c = A.col[j];
            double v = A.val[j];
            if (!S[j] || cf[c] != 'C') continue;
            if (do_trunc && Amin[i] <= v && v <= Amax[i]) continue;
            PC[head] = cidx[c];
            PV[head] = (v < 0 ? alpha : beta) * v;
            ++head;
        }
    }
    return py::make_tuple(pptr, pcol, pval, nc);
}

// ---------------------------------------------------------------------------
// SPAI-0 weights (parity: amgcl/relaxation/spai0.hpp:66-77): m_i = a_ii / sum_j a_ij^2
// ---------------------------------------------------------------------------
static arr<double> spai0(i64 nrows, arr<i32> ptr, arr<i32> col, arr<double> val) {
    auto A = view(nrows, nrows, ptr, col, val);
    arr<double> m(nrows);
    double *M = m.mutable_data();
#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < nrows; ++i) {
        double num = 0, den = 0;
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
            double v = A.val[j];
            den += v * v;
            if (A.col[j] == (i32)i) num = v;
        }
        M[i] = den > 0 ? num / den : 0.0;
    }
    return m;
}

// ---------------------------------------------------------------------------
// Solve-phase reference primitives (OpenMP). These define the numerics the
// HIP kernels are tested against. y = alpha*A*x + beta*y.
// ---------------------------------------------------------------------------
static void spmv(double alpha, i64 nrows, arr<i32> ptr, arr<i32> col, arr<double> val,
                 arr<double> x, double beta, arr<double> y) {
    auto A = view(nrows, 0, ptr, col, val);
    const double *X = x.data();
    double *Y = y.mutable_data();
#pragma omp parallel for schedule(static) if (nrows > 8192)
    for (i64 i = 0; i < nrows; ++i) {
        double s = 0;
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) s += A.val[j] * X[A.col[j]];
        Y[i] = beta == 0.0 ? alpha * s : alpha * s + beta * Y[i];
    }
}

// r = b - A x
static void residual(i64 nrows, arr<i32> ptr, arr<i32> col, arr<double> val,
                     arr<double> b, arr<double> x, arr<double> r) {
    auto A = view(nrows, 0, ptr, col, val);
    const double *X = x.data(), *B = b.data();
    double *R = r.mutable_data();
#pragma omp parallel for schedule(static) if (nrows > 8192)
    for (i64 i = 0; i < nrows; ++i) {
        double s = B[i];
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) s -= A.val[j] * X[A.col[j]];
        R[i] = s;
    }
}

// Gauss-Seidel serial sweep (parity: amgcl/relaxation/gauss_seidel.hpp:58).
static void gauss_seidel(i64 nrows, arr<i32> ptr, arr<i32> col, arr<double> val,
                         arr<double> b, arr<double> x, bool forward) {
    auto A = view(nrows, 0, ptr, col, val);
    const double *B = b.data();
    double *X = x.mutable_data();
    auto sweep_row = [&](i64 i) {
        double s = B[i], d = 1.0;
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
            i32 c = A.col[j];
            if (c == (i32)i) d = A.val[j];
            else s -= A.val[j] * X[c];
        }
        X[i] = s / d;
    };
    if (forward) for (i64 i = 0; i < nrows; ++i) sweep_row(i);
    else         for (i64 i = nrows - 1; i >= 0; --i) sweep_row(i);
}

// Block-valued ILU(0): IKJ factorization over bxb blocks (parity:
// amgcl/relaxation/ilu0.hpp:51 instantiated over static_matrix<double,B,B>
// via value_type/static_matrix.hpp — the reference's block-valued route,
// tutorial CoupCons3D).  Input is BSR (ptr/col over block rows, val
// flattened bxb row-major per block, sorted rows).  Diagonal blocks are
// stored INVERTED (Gauss-Jordan with partial pivoting, the reference's
// detail/inverse.hpp:45) so the sweeps multiply instead of solving.
// Returns (lu_blocks, dia_idx).
static void gj_invert(double *m, double *inv, int b) {
    // inv = m^-1 by Gauss-Jordan with partial pivoting (b <= 8)
    double a[64 * 2];
    for (int r = 0; r < b; ++r) {
        for (int c = 0; c < b; ++c) {
            a[r * 2 * b + c] = m[r * b + c];
            a[r * 2 * b + b + c] = (r == c) ? 1.0 : 0.0;
        }
    }
    for (int k = 0; k < b; ++k) {
        int p = k;
        for (int r = k + 1; r < b; ++r)
            if (std::fabs(a[r * 2 * b + k]) > std::fabs(a[p * 2 * b + k])) p = r;
        if (p != k)
            for (int c = 0; c < 2 * b; ++c) std::swap(a[k * 2 * b + c], a[p * 2 * b + c]);
        double d = a[k * 2 * b + k];
        if (d == 0.0) throw std::runtime_error("block_ilu0: singular diagonal block");
        for (int c = 0; c < 2 * b; ++c) a[k * 2 * b + c] /= d;
        for (int r = 0; r < b; ++r) {
            if (r == k) continue;
            double f = a[r * 2 * b + k];
            if (f == 0.0) continue;
            for (int c = 0; c < 2 * b; ++c) a[r * 2 * b + c] -= f * a[k * 2 * b + c];
        }
    }
    for (int r = 0; r < b; ++r)
        for (int c = 0; c < b; ++c) inv[r * b + c] = a[r * 2 * b + b + c];
}

static py::tuple block_ilu0_factor(i64 nb, i64 bs, arr<i32> ptr_a, arr<i32> col_a,
                                   arr<double> val_a) {
    const int b = (int)bs;
    if (b < 1 || b > 8) throw std::runtime_error("block_ilu0: 1 <= b <= 8");
    const i32 *ptr = ptr_a.data();
    const i32 *col = col_a.data();
    const i64 bb = (i64)b * b;
    const i64 nnzb = (i64)col_a.size();
    arr<double> lu_a(nnzb * bb);
    double *LU = lu_a.mutable_data();
    std::memcpy(LU, val_a.data(), (size_t)nnzb * bb * sizeof(double));
    arr<i32> dia_a(nb);
    i32 *dia = dia_a.mutable_data();
    for (i64 i = 0; i < nb; ++i) {
        dia[i] = -1;
        for (i32 j = ptr[i]; j < ptr[i + 1]; ++j)
            if (col[j] == (i32)i) { dia[i] = j; break; }
        if (dia[i] < 0) throw std::runtime_error("block_ilu0: missing diagonal block");
    }
    std::vector<i32> work(nb, -1);
    double lik[64], t[64];
    for (i64 i = 0; i < nb; ++i) {
        i32 rb = ptr[i], re = ptr[i + 1];
        for (i32 j = rb; j < re; ++j) work[col[j]] = j;
        for (i32 j = rb; j < re && col[j] < (i32)i; ++j) {
            i32 k = col[j];
            // lik = LU[j] * Dinv[k] (diagonal of row k already inverted)
            const double *A_ = LU + (i64)j * bb, *B_ = LU + (i64)dia[k] * bb;
            for (int r = 0; r < b; ++r)
                for (int c = 0; c < b; ++c) {
                    double s = 0.0;
                    for (int q = 0; q < b; ++q) s += A_[r * b + q] * B_[q * b + c];
                    lik[r * b + c] = s;
                }
            std::memcpy(LU + (i64)j * bb, lik, (size_t)bb * sizeof(double));
            for (i32 jk = dia[k] + 1; jk < ptr[k + 1]; ++jk) {
                i32 w = work[col[jk]];
                if (w < 0) continue;
                const double *U_ = LU + (i64)jk * bb;
                double *W_ = LU + (i64)w * bb;
                for (int r = 0; r < b; ++r)
                    for (int c = 0; c < b; ++c) {
                        double s = 0.0;
                        for (int q = 0; q < b; ++q) s += lik[r * b + q] * U_[q * b + c];
                        W_[r * b + c] -= s;
                    }
            }
        }
        gj_invert(LU + (i64)dia[i] * bb, t, b);
        std::memcpy(LU + (i64)dia[i] * bb, t, (size_t)bb * sizeof(double));
        for (i32 j = rb; j < re; ++j) work[col[j]] = -1;
    }
    return py::make_tuple(lu_a, dia_a);
}

// Serial block forward/backward sweeps over the combined factor
// (z := M^-1 z; unit block diagonal in L, inverted diagonal blocks in U).
static void block_ilu0_solve(i64 nb, i64 bs, arr<i32> ptr_a, arr<i32> col_a,
                             arr<double> lu_a, arr<i32> dia_a,
                             py::array_t<double> z_a) {
    const int b = (int)bs;
    const i64 bb = (i64)b * b;
    const i32 *ptr = ptr_a.data(), *col = col_a.data(), *dia = dia_a.data();
    const double *LU = lu_a.data();
    double *Z = z_a.mutable_data();
    double acc[8];
    for (i64 i = 0; i < nb; ++i) {  // forward: z_i -= sum L_ik z_k
        double *zi = Z + i * b;
        for (i32 j = ptr[i]; j < dia[i]; ++j) {
            const double *M = LU + (i64)j * bb;
            const double *zk = Z + (i64)col[j] * b;
            for (int r = 0; r < b; ++r) {
                double s = 0.0;
                for (int c = 0; c < b; ++c) s += M[r * b + c] * zk[c];
                zi[r] -= s;
            }
        }
    }
    for (i64 i = nb - 1; i >= 0; --i) {  // backward: z_i = Dinv (z_i - sum U z_k)
        double *zi = Z + i * b;
        for (i32 j = dia[i] + 1; j < ptr[i + 1]; ++j) {
            const double *M = LU + (i64)j * bb;
            const double *zk = Z + (i64)col[j] * b;
            for (int r = 0; r < b; ++r) {
                double s = 0.0;
                for (int c = 0; c < b; ++c) s += M[r * b + c] * zk[c];
                zi[r] -= s;
            }
        }
        const double *D = LU + (i64)dia[i] * bb;
        for (int r = 0; r < b; ++r) {
            double s = 0.0;
            for (int c = 0; c < b; ++c) s += D[r * b + c] * zi[c];
            acc[r] = s;
        }
        for (int r = 0; r < b; ++r) zi[r] = acc[r];
    }
}

// ILU(0) factorization, in-place IKJ (parity: amgcl/relaxation/ilu0.hpp:51).
// Requires sorted rows. Returns (luval, dia_idx) where dia_idx[i] points at
// the diagonal entry of row i inside the CSR arrays; U's diagonal is stored
// inverted for the solve.
static py::tuple ilu0_factor(i64 nrows, arr<i32> ptr, arr<i32> col, arr<double> val) {
    auto A = view(nrows, 0, ptr, col, val);
    const i64 nnz = (i64)col.size();
    arr<double> lu(nnz);
    double *LU = lu.mutable_data();
    std::memcpy(LU, A.val, nnz * sizeof(double));
    arr<i32> dia_a(nrows);
    i32 *dia = dia_a.mutable_data();
    for (i64 i = 0; i < nrows; ++i) {
        dia[i] = -1;
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j)
            if (A.col[j] == (i32)i) { dia[i] = j; break; }
        if (dia[i] < 0) throw std::runtime_error("ilu0: missing diagonal");
    }
    std::vector<i32> work(nrows, -1);
    for (i64 i = 0; i < nrows; ++i) {
        i32 rb = A.ptr[i], re = A.ptr[i + 1];
        for (i32 j = rb; j < re; ++j) work[A.col[j]] = j;
        for (i32 j = rb; j < re && A.col[j] < (i32)i; ++j) {
            i32 k = A.col[j];
            double lik = LU[j] * LU[dia[k]];  // dia stored inverted below
            LU[j] = lik;
            for (i32 jk = dia[k] + 1; jk < A.ptr[k + 1]; ++jk) {
                i32 w = work[A.col[jk]];
                if (w >= 0) LU[w] -= lik * LU[jk];
            }
        }
        if (LU[dia[i]] == 0.0) throw std::runtime_error("ilu0: zero pivot");
        LU[dia[i]] = 1.0 / LU[dia[i]];
        for (i32 j = rb; j < re; ++j) work[A.col[j]] = -1;
    }
    return py::make_tuple(lu, dia_a);
}

// ILU(k): level-of-fill factorization (parity: amgcl/relaxation/iluk.hpp:49).
// Returns (ptr, col, lu, dia) of the combined LU factor (unit L implied,
// U diagonal stored inverted) with rows sorted — compatible with ilu0_solve.
static py::tuple iluk_factor(i64 n, arr<i32> ptr_a, arr<i32> col_a, arr<double> val_a,
                             i64 kfill) {
    auto A = view(n, n, ptr_a, col_a, val_a);
    std::vector<std::vector<i32>> cols(n);
    std::vector<std::vector<double>> vals(n);
    std::vector<std::vector<i32>> levs(n);
    std::vector<i32> dia_pos(n, -1);

    std::vector<double> w(n, 0.0);
    std::vector<i32> wl(n, 0);
    std::vector<uint8_t> inrow(n, 0);
    std::vector<i32> touched;

    for (i64 i = 0; i < n; ++i) {
        touched.clear();
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
            i32 c = A.col[j];
            w[c] = A.val[j];
            wl[c] = 0;
            inrow[c] = 1;
            touched.push_back(c);
        }
        // IKJ elimination over k < i present in the row
        std::sort(touched.begin(), touched.end());
        for (size_t t = 0; t < touched.size(); ++t) {
            i32 k = touched[t];
            if (k >= (i64)i) break;
            if (!inrow[k]) continue;
            i32 dk = dia_pos[k];
            double lik = w[k] * vals[k][dk];  // dia stored inverted
            w[k] = lik;
            i32 lev_ik = wl[k];
            for (size_t jj = dk + 1; jj < cols[k].size(); ++jj) {
                i32 c = cols[k][jj];
                i32 lev_new = lev_ik + levs[k][jj] + 1;
                if (inrow[c]) {
                    w[c] -= lik * vals[k][jj];
                    if (lev_new < wl[c]) wl[c] = lev_new;
                } else if (lev_new <= kfill) {
                    w[c] = -lik * vals[k][jj];
                    wl[c] = lev_new;
                    inrow[c] = 1;
                    touched.insert(std::upper_bound(touched.begin() + t + 1,
                                                    touched.end(), c), c);
                }
            }
        }
        // store the row (sorted)
        for (i32 c : touched) {
            if (!inrow[c]) continue;
            if (c == (i64)i) dia_pos[i] = (i32)cols[i].size();
            cols[i].push_back(c);
            vals[i].push_back(w[c]);
            levs[i].push_back(wl[c]);
            inrow[c] = 0;
        }
        if (dia_pos[i] < 0 || vals[i][dia_pos[i]] == 0.0)
            throw std::runtime_error("iluk: zero pivot");
        vals[i][dia_pos[i]] = 1.0 / vals[i][dia_pos[i]];
    }
    i64 nnz = 0;
    for (i64 i = 0; i < n; ++i) nnz += cols[i].size();
    arr<i32> optr(n + 1), ocol(nnz), odia(n);
    arr<double> oval(nnz);
    i32 *OP = optr.mutable_data();
    i32 *OC = ocol.mutable_data();
    double *OV = oval.mutable_data();
    i32 *OD = odia.mutable_data();
    OP[0] = 0;
    i64 h = 0;
    for (i64 i = 0; i < n; ++i) {
        OD[i] = (i32)(h + dia_pos[i]);
        for (size_t j = 0; j < cols[i].size(); ++j, ++h) {
            OC[h] = cols[i][j];
            OV[h] = vals[i][j];
        }
        OP[i + 1] = (i32)h;
    }
    return py::make_tuple(optr, ocol, oval, odia);
}

// ILUT(p, tau): threshold ILU (parity: amgcl/relaxation/ilut.hpp:56).
// Keeps the p*row_nnz largest entries per L/U part above tau*row_norm.
static py::tuple ilut_factor(i64 n, arr<i32> ptr_a, arr<i32> col_a, arr<double> val_a,
                             double pfactor, double tau) {
    auto A = view(n, n, ptr_a, col_a, val_a);
    std::vector<std::vector<i32>> cols(n);
    std::vector<std::vector<double>> vals(n);
    std::vector<i32> dia_pos(n, -1);
    std::vector<double> w(n, 0.0);
    std::vector<uint8_t> inrow(n, 0);
    std::vector<i32> touched;

    for (i64 i = 0; i < n; ++i) {
        touched.clear();
        double nrm = 0.0;
        i32 lcount = 0, ucount = 0;
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
            i32 c = A.col[j];
            w[c] = A.val[j];
            inrow[c] = 1;
            touched.push_back(c);
            nrm += A.val[j] * A.val[j];
            if (c < (i64)i) ++lcount;
            else if (c > (i64)i) ++ucount;
        }
        nrm = std::sqrt(nrm);
        const double drop = tau * nrm;
        const i32 lmax = (i32)(pfactor * lcount) + 1;
        const i32 umax = (i32)(pfactor * ucount) + 1;
        std::sort(touched.begin(), touched.end());
        for (size_t t = 0; t < touched.size(); ++t) {
            i32 k = touched[t];
            if (k >= (i64)i) break;
            if (!inrow[k]) continue;
            i32 dk = dia_pos[k];
            double lik = w[k] * vals[k][dk];
            if (std::abs(lik) < drop) {
                w[k] = 0.0;
                inrow[k] = 0;
                continue;
            }
            w[k] = lik;
            for (size_t jj = dk + 1; jj < cols[k].size(); ++jj) {
                i32 c = cols[k][jj];
                double upd = lik * vals[k][jj];
                if (inrow[c]) {
                    w[c] -= upd;
                } else if (std::abs(upd) >= drop) {
                    w[c] = -upd;
                    inrow[c] = 1;
                    touched.insert(std::upper_bound(touched.begin() + t + 1,
                                                    touched.end(), c), c);
                }
            }
        }
        // select entries: diagonal always; p largest in each part above drop
        std::vector<std::pair<double, i32>> lpart, upart;
        double dval = 0.0;
        for (i32 c : touched) {
            if (!inrow[c]) continue;
            inrow[c] = 0;
            if (c == (i64)i) {
                dval = w[c];
                continue;
            }
            if (std::abs(w[c]) < drop) continue;
            (c < (i64)i ? lpart : upart).push_back({std::abs(w[c]), c});
        }
        auto keep = [&](std::vector<std::pair<double, i32>> &part, i32 cap) {
            if ((i32)part.size() > cap) {
                std::nth_element(part.begin(), part.begin() + cap, part.end(),
                                 [](auto &a, auto &b) { return a.first > b.first; });
                part.resize(cap);
            }
            std::sort(part.begin(), part.end(),
                      [](auto &a, auto &b) { return a.second < b.second; });
        };
        keep(lpart, lmax);
        keep(upart, umax);
        if (dval == 0.0) throw std::runtime_error("ilut: zero pivot");
        for (auto &pr : lpart) {
            cols[i].push_back(pr.second);
            vals[i].push_back(w[pr.second]);
        }
        dia_pos[i] = (i32)cols[i].size();
        cols[i].push_back((i32)i);
        vals[i].push_back(1.0 / dval);
        for (auto &pr : upart) {
            cols[i].push_back(pr.second);
            vals[i].push_back(w[pr.second]);
        }
    }
    i64 nnz = 0;
    for (i64 i = 0; i < n; ++i) nnz += cols[i].size();
    arr<i32> optr(n + 1), ocol(nnz), odia(n);
    arr<double> oval(nnz);
    i32 *OP = optr.mutable_data();
    i32 *OC = ocol.mutable_data();
    double *OV = oval.mutable_data();
    i32 *OD = odia.mutable_data();
    OP[0] = 0;
    i64 h = 0;
    for (i64 i = 0; i < n; ++i) {
        OD[i] = (i32)(h + dia_pos[i]);
        for (size_t j = 0; j < cols[i].size(); ++j, ++h) {
            OC[h] = cols[i][j];
            OV[h] = vals[i][j];
        }
        OP[i + 1] = (i32)h;
    }
    return py::make_tuple(optr, ocol, oval, odia);
}

// Chow-Patel fine-grained parallel ILU(0)
// (parity: amgcl/relaxation/ilu0_chow_patel.hpp:87): fixed-point sweeps
//   l_ij = (a_ij - sum_{k<j} l_ik u_kj) / u_jj     (i > j)
//   u_ij =  a_ij - sum_{k<i} l_ik u_kj             (i <= j)
// updating all nonzeros in parallel with asynchronous reads. Returns the
// same (ptr, col, lu, dia) layout as ilu0_factor (U diagonal inverted).
static py::tuple ilu0_chow_patel(i64 n, arr<i32> ptr_a, arr<i32> col_a,
                                 arr<double> val_a, i64 sweeps) {
    auto A = view(n, n, ptr_a, col_a, val_a);
    const i64 nnz = (i64)col_a.size();
    arr<double> lu(nnz);
    double *LU = lu.mutable_data();
    arr<i32> dia_a(n);
    i32 *dia = dia_a.mutable_data();
#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < n; ++i) {
        dia[i] = -1;
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
            if (A.col[j] == (i32)i) dia[i] = j;
        }
        if (dia[i] < 0) {
            // flagged below (cannot throw inside the parallel region cleanly)
        }
    }
    for (i64 i = 0; i < n; ++i)
        if (dia[i] < 0) throw std::runtime_error("chow_patel: missing diagonal");

    // initial guess: L = strictly-lower(A) scaled by diag, U = upper(A)
    const double *D0 = nullptr;
    std::vector<double> dvals(n);
#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < n; ++i) dvals[i] = A.val[dia[i]];
    D0 = dvals.data();
#pragma omp parallel for schedule(static)
    for (i64 i = 0; i < n; ++i)
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
            i32 c = A.col[j];
            LU[j] = (c < (i32)i) ? A.val[j] / D0[c] : A.val[j];
        }

    auto find_entry = [&](i32 row, i32 want) -> i64 {
        i32 lo = A.ptr[row], hi = A.ptr[row + 1] - 1;
        while (lo <= hi) {
            i32 mid = (lo + hi) / 2;
            if (A.col[mid] == want) return mid;
            if (A.col[mid] < want) lo = mid + 1;
            else hi = mid - 1;
        }
        return -1;
    };

    for (i64 s = 0; s < sweeps; ++s) {
#pragma omp parallel for schedule(dynamic, 2048)
        for (i64 i = 0; i < n; ++i) {
            for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
                i32 cj = A.col[j];
                // sum_{k < min(i, cj)} l_ik u_kcj over the row i pattern
                double sum = 0.0;
                i32 kmax = std::min((i32)i, cj);
                for (i32 t = A.ptr[i]; t < A.ptr[i + 1]; ++t) {
                    i32 k = A.col[t];
                    if (k >= kmax) break;  // rows sorted
                    i64 ku = find_entry(k, cj);
                    if (ku >= 0) sum += LU[t] * LU[ku];
                }
                if (cj < (i32)i) {
                    double ujj = LU[dia[cj]];
                    LU[j] = (ujj != 0.0) ? (A.val[j] - sum) / ujj : 0.0;
                } else {
                    LU[j] = A.val[j] - sum;
                }
            }
        }
    }
    // invert the U diagonal (solve convention)
    for (i64 i = 0; i < n; ++i) {
        if (LU[dia[i]] == 0.0) throw std::runtime_error("chow_patel: zero pivot");
        LU[dia[i]] = 1.0 / LU[dia[i]];
    }
    return py::make_tuple(ptr_a, col_a, lu, dia_a);
}

// SPAI-1: sparse approximate inverse with A's sparsity pattern
// (parity: amgcl/relaxation/spai1.hpp:54). Per row: least squares
// min || A(:,J) m - e_i || over J = pattern(i), solved via normal equations
// (small dense SPD system per row).
static py::tuple spai1(i64 n, arr<i32> ptr_a, arr<i32> col_a, arr<double> val_a) {
    auto A = view(n, n, ptr_a, col_a, val_a);
    arr<i32> mptr(n + 1);
    std::memcpy(mptr.mutable_data(), A.ptr, (n + 1) * sizeof(i32));
    const i64 nnz = (i64)col_a.size();
    arr<i32> mcol(nnz);
    std::memcpy(mcol.mutable_data(), A.col, nnz * sizeof(i32));
    arr<double> mval(nnz);
    double *MV = mval.mutable_data();

    // A^T pattern access for gathering columns: build transpose once
    std::vector<i32> tp(n + 1, 0), tc(nnz), tj(nnz);
    std::vector<double> tv(nnz);
    for (i64 j = 0; j < nnz; ++j) ++tp[A.col[j] + 1];
    for (i64 c = 0; c < n; ++c) tp[c + 1] += tp[c];
    {
        std::vector<i32> cur(tp.begin(), tp.end() - 1);
        for (i64 i = 0; i < n; ++i)
            for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
                i32 h = cur[A.col[j]]++;
                tc[h] = (i32)i;
                tv[h] = A.val[j];
            }
    }

#pragma omp parallel
    {
        std::vector<double> G, rhs, sol;
#pragma omp for schedule(dynamic, 512)
        for (i64 i = 0; i < n; ++i) {
            i32 jb = A.ptr[i], je = A.ptr[i + 1];
            int k = je - jb;
            if (k <= 0) continue;
            // G = (A_J)^T A_J  via sparse column dot products, rhs = (A_J)^T e_i
            G.assign((size_t)k * k, 0.0);
            rhs.assign(k, 0.0);
            for (int a = 0; a < k; ++a) {
                i32 ca = A.col[jb + a];
                for (i32 t = tp[ca]; t < tp[ca + 1]; ++t) {
                    i32 row = tc[t];
                    double va = tv[t];
                    if (row == (i32)i) rhs[a] += va;
                    // dot with other columns: binary search row in each col?
                    // accumulate via the row's pattern instead:
                    for (i32 jj = A.ptr[row]; jj < A.ptr[row + 1]; ++jj) {
                        i32 cb = A.col[jj];
                        // find cb's position in J (row i's sorted pattern)
                        i32 lo = jb, hi = je - 1, pos = -1;
                        while (lo <= hi) {
                            i32 mid = (lo + hi) / 2;
                            if (A.col[mid] == cb) { pos = mid; break; }
                            if (A.col[mid] < cb) lo = mid + 1;
                            else hi = mid - 1;
                        }
                        if (pos >= 0) G[(size_t)a * k + (pos - jb)] += va * A.val[jj];
                    }
                }
            }
            // solve G m = rhs (Cholesky-free: Gauss with partial pivot)
            sol = rhs;
            for (int c = 0; c < k; ++c) {
                int piv = c;
                for (int r2 = c + 1; r2 < k; ++r2)
                    if (std::abs(G[(size_t)r2 * k + c]) > std::abs(G[(size_t)piv * k + c]))
                        piv = r2;
                if (piv != c) {
                    for (int cc = 0; cc < k; ++cc)
                        std::swap(G[(size_t)c * k + cc], G[(size_t)piv * k + cc]);
                    std::swap(sol[c], sol[piv]);
                }
                double d = G[(size_t)c * k + c];
                if (d == 0.0) d = 1e-300;
                for (int r2 = c + 1; r2 < k; ++r2) {
                    double f = G[(size_t)r2 * k + c] / d;
                    if (f == 0.0) continue;
                    for (int cc = c; cc < k; ++cc)
                        G[(size_t)r2 * k + cc] -= f * G[(size_t)c * k + cc];
                    sol[r2] -= f * sol[c];
                }
            }
            for (int c = k - 1; c >= 0; --c) {
                double s = sol[c];
                for (int cc = c + 1; cc < k; ++cc)
                    s -= G[(size_t)c * k + cc] * sol[cc];
                sol[c] = s / (G[(size_t)c * k + c] == 0.0 ? 1e-300 : G[(size_t)c * k + c]);
            }
            for (int a = 0; a < k; ++a) MV[jb + a] = sol[a];
        }
    }
    return py::make_tuple(mptr, mcol, mval);
}

// serial L/U sweeps: solve (LU) z = r with unit L, inverted-diagonal U.
static void ilu0_solve(i64 nrows, arr<i32> ptr, arr<i32> col, arr<double> lu,
                       arr<i32> dia_a, arr<double> z) {
    const i32 *P = ptr.data(), *C = col.data(), *dia = dia_a.data();
    const double *LU = lu.data();
    double *Z = z.mutable_data();
    for (i64 i = 0; i < nrows; ++i) {
        double s = Z[i];
        for (i32 j = P[i]; j < dia[i]; ++j) s -= LU[j] * Z[C[j]];
        Z[i] = s;
    }
    for (i64 i = nrows - 1; i >= 0; --i) {
        double s = Z[i];
        for (i32 j = dia[i] + 1; j < P[i + 1]; ++j) s -= LU[j] * Z[C[j]];
        Z[i] = s * LU[dia[i]];
    }
}

// ---------------------------------------------------------------------------
// Level schedule of a triangular dependency DAG (parity: relaxation/detail/
// ilu_solve.hpp:257 sptr_solve level scheduling). For the LOWER solve row i
// depends on rows c < i among its strictly-lower nonzeros; for the UPPER
// solve on rows c > i. Rows of one level are independent, so the parallel
// sweep is BITWISE identical to the serial one (per-row accumulation order
// unchanged). Returns (lvl_ptr, rows) with rows bucketed by level, ascending
// row index inside each level.
// ---------------------------------------------------------------------------

// ---------------------------------------------------------------------------
// Skyline (profile) LU coarse solver.
// Crout factorization inside the symmetric envelope of the (CM-permuted)
// matrix: L stored by rows (non-unit, diagonal kept inverted in D), U by
// columns (unit diagonal), sharing one profile pointer array.  Role parity:
// amgcl/solver/skyline_lu.hpp:85 (same storage scheme, the classic skyline
// algorithm); memory O(profile) instead of the dense inverse's O(n^2).
// The caller applies a bandwidth-reducing permutation first.
static py::tuple skyline_factor(i64 n, arr<i32> ptr_a, arr<i32> col_a,
                                arr<double> val_a) {
    auto A = view(n, n, ptr_a, col_a, val_a);
    // envelope: len[i] = max reach below the diagonal in row i / above in col i
    std::vector<i64> len(n + 1, 0);
    for (i64 i = 0; i < n; ++i)
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
            i64 c = A.col[j];
            i64 d = i > c ? i - c : c - i;
            i64 t = i > c ? i : c;
            if (len[t + 1] < d) len[t + 1] = d;  // profile of index t
        }
    arr<i64> sp_a(n + 1);
    i64 *sp = sp_a.mutable_data();
    sp[0] = 0;
    for (i64 i = 0; i < n; ++i) sp[i + 1] = sp[i] + len[i + 1];
    const i64 total = sp[n];
    arr<double> L_a(total), U_a(total), D_a(n);
    double *L = L_a.mutable_data();
    double *U = U_a.mutable_data();
    double *D = D_a.mutable_data();
    std::fill(L, L + total, 0.0);
    std::fill(U, U + total, 0.0);
    std::fill(D, D + n, 0.0);
    // scatter CSR entries into the profile
    auto lo = [&](i64 i) { return i - (sp[i + 1] - sp[i]); };
    for (i64 i = 0; i < n; ++i)
        for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
            i64 c = A.col[j];
            double v = A.val[j];
            if (c == i) D[i] = v;
            else if (c < i) L[sp[i] + (c - lo(i))] = v;       // row i of L
            else U[sp[c] + (i - lo(c))] = v;                  // column c of U
        }
    // Crout: advance index m; compute U column m, L row m, then D[m]
    if (D[0] == 0.0) throw std::runtime_error("zero pivot in skyline_lu");
    D[0] = 1.0 / D[0];
    for (i64 m = 1; m < n; ++m) {
        const i64 lom = lo(m);
        // column m of U: U[i,m] = (A[i,m] - sum_j L[i,j] U[j,m]) * D[i]
        for (i64 i = lom; i < m; ++i) {
            const i64 loi = lo(i);
            i64 jb = lom > loi ? lom : loi;
            double s = U[sp[m] + (i - lom)];
            const double *Lr = L + sp[i] - loi;   // Lr[j] = L[i,j]
            const double *Uc = U + sp[m] - lom;   // Uc[j] = U[j,m]
            for (i64 j = jb; j < i; ++j) s -= Lr[j] * Uc[j];
            U[sp[m] + (i - lom)] = s * D[i];
        }
        // row m of L: L[m,i] = A[m,i] - sum_j L[m,j] U[j,i]
        for (i64 i = lom; i < m; ++i) {
            const i64 loi = lo(i);
            i64 jb = lom > loi ? lom : loi;
            double s = L[sp[m] + (i - lom)];
            const double *Lr = L + sp[m] - lom;   // Lr[j] = L[m,j]
            const double *Uc = U + sp[i] - loi;   // Uc[j] = U[j,i]
            for (i64 j = jb; j < i; ++j) s -= Lr[j] * Uc[j];
            L[sp[m] + (i - lom)] = s;
        }
        // pivot
        double s = D[m];
        const double *Lr = L + sp[m] - lom;
        const double *Uc = U + sp[m] - lom;
        for (i64 j = lom; j < m; ++j) s -= Lr[j] * Uc[j];
        if (s == 0.0) throw std::runtime_error("zero pivot in skyline_lu");
        D[m] = 1.0 / s;
    }
    return py::make_tuple(sp_a, L_a, U_a, D_a);
}

static arr<double> skyline_solve(arr<i64> sp_a, arr<double> L_a, arr<double> U_a,
                                 arr<double> D_a, arr<double> b_a) {
    const i64 n = (i64)D_a.size();
    const i64 *sp = sp_a.data();
    const double *L = L_a.data();
    const double *U = U_a.data();
    const double *D = D_a.data();
    const double *b = b_a.data();
    arr<double> y_a(n);
    double *y = y_a.mutable_data();
    auto lo = [&](i64 i) { return i - (sp[i + 1] - sp[i]); };
    // forward: y = L^{-1} b (L non-unit, D = inverted diagonal)
    for (i64 i = 0; i < n; ++i) {
        double s = b[i];
        const i64 loi = lo(i);
        const double *Lr = L + sp[i] - loi;
        for (i64 j = loi; j < i; ++j) s -= Lr[j] * y[j];
        y[i] = s * D[i];
    }
    // backward: y = U^{-1} y (U unit diagonal, stored by columns)
    for (i64 c = n - 1; c >= 0; --c) {
        const i64 loc = lo(c);
        const double *Uc = U + sp[c] - loc;
        const double yc = y[c];
        for (i64 i = loc; i < c; ++i) y[i] -= Uc[i] * yc;
    }
    return y_a;
}

static py::tuple tri_levels(i64 nrows, arr<i32> ptr_a, arr<i32> col_a,
                            arr<i32> dia_a, bool lower) {
    const i32 *P = ptr_a.data(), *C = col_a.data(), *dia = dia_a.data();
    std::vector<i32> lvl(nrows, 0);
    i32 nlvl = nrows ? 1 : 0;
    if (lower) {
        for (i64 i = 0; i < nrows; ++i) {
            i32 L = 0;
            for (i32 j = P[i]; j < dia[i]; ++j)
                if (lvl[C[j]] + 1 > L) L = lvl[C[j]] + 1;
            lvl[i] = L;
            if (L + 1 > nlvl) nlvl = L + 1;
        }
    } else {
        for (i64 i = nrows - 1; i >= 0; --i) {
            i32 L = 0;
            for (i32 j = dia[i] + 1; j < P[i + 1]; ++j)
                if (lvl[C[j]] + 1 > L) L = lvl[C[j]] + 1;
            lvl[i] = L;
            if (L + 1 > nlvl) nlvl = L + 1;
        }
    }
    arr<i32> lvl_ptr(nlvl + 1);
    i32 *LP = lvl_ptr.mutable_data();
    std::fill(LP, LP + nlvl + 1, 0);
    for (i64 i = 0; i < nrows; ++i) ++LP[lvl[i] + 1];
    for (i32 l = 0; l < nlvl; ++l) LP[l + 1] += LP[l];
    arr<i32> rows(nrows);
    i32 *R = rows.mutable_data();
    std::vector<i32> head(LP, LP + nlvl);
    for (i64 i = 0; i < nrows; ++i) R[head[lvl[i]]++] = (i32)i;
    return py::make_tuple(lvl_ptr, rows);
}

// OpenMP level-scheduled ILU(0/k/T) triangular solves — same results as the
// serial ilu0_solve, levels run in parallel.
static void ilu0_solve_parallel(i64 nrows, arr<i32> ptr, arr<i32> col,
                                arr<double> lu, arr<i32> dia_a,
                                arr<i32> low_ptr_a, arr<i32> low_rows_a,
                                arr<i32> up_ptr_a, arr<i32> up_rows_a,
                                arr<double> z) {
    const i32 *P = ptr.data(), *C = col.data(), *dia = dia_a.data();
    const double *LU = lu.data();
    double *Z = z.mutable_data();
    const i32 *LL = low_ptr_a.data(), *LR = low_rows_a.data();
    const i32 *UL = up_ptr_a.data(), *UR = up_rows_a.data();
    const i64 nlow = (i64)low_ptr_a.size() - 1;
    const i64 nup = (i64)up_ptr_a.size() - 1;
    for (i64 l = 0; l < nlow; ++l) {
#pragma omp parallel for schedule(static) if (LL[l + 1] - LL[l] > 512)
        for (i32 t = LL[l]; t < LL[l + 1]; ++t) {
            i32 i = LR[t];
            double s = Z[i];
            for (i32 j = P[i]; j < dia[i]; ++j) s -= LU[j] * Z[C[j]];
            Z[i] = s;
        }
    }
    for (i64 l = 0; l < nup; ++l) {
#pragma omp parallel for schedule(static) if (UL[l + 1] - UL[l] > 512)
        for (i32 t = UL[l]; t < UL[l + 1]; ++t) {
            i32 i = UR[t];
            double s = Z[i];
            for (i32 j = dia[i] + 1; j < P[i + 1]; ++j) s -= LU[j] * Z[C[j]];
            Z[i] = s * LU[dia[i]];
        }
    }
}

// Deterministic parallel Gauss-Seidel: multicolor sweep (rows of one color
// are independent). The reference's level-scheduled parallel GS
// (gauss_seidel.hpp:185) races on same-level upper reads and is therefore
// run-to-run nondeterministic; the multicolor ordering (the same one the
// GPU path uses) is the deterministic CPU-parallel choice.
static void gauss_seidel_colored(i64 nrows, arr<i32> ptr, arr<i32> col,
                                 arr<double> val, arr<double> b, arr<double> x,
                                 arr<i32> order_a, arr<i32> cptr_a, bool forward) {
    auto A = view(nrows, 0, ptr, col, val);
    const double *B = b.data();
    double *X = x.mutable_data();
    const i32 *order = order_a.data();
    const i32 *CP = cptr_a.data();
    const i64 nc = (i64)cptr_a.size() - 1;
    auto sweep_color = [&](i64 c) {
#pragma omp parallel for schedule(static) if (CP[c + 1] - CP[c] > 512)
        for (i32 t = CP[c]; t < CP[c + 1]; ++t) {
            i32 i = order[t];
            double s = B[i], d = 1.0;
            for (i32 j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
                i32 cc = A.col[j];
                if (cc == i) d = A.val[j];
                else s -= A.val[j] * X[cc];
            }
            X[i] = s / d;
        }
    };
    if (forward) for (i64 c = 0; c < nc; ++c) sweep_color(c);
    else         for (i64 c = nc - 1; c >= 0; --c) sweep_color(c);
}

PYBIND11_MODULE(_core, m) {
    m.doc() = "amgcl_amd host setup engine (OpenMP)";
    m.def("poisson3d", &poisson3d, py::arg("n"), py::arg("anisotropy") = 1.0);
    m.def("poisson3d_strip", &poisson3d_strip);
    m.def("poisson3d_box_strip", &poisson3d_box_strip);
    m.def("split_strip", &split_strip);
    m.def("diagonal", &diagonal);
    m.def("transpose", &transpose);
    m.def("spgemm", &spgemm);
    m.def("aggregates", &aggregates);
    m.def("aggregates_parallel", &aggregates_parallel);
    m.def("smoothed_prolongation", &smoothed_prolongation);
    m.def("tentative_prolongation", &tentative_prolongation);
    m.def("spai0", &spai0);
    m.def("ruge_stuben", &ruge_stuben);
    m.def("tentative_nullspace", &tentative_nullspace);
    m.def("pointwise_matrix", &pointwise_matrix);
    m.def("csr_to_bsr", &csr_to_bsr);
    m.def("color_graph", &color_graph);
    m.def("expand_strong", &expand_strong);
    m.def("filtered_smoother_matrix", &filtered_smoother_matrix);
    m.def("spmv", &spmv);
    m.def("residual", &residual);
    m.def("gauss_seidel", &gauss_seidel);
    m.def("ilu0_factor", &ilu0_factor);
    m.def("block_ilu0_factor", &block_ilu0_factor);
    m.def("block_ilu0_solve", &block_ilu0_solve);
    m.def("iluk_factor", &iluk_factor);
    m.def("ilut_factor", &ilut_factor);
    m.def("ilu0_chow_patel", &ilu0_chow_patel);
    m.def("spai1", &spai1);
    m.def("ilu0_solve", &ilu0_solve);
    m.def("tri_levels", &tri_levels);
    m.def("skyline_factor", &skyline_factor);
    m.def("skyline_solve", &skyline_solve);
    m.def("ilu0_solve_parallel", &ilu0_solve_parallel);
    m.def("gauss_seidel_colored", &gauss_seidel_colored);
    m.def("omp_threads", []() { return omp_get_max_threads(); });
}
