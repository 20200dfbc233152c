// amgcl_amd — torch-free GPU C API.
//
// VERDICT r01 next-step #3 ("a solve that runs with torch absent on a single
// GPU"): the hierarchy is built by the self-contained host engine
// (csrc/capi/amgcl_host.hpp, plain C++/OpenMP), uploaded with raw
// hipMalloc/hipMemcpy, and the whole CG/BiCGStab solve runs through the
// native driver (driver.hip) — no Python, no torch, anywhere.  C and
// Fortran programs get the GPU path through these entry points.
//
// Config string: "key=value;key=value" with the same keys as the C API
// Params (solver.type, solver.tol, solver.maxiter, precond.coarse_enough,
// precond.npre/npost/ncycle, precond.relax.type, ...).

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstring>
#include <string>
#include <vector>

#include "../capi/amgcl_host.hpp"
#include "amg_common.h"

extern "C" void *amg_driver_create(const LevelDesc *levels, int nlevels,
                                   const void *coarse_inv, int64_t ncoarse, int npre,
                                   int npost, int ncycle, int pre_cycles, int f32,
                                   int64_t a64_nnz, const int *a64_ptr,
                                   const int *a64_col, const double *a64_val,
                                   int a64_subw, hipStream_t stream);
extern "C" void amg_driver_destroy(void *h);
// device setup engine (setup.hip)
extern "C" int amg_setup_diag(int64_t n, const int *ptr, const int *col,
                              const double *val, double *d, hipStream_t s);
extern "C" int amg_setup_strong(int64_t n, const int *ptr, const int *col,
                                const double *val, const double *d, double eps2,
                                uint8_t *S, hipStream_t s);
extern "C" int amg_setup_spai0(int64_t n, const int *ptr, const int *col,
                               const double *val, double *m, hipStream_t s);
extern "C" int amg_agg_init(int64_t n, const int *ptr, const uint8_t *S, int *id,
                            hipStream_t s);
extern "C" int amg_agg_run(int64_t n, const int *ptr, const int *col, const uint8_t *S,
                           int *id, uint8_t *prov, uint64_t *m1, uint8_t *newroot,
                           uint8_t *near, int *remaining, int sync_stride,
                           int max_rounds, int *rounds_out, int *lists, hipStream_t s);
extern "C" int amg_agg_renumber(int64_t n, int *id, int *mark, hipStream_t s);
extern "C" int amg_psmooth_count(int64_t n, const int *ptr, const int *col,
                                 const uint8_t *S, const int *id, int *cnt,
                                 int *overflow, hipStream_t s);
extern "C" int amg_psmooth_fill(int64_t n, const int *ptr, const int *col,
                                const double *val, const uint8_t *S, const int *id,
                                double omega, const int *pptr_scanned, int *pcol,
                                double *pval, hipStream_t s);
extern "C" int amg_transpose_count(int64_t nnz, const int *col, int *tcnt,
                                   hipStream_t s);
extern "C" int amg_transpose_scatter(int64_t n, const int *ptr, const int *col,
                                     const double *val, int *cursor, int *tcol,
                                     double *tval, hipStream_t s);
extern "C" int amg_spgemm_count(int64_t an, const int *aptr, const int *acol,
                                const int *bptr, const int *bcol, int *ub, int *cnt,
                                int *overflow, int *bigscratch, hipStream_t s);
extern "C" int amg_spgemm_fill(int64_t an, const int *aptr, const int *acol,
                               const double *aval, const int *bptr, const int *bcol,
                               const double *bval, const int *ub,
                               const int *cptr_scanned, int *ccol, double *cval,
                               int do_sort, const int *bigscratch, hipStream_t s);
extern "C" int amg_scan_i32(int *a, int64_t n, hipStream_t s);
extern "C" int amg_sell_fill_f64(int64_t nrows, int64_t nslice, const int *ptr,
                                 const int *col, const double *val,
                                 const int64_t *soff, const int *srows, int *scol,
                                 double *sval, hipStream_t stream);
extern "C" int amg_driver_cg(void *h, const double *rhs, double *x, double *r,
                             double *s, double *p, double *q, double *s_swap,
                             double tol, double abstol, int maxiter,
                             int64_t *iters_out, double *resid_out);
extern "C" int amg_driver_bicgstab(void *h, const double *rhs, double *x, double *r,
                                   double *p, double *v, double *s2, double *t2,
                                   double *rh, double *T, double *T_swap, double tol,
                                   double abstol, int maxiter, int64_t *iters_out,
                                   double *resid_out);

namespace {

using amgclamd_host::Csr;
using amgclamd_host::DenseLU;
using amgclamd_host::Params;
using amgclamd_host::Precond;

// Pageable hipMemcpy H2D runs at ~3.5 GB/s (a 512^3 operator costs ~3.2 s
// of the create call); staging through two pinned bounce buffers overlaps
// the host memcpy with the DMA and reaches ~8-12 GB/s.
static bool upload_bytes(void *dst, const void *src, size_t bytes) {
    constexpr size_t CH = 64ull << 20;
    static void *pin[2] = {nullptr, nullptr};
    static hipEvent_t ev[2] = {nullptr, nullptr};
    if (bytes < CH) {
        return hipMemcpy(dst, src, bytes, hipMemcpyHostToDevice) == hipSuccess;
    }
    if (!pin[0]) {
        if (hipHostMalloc(&pin[0], CH, 0) != hipSuccess ||
            hipHostMalloc(&pin[1], CH, 0) != hipSuccess ||
            hipEventCreateWithFlags(&ev[0], hipEventDisableTiming) != hipSuccess ||
            hipEventCreateWithFlags(&ev[1], hipEventDisableTiming) != hipSuccess) {
            pin[0] = nullptr;
            return hipMemcpy(dst, src, bytes, hipMemcpyHostToDevice) == hipSuccess;
        }
        (void)hipEventRecord(ev[0], 0);
        (void)hipEventRecord(ev[1], 0);
    }
    size_t off = 0;
    int slot = 0;
    while (off < bytes) {
        size_t sz = bytes - off < CH ? bytes - off : CH;
        if (hipEventSynchronize(ev[slot]) != hipSuccess) return false;
        std::memcpy(pin[slot], (const char *)src + off, sz);
        if (hipMemcpyAsync((char *)dst + off, pin[slot], sz,
                           hipMemcpyHostToDevice, 0) != hipSuccess)
            return false;
        if (hipEventRecord(ev[slot], 0) != hipSuccess) return false;
        off += sz;
        slot ^= 1;
    }
    return hipStreamSynchronize(0) == hipSuccess;
}

template <typename T>
T *upload(const std::vector<T> &v) {
    if (v.empty()) return nullptr;
    T *d = nullptr;
    if (hipMalloc(&d, v.size() * sizeof(T)) != hipSuccess) return nullptr;
    if (!upload_bytes(d, v.data(), v.size() * sizeof(T))) {
        (void)hipFree(d);
        return nullptr;
    }
    return d;
}

double *dalloc(size_t n) {
    double *d = nullptr;
    if (hipMalloc(&d, n * sizeof(double)) != hipSuccess) return nullptr;
    (void)hipMemset(d, 0, n * sizeof(double));
    return d;
}

static int pick_subw(int64_t nrows, int64_t nnz) {
    double m = nrows ? (double)nnz / (double)nrows : 1.0;
    if (m <= 4) return 2;
    if (m <= 10) return 4;
    if (m <= 24) return 8;
    if (m <= 128) return 16;
    return 32;
}

struct GpuSolver {
    void *driver = nullptr;
    std::vector<void *> blobs;       // hipMalloc'd
    std::vector<void *> pool_blobs;  // hipMallocAsync'd (device-setup levels)
    int64_t n = 0;
    std::string type = "cg";
    double tol = 1e-8;
    int maxiter = 200;
    std::vector<double *> work;  // solver work vectors
    double *rhs_d = nullptr, *x_d = nullptr;

    ~GpuSolver() {
        if (driver) amg_driver_destroy(driver);
        for (void *p : blobs) (void)hipFree(p);
        for (void *p : pool_blobs) (void)hipFreeAsync(p, 0);
        if (!pool_blobs.empty()) (void)hipStreamSynchronize(0);
    }

    void *keep(void *p) {
        if (p) blobs.push_back(p);
        return p;
    }

    void *keep_pool(void *p) {
        if (p) pool_blobs.push_back(p);
        return p;
    }
};

Params parse_config(const char *cfg) {
    Params p;
    if (!cfg) return p;
    std::string s(cfg);
    size_t pos = 0;
    while (pos < s.size()) {
        size_t semi = s.find(';', pos);
        if (semi == std::string::npos) semi = s.size();
        std::string kv = s.substr(pos, semi - pos);
        size_t eq = kv.find('=');
        if (eq != std::string::npos)
            p.kv[kv.substr(0, eq)] = kv.substr(eq + 1);
        pos = semi + 1;
    }
    return p;
}

// ---------------------------------------------------------------------------
// Device-resident setup: the same SA pipeline the Python flagship runs
// (strong -> MIS aggregation -> smoothed P -> R = P^T -> Galerkin), driven
// from C++ over the setup.hip kernels with raw hipMalloc buffers — no torch,
// no Python.  The tail below `precond.device_handoff` rows (default 20000)
// is downloaded and finished by the host engine (same policy and rationale
// as backend/hip_setup.py: tiny levels are launch-bound on the GPU and the
// host engine supports every smoother/coarse solver).
// ---------------------------------------------------------------------------

__global__ void capi_dj_k(int64_t n, const double *__restrict__ d, double damping,
                          double *__restrict__ m) {
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) m[i] = d[i] != 0.0 ? damping / d[i] : 0.0;
}

__global__ void capi_slicew_k(int64_t n, int64_t nslice, const int *__restrict__ ptr,
                              int *__restrict__ w) {
    int64_t s = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; s < nslice; s += stride) {
        int mx = 0;
        int64_t r0 = s * 64, r1 = r0 + 64 < n ? r0 + 64 : n;
        for (int64_t r = r0; r < r1; ++r) {
            int l = ptr[r + 1] - ptr[r];
            if (l > mx) mx = l;
        }
        w[s] = mx;
    }
}

namespace {

struct DevSetupFail {};  // -> host fallback

// tracked hipMalloc: `keep` buffers live with the solver, `tmp` buffers are
// freed when the Tracker goes out of scope (per-level scratch)
struct Tracker {
    std::vector<void *> bufs;
    ~Tracker() {
        // stream-ordered frees: ordered after the kernels that read this
        // scratch (SpGEMM fill, the AP intermediate) with no device sync,
        // and the pool recycles the pages instead of unmapping them (raw
        // hipFree faulted async at 256^3 and its unmap churn showed up as
        // ~90 ms launch stalls later in the process)
        for (void *p : bufs) (void)hipFreeAsync(p, 0);
    }
    template <typename T>
    T *alloc(size_t count, bool zero = false) {
        void *p = nullptr;
        if (hipMallocAsync(&p, count * sizeof(T), 0) != hipSuccess)
            throw DevSetupFail{};
        if (zero) (void)hipMemsetAsync(p, 0, count * sizeof(T), 0);
        bufs.push_back(p);
        return (T *)p;
    }
    void *release(void *p) {  // transfer ownership to the caller
        for (size_t i = 0; i < bufs.size(); ++i)
            if (bufs[i] == p) {
                bufs.erase(bufs.begin() + i);
                return p;
            }
        return p;
    }
};

static void dev_check(int rc) {
    if (rc != 0) throw DevSetupFail{};
}

static void trace(const char *what, long long a = -1, long long b = -1) {
    if (!getenv("AMGCL_CAPI_TRACE")) return;
    fprintf(stderr, "[capi setup] %s %lld %lld\n", what, a, b);
    (void)hipDeviceSynchronize();
    fflush(stderr);
}

struct DevCsr {
    int64_t n = 0, ncols = 0, nnz = 0;
    int *ptr = nullptr;
    int *col = nullptr;
    double *val = nullptr;
};

static int read_i32(const int *dev_p) {
    int v = 0;
    if (hipMemcpy(&v, dev_p, sizeof(int), hipMemcpyDeviceToHost) != hipSuccess)
        throw DevSetupFail{};
    return v;
}

// scan counts stored at p[1..n] in place, p[0] = 0; returns total
static int scan_counts(int *p, int64_t n) {
    dev_check(amg_scan_i32(p + 1, n, 0));
    return read_i32(p + n);
}

// SELL-64 image built on the device (same layout as DeviceCSR.build_sell)
static void build_sell_capi(GpuSolver *S, int64_t n, const int *ptr, const int *col,
                            const double *val, int64_t &nslice_out,
                            const int64_t *&soff_out, const int *&scol_out,
                            const void *&sval_out) {
    int64_t nslice = (n + 63) / 64;
    Tracker tmp;
    int *w = tmp.alloc<int>(nslice);
    capi_slicew_k<<<amg_nblocks(nslice), 256, 0, 0>>>(n, nslice, ptr, w);
    dev_check((int)hipGetLastError());
    std::vector<int> wh(nslice);
    if (hipMemcpy(wh.data(), w, nslice * sizeof(int), hipMemcpyDeviceToHost) !=
        hipSuccess)
        throw DevSetupFail{};
    std::vector<int64_t> soff(nslice + 1, 0);
    for (int64_t s = 0; s < nslice; ++s) soff[s + 1] = soff[s] + (int64_t)wh[s] * 64;
    int64_t total = soff[nslice];
    int64_t *soff_d = (int64_t *)S->keep(upload(soff));
    if (!soff_d) throw DevSetupFail{};
    Tracker out;
    int *scol = out.alloc<int>(total);
    double *sval = out.alloc<double>(total);
    dev_check(amg_sell_fill_f64(n, nslice, ptr, col, val, soff_d, nullptr, scol,
                                sval, 0));
    S->keep_pool(out.release(scol));
    S->keep_pool(out.release(sval));
    nslice_out = nslice;
    soff_out = soff_d;
    scol_out = scol;
    sval_out = sval;
}

// One SpGEMM product C = A*B on the device (ownership of C's arrays moves to
// the caller's tracker)
static DevCsr spgemm_dev(Tracker &own, const DevCsr &A, const DevCsr &B, int sort) {
    Tracker tmp;
    int *ub = tmp.alloc<int>(A.n);
    int *cptr = own.alloc<int>(A.n + 1, true);
    int *overflow = tmp.alloc<int>(1, true);
    int *bigscratch = tmp.alloc<int>(2 * A.n + 1);
    dev_check(amg_spgemm_count(A.n, A.ptr, A.col, B.ptr, B.col, ub, cptr + 1,
                               overflow, bigscratch, 0));
    int nnz = scan_counts(cptr, A.n);
    if (read_i32(overflow) != 0) throw DevSetupFail{};
    DevCsr C;
    C.n = A.n;
    C.ncols = B.ncols;
    C.nnz = nnz;
    C.ptr = cptr;
    C.col = own.alloc<int>(nnz);
    C.val = own.alloc<double>(nnz);
    dev_check(amg_spgemm_fill(A.n, A.ptr, A.col, A.val, B.ptr, B.col, B.val, ub,
                              cptr + 1, C.col, C.val, sort, bigscratch, 0));
    return C;
}

// Build the device part of the hierarchy; returns the handoff matrix
// downloaded to the host (tail continues in the host engine).
static Csr build_device_levels(GpuSolver *S, int n, const int *ptr, const int *col,
                               const double *val, const Params &p,
                               std::vector<LevelDesc> &descs, double &eps_inout) {
    const int handoff = p.geti("precond.device_handoff", 20000);
    const int coarse_enough = p.geti("precond.coarse_enough", 1000);
    const int max_levels = p.geti("precond.max_levels", 20);
    const std::string relax = p.gets("precond.relax.type", "spai0");
    const double damping = p.getf("precond.relax.damping", 0.72);
    const int64_t sell_min = 500000;
    if (relax != "spai0" && relax != "damped_jacobi") throw DevSetupFail{};

    Tracker own;  // per-hierarchy ownership; transferred to S->keep on success
    DevCsr A;
    A.n = n;
    A.ncols = n;
    A.nnz = ptr[n];
    A.ptr = own.alloc<int>(n + 1);
    A.col = own.alloc<int>(A.nnz);
    A.val = own.alloc<double>(A.nnz);
    if (!upload_bytes(A.ptr, ptr, (n + 1) * sizeof(int)) ||
        !upload_bytes(A.col, col, A.nnz * sizeof(int)) ||
        !upload_bytes(A.val, val, A.nnz * sizeof(double)))
        throw DevSetupFail{};

    double eps = eps_inout;
    while (A.n > handoff && A.n > coarse_enough &&
           (int)descs.size() + 1 < max_levels) {
        Tracker tmp;
        // smoother weights
        double *M = own.alloc<double>(A.n);
        if (relax == "spai0") {
            dev_check(amg_setup_spai0(A.n, A.ptr, A.col, A.val, M, 0));
        } else {
            double *d0 = tmp.alloc<double>(A.n);
            dev_check(amg_setup_diag(A.n, A.ptr, A.col, A.val, d0, 0));
            capi_dj_k<<<amg_nblocks(A.n), 256, 0, 0>>>(A.n, d0, damping, M);
            dev_check((int)hipGetLastError());
        }
        trace("smoother", A.n, A.nnz);
        // strong connections + MIS aggregation
        double *d = tmp.alloc<double>(A.n);
        dev_check(amg_setup_diag(A.n, A.ptr, A.col, A.val, d, 0));
        uint8_t *Sg = tmp.alloc<uint8_t>(A.nnz);
        dev_check(amg_setup_strong(A.n, A.ptr, A.col, A.val, d, eps * eps, Sg, 0));
        int *ids = tmp.alloc<int>(A.n);
        dev_check(amg_agg_init(A.n, A.ptr, Sg, ids, 0));
        uint8_t *prov = tmp.alloc<uint8_t>(A.n, true);
        uint64_t *m1 = tmp.alloc<uint64_t>(A.n);
        uint8_t *newroot = tmp.alloc<uint8_t>(A.n);
        uint8_t *near = tmp.alloc<uint8_t>(A.n);
        int *remaining = tmp.alloc<int>(1, true);
        int *lists = tmp.alloc<int>(3 * (size_t)A.n);
        int rounds = 0;
        dev_check(amg_agg_run(A.n, A.ptr, A.col, Sg, ids, prov, m1, newroot, near,
                              remaining, 2, 64, &rounds, lists, 0));
        trace("agg_run");
        int *mark = tmp.alloc<int>(A.n);
        dev_check(amg_agg_renumber(A.n, ids, mark, 0));
        int naggr = read_i32(mark + A.n - 1);
        trace("naggr", naggr);
        if (naggr <= 0 || naggr >= A.n) break;  // no progress: host tail
        eps *= 0.5;
        // smoothed prolongation P = (I - omega D^-1 A_F) T, omega = 2/3
        DevCsr P;
        P.n = A.n;
        P.ncols = naggr;
        P.ptr = own.alloc<int>(A.n + 1, true);
        int *pov = tmp.alloc<int>(1, true);
        dev_check(amg_psmooth_count(A.n, A.ptr, A.col, Sg, ids, P.ptr + 1, pov, 0));
        P.nnz = scan_counts(P.ptr, A.n);
        if (read_i32(pov) != 0) throw DevSetupFail{};
        P.col = own.alloc<int>(P.nnz);
        P.val = own.alloc<double>(P.nnz);
        dev_check(amg_psmooth_fill(A.n, A.ptr, A.col, A.val, Sg, ids, 2.0 / 3.0,
                                   P.ptr + 1, P.col, P.val, 0));
        trace("psmooth", P.nnz);
        // R = P^T
        DevCsr R;
        R.n = naggr;
        R.ncols = A.n;
        R.nnz = P.nnz;
        R.ptr = own.alloc<int>(naggr + 1, true);
        dev_check(amg_transpose_count(P.nnz, P.col, R.ptr + 1, 0));
        (void)scan_counts(R.ptr, naggr);
        int *cursor = tmp.alloc<int>(naggr);
        if (hipMemcpy(cursor, R.ptr, naggr * sizeof(int),
                      hipMemcpyDeviceToDevice) != hipSuccess)
            throw DevSetupFail{};
        R.col = own.alloc<int>(R.nnz);
        R.val = own.alloc<double>(R.nnz);
        dev_check(amg_transpose_scatter(A.n, P.ptr, P.col, P.val, cursor, R.col,
                                        R.val, 0));
        trace("transpose");
        // Galerkin Ac = R*(A*P) (association measured in matrix.py galerkin)
        DevCsr Ac;
        {
            Tracker mid;
            DevCsr AP = spgemm_dev(mid, A, P, 0);
            Ac = spgemm_dev(own, R, AP, 1);
        }
        trace("galerkin", Ac.n, Ac.nnz);
        // record this level
        LevelDesc dsc;
        std::memset(&dsc, 0, sizeof dsc);
        dsc.nrows = A.n;
        dsc.nnz = A.nnz;
        dsc.ptr = A.ptr;
        dsc.col = A.col;
        dsc.val = A.val;
        dsc.subw = pick_subw(A.n, A.nnz);
        dsc.pnnz = P.nnz;
        dsc.pptr = P.ptr;
        dsc.pcol = P.col;
        dsc.pval = P.val;
        dsc.psubw = pick_subw(P.n, P.nnz);
        dsc.rnnz = R.nnz;
        dsc.rptr = R.ptr;
        dsc.rcol = R.col;
        dsc.rval = R.val;
        dsc.rsubw = pick_subw(R.n, R.nnz);
        dsc.M = M;
        dsc.f = (double *)own.alloc<double>(A.n, true);
        dsc.u = (double *)own.alloc<double>(A.n, true);
        dsc.t = (double *)own.alloc<double>(A.n, true);
        trace("record");
        if (getenv("AMGCL_CAPI_NO_SELL")) {
            descs.push_back(dsc);
            A = Ac;
            continue;
        }
        if (A.n >= sell_min)
            build_sell_capi(S, A.n, A.ptr, A.col, A.val, dsc.nslice, dsc.soff,
                            dsc.scol, dsc.sval);
        if (P.n >= sell_min) {
            build_sell_capi(S, P.n, P.ptr, P.col, P.val, dsc.pnslice, dsc.psoff,
                            dsc.pscol, dsc.psval);
            build_sell_capi(S, R.n, R.ptr, R.col, R.val, dsc.rnslice, dsc.rsoff,
                            dsc.rscol, dsc.rsval);
        }
        trace("sell");
        descs.push_back(dsc);
        A = Ac;
    }
    trace("loop done", A.n);
    // download the handoff matrix for the host tail
    std::vector<int> hp(A.n + 1), hc(A.nnz);
    std::vector<double> hv(A.nnz);
    if (hipMemcpy(hp.data(), A.ptr, (A.n + 1) * sizeof(int),
                  hipMemcpyDeviceToHost) != hipSuccess ||
        hipMemcpy(hc.data(), A.col, A.nnz * sizeof(int), hipMemcpyDeviceToHost) !=
            hipSuccess ||
        hipMemcpy(hv.data(), A.val, A.nnz * sizeof(double),
                  hipMemcpyDeviceToHost) != hipSuccess)
        throw DevSetupFail{};
    Csr tail;
    tail.n = tail.m = A.n;
    tail.ptr = std::move(hp);
    tail.col = std::move(hc);
    tail.val = std::move(hv);
    // success: everything still owned by `own` now belongs to the solver
    // (hipMallocAsync memory -> freed with hipFreeAsync at destroy)
    for (void *b : own.bufs) S->keep_pool(b);
    own.bufs.clear();
    eps_inout = eps;
    return tail;
}

}  // namespace (inner)

}  // namespace

extern "C" void *amgcl_amd_gpu_solver_create(int n, const int *ptr, const int *col,
                                             const double *val, const char *config) {
    Params p = parse_config(config);
    auto *S = new GpuSolver;
    S->n = n;
    S->type = p.gets("solver.type", "cg");
    if (S->type != "cg" && S->type != "bicgstab") {
        // the native driver carries CG and BiCGStab; fail loudly instead of
        // silently substituting (the Python API has the full Krylov set)
        delete S;
        return nullptr;
    }
    S->tol = p.getf("solver.tol", 1e-8);
    S->maxiter = p.geti("solver.maxiter", 200);

    // device levels first (precond.setup=device, or auto above 200k rows),
    // then the host engine finishes the tail below precond.device_handoff
    std::vector<LevelDesc> descs;
    double eps = p.getf("precond.coarsening.eps_strong", 0.08);
    const std::string setup = p.gets("precond.setup", "auto");
    Csr tail;
    bool have_tail = false;
    if (setup == "device" || (setup == "auto" && n > 200000)) {
        try {
            tail = build_device_levels(S, n, ptr, col, val, p, descs, eps);
            have_tail = true;
        } catch (DevSetupFail &) {
            descs.clear();  // partial buffers stay in S->blobs until destroy
        }
    }

    // host hierarchy (same engine as the CPU C API) — the whole thing, or
    // just the tail the device setup handed off
    Precond P;
    {
        Params p2 = p;
        char ebuf[32];
        std::snprintf(ebuf, sizeof ebuf, "%.17g", eps);
        p2.kv["precond.coarsening.eps_strong"] = ebuf;
        if (have_tail)
            P.build(std::move(tail), p2);
        else {
            Csr A = amgclamd_host::make_csr(n, ptr, col, val, 0);
            P.build(std::move(A), p2);
        }
    }

    // upload the host-built levels, appended after the device-built ones
    size_t base0 = descs.size();
    descs.resize(base0 + P.lvl.size());
    bool oom = false;
    for (size_t i = 0; i < P.lvl.size(); ++i) {
        auto &L = P.lvl[i];
        LevelDesc &d = descs[base0 + i];
        std::memset(&d, 0, sizeof d);
        d.nrows = L.A.n;
        d.nnz = L.A.nnz();
        d.ptr = (const int *)S->keep(upload(L.A.ptr));
        d.col = (const int *)S->keep(upload(L.A.col));
        d.val = (const double *)S->keep(upload(L.A.val));
        d.subw = pick_subw(d.nrows, d.nnz);
        if (L.P.n) {
            d.pnnz = L.P.nnz();
            d.pptr = (const int *)S->keep(upload(L.P.ptr));
            d.pcol = (const int *)S->keep(upload(L.P.col));
            d.pval = (const double *)S->keep(upload(L.P.val));
            d.psubw = pick_subw(L.P.n, d.pnnz);
            d.rnnz = L.R.nnz();
            d.rptr = (const int *)S->keep(upload(L.R.ptr));
            d.rcol = (const int *)S->keep(upload(L.R.col));
            d.rval = (const double *)S->keep(upload(L.R.val));
            d.rsubw = pick_subw(L.R.n, d.rnnz);
        }
        if (!L.M.empty()) d.M = (const double *)S->keep(upload(L.M));
        d.f = (double *)S->keep(dalloc(L.A.n));
        d.u = (double *)S->keep(dalloc(L.A.n));
        d.t = (double *)S->keep(dalloc(L.A.n));
        // hipMalloc failure shows up as a null field: fail the create
        // cleanly instead of handing the driver dangling level pointers
        oom |= !d.ptr || !d.col || !d.val || !d.f || !d.u || !d.t;
        oom |= (L.P.n != 0) && (!d.pptr || !d.pcol || !d.pval ||
                                !d.rptr || !d.rcol || !d.rval);
        oom |= !L.M.empty() && !d.M;
    }
    if (oom) {
        delete S;
        return nullptr;
    }

    // coarsest: dense inverse on device (solve LU against identity columns)
    const double *inv_d = nullptr;
    int64_t ncoarse = 0;
    if (P.coarse.n > 0) {
        int m = P.coarse.n;
        ncoarse = m;
        std::vector<double> inv((size_t)m * m);
        // columns are independent solves; serial this is O(m^3)-ish and was
        // the 3.4 s tail of a 512^3 create (m=846)
#pragma omp parallel
        {
            std::vector<double> e(m, 0.0), x1(m);
#pragma omp for schedule(static)
            for (int c = 0; c < m; ++c) {
                e[c] = 1.0;
                P.coarse.solve(e.data(), x1.data());
                e[c] = 0.0;
                for (int r = 0; r < m; ++r) inv[(size_t)r * m + c] = x1[r];
            }
        }
        inv_d = (const double *)S->keep(upload(inv));
    }

    S->driver = amg_driver_create(descs.data(), (int)descs.size(), inv_d, ncoarse,
                                  P.npre, P.npost, P.ncycle, 1, 0, 0, nullptr,
                                  nullptr, nullptr, 0, (hipStream_t)0);
    if (!S->driver) {
        delete S;
        return nullptr;
    }
    int nwork = S->type == "cg" ? 5 : 8;
    for (int i = 0; i < nwork; ++i) S->work.push_back((double *)S->keep(dalloc(n)));
    S->rhs_d = (double *)S->keep(dalloc(n));
    S->x_d = (double *)S->keep(dalloc(n));
    // setup kernels (SELL fills etc.) may still be in flight; finish them so
    // create/solve wall times split honestly for callers that time them
    (void)hipDeviceSynchronize();
    return S;
}

extern "C" int amgcl_amd_gpu_solver_solve(void *h, const double *rhs, double *x,
                                          int *iters, double *resid) {
    auto *S = (GpuSolver *)h;
    const int64_t n = S->n;
    (void)hipMemcpy(S->rhs_d, rhs, n * sizeof(double), hipMemcpyHostToDevice);
    (void)hipMemcpy(S->x_d, x, n * sizeof(double), hipMemcpyHostToDevice);
    int64_t it = 0;
    double res = 0.0;
    int rc;
    auto &w = S->work;
    if (S->type == "cg")
        rc = amg_driver_cg(S->driver, S->rhs_d, S->x_d, w[0], w[1], w[2], w[3], w[4],
                           S->tol, 0.0, S->maxiter, &it, &res);
    else
        rc = amg_driver_bicgstab(S->driver, S->rhs_d, S->x_d, w[0], w[1], w[2], w[3],
                                 w[4], w[5], w[6], w[7], S->tol, 0.0, S->maxiter,
                                 &it, &res);
    (void)hipMemcpy(x, S->x_d, n * sizeof(double), hipMemcpyDeviceToHost);
    if (iters) *iters = (int)it;
    if (resid) *resid = res;
    return rc;
}

extern "C" void amgcl_amd_gpu_solver_destroy(void *h) { delete (GpuSolver *)h; }
