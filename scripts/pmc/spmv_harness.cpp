// Torch-free SpMV harness for rocprofv3 PMC counter runs (rocprofv3 --pmc
// crashes when tracing a torch workload on this pool; a plain HIP binary is
// the supported combination). Links the in-tree libamghip.so.
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <vector>

extern "C" int amg_spmv_f64(int64_t, int64_t, const int *, const int *, const double *,
                            const double *, double, double, double *, int, hipStream_t);

#define CHK(x)                                                        \
    do {                                                              \
        hipError_t e = (x);                                           \
        if (e != hipSuccess) {                                        \
            fprintf(stderr, "HIP error %d at %d\n", e, __LINE__);     \
            return 1;                                                 \
        }                                                             \
    } while (0)

int main(int argc, char **argv) {
    if (argc < 3) {
        fprintf(stderr, "usage: %s level.bin subw [iters]\n", argv[0]);
        return 2;
    }
    FILE *f = fopen(argv[1], "rb");
    if (!f) return 2;
    int64_t n, nnz;
    if (fread(&n, 8, 1, f) != 1 || fread(&nnz, 8, 1, f) != 1) return 2;
    std::vector<int> ptr(n + 1), col(nnz);
    std::vector<double> val(nnz);
    if (fread(ptr.data(), 4, n + 1, f) != (size_t)(n + 1)) return 2;
    if (fread(col.data(), 4, nnz, f) != (size_t)nnz) return 2;
    if (fread(val.data(), 8, nnz, f) != (size_t)nnz) return 2;
    fclose(f);
    int subw = atoi(argv[2]);
    int iters = argc > 3 ? atoi(argv[3]) : 20;

    int *dptr, *dcol;
    double *dval, *dx, *dy;
    CHK(hipMalloc(&dptr, (n + 1) * 4));
    CHK(hipMalloc(&dcol, nnz * 4));
    CHK(hipMalloc(&dval, nnz * 8));
    CHK(hipMalloc(&dx, n * 8));
    CHK(hipMalloc(&dy, n * 8));
    CHK(hipMemcpy(dptr, ptr.data(), (n + 1) * 4, hipMemcpyHostToDevice));
    CHK(hipMemcpy(dcol, col.data(), nnz * 4, hipMemcpyHostToDevice));
    CHK(hipMemcpy(dval, val.data(), nnz * 8, hipMemcpyHostToDevice));
    std::vector<double> x(n);
    for (int64_t i = 0; i < n; ++i) x[i] = 1.0 + (double)(i % 97) / 97.0;
    CHK(hipMemcpy(dx, x.data(), n * 8, hipMemcpyHostToDevice));

    amg_spmv_f64(n, nnz, dptr, dcol, dval, dx, 1.0, 0.0, dy, subw, 0);
    CHK(hipDeviceSynchronize());
    hipEvent_t e0, e1;
    CHK(hipEventCreate(&e0));
    CHK(hipEventCreate(&e1));
    CHK(hipEventRecord(e0));
    for (int it = 0; it < iters; ++it)
        amg_spmv_f64(n, nnz, dptr, dcol, dval, dx, 1.0, 0.0, dy, subw, 0);
    CHK(hipEventRecord(e1));
    CHK(hipEventSynchronize(e1));
    float ms = 0;
    CHK(hipEventElapsedTime(&ms, e0, e1));
    double gb = (nnz * 12.0 + (n + 1) * 4.0 + 2.0 * n * 8.0) / 1e9;
    printf("n=%ld nnz=%ld subw=%d: %.3f ms/spmv, %.0f GB/s nominal\n",
           (long)n, (long)nnz, subw, ms / iters, gb / (ms / iters / 1e3));
    return 0;
}
