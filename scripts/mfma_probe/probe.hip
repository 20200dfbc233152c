// Empirical fragment-layout probe for v_mfma_f64_16x16x4_f64 on gfx950.
// For each source lane e: a[l] = l+1, b[l] = (l==e), one MFMA, dump all
// 4 acc items of all 64 lanes.  Host-side analysis reconstructs the A/B
// lane->element mappings given the guide-verified D mapping.
#include <hip/hip_runtime.h>
#include <cstdio>

typedef double d4_t __attribute__((ext_vector_type(4)));

__global__ void probe_k(double *out /* 64 runs x 64 lanes x 4 */) {
    int l = threadIdx.x;
    for (int e = 0; e < 64; ++e) {
        double a = (double)(l + 1);
        double b = (l == e) ? 1.0 : 0.0;
        d4_t acc = {0, 0, 0, 0};
        acc = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc, 0, 0, 0);
        for (int i = 0; i < 4; ++i) out[(e * 64 + l) * 4 + i] = acc[i];
    }
}

int main() {
    double *d;
    (void)hipMalloc(&d, 64 * 64 * 4 * sizeof(double));
    probe_k<<<1, 64>>>(d);
    (void)hipDeviceSynchronize();
    double *h = new double[64 * 64 * 4];
    (void)hipMemcpy(h, d, 64 * 64 * 4 * sizeof(double), hipMemcpyDeviceToHost);
    // D mapping assumed: lane l item i -> D[4*(l/16)+i][l%16]
    for (int e = 0; e < 64; ++e) {
        printf("e=%2d :", e);
        for (int l = 0; l < 64; ++l)
            for (int i = 0; i < 4; ++i) {
                double v = h[(e * 64 + l) * 4 + i];
                if (v != 0.0)
                    printf(" D[%d][%d]=%g(src a-lane %d)", 4 * (l / 16) + i,
                           l % 16, v, (int)v - 1);
            }
        printf("\n");
    }
    return 0;
}
