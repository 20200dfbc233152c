/* Torch-free GPU solve from plain C (parity role: the reference's lib/
 * examples; see also examples/poisson_f.f90 for the Fortran twin).
 *
 * Build (no Python, no torch anywhere):
 *   hipcc gpu_capi.c -o gpu_capi \
 *     -L../amgcl_amd/_hip -lamghip -Wl,-rpath,$PWD/../amgcl_amd/_hip
 * (or gcc, linking the same libamghip.so and hip runtime)
 */
#include <stdio.h>
#include <stdlib.h>

#include "../amgcl_amd/csrc/capi/amgcl_amd_c.h"

int main(void) {
    const int m = 32, n = m * m * m;
    int *ptr = malloc((n + 1) * sizeof(int));
    int *col = malloc(7 * (size_t)n * sizeof(int));
    double *val = malloc(7 * (size_t)n * sizeof(double));
    double *rhs = malloc(n * sizeof(double));
    double *x = calloc(n, sizeof(double));
    int idx = 0, row = 0;
    ptr[0] = 0;
    for (int k = 0; k < m; ++k)
        for (int j = 0; j < m; ++j)
            for (int i = 0; i < m; ++i, ++row) {
                if (k) { col[idx] = row - m * m; val[idx++] = -1.0; }
                if (j) { col[idx] = row - m; val[idx++] = -1.0; }
                if (i) { col[idx] = row - 1; val[idx++] = -1.0; }
                col[idx] = row; val[idx++] = 6.0;
                if (i + 1 < m) { col[idx] = row + 1; val[idx++] = -1.0; }
                if (j + 1 < m) { col[idx] = row + m; val[idx++] = -1.0; }
                if (k + 1 < m) { col[idx] = row + m * m; val[idx++] = -1.0; }
                ptr[row + 1] = idx;
                rhs[row] = 1.0;
            }

    amgcl_amd_handle s = amgcl_amd_gpu_solver_create(
        n, ptr, col, val, "solver.type=cg;solver.tol=1e-8");
    if (!s) { fprintf(stderr, "create failed\n"); return 1; }
    int iters = 0;
    double resid = 0.0;
    int rc = amgcl_amd_gpu_solver_solve(s, rhs, x, &iters, &resid);
    printf("rc=%d iters=%d resid=%.3e\n", rc, iters, resid);
    amgcl_amd_gpu_solver_destroy(s);
    return rc == 0 && resid < 1e-8 ? 0 : 1;
}
