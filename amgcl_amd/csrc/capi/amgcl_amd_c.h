/* amgcl_amd — C API (host engine).
 *
 * Parity: the reference ships a C-callable shared library wrapping its
 * builtin (OpenMP host) backend with runtime-configurable parameters and
 * 1-based entry points for Fortran (lib/amgcl.h, lib/amgcl.cpp). This is
 * the equivalent surface for amgcl_amd: a self-contained plain-C++/OpenMP
 * AMG-preconditioned Krylov solver with no Python dependency.
 *
 *   h = amgcl_amd_params_create();
 *   amgcl_amd_params_sets(h, "solver.type", "bicgstab");
 *   amgcl_amd_params_setf(h, "solver.tol", 1e-8);
 *   s = amgcl_amd_solver_create(n, ptr, col, val, h);
 *   amgcl_amd_solver_solve(s, rhs, x, &iters, &resid);
 *   amgcl_amd_solver_destroy(s);
 *   amgcl_amd_params_destroy(h);
 *
 * Recognized parameters:
 *   solver.type            cg | bicgstab            (default cg)
 *   solver.tol             relative tolerance        (default 1e-8)
 *   solver.maxiter         iteration cap             (default 200)
 *   precond.relax.type     spai0 | damped_jacobi     (default spai0)
 *   precond.relax.damping  Jacobi damping            (default 0.72)
 *   precond.coarsening.eps_strong                    (default 0.08)
 *   precond.coarse_enough  direct-solve threshold    (default 1000)
 *   precond.max_levels     hierarchy depth cap       (default 20)
 *   precond.npre / npost / ncycle                    (default 1 / 1 / 1)
 */
#ifndef AMGCL_AMD_C_H
#define AMGCL_AMD_C_H

#ifdef __cplusplus
extern "C" {
#endif

typedef void *amgcl_amd_handle;

/* parameter list ---------------------------------------------------------- */
amgcl_amd_handle amgcl_amd_params_create(void);
void amgcl_amd_params_seti(amgcl_amd_handle prm, const char *name, int value);
void amgcl_amd_params_setf(amgcl_amd_handle prm, const char *name, double value);
void amgcl_amd_params_sets(amgcl_amd_handle prm, const char *name, const char *value);
void amgcl_amd_params_destroy(amgcl_amd_handle prm);

/* AMG preconditioner ------------------------------------------------------ */
amgcl_amd_handle amgcl_amd_precond_create(int n, const int *ptr, const int *col,
                                          const double *val, amgcl_amd_handle prm);
/* 1-based (Fortran) index variant */
amgcl_amd_handle amgcl_amd_precond_create_f(int n, const int *ptr, const int *col,
                                            const double *val, amgcl_amd_handle prm);
void amgcl_amd_precond_apply(amgcl_amd_handle amg, const double *rhs, double *x);
/* writes a per-level summary into buf (truncated to len); returns needed size */
int amgcl_amd_precond_report(amgcl_amd_handle amg, char *buf, int len);
void amgcl_amd_precond_destroy(amgcl_amd_handle amg);

/* AMG-preconditioned iterative solver ------------------------------------- */
amgcl_amd_handle amgcl_amd_solver_create(int n, const int *ptr, const int *col,
                                         const double *val, amgcl_amd_handle prm);
amgcl_amd_handle amgcl_amd_solver_create_f(int n, const int *ptr, const int *col,
                                           const double *val, amgcl_amd_handle prm);
/* returns 0 on convergence, 1 when maxiter was reached */
int amgcl_amd_solver_solve(amgcl_amd_handle solver, const double *rhs, double *x,
                           int *iters, double *resid);
/* solve with a different matrix of the same pattern class (lagged precond) */
int amgcl_amd_solver_solve_mtx(amgcl_amd_handle solver, const int *A_ptr,
                               const int *A_col, const double *A_val,
                               const double *rhs, double *x, int *iters,
                               double *resid);
int amgcl_amd_solver_report(amgcl_amd_handle solver, char *buf, int len);
void amgcl_amd_solver_destroy(amgcl_amd_handle solver);

/* Torch-free GPU solver (exported by libamghip.so, NOT libamgclamd_c.so):
 * hierarchy assembly + the native gfx950 solve driver, no Python/torch in
 * the process.  `config` is "key=value;key=value" with the same keys as the
 * Params object (solver.type/tol/maxiter, precond.coarse_enough, ...).
 * `precond.setup` selects where the hierarchy is built: "device" runs the
 * whole SA setup (strong/MIS aggregation/smoothed P/Galerkin + SELL-64
 * images) on the GPU via the setup.hip kernels, "host" uses the C++/OpenMP
 * engine + upload, "auto" (default) picks device above 200k rows.  The
 * device setup hands the tail below `precond.device_handoff` (20000) rows
 * to the host engine, and falls back to the host path entirely if a row
 * overflows its LDS buffers. */
amgcl_amd_handle amgcl_amd_gpu_solver_create(int n, const int *ptr, const int *col,
                                             const double *val, const char *config);
int amgcl_amd_gpu_solver_solve(amgcl_amd_handle solver, const double *rhs,
                               double *x, int *iters, double *resid);
void amgcl_amd_gpu_solver_destroy(amgcl_amd_handle solver);

#ifdef __cplusplus
}
#endif

#endif
