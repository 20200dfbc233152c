// amgcl_amd — C API implementation: a self-contained plain-C++17/OpenMP
// AMG(smoothed aggregation) + CG/BiCGStab solver with no Python dependency.
//
// Parity: lib/amgcl.cpp of the reference (C wrapper over the builtin host
// backend with runtime parameters + Fortran 1-based entries). The algorithms
// match the framework's host engine semantics (csrc/core/core.cpp): strong
// coupling eps^2*|a_ii*a_jj| < a_ij^2, greedy aggregation, filtered-Jacobi
// smoothed prolongation P = (I - omega Df^-1 Af) Ptent with omega = 2/3,
// eps halved per level, Galerkin coarse operator, SPAI0/damped-Jacobi
// smoothing, dense-LU coarsest solve.

#include "amgcl_amd_c.h"

#include <algorithm>
#include <cmath>
#include <cstdio>
#include <cstring>
#include <map>
#include <memory>
#include <string>
#include <vector>

#ifdef _OPENMP
#include <omp.h>
#endif

#include "amgcl_host.hpp"

using namespace amgclamd_host;


extern "C" {

amgcl_amd_handle amgcl_amd_params_create(void) { return new Params; }

void amgcl_amd_params_seti(amgcl_amd_handle prm, const char *name, int value) {
    static_cast<Params *>(prm)->kv[name] = std::to_string(value);
}

void amgcl_amd_params_setf(amgcl_amd_handle prm, const char *name, double value) {
    char b[64];
    std::snprintf(b, sizeof b, "%.17g", value);
    static_cast<Params *>(prm)->kv[name] = b;
}

void amgcl_amd_params_sets(amgcl_amd_handle prm, const char *name,
                           const char *value) {
    static_cast<Params *>(prm)->kv[name] = value;
}

void amgcl_amd_params_destroy(amgcl_amd_handle prm) {
    delete static_cast<Params *>(prm);
}

static amgcl_amd_handle precond_create(int n, const int *ptr, const int *col,
                                       const double *val, amgcl_amd_handle prm,
                                       int base) {
    Params dflt;
    const Params &p = prm ? *static_cast<Params *>(prm) : dflt;
    auto *P = new Precond;
    P->build(make_csr(n, ptr, col, val, base), p);
    return P;
}

amgcl_amd_handle amgcl_amd_precond_create(int n, const int *ptr, const int *col,
                                          const double *val, amgcl_amd_handle prm) {
    return precond_create(n, ptr, col, val, prm, 0);
}

amgcl_amd_handle amgcl_amd_precond_create_f(int n, const int *ptr, const int *col,
                                            const double *val,
                                            amgcl_amd_handle prm) {
    return precond_create(n, ptr, col, val, prm, 1);
}

void amgcl_amd_precond_apply(amgcl_amd_handle amg, const double *rhs, double *x) {
    static_cast<Precond *>(amg)->apply(rhs, x);
}

int amgcl_amd_precond_report(amgcl_amd_handle amg, char *buf, int len) {
    return static_cast<Precond *>(amg)->report(buf, len);
}

void amgcl_amd_precond_destroy(amgcl_amd_handle amg) {
    delete static_cast<Precond *>(amg);
}

static amgcl_amd_handle solver_create(int n, const int *ptr, const int *col,
                                      const double *val, amgcl_amd_handle prm,
                                      int base) {
    Params dflt;
    const Params &p = prm ? *static_cast<Params *>(prm) : dflt;
    auto *S = new Solver;
    S->type = p.gets("solver.type", "cg");
    if (S->type != "cg" && S->type != "bicgstab") {
        // the compiled engine carries CG and BiCGStab; fail loudly instead
        // of silently substituting (the Python API has the full Krylov set)
        delete S;
        return nullptr;
    }
    S->A = make_csr(n, ptr, col, val, base);
    S->tol = p.getf("solver.tol", 1e-8);
    S->maxiter = p.geti("solver.maxiter", 200);
    S->prec.build(S->A, p);
    return S;
}

amgcl_amd_handle amgcl_amd_solver_create(int n, const int *ptr, const int *col,
                                         const double *val, amgcl_amd_handle prm) {
    return solver_create(n, ptr, col, val, prm, 0);
}

amgcl_amd_handle amgcl_amd_solver_create_f(int n, const int *ptr, const int *col,
                                           const double *val,
                                           amgcl_amd_handle prm) {
    return solver_create(n, ptr, col, val, prm, 1);
}

int amgcl_amd_solver_solve(amgcl_amd_handle solver, const double *rhs, double *x,
                           int *iters, double *resid) {
    auto *S = static_cast<Solver *>(solver);
    return S->solve(S->A, rhs, x, iters, resid);
}

int amgcl_amd_solver_solve_mtx(amgcl_amd_handle solver, const int *A_ptr,
                               const int *A_col, const double *A_val,
                               const double *rhs, double *x, int *iters,
                               double *resid) {
    auto *S = static_cast<Solver *>(solver);
    Csr M = make_csr(S->A.n, A_ptr, A_col, A_val, 0);
    return S->solve(M, rhs, x, iters, resid);
}

int amgcl_amd_solver_report(amgcl_amd_handle solver, char *buf, int len) {
    return static_cast<Solver *>(solver)->prec.report(buf, len);
}

void amgcl_amd_solver_destroy(amgcl_amd_handle solver) {
    delete static_cast<Solver *>(solver);
}

}  // extern "C"
