// amgcl_amd — self-contained host AMG engine shared by the C API
// (amgcl_amd_c.cpp, plain C++/OpenMP) and the torch-free GPU C API
// (csrc/hip/capi_gpu.hip).  Factored out of amgcl_amd_c.cpp verbatim.
#pragma once

#include <algorithm>
#include <cmath>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <map>
#include <string>
#include <vector>

#ifdef _OPENMP
#include <omp.h>
#endif

namespace amgclamd_host {

struct Csr {
    int n = 0, m = 0;
    std::vector<int> ptr, col;
    std::vector<double> val;
    long long nnz() const { return (long long)col.size(); }
};

void spmv(const Csr &A, const double *x, double *y, double alpha = 1.0,
          double beta = 0.0) {
#pragma omp parallel for schedule(static) if (A.n > 8192)
    for (int i = 0; i < A.n; ++i) {
        double s = 0.0;
        for (int j = A.ptr[i]; j < A.ptr[i + 1]; ++j) s += A.val[j] * x[A.col[j]];
        y[i] = beta == 0.0 ? alpha * s : alpha * s + beta * y[i];
    }
}

void residual(const Csr &A, const double *rhs, const double *x, double *r) {
#pragma omp parallel for schedule(static) if (A.n > 8192)
    for (int i = 0; i < A.n; ++i) {
        double s = rhs[i];
        for (int j = A.ptr[i]; j < A.ptr[i + 1]; ++j) s -= A.val[j] * x[A.col[j]];
        r[i] = s;
    }
}

double dot(int n, const double *a, const double *b) {
    double s = 0.0;
#pragma omp parallel for schedule(static) reduction(+ : s) if (n > 8192)
    for (int i = 0; i < n; ++i) s += a[i] * b[i];
    return s;
}

std::vector<double> diagonal(const Csr &A) {
    std::vector<double> d(A.n, 0.0);
#pragma omp parallel for schedule(static) if (A.n > 8192)
    for (int i = 0; i < A.n; ++i)
        for (int j = A.ptr[i]; j < A.ptr[i + 1]; ++j)
            if (A.col[j] == i) d[i] = A.val[j];
    return d;
}

Csr transpose(const Csr &A) {
    Csr T;
    T.n = A.m;
    T.m = A.n;
    T.ptr.assign(T.n + 1, 0);
    for (long long j = 0; j < A.nnz(); ++j) ++T.ptr[A.col[j] + 1];
    for (int i = 0; i < T.n; ++i) T.ptr[i + 1] += T.ptr[i];
    T.col.resize(A.nnz());
    T.val.resize(A.nnz());
    std::vector<int> head(T.ptr.begin(), T.ptr.end() - 1);
    for (int i = 0; i < A.n; ++i)
        for (int j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
            int p = head[A.col[j]]++;
            T.col[p] = i;
            T.val[p] = A.val[j];
        }
    return T;
}

// C = A*B, marker-based two-pass SpGEMM, rows of C sorted.
Csr spgemm(const Csr &A, const Csr &B) {
    Csr C;
    C.n = A.n;
    C.m = B.m;
    C.ptr.assign(C.n + 1, 0);
#pragma omp parallel
    {
        std::vector<int> marker(B.m, -1);
#pragma omp for schedule(dynamic, 256)
        for (int i = 0; i < A.n; ++i) {
            int cnt = 0;
            for (int ja = A.ptr[i]; ja < A.ptr[i + 1]; ++ja) {
                int ca = A.col[ja];
                for (int jb = B.ptr[ca]; jb < B.ptr[ca + 1]; ++jb)
                    if (marker[B.col[jb]] != i) {
                        marker[B.col[jb]] = i;
                        ++cnt;
                    }
            }
            C.ptr[i + 1] = cnt;
        }
    }
    for (int i = 0; i < C.n; ++i) C.ptr[i + 1] += C.ptr[i];
    C.col.resize(C.ptr[C.n]);
    C.val.resize(C.ptr[C.n]);
#pragma omp parallel
    {
        // Position marker plus a row-ownership marker: the position test alone
        // (marker[cb] < beg) is only safe when each thread visits rows in
        // increasing order, which OpenMP>=5 dynamic schedules do not guarantee
        // (non-monotonic by default).  marker_row pins validity to this row.
        std::vector<int> marker(B.m, -1);
        std::vector<int> marker_row(B.m, -1);
#pragma omp for schedule(dynamic, 256)
        for (int i = 0; i < A.n; ++i) {
            int beg = C.ptr[i], end = beg;
            for (int ja = A.ptr[i]; ja < A.ptr[i + 1]; ++ja) {
                int ca = A.col[ja];
                double va = A.val[ja];
                for (int jb = B.ptr[ca]; jb < B.ptr[ca + 1]; ++jb) {
                    int cb = B.col[jb];
                    if (marker_row[cb] != i) {
                        marker_row[cb] = i;
                        marker[cb] = end;
                        C.col[end] = cb;
                        C.val[end] = va * B.val[jb];
                        ++end;
                    } else {
                        C.val[marker[cb]] += va * B.val[jb];
                    }
                }
            }
            // insertion sort the row (rows are short for AMG operators)
            for (int a = beg + 1; a < end; ++a) {
                int c = C.col[a];
                double v = C.val[a];
                int b = a - 1;
                for (; b >= beg && C.col[b] > c; --b) {
                    C.col[b + 1] = C.col[b];
                    C.val[b + 1] = C.val[b];
                }
                C.col[b + 1] = c;
                C.val[b + 1] = v;
            }
        }
    }
    return C;
}

// Greedy aggregation over strong connections; id -2 marks isolated points.
int aggregates(const Csr &A, double eps, std::vector<int> &id,
               std::vector<unsigned char> &strong) {
    const double eps2 = eps * eps;
    std::vector<double> D = diagonal(A);
    strong.assign(A.nnz(), 0);
#pragma omp parallel for schedule(static) if (A.n > 8192)
    for (int i = 0; i < A.n; ++i) {
        double edi = eps2 * D[i];
        for (int j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
            int c = A.col[j];
            double v = A.val[j];
            strong[j] = (c != i) && (edi * D[c] < v * v);
        }
    }
    const int UNDEF = -1, REMOVED = -2;
    id.assign(A.n, REMOVED);
    for (int i = 0; i < A.n; ++i)
        for (int j = A.ptr[i]; j < A.ptr[i + 1]; ++j)
            if (strong[j]) {
                id[i] = UNDEF;
                break;
            }
    int naggr = 0;
    for (int i = 0; i < A.n; ++i) {
        if (id[i] != UNDEF) continue;
        // become a root only if the whole strong 1-ring is unclaimed or
        // provisional-free (greedy seed), then claim the ring
        bool ok = true;
        for (int j = A.ptr[i]; j < A.ptr[i + 1] && ok; ++j)
            if (strong[j] && id[A.col[j]] >= 0) ok = false;
        if (!ok) continue;
        id[i] = naggr;
        for (int j = A.ptr[i]; j < A.ptr[i + 1]; ++j)
            if (strong[j]) id[A.col[j]] = naggr;
        ++naggr;
    }
    // leftovers join any aggregated strong neighbor
    for (int i = 0; i < A.n; ++i) {
        if (id[i] != UNDEF) continue;
        for (int j = A.ptr[i]; j < A.ptr[i + 1]; ++j)
            if (strong[j] && id[A.col[j]] >= 0) {
                id[i] = id[A.col[j]];
                break;
            }
        if (id[i] == UNDEF) {  // strong neighbors all isolated: own aggregate
            id[i] = naggr++;
        }
    }
    return naggr;
}

// P = (I - omega Df^-1 Af) Ptent, built in one fused pass per row.
Csr smoothed_prolongation(const Csr &A, const std::vector<unsigned char> &strong,
                          const std::vector<int> &id, int naggr, double omega) {
    Csr P;
    P.n = A.n;
    P.m = naggr;
    P.ptr.assign(A.n + 1, 0);
#pragma omp parallel
    {
        std::vector<int> marker(naggr, -1);
#pragma omp for schedule(static)
        for (int i = 0; i < A.n; ++i) {
            int cnt = 0;
            if (id[i] >= 0 && marker[id[i]] != i) {
                marker[id[i]] = i;
                ++cnt;
            }
            for (int j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
                if (!strong[j]) continue;
                int a = id[A.col[j]];
                if (a >= 0 && marker[a] != i) {
                    marker[a] = i;
                    ++cnt;
                }
            }
            P.ptr[i + 1] = cnt;
        }
    }
    for (int i = 0; i < A.n; ++i) P.ptr[i + 1] += P.ptr[i];
    P.col.resize(P.ptr[A.n]);
    P.val.resize(P.ptr[A.n]);
#pragma omp parallel
    {
        std::vector<int> marker(naggr, -1);
#pragma omp for schedule(static)
        for (int i = 0; i < A.n; ++i) {
            // filtered diagonal: diagonal plus weak off-diagonal entries
            double dia = 0.0;
            for (int j = A.ptr[i]; j < A.ptr[i + 1]; ++j)
                if (A.col[j] == i || !strong[j]) dia += A.val[j];
            double w = dia != 0.0 ? -omega / dia : 0.0;
            int beg = P.ptr[i], end = beg;
            if (id[i] >= 0) {
                marker[id[i]] = end;
                P.col[end] = id[i];
                P.val[end] = 1.0 - omega;
                ++end;
            }
            for (int j = A.ptr[i]; j < A.ptr[i + 1]; ++j) {
                if (!strong[j]) continue;
                int a = id[A.col[j]];
                if (a < 0) continue;
                double v = w * A.val[j];
                if (marker[a] < beg) {
                    marker[a] = end;
                    P.col[end] = a;
                    P.val[end] = v;
                    ++end;
                } else {
                    P.val[marker[a]] += v;
                }
            }
        }
    }
    return P;
}

// Dense LU with partial pivoting for the coarsest level.
struct DenseLU {
    int n = 0;
    std::vector<double> lu;
    std::vector<int> piv;

    void factor(const Csr &A) {
        n = A.n;
        lu.assign((size_t)n * n, 0.0);
        piv.resize(n);
        for (int i = 0; i < n; ++i)
            for (int j = A.ptr[i]; j < A.ptr[i + 1]; ++j)
                lu[(size_t)i * n + A.col[j]] = A.val[j];
        for (int k = 0; k < n; ++k) {
            int p = k;
            for (int i = k + 1; i < n; ++i)
                if (std::fabs(lu[(size_t)i * n + k]) > std::fabs(lu[(size_t)p * n + k]))
                    p = i;
            piv[k] = p;
            if (p != k)
                for (int j = 0; j < n; ++j)
                    std::swap(lu[(size_t)k * n + j], lu[(size_t)p * n + j]);
            double d = lu[(size_t)k * n + k];
            if (d == 0.0) continue;
            for (int i = k + 1; i < n; ++i) {
                double f = lu[(size_t)i * n + k] / d;
                lu[(size_t)i * n + k] = f;
                for (int j = k + 1; j < n; ++j)
                    lu[(size_t)i * n + j] -= f * lu[(size_t)k * n + j];
            }
        }
    }

    void solve(const double *b, double *x) const {
        std::vector<double> y(b, b + n);
        for (int k = 0; k < n; ++k) {
            if (piv[k] != k) std::swap(y[k], y[piv[k]]);
            for (int i = k + 1; i < n; ++i) y[i] -= lu[(size_t)i * n + k] * y[k];
        }
        for (int i = n - 1; i >= 0; --i) {
            double s = y[i];
            for (int j = i + 1; j < n; ++j) s -= lu[(size_t)i * n + j] * x[j];
            double d = lu[(size_t)i * n + i];
            x[i] = d != 0.0 ? s / d : 0.0;
        }
    }
};

struct Params {
    std::map<std::string, std::string> kv;

    std::string gets(const char *k, const char *dflt) const {
        auto it = kv.find(k);
        return it == kv.end() ? dflt : it->second;
    }
    double getf(const char *k, double dflt) const {
        auto it = kv.find(k);
        return it == kv.end() ? dflt : std::stod(it->second);
    }
    int geti(const char *k, int dflt) const {
        auto it = kv.find(k);
        return it == kv.end() ? dflt : std::stoi(it->second);
    }
};

struct Level {
    Csr A, P, R;
    std::vector<double> M;          // diagonal smoother weights
    std::vector<double> f, u, t;    // work vectors
};

struct Precond {
    std::vector<Level> lvl;
    DenseLU coarse;
    int npre = 1, npost = 1, ncycle = 1;

    void build(Csr A, const Params &p) {
        double eps = p.getf("precond.coarsening.eps_strong", 0.08);
        const int coarse_enough = p.geti("precond.coarse_enough", 1000);
        const int max_levels = p.geti("precond.max_levels", 20);
        npre = p.geti("precond.npre", 1);
        npost = p.geti("precond.npost", 1);
        ncycle = p.geti("precond.ncycle", 1);
        const std::string relax = p.gets("precond.relax.type", "spai0");
        const double damping = p.getf("precond.relax.damping", 0.72);

        while ((int)lvl.size() < max_levels) {
            Level L;
            L.A = std::move(A);
            L.f.resize(L.A.n);
            L.u.resize(L.A.n);
            L.t.resize(L.A.n);
            if (L.A.n <= coarse_enough) {
                coarse.factor(L.A);
                lvl.push_back(std::move(L));
                break;
            }
            // smoother weights
            L.M.resize(L.A.n);
            if (relax == "spai0") {
#pragma omp parallel for schedule(static)
                for (int i = 0; i < L.A.n; ++i) {
                    double num = 0.0, den = 0.0;
                    for (int j = L.A.ptr[i]; j < L.A.ptr[i + 1]; ++j) {
                        double v = L.A.val[j];
                        den += v * v;
                        if (L.A.col[j] == i) num = v;
                    }
                    L.M[i] = den != 0.0 ? num / den : 0.0;
                }
            } else {  // damped_jacobi
                std::vector<double> d = diagonal(L.A);
#pragma omp parallel for schedule(static)
                for (int i = 0; i < L.A.n; ++i)
                    L.M[i] = d[i] != 0.0 ? damping / d[i] : 0.0;
            }
            std::vector<int> id;
            std::vector<unsigned char> strong;
            int naggr = aggregates(L.A, eps, id, strong);
            eps *= 0.5;
            if (naggr == 0 || naggr >= L.A.n) {  // no coarsening progress
                coarse.factor(L.A);
                lvl.push_back(std::move(L));
                break;
            }
            L.P = smoothed_prolongation(L.A, strong, id, naggr, 2.0 / 3.0);
            L.R = transpose(L.P);
            Csr AP = spgemm(L.A, L.P);
            A = spgemm(L.R, AP);
            lvl.push_back(std::move(L));
        }
    }

    void relax_step(Level &L, const double *rhs, double *x) const {
        residual(L.A, rhs, x, L.t.data());
#pragma omp parallel for schedule(static) if (L.A.n > 8192)
        for (int i = 0; i < L.A.n; ++i) x[i] += L.M[i] * L.t[i];
    }

    void cycle(int k) const {
        Level &L = const_cast<Level &>(lvl[k]);
        if (k == (int)lvl.size() - 1) {
            if (coarse.n == L.A.n) {
                coarse.solve(L.f.data(), L.u.data());
            } else {
                std::fill(L.u.begin(), L.u.end(), 0.0);
                for (int s = 0; s < 4; ++s) relax_step(L, L.f.data(), L.u.data());
            }
            return;
        }
        Level &C = const_cast<Level &>(lvl[k + 1]);
        std::fill(L.u.begin(), L.u.end(), 0.0);
        for (int c = 0; c < ncycle; ++c) {
            for (int s = 0; s < npre; ++s) relax_step(L, L.f.data(), L.u.data());
            residual(L.A, L.f.data(), L.u.data(), L.t.data());
            spmv(L.R, L.t.data(), C.f.data());
            cycle(k + 1);
            spmv(L.P, C.u.data(), L.t.data());
#pragma omp parallel for schedule(static) if (L.A.n > 8192)
            for (int i = 0; i < L.A.n; ++i) L.u[i] += L.t[i];
            for (int s = 0; s < npost; ++s) relax_step(L, L.f.data(), L.u.data());
        }
    }

    void apply(const double *rhs, double *x) const {
        Level &L = const_cast<Level &>(lvl[0]);
        std::copy(rhs, rhs + L.A.n, L.f.begin());
        cycle(0);
        std::copy(L.u.begin(), L.u.end(), x);
    }

    int report(char *buf, int len) const {
        std::string s = "level   unknowns    nonzeros\n";
        long long nnz0 = lvl.empty() ? 1 : lvl[0].A.nnz();
        double opc = 0.0;
        char line[128];
        for (size_t k = 0; k < lvl.size(); ++k) {
            std::snprintf(line, sizeof line, "%5zu %10d %11lld\n", k, lvl[k].A.n,
                          lvl[k].A.nnz());
            s += line;
            opc += (double)lvl[k].A.nnz();
        }
        std::snprintf(line, sizeof line, "operator complexity: %.2f\n",
                      opc / (double)nnz0);
        s += line;
        if (buf && len > 0) {
            std::strncpy(buf, s.c_str(), (size_t)len - 1);
            buf[len - 1] = '\0';
        }
        return (int)s.size() + 1;
    }
};

struct Solver {
    Csr A;
    Precond prec;
    std::string type = "cg";
    double tol = 1e-8;
    int maxiter = 200;

    int solve(const Csr &M, const double *rhs, double *x, int *iters,
              double *resid) const {
        return type == "bicgstab" ? bicgstab(M, rhs, x, iters, resid)
                                  : cg(M, rhs, x, iters, resid);
    }

    int cg(const Csr &M, const double *rhs, double *x, int *out_it,
           double *out_res) const {
        const int n = M.n;
        std::vector<double> r(n), z(n), p(n), q(n);
        std::fill(x, x + n, 0.0);
        std::copy(rhs, rhs + n, r.begin());
        double norm_rhs = std::sqrt(dot(n, rhs, rhs));
        if (norm_rhs == 0.0) {
            *out_it = 0;
            *out_res = 0.0;
            return 0;
        }
        double rho1 = 0.0, rho2 = 0.0, res = 1.0;
        int it = 0;
        for (; it < maxiter; ++it) {
            res = std::sqrt(dot(n, r.data(), r.data())) / norm_rhs;
            if (res < tol) break;
            prec.apply(r.data(), z.data());
            rho2 = rho1;
            rho1 = dot(n, r.data(), z.data());
            if (it == 0)
                std::copy(z.begin(), z.end(), p.begin());
            else {
                double beta = rho1 / rho2;
                for (int i = 0; i < n; ++i) p[i] = z[i] + beta * p[i];
            }
            spmv(M, p.data(), q.data());
            double alpha = rho1 / dot(n, q.data(), p.data());
            for (int i = 0; i < n; ++i) {
                x[i] += alpha * p[i];
                r[i] -= alpha * q[i];
            }
        }
        *out_it = it;
        *out_res = res;
        return res < tol ? 0 : 1;
    }

    int bicgstab(const Csr &M, const double *rhs, double *x, int *out_it,
                 double *out_res) const {
        const int n = M.n;
        std::vector<double> r(n), rh(n), p(n), ph(n), v(n), s(n), sh(n), t(n);
        std::fill(x, x + n, 0.0);
        std::copy(rhs, rhs + n, r.begin());
        std::copy(rhs, rhs + n, rh.begin());
        double norm_rhs = std::sqrt(dot(n, rhs, rhs));
        if (norm_rhs == 0.0) {
            *out_it = 0;
            *out_res = 0.0;
            return 0;
        }
        double rho1 = 0, rho2 = 0, alpha = 0, omega = 0, res = 1.0;
        int it = 0;
        for (; it < maxiter; ++it) {
            res = std::sqrt(dot(n, r.data(), r.data())) / norm_rhs;
            if (res < tol) break;
            rho2 = rho1;
            rho1 = dot(n, rh.data(), r.data());
            if (rho1 == 0.0) break;
            if (it == 0)
                std::copy(r.begin(), r.end(), p.begin());
            else {
                double beta = (rho1 / rho2) * (alpha / omega);
                for (int i = 0; i < n; ++i) p[i] = r[i] + beta * (p[i] - omega * v[i]);
            }
            prec.apply(p.data(), ph.data());
            spmv(M, ph.data(), v.data());
            alpha = rho1 / dot(n, rh.data(), v.data());
            for (int i = 0; i < n; ++i) s[i] = r[i] - alpha * v[i];
            double norm_s = std::sqrt(dot(n, s.data(), s.data()));
            if (norm_s / norm_rhs < tol) {
                for (int i = 0; i < n; ++i) x[i] += alpha * ph[i];
                ++it;
                res = norm_s / norm_rhs;
                break;
            }
            prec.apply(s.data(), sh.data());
            spmv(M, sh.data(), t.data());
            omega = dot(n, t.data(), s.data()) / dot(n, t.data(), t.data());
            for (int i = 0; i < n; ++i) {
                x[i] += alpha * ph[i] + omega * sh[i];
                r[i] = s[i] - omega * t[i];
            }
            if (omega == 0.0) break;
        }
        *out_it = it;
        *out_res = res;
        return res < tol ? 0 : 1;
    }
};

Csr make_csr(int n, const int *ptr, const int *col, const double *val, int base) {
    Csr A;
    A.n = A.m = n;
    A.ptr.assign(ptr, ptr + n + 1);
    A.col.assign(col, col + (ptr[n] - base));
    A.val.assign(val, val + (ptr[n] - base));
    if (base) {
        for (auto &x : A.ptr) x -= base;
        for (auto &x : A.col) x -= base;
    }
    return A;
}

}  // namespace amgclamd_host
