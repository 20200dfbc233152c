"""Multi-process distributed tests (gloo backend, CPU, world_size 2/4).

Replaces the reference's mpirun-based example checks
(examples/mpi/check_direct.cpp, runtime_sdd.cpp) with spawn-based
torch.distributed tests that run on the CPU-only container.
"""
import os

import numpy as np
import pytest
import torch.multiprocessing as mp


def _run_dist(rank, world, fn, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        out = fn(rank, world)
        results[rank] = out
    finally:
        dist.destroy_process_group()


def spawn(world, fn, port):
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        procs = [
            ctx.Process(target=_run_dist, args=(r, world, fn, port, results))
            for r in range(world)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(300)
        for p in procs:
            assert p.exitcode == 0, f"rank process failed: {p.exitcode}"
        return dict(results)


def _solve_poisson(rank, world):
    import amgcl_amd as am
    from amgcl_amd.parallel import make_dist_solver

    n = 20
    strip, b, row_beg, row_end = am.poisson3d_strip(n, rank, world, rhs="ones")
    solve = make_dist_solver(
        strip,
        {"precond": {"class": "amg", "coarse_enough": 200},
         "solver": {"type": "cg", "tol": 1e-8, "maxiter": 200}},
        backend="cpu",
    )
    x, iters, resid = solve(b)
    xg = solve.gather_solution(x)
    return iters, resid, None if xg is None else xg.tolist()


def _spmv_check(rank, world):
    import amgcl_amd as am
    from amgcl_amd.parallel import DistBackend
    from amgcl_amd.backend import make_backend

    n = 12
    strip, _, row_beg, row_end = am.poisson3d_strip(n, rank, world)
    backend = DistBackend(make_backend("cpu"))
    A = backend.matrix(strip)
    rng = np.random.default_rng(5)
    x_global = rng.standard_normal(n**3)
    x = backend.from_host(x_global[row_beg:row_end])
    y = backend.vector(row_end - row_beg)
    backend.spmv(1.0, A, x, 0.0, y)
    return row_beg, row_end, backend.to_host(y).tolist()


@pytest.mark.parametrize("world", [2, 4])
def test_dist_spmv_matches_serial(world):
    results = spawn(world, _spmv_check, 29511 + world)
    import amgcl_amd as am

    n = 12
    A, _ = am.poisson3d(n)
    rng = np.random.default_rng(5)
    x = rng.standard_normal(n**3)
    ref = A @ x
    for rank, (rb, re_, y) in results.items():
        np.testing.assert_allclose(np.array(y), ref[rb:re_], rtol=1e-13, atol=1e-13)


@pytest.mark.parametrize("world", [2])
def test_dist_solve_converges_and_matches(world):
    results = spawn(world, _solve_poisson, 29611 + world)
    import amgcl_amd as am

    n = 20
    A, b = am.poisson3d(n, rhs="ones")
    iters, resid, xg = results[0]
    assert resid < 1e-8
    assert iters < 120
    xg = np.array(xg)
    r = b - A @ xg
    assert np.linalg.norm(r) / np.linalg.norm(b) < 1e-7


def test_dist_inner_product_single_rank_passthrough():
    # world_size=1 path: DistMatrix with no neighbors must behave like local
    # (exercised via bench --gpus 1; here just the mask split logic)
    import amgcl_amd as am
    from amgcl_amd.matrix import CSR

    n = 8
    strip, b, rb, re_ = am.poisson3d_strip(n, 0, 2)
    assert strip.nrows == n**3 // 2
    assert strip.ncols == n**3


def _solve_deflated(rank, world):
    import numpy as np

    import amgcl_amd as am
    from amgcl_amd.parallel import make_dist_solver

    n = 20
    strip, b, row_beg, row_end = am.poisson3d_strip(n, rank, world, rhs="ones")
    # node coordinates of the local rows (for linear deflation)
    idx = np.arange(row_beg, row_end)
    coords = np.stack([idx % n, (idx // n) % n, idx // (n * n)], axis=1)
    results = {}
    for defl in (None, {"type": "constant"}, {"type": "linear", "coords_raw": coords}):
        prm = {"precond": {"class": "amg", "coarse_enough": 200},
               "solver": {"type": "cg", "tol": 1e-8, "maxiter": 300}}
        if defl:
            prm["deflation"] = defl
        solve = make_dist_solver(strip, prm, backend="cpu")
        x, iters, resid = solve(b)
        xg = solve.gather_solution(x)
        key = defl["type"] if defl else "none"
        results[key] = (iters, resid, None if xg is None else xg.tolist())
    return results


@pytest.mark.parametrize("world", [2, 4])
def test_subdomain_deflation(world):
    out = spawn(world, _solve_deflated, 29811 + world)
    import amgcl_amd as am

    n = 20
    A, b = am.poisson3d(n, rhs="ones")
    r0 = out[0]
    for kind in ("none", "constant", "linear"):
        iters, resid, xg = r0[kind]
        assert resid < 1e-7, kind
        x = np.array(xg)
        assert np.linalg.norm(b - A @ x) / np.linalg.norm(b) < 1e-6, kind
    # deflation must not be worse than plain block-Jacobi
    assert r0["constant"][0] <= r0["none"][0] + 2
    assert r0["linear"][0] <= r0["constant"][0] + 2


def _solve_dist_schur(rank, world):
    import numpy as np
    import scipy.sparse as sp

    import amgcl_amd as am
    from amgcl_amd.matrix import CSR
    from amgcl_amd.parallel import make_dist_solver

    # global stokes-like system with INTERLEAVED (u, p) per node so row strips
    # contain both fields; local Schur field split per rank (additive Schwarz)
    n = 12
    Ap, _ = am.poisson3d(n)
    a = Ap.to_scipy()
    nv = a.shape[0]
    B = 0.1 * (sp.identity(nv) - sp.diags(np.ones(nv - 1), 1)).tocsr()
    K = sp.bmat([[a, B], [B.T, a + sp.identity(nv)]], format="csr")
    perm = np.arange(2 * nv).reshape(2, nv).T.ravel()  # interleave
    K = K[perm][:, perm].tocsr()
    K.sort_indices()
    n_glob = 2 * nv
    row_beg = (n_glob * rank) // world
    row_end = (n_glob * (rank + 1)) // world
    lo, hi = K.indptr[row_beg], K.indptr[row_end]
    strip = CSR(row_end - row_beg, n_glob,
                K.indptr[row_beg:row_end + 1] - lo, K.indices[lo:hi], K.data[lo:hi])
    rng = np.random.default_rng(1)
    b_glob = rng.standard_normal(n_glob)
    pmask_local = (np.arange(row_beg, row_end) % 2) == 1
    solve = make_dist_solver(
        strip,
        {"precond": {"class": "schur_pressure_correction",
                     "pmask_raw": pmask_local,
                     "psolver": {"precond": {"class": "amg", "coarse_enough": 300},
                                 "solver": {"type": "preonly"}}},
         "solver": {"type": "fgmres", "tol": 1e-8, "maxiter": 300}},
        backend="cpu",
    )
    x, iters, resid = solve(b_glob[row_beg:row_end])
    xg = solve.gather_solution(x)
    return iters, resid, None if xg is None else xg.tolist(), K.toarray().tolist() if rank == -1 else None


@pytest.mark.parametrize("world", [2])
def test_distributed_schur_field_split(world):
    out = spawn(world, _solve_dist_schur, 29911 + world)
    import scipy.sparse as sp

    import amgcl_amd as am

    iters, resid, xg, _ = out[0]
    assert resid < 1e-7
    # rebuild the global matrix to verify the true residual
    n = 12
    Ap, _ = am.poisson3d(n)
    a = Ap.to_scipy()
    nv = a.shape[0]
    B = 0.1 * (sp.identity(nv) - sp.diags(np.ones(nv - 1), 1)).tocsr()
    K = sp.bmat([[a, B], [B.T, a + sp.identity(nv)]], format="csr")
    perm = np.arange(2 * nv).reshape(2, nv).T.ravel()
    K = K[perm][:, perm].tocsr()
    rng = np.random.default_rng(1)
    b = rng.standard_normal(2 * nv)
    x = np.array(xg)
    assert np.linalg.norm(b - K @ x) / np.linalg.norm(b) < 1e-6


def _solve_bicgstab(rank, world):
    import amgcl_amd as am
    from amgcl_amd.parallel import make_dist_solver

    n = 16
    strip, b, rb, re_ = am.poisson3d_strip(n, rank, world, rhs="ones")
    solve = make_dist_solver(
        strip,
        {"precond": {"class": "amg", "coarse_enough": 300},
         "solver": {"type": "bicgstab", "tol": 1e-8, "maxiter": 200}},
        backend="cpu",
    )
    x, iters, resid = solve(b)
    xg = solve.gather_solution(x)
    return iters, resid, None if xg is None else xg.tolist()


@pytest.mark.parametrize("world", [2])
def test_dist_bicgstab_fused_dots(world):
    """Distributed BiCGStab exercises the fused 2-dot allreduce path."""
    out = spawn(world, _solve_bicgstab, 29961 + world)
    import amgcl_amd as am

    iters, resid, xg = out[0]
    assert resid < 1e-7
    n = 16
    A, b = am.poisson3d(n, rhs="ones")
    x = np.array(xg)
    assert np.linalg.norm(b - A @ x) / np.linalg.norm(b) < 1e-6


def _solve_dist_amg(rank, world):
    import amgcl_amd as am
    from amgcl_amd.parallel import make_dist_solver

    n = 20
    strip, b, row_beg, row_end = am.poisson3d_strip(n, rank, world, rhs="ones")
    solve = make_dist_solver(
        strip,
        {"precond": {"class": "dist_amg", "coarse_enough": 300,
                     "repart_threshold": 500},
         "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}},
        backend="cpu",
    )
    nlevels = len(solve.P.levels)
    x, iters, resid = solve(b)
    xg = solve.gather_solution(x)
    return iters, resid, nlevels, None if xg is None else xg.tolist()


@pytest.mark.parametrize("world", [2, 3])
def test_dist_amg_hierarchy(world):
    """Cross-rank AMG hierarchy: distributed Galerkin coarse operators +
    replicated coarse solve (parity: mpi/amg.hpp)."""
    out = spawn(world, _solve_dist_amg, 30011 + world)
    import amgcl_amd as am

    n = 20
    A, b = am.poisson3d(n, rhs="ones")
    it0, res0, nlev, xg = out[0]
    assert res0 < 1e-8
    assert nlev >= 2  # a real multilevel hierarchy was built
    x = np.asarray(xg)
    assert np.linalg.norm(b - A @ x) / np.linalg.norm(b) < 1e-7
    # iteration counts must agree across ranks and stay in the serial class
    s = am.make_solver(A, {"solver": {"type": "cg", "tol": 1e-8, "maxiter": 100},
                           "precond": {"class": "amg", "coarse_enough": 300}})
    _, it_serial, _ = s(b)
    for r in range(1, world):
        assert out[r][0] == it0
    assert it0 <= 2 * it_serial + 5


def _galerkin_consistency(rank, world):
    """The distributed coarse operator must equal the explicitly assembled
    global P^T A P (exactness of the P-row halo exchange)."""
    import scipy.sparse as sp

    import amgcl_amd as am
    import torch.distributed as dist
    from amgcl_amd.backend import make_backend
    from amgcl_amd.parallel.dist_amg import DistAMG
    from amgcl_amd.parallel.dist_backend import DistBackend

    n = 12
    strip, _, row_beg, row_end = am.poisson3d_strip(n, rank, world)
    backend = DistBackend(make_backend("cpu"))
    A = backend.matrix(strip)
    amg = DistAMG(A, {"coarse_enough": 100, "repart_threshold": 150}, backend)
    L1 = amg.levels[1].A  # first coarse DistMatrix

    # assemble the global coarse matrix from the per-rank strips
    loc = L1.A_loc_host.to_scipy()
    cols_g = np.asarray(L1.A_loc_host.col, dtype=np.int64) + L1.row_beg
    parts = sp.csr_matrix((L1.A_loc_host.val, cols_g, L1.A_loc_host.ptr),
                          shape=(L1.n_loc, L1.n_global))
    if L1.A_rem_host is not None:
        gg = np.asarray(L1.ghost_global, dtype=np.int64)
        parts = parts + sp.csr_matrix(
            (L1.A_rem_host.val, gg[np.asarray(L1.A_rem_host.col)],
             L1.A_rem_host.ptr), shape=(L1.n_loc, L1.n_global))
    got = [None] * world
    dist.all_gather_object(got, (parts.indptr, parts.indices, parts.data))

    # the reference computation: global block-diagonal P, global A
    P_strip = amg.levels[0].P
    pg = [None] * world
    dist.all_gather_object(
        pg, (P_strip.ptr, P_strip.col, P_strip.val, P_strip.nrows, P_strip.ncols))
    if rank != 0:
        return True
    Ac = sp.vstack([sp.csr_matrix((v, c, p), shape=(len(p) - 1, L1.n_global))
                    for p, c, v in got], format="csr")
    A_glob, _ = am.poisson3d(n)
    Pg = sp.block_diag(
        [sp.csr_matrix((v, c, p), shape=(nr, nc)) for p, c, v, nr, nc in pg],
        format="csr")
    ref = (Pg.T @ A_glob.to_scipy() @ Pg).tocsr()
    return float(abs(Ac - ref).max())


@pytest.mark.parametrize("world", [2, 3])
def test_dist_galerkin_exact(world):
    out = spawn(world, _galerkin_consistency, 30061 + world)
    assert out[0] < 1e-12


def _schur_S_exactness(rank, world):
    """Distributed explicit S strips must equal the serial S = Kpp - Kpu
    D^-1 Kup (cross-rank product terms included)."""
    import scipy.sparse as sp

    import amgcl_amd as am
    import torch.distributed as dist
    from amgcl_amd.matrix import CSR
    from amgcl_amd.backend import make_backend
    from amgcl_amd.parallel.dist_backend import DistBackend
    from amgcl_amd.parallel.schur import DistSchurPressureCorrection

    n = 10
    Ap, _ = am.poisson3d(n)
    a = Ap.to_scipy()
    nv = a.shape[0]
    B = 0.1 * (sp.identity(nv) - sp.diags(np.ones(nv - 1), 1)).tocsr()
    K = sp.bmat([[a, B], [B.T, a + sp.identity(nv)]], format="csr")
    perm = np.arange(2 * nv).reshape(2, nv).T.ravel()
    K = K[perm][:, perm].tocsr()
    K.sort_indices()
    n_glob = 2 * nv
    row_beg = (n_glob * rank) // world
    row_end = (n_glob * (rank + 1)) // world
    lo, hi = K.indptr[row_beg], K.indptr[row_end]
    strip = CSR(row_end - row_beg, n_glob,
                K.indptr[row_beg:row_end + 1] - lo, K.indices[lo:hi], K.data[lo:hi])
    backend = DistBackend(make_backend("cpu"))
    P = DistSchurPressureCorrection(strip, {"pmask_pattern": "%2"},
                                    backend, None)
    # gather the distributed S strips and the dinv pieces
    Ssp = P.psolve.A  # DistMatrix of S
    loc = Ssp.A_loc_host.to_scipy()
    cols_g = np.asarray(Ssp.A_loc_host.col, dtype=np.int64) + Ssp.row_beg
    part = sp.csr_matrix((Ssp.A_loc_host.val, cols_g, Ssp.A_loc_host.ptr),
                         shape=(Ssp.n_loc, Ssp.n_global))
    if Ssp.A_rem_host is not None:
        gg = np.asarray(Ssp.ghost_global, dtype=np.int64)
        part = part + sp.csr_matrix(
            (Ssp.A_rem_host.val, gg[np.asarray(Ssp.A_rem_host.col)],
             Ssp.A_rem_host.ptr), shape=(Ssp.n_loc, Ssp.n_global))
    got = [None] * world
    dist.all_gather_object(got, (part.indptr, part.indices, part.data))
    dv = [None] * world
    dist.all_gather_object(dv, P.dinv_host)
    if rank != 0:
        return 0.0
    S_dist = sp.vstack([sp.csr_matrix((v, c, p), shape=(len(p) - 1, Ssp.n_global))
                        for p, c, v in got], format="csr")
    # serial reference
    pmask = (np.arange(n_glob) % 2) == 1
    um = ~pmask
    Kuu = K[um][:, um]
    Kup = K[um][:, pmask]
    Kpu = K[pmask][:, um]
    Kpp = K[pmask][:, pmask]
    dinv = np.concatenate(dv)
    S_ref = (Kpp - Kpu @ sp.diags(dinv) @ Kup).tocsr()
    return float(abs(S_dist - S_ref).max())


@pytest.mark.parametrize("world", [2, 3])
def test_dist_schur_S_exact(world):
    """The fully-coupled distributed Schur complement equals the serial one
    (parity: mpi/schur_pressure_correction.hpp)."""
    out = spawn(world, _schur_S_exactness, 30111 + world)
    assert out[0] < 1e-12


def _cpr_problem(n=10, b=2):
    import scipy.sparse as sp

    import amgcl_amd as am

    Ap, _ = am.poisson3d(n)
    a = Ap.to_scipy()
    nv = a.shape[0]
    # cell-interleaved 2x2 blocks: pressure first, coupled saturation
    C = sp.csr_matrix(np.array([[1.0, 0.2], [0.3, 1.0]]))
    K = sp.kron(a, C, format="csr") + sp.kron(sp.identity(nv), 0.5 * sp.identity(b),
                                              format="csr")
    K.sort_indices()
    return K.tocsr(), nv * b


def _solve_dist_cpr(rank, world):
    import numpy as np

    import amgcl_amd as am
    from amgcl_amd.matrix import CSR
    from amgcl_amd.parallel import make_dist_solver

    K, n_glob = _cpr_problem()
    b = 2
    ncells = n_glob // b
    cell_beg = (ncells * rank) // world
    cell_end = (ncells * (rank + 1)) // world
    row_beg, row_end = cell_beg * b, cell_end * b
    lo, hi = K.indptr[row_beg], K.indptr[row_end]
    strip = CSR(row_end - row_beg, n_glob,
                K.indptr[row_beg:row_end + 1] - lo, K.indices[lo:hi], K.data[lo:hi])
    rng = np.random.default_rng(2)
    b_glob = rng.standard_normal(n_glob)
    solve = make_dist_solver(
        strip,
        {"precond": {"class": "cpr", "block_size": 2,
                     "psolver": {"precond": {"class": "amg", "coarse_enough": 200},
                                 "solver": {"type": "preonly"}}},
         "solver": {"type": "bicgstab", "tol": 1e-8, "maxiter": 200}},
        backend="cpu",
    )
    x, iters, resid = solve(b_glob[row_beg:row_end])
    xg = solve.gather_solution(x)
    return iters, resid, None if xg is None else xg.tolist()


@pytest.mark.parametrize("world", [2, 3])
def test_dist_cpr(world):
    """Fully-coupled distributed CPR (parity: mpi/cpr.hpp): distributed
    quasi-IMPES pressure stage + global smoother."""
    out = spawn(world, _solve_dist_cpr, 30161 + world)
    iters, resid, xg = out[0]
    assert resid < 1e-7
    K, n_glob = _cpr_problem()
    rng = np.random.default_rng(2)
    b = rng.standard_normal(n_glob)
    x = np.asarray(xg)
    assert np.linalg.norm(b - K @ x) / np.linalg.norm(b) < 1e-6
    # serial CPR on the same problem stays in the same iteration class
    import amgcl_amd as am
    from amgcl_amd.matrix import CSR

    s = am.make_solver(
        CSR.from_scipy(K),
        {"precond": {"class": "cpr", "block_size": 2,
                     "pprecond": {"class": "amg", "coarse_enough": 200}},
         "solver": {"type": "bicgstab", "tol": 1e-8, "maxiter": 200}})
    _, it_serial, _ = s(b)
    assert iters <= 2 * it_serial + 5


def _solve_dist_amg_repart(rank, world):
    import amgcl_amd as am
    from amgcl_amd.parallel import make_dist_solver

    n = 20
    strip, b, row_beg, row_end = am.poisson3d_strip(n, rank, world, rhs="ones")
    solve = make_dist_solver(
        strip,
        {"precond": {"class": "dist_amg", "coarse_enough": 200,
                     "repart_threshold": 2000},
         "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}},
        backend="cpu",
    )
    has_tail = solve.P.tail is not None
    x, iters, resid = solve(b)
    xg = solve.gather_solution(x)
    return iters, resid, has_tail, None if xg is None else xg.tolist()


@pytest.mark.parametrize("world", [2])
def test_dist_amg_replicated_tail(world):
    """Merge-style coarse repartition: below repart_threshold the hierarchy
    is replicated and continued serially per rank (mpi/partition/merge.hpp
    semantics, redundant-apply variant)."""
    out = spawn(world, _solve_dist_amg_repart, 30211 + world)
    iters, resid, has_tail, xg = out[0]
    assert has_tail
    assert resid < 1e-8
    import amgcl_amd as am

    A, b = am.poisson3d(20, rhs="ones")
    x = np.asarray(xg)
    assert np.linalg.norm(b - A @ x) / np.linalg.norm(b) < 1e-7
    # the replicated tail continues the same hierarchy: iterations stay in
    # the serial class
    s = am.make_solver(A, {"solver": {"type": "cg", "tol": 1e-8, "maxiter": 100},
                           "precond": {"class": "amg", "coarse_enough": 200}})
    _, it_serial, _ = s(b)
    assert iters <= it_serial + 6


def _pmis_equality(rank, world):
    """Cross-rank pmis must reproduce the single-process deterministic MIS
    aggregation EXACTLY (same hash keys, same round structure)."""
    import amgcl_amd as am
    import torch.distributed as dist
    from amgcl_amd import _core
    from amgcl_amd.backend import make_backend
    from amgcl_amd.parallel import pmis
    from amgcl_amd.parallel.dist_backend import DistBackend

    n = 14
    strip, _, row_beg, row_end = am.poisson3d_strip(n, rank, world)
    backend = DistBackend(make_backend("cpu"))
    A = backend.matrix(strip)
    ids, _, _ = pmis.pmis_aggregates(A, 0.08, dist, None)
    got = [None] * world
    dist.all_gather_object(got, ids.tolist())
    if rank != 0:
        return None
    return np.concatenate([np.asarray(g) for g in got]).tolist()


@pytest.mark.parametrize("world", [1, 2, 3, 5])
def test_pmis_matches_serial(world):
    out = spawn(world, _pmis_equality, 30261 + world)
    import amgcl_amd as am
    from amgcl_amd import _core

    n = 14
    A, _ = am.poisson3d(n)
    naggr, ids_s, _ = _core.aggregates_parallel(A.nrows, A.ptr, A.col, A.val, 0.08)
    ids_d = np.asarray(out[0], dtype=np.int64)
    ids_s = np.asarray(ids_s, dtype=np.int64)
    # serial ids are compact aggregate numbers; distributed ids are global
    # root node ids. The partitions must be identical: same removed set and a
    # bijection between labels.
    assert np.array_equal(ids_d == -2, ids_s == -2)
    m = ids_s >= 0
    pairs = np.unique(np.stack([ids_s[m], ids_d[m]]), axis=1)
    assert pairs.shape[1] == naggr                      # serial label -> one root
    assert len(np.unique(pairs[1])) == naggr            # roots distinct
    assert len(np.unique(ids_d[m])) == naggr


def _solve_dist_amg_pmis(rank, world):
    import amgcl_amd as am
    from amgcl_amd.parallel import make_dist_solver

    n = 20
    strip, b, row_beg, row_end = am.poisson3d_strip(n, rank, world, rhs="ones")
    solve = make_dist_solver(
        strip,
        {"precond": {"class": "dist_amg", "coarse_enough": 300,
                     "repart_threshold": 500,
                     "coarsening": {"cross_rank": True}},
         "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}},
        backend="cpu",
    )
    x, iters, resid = solve(b)
    xg = solve.gather_solution(x)
    return iters, resid, None if xg is None else xg.tolist()


@pytest.mark.parametrize("world", [2, 3])
def test_dist_amg_cross_rank_pmis(world):
    """DistAMG with cross-rank aggregation: aggregates cross boundaries, the
    hierarchy matches the serial iteration count (same deterministic MIS)."""
    out = spawn(world, _solve_dist_amg_pmis, 30311 + world)
    import amgcl_amd as am

    n = 20
    A, b = am.poisson3d(n, rhs="ones")
    iters, resid, xg = out[0]
    assert resid < 1e-8
    x = np.asarray(xg)
    assert np.linalg.norm(b - A @ x) / np.linalg.norm(b) < 1e-7
    s = am.make_solver(A, {"solver": {"type": "cg", "tol": 1e-8, "maxiter": 100},
                           "precond": {"class": "amg", "coarse_enough": 300}})
    _, it_serial, _ = s(b)
    assert iters <= it_serial + 4  # cross-rank hierarchy ~= serial quality


def _dist_amg_jacobi_wcycle(rank, world):
    import amgcl_amd as am
    from amgcl_amd.parallel import make_dist_solver

    strip, b, rb, re_ = am.poisson3d_strip(16, rank, world, rhs="ones")
    solve = make_dist_solver(
        strip,
        {"precond": {"class": "dist_amg", "coarse_enough": 200,
                     "repart_threshold": 400, "npre": 2, "npost": 2,
                     "ncycle": 2, "relax": {"type": "damped_jacobi",
                                            "damping": 0.72}},
         "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}},
        backend="cpu")
    x, iters, resid = solve(b)
    xg = solve.gather_solution(x)
    return iters, resid, None if xg is None else xg.tolist()


@pytest.mark.parametrize("world", [2])
def test_dist_amg_jacobi_wcycle(world):
    """DistAMG parameter space: damped-Jacobi smoothing, npre/npost=2, W-cycle."""
    out = spawn(world, _dist_amg_jacobi_wcycle, 30361 + world)
    iters, resid, xg = out[0]
    assert resid < 1e-8
    import amgcl_amd as am

    A, b = am.poisson3d(16, rhs="ones")
    x = np.asarray(xg)
    assert np.linalg.norm(b - A @ x) / np.linalg.norm(b) < 1e-7


def test_rcb_partition_quality():
    """Geometric RCB partitioner (parity: mpi/partition class): balanced
    parts and a lower halo surface than 1-D strips on a 3D grid."""
    import amgcl_amd as am
    from amgcl_amd.parallel.partition import (edge_cut, partition_permutation,
                                              rcb_partition, permute_system)

    n = 16
    A, b = am.poisson3d(n, rhs="random")
    idx = np.arange(n ** 3)
    coords = np.stack([idx % n, (idx // n) % n, idx // (n * n)], axis=1)
    for world in (2, 4, 8, 6):
        part = rcb_partition(coords, world)
        sizes = np.bincount(part, minlength=world)
        assert sizes.sum() == n ** 3
        assert sizes.max() - sizes.min() <= world  # balanced
        # 1-D strips as the baseline decomposition
        strip_part = (idx * world) // (n ** 3)
        if world > 2:  # at world 2 both are a single plane cut
            assert edge_cut(A, part) <= edge_cut(A, strip_part)
    # the permuted system solves identically
    part = rcb_partition(coords, 4)
    perm, sizes = partition_permutation(part)
    assert sorted(perm.tolist()) == list(range(n ** 3))
    Ap, bp = permute_system(A, perm, b)
    s = am.make_solver(Ap, {"solver": {"type": "cg", "tol": 1e-8, "maxiter": 100},
                            "precond": {"class": "amg", "coarse_enough": 300}})
    x, iters, resid = s(bp)
    assert resid < 1e-8


def _solve_rcb_partitioned(rank, world):
    import amgcl_amd as am
    from amgcl_amd.matrix import CSR
    from amgcl_amd.parallel import make_dist_solver
    from amgcl_amd.parallel.partition import (partition_permutation,
                                              rcb_partition, permute_system)

    n = 14
    A, b = am.poisson3d(n, rhs="ones")
    idx = np.arange(n ** 3)
    coords = np.stack([idx % n, (idx // n) % n, idx // (n * n)], axis=1)
    part = rcb_partition(coords, world)
    perm, sizes = partition_permutation(part)
    Ap, bp = permute_system(A, perm, b)
    beg = int(np.sum(sizes[:rank]))
    end = beg + sizes[rank]
    m = Ap.to_scipy()
    lo, hi = m.indptr[beg], m.indptr[end]
    strip = CSR(end - beg, n ** 3, m.indptr[beg:end + 1] - lo,
                m.indices[lo:hi], m.data[lo:hi])
    solve = make_dist_solver(
        strip, {"precond": {"class": "amg", "coarse_enough": 300},
                "solver": {"type": "cg", "tol": 1e-8, "maxiter": 200}},
        backend="cpu")
    x, iters, resid = solve(bp[beg:end])
    xg = solve.gather_solution(x)
    return iters, resid, None if xg is None else xg.tolist(), perm.tolist()


@pytest.mark.parametrize("world", [4])
def test_rcb_partitioned_distributed_solve(world):
    """End-to-end: RCB decomposition feeding the distributed solver."""
    out = spawn(world, _solve_rcb_partitioned, 30411 + world)
    import amgcl_amd as am

    iters, resid, xg, perm = out[0]
    assert resid < 1e-8
    n = 14
    A, b = am.poisson3d(n, rhs="ones")
    perm = np.asarray(perm)
    x = np.empty(n ** 3)
    x[perm] = np.asarray(xg)  # un-permute
    assert np.linalg.norm(b - A @ x) / np.linalg.norm(b) < 1e-7


def _solve_field_blocked_schur(rank, world):
    """Extreme split: the global ordering is FIELD-blocked, so one rank owns
    only velocity rows and another only pressure rows (zero-row field blocks
    in the rectangular distributed machinery)."""
    import scipy.sparse as sp

    import amgcl_amd as am
    from amgcl_amd.matrix import CSR
    from amgcl_amd.parallel import make_dist_solver

    n = 8
    Ap, _ = am.poisson3d(n)
    a = Ap.to_scipy()
    nv = a.shape[0]
    B = 0.1 * (sp.identity(nv) - sp.diags(np.ones(nv - 1), 1)).tocsr()
    K = sp.bmat([[a, B], [B.T, a + sp.identity(nv)]], format="csr")
    K.sort_indices()
    n_glob = 2 * nv
    row_beg = (n_glob * rank) // world
    row_end = (n_glob * (rank + 1)) // world
    lo, hi = K.indptr[row_beg], K.indptr[row_end]
    strip = CSR(row_end - row_beg, n_glob,
                K.indptr[row_beg:row_end + 1] - lo, K.indices[lo:hi], K.data[lo:hi])
    pmask_local = np.arange(row_beg, row_end) >= nv
    b = np.random.default_rng(1).standard_normal(n_glob)
    solve = make_dist_solver(
        strip,
        {"precond": {"class": "schur_pressure_correction", "pmask_raw": pmask_local,
                     "psolver": {"precond": {"class": "amg", "coarse_enough": 200},
                                 "solver": {"type": "preonly"}}},
         "solver": {"type": "fgmres", "tol": 1e-8, "maxiter": 300}},
        backend="cpu")
    x, iters, resid = solve(b[row_beg:row_end])
    xg = solve.gather_solution(x)
    return iters, resid, None if xg is None else xg.tolist()


@pytest.mark.parametrize("world", [2])
def test_dist_schur_field_blocked_ordering(world):
    import scipy.sparse as sp

    import amgcl_amd as am

    out = spawn(world, _solve_field_blocked_schur, 30461 + world)
    iters, resid, xg = out[0]
    assert resid < 1e-7
    n = 8
    Ap, _ = am.poisson3d(n)
    a = Ap.to_scipy()
    nv = a.shape[0]
    B = 0.1 * (sp.identity(nv) - sp.diags(np.ones(nv - 1), 1)).tocsr()
    K = sp.bmat([[a, B], [B.T, a + sp.identity(nv)]], format="csr")
    b = np.random.default_rng(1).standard_normal(2 * nv)
    x = np.asarray(xg)
    assert np.linalg.norm(b - K @ x) / np.linalg.norm(b) < 1e-6


def _solve_dist_elasticity(rank, world):
    """Distributed elasticity with per-rank rigid-body nullspace (the
    reference's NullspaceMPI tutorial shape: strips cut on node boundaries,
    RBMs from the local coordinates)."""
    import amgcl_amd as am
    from amgcl_amd.generators import elasticity3d, rigid_body_modes
    from amgcl_amd.matrix import CSR
    from amgcl_amd.parallel import make_dist_solver

    A, b, coords = elasticity3d(10)
    n = A.nrows
    nnodes = n // 3
    node_beg = (nnodes * rank) // world
    node_end = (nnodes * (rank + 1)) // world
    row_beg, row_end = node_beg * 3, node_end * 3
    m = A.to_scipy()
    lo, hi = m.indptr[row_beg], m.indptr[row_end]
    strip = CSR(row_end - row_beg, n, m.indptr[row_beg:row_end + 1] - lo,
                m.indices[lo:hi], m.data[lo:hi])
    B_loc = rigid_body_modes(coords[node_beg:node_end])
    solve = make_dist_solver(
        strip,
        {"precond": {"class": "amg", "coarse_enough": 300,
                     "coarsening": {"type": "smoothed_aggregation",
                                    "block_size": 3, "nullspace_raw": B_loc,
                                    "estimate_spectral_radius": True,
                                    "power_iters": 10},
                     "relax": {"type": "chebyshev"}},
         "solver": {"type": "cg", "tol": 1e-8, "maxiter": 300}},
        backend="cpu")
    x, iters, resid = solve(b[row_beg:row_end])
    xg = solve.gather_solution(x)
    return iters, resid, None if xg is None else xg.tolist()


@pytest.mark.parametrize("world", [2])
def test_dist_elasticity_nullspace(world):
    out = spawn(world, _solve_dist_elasticity, 30511 + world)
    import amgcl_amd as am
    from amgcl_amd.generators import elasticity3d

    iters, resid, xg = out[0]
    assert resid < 1e-8
    assert iters < 120
    A, b, _ = elasticity3d(10)
    x = np.asarray(xg)
    assert np.linalg.norm(b - A @ x) / np.linalg.norm(b) < 1e-7


def test_graph_partition_quality():
    """Coordinate-free greedy graph growing: balanced, connected-ish parts
    with an edge cut within 3x of geometric RCB on the same problem
    (parity role: amgcl/mpi/partition/ptscotch.hpp, parmetis.hpp)."""
    from amgcl_amd.parallel.partition import (edge_cut, graph_partition,
                                              rcb_partition)

    import amgcl_amd as am

    n = 14
    A, _ = am.poisson3d(n)
    part = graph_partition(A, 4)
    sizes = np.bincount(part, minlength=4)
    assert sizes.min() > 0
    assert sizes.max() - sizes.min() <= 2
    idx = np.arange(n ** 3)
    coords = np.stack([idx % n, (idx // n) % n, idx // (n * n)], axis=1)
    cut_rcb = edge_cut(A, rcb_partition(coords, 4))
    cut_graph = edge_cut(A, part)
    assert cut_graph <= 3 * cut_rcb


def _solve_graph_partitioned(rank, world):
    import amgcl_amd as am
    from amgcl_amd.matrix import CSR
    from amgcl_amd.parallel import make_dist_solver
    from amgcl_amd.parallel.partition import (graph_partition,
                                              partition_permutation,
                                              permute_system)

    n = 12
    A, b = am.poisson3d(n, rhs="ones")
    part = graph_partition(A, world)
    perm, sizes = partition_permutation(part)
    Ap, bp = permute_system(A, perm, b)
    beg = int(np.sum(sizes[:rank]))
    end = beg + sizes[rank]
    m = Ap.to_scipy()
    lo, hi = m.indptr[beg], m.indptr[end]
    strip = CSR(end - beg, n ** 3, m.indptr[beg:end + 1] - lo,
                m.indices[lo:hi], m.data[lo:hi])
    solve = make_dist_solver(
        strip, {"precond": {"class": "amg", "coarse_enough": 300},
                "solver": {"type": "cg", "tol": 1e-8, "maxiter": 200}},
        backend="cpu")
    x, iters, resid = solve(bp[beg:end])
    xg = solve.gather_solution(x)
    return iters, resid, None if xg is None else xg.tolist(), perm.tolist()


@pytest.mark.parametrize("world", [3])
def test_graph_partitioned_distributed_solve(world):
    """End-to-end: coordinate-free partition feeding the distributed solver."""
    out = spawn(world, _solve_graph_partitioned, 30461 + world)
    import amgcl_amd as am

    iters, resid, xg, perm = out[0]
    assert resid < 1e-8
    n = 12
    A, b = am.poisson3d(n, rhs="ones")
    perm = np.asarray(perm)
    x = np.empty(n ** 3)
    x[perm] = np.asarray(xg)
    assert np.linalg.norm(b - A @ x) / np.linalg.norm(b) < 1e-7


def _solve_dist_amg_group_coarse(rank, world):
    import amgcl_amd as am
    from amgcl_amd.parallel import make_dist_solver

    n = 16
    strip, b, row_beg, row_end = am.poisson3d_strip(n, rank, world, rhs="ones")
    solve = make_dist_solver(
        strip,
        {"precond": {"class": "dist_amg", "coarse_enough": 400,
                     "coarse_group_size": 2},
         "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}},
        backend="cpu")
    x, iters, resid = solve(b)
    xg = solve.gather_solution(x)
    P = solve.P
    has_tail = P.coarse is not None or P.tail is not None
    return iters, resid, None if xg is None else xg.tolist(), has_tail


@pytest.mark.parametrize("world", [4])
def test_dist_amg_subcommunicator_coarse(world):
    """Subcommunicator coarse solve (solver_base.hpp shape): only group
    masters hold the coarse tail; slaves receive the solution by an
    intra-group broadcast.  Same convergence as the replicated path."""
    out = spawn(world, _solve_dist_amg_group_coarse, 30511 + world)
    import amgcl_amd as am

    n = 16
    A, b = am.poisson3d(n, rhs="ones")
    iters, resid, xg, _ = out[0]
    assert resid < 1e-8
    xg = np.array(xg)
    assert np.linalg.norm(b - A @ xg) / np.linalg.norm(b) < 1e-7
    # masters (ranks 0, 2) hold the tail, slaves (1, 3) hold nothing
    tails = {r: out[r][3] for r in range(world)}
    assert tails[0] and tails[2]
    assert not tails[1] and not tails[3]


def _solve_dist_amg_skyline(rank, world):
    import amgcl_amd as am
    from amgcl_amd.parallel import make_dist_solver

    n = 14
    strip, b, row_beg, row_end = am.poisson3d_strip(n, rank, world, rhs="ones")
    solve = make_dist_solver(
        strip,
        {"precond": {"class": "dist_amg", "coarse_enough": 400,
                     "direct_solver": "skyline"},
         "solver": {"type": "cg", "tol": 1e-8, "maxiter": 100}},
        backend="cpu")
    x, iters, resid = solve(b)
    xg = solve.gather_solution(x)
    return iters, resid, None if xg is None else xg.tolist()


@pytest.mark.parametrize("world", [2])
def test_dist_amg_skyline_coarse(world):
    """Distributed coarse solve through the skyline (profile) LU instead of
    the dense inverse (parity: mpi/direct_solver/skyline_lu.hpp)."""
    out = spawn(world, _solve_dist_amg_skyline, 30561 + world)
    import amgcl_amd as am

    n = 14
    A, b = am.poisson3d(n, rhs="ones")
    iters, resid, xg = out[0]
    assert resid < 1e-8
    xg = np.array(xg)
    assert np.linalg.norm(b - A @ xg) / np.linalg.norm(b) < 1e-7
