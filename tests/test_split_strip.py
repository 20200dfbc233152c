"""CPU-side parity tests for the torch strip split used by the
device-resident distributed setup (no GPU marker: pure torch-cpu)."""
import numpy as np


def test_split_strip_torch_matches_core():
    """The torch local/remote strip split (device path of DistMatrix) agrees
    with the C++ _core.split_strip on CPU tensors — the CPU-testable parity
    evidence for the GPU-resident distributed setup (parity:
    amgcl/mpi/distributed_matrix.hpp:370-430)."""
    import torch

    import amgcl_amd as am
    from amgcl_amd import _core
    from amgcl_amd.backend.hip_setup import split_strip_torch

    n = 12
    world = 3
    for rank in range(world):
        row_beg = rank * n**3 // world
        row_end = (rank + 1) * n**3 // world
        A, _, rb, re = am.poisson3d_strip(n, rank, world, rhs=None)
        col_beg, col_end = rb, re
        lp, lc, lv, rp, rc, rv, gg = _core.split_strip(
            A.nrows, col_beg, col_end, A.ptr, A.col, A.val)
        tp = torch.from_numpy(np.asarray(A.ptr, dtype=np.int32))
        tc = torch.from_numpy(np.asarray(A.col, dtype=np.int32))
        tv = torch.from_numpy(np.asarray(A.val))
        lp2, lc2, lv2, rp2, rc2, rv2, gg2 = split_strip_torch(
            tp, tc, tv, col_beg, col_end)
        np.testing.assert_array_equal(np.asarray(lp), lp2.numpy())
        np.testing.assert_array_equal(np.asarray(lc), lc2.numpy())
        np.testing.assert_array_equal(np.asarray(lv), lv2.numpy())
        np.testing.assert_array_equal(np.asarray(rp), rp2.numpy())
        np.testing.assert_array_equal(np.asarray(rc), rc2.numpy())
        np.testing.assert_array_equal(np.asarray(rv), rv2.numpy())
        np.testing.assert_array_equal(np.asarray(gg), gg2.numpy())


def test_split_strip_torch_random():
    """Same parity on a random rectangular-ish strip (ghosts on both sides,
    empty rows, rows with no local entries)."""
    import scipy.sparse as sp
    import torch

    from amgcl_amd import _core
    from amgcl_amd.backend.hip_setup import split_strip_torch

    rng = np.random.default_rng(7)
    nglob, nloc, beg = 200, 37, 80
    m = sp.random(nloc, nglob, density=0.08, random_state=rng, format="csr")
    m.sort_indices()
    ptr = m.indptr.astype(np.int32)
    col = m.indices.astype(np.int32)
    val = m.data
    ref = _core.split_strip(nloc, beg, beg + nloc, ptr, col, val)
    got = split_strip_torch(torch.from_numpy(ptr), torch.from_numpy(col),
                            torch.from_numpy(val), beg, beg + nloc)
    for a, b in zip(ref, got):
        np.testing.assert_array_equal(np.asarray(a), b.numpy())
