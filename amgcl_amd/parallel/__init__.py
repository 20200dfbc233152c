"""Distributed (multi-GPU) subsystem: one process per GPU over
torch.distributed — RCCL/xGMI on GPUs, gloo for CPU tests.

MI355X-native re-design of the reference's MPI layer (amgcl/mpi/):
  - comm_pattern / distributed_matrix (mpi/distributed_matrix.hpp:51/:317)
    -> DistMatrix: local/remote split, device-resident halo exchange via
       batched isend/irecv packed by gather kernels, overlapped with the
       local SpMV.
  - mpi::inner_product (mpi/inner_product.hpp:44) -> DistInnerProduct:
    local device dot + all_reduce.
  - mpi::block_preconditioner (mpi/block_preconditioner.hpp:49) and
    mpi::subdomain_deflation (mpi/subdomain_deflation.hpp:113)
    -> LocalBlockPrecond / SubdomainDeflation.
  - mpi::amg + mpi::coarsening::pmis (mpi/amg.hpp:56, mpi/coarsening/
    pmis.hpp:50) -> DistAMG (one hierarchy over the distributed operator;
    decoupled or cross-rank pmis aggregation, replicated-tail repartition).
  - mpi::schur_pressure_correction / mpi::cpr -> DistSchurPressureCorrection
    / DistCPR (fully-coupled distributed field splits).
  - mpi::partition (ptscotch/parmetis class) -> partition.rcb_partition
    (geometric recursive coordinate bisection).
"""
from .dist_matrix import DistMatrix
from .dist_backend import DistBackend, DistInnerProduct
from .precond import LocalBlockPrecond
from .solver import make_dist_solver, DistSolver
from .dist_amg import DistAMG
from .schur import DistSchurPressureCorrection
from .cpr import DistCPR
from .partition import rcb_partition, partition_permutation
