! amgcl_amd — Fortran interface to the C API (libamgclamd_c.so).
!
! Parity: the reference's C library exposes 1-based entry points for Fortran
! callers; this module binds them through iso_c_binding. Matrix arrays use
! 1-based indices (the *_f entry points convert internally).
!
! Note: the build image for this repository carries no Fortran compiler, so
! this module ships as an interface source verified against the C header;
! the 1-based entry points themselves are covered by the ctypes test suite
! (tests/test_capi.py::test_capi_precond_and_fortran_indexing).
module amgcl_amd
  use iso_c_binding
  implicit none

  interface
     function amgcl_amd_params_create() bind(c, name="amgcl_amd_params_create")
       import c_ptr
       type(c_ptr) :: amgcl_amd_params_create
     end function

     subroutine amgcl_amd_params_seti(prm, name, val) &
          bind(c, name="amgcl_amd_params_seti")
       import c_ptr, c_char, c_int
       type(c_ptr), value :: prm
       character(kind=c_char) :: name(*)
       integer(c_int), value :: val
     end subroutine

     subroutine amgcl_amd_params_setf(prm, name, val) &
          bind(c, name="amgcl_amd_params_setf")
       import c_ptr, c_char, c_double
       type(c_ptr), value :: prm
       character(kind=c_char) :: name(*)
       real(c_double), value :: val
     end subroutine

     subroutine amgcl_amd_params_sets(prm, name, val) &
          bind(c, name="amgcl_amd_params_sets")
       import c_ptr, c_char
       type(c_ptr), value :: prm
       character(kind=c_char) :: name(*), val(*)
     end subroutine

     subroutine amgcl_amd_params_destroy(prm) &
          bind(c, name="amgcl_amd_params_destroy")
       import c_ptr
       type(c_ptr), value :: prm
     end subroutine

     function amgcl_amd_solver_create_f(n, ptr, col, val, prm) &
          bind(c, name="amgcl_amd_solver_create_f")
       import c_ptr, c_int, c_double
       type(c_ptr) :: amgcl_amd_solver_create_f
       integer(c_int), value :: n
       integer(c_int) :: ptr(*), col(*)
       real(c_double) :: val(*)
       type(c_ptr), value :: prm
     end function

     function amgcl_amd_solver_solve(solver, rhs, x, iters, resid) &
          bind(c, name="amgcl_amd_solver_solve")
       import c_ptr, c_int, c_double
       integer(c_int) :: amgcl_amd_solver_solve
       type(c_ptr), value :: solver
       real(c_double) :: rhs(*), x(*)
       integer(c_int) :: iters
       real(c_double) :: resid
     end function

     subroutine amgcl_amd_solver_destroy(solver) &
          bind(c, name="amgcl_amd_solver_destroy")
       import c_ptr
       type(c_ptr), value :: solver
     end subroutine

     function amgcl_amd_precond_create_f(n, ptr, col, val, prm) &
          bind(c, name="amgcl_amd_precond_create_f")
       import c_ptr, c_int, c_double
       type(c_ptr) :: amgcl_amd_precond_create_f
       integer(c_int), value :: n
       integer(c_int) :: ptr(*), col(*)
       real(c_double) :: val(*)
       type(c_ptr), value :: prm
     end function

     subroutine amgcl_amd_precond_apply(amg, rhs, x) &
          bind(c, name="amgcl_amd_precond_apply")
       import c_ptr, c_double
       type(c_ptr), value :: amg
       real(c_double) :: rhs(*), x(*)
     end subroutine

     subroutine amgcl_amd_precond_destroy(amg) &
          bind(c, name="amgcl_amd_precond_destroy")
       import c_ptr
       type(c_ptr), value :: amg
     end subroutine

     ! torch-free GPU solver (link libamghip.so; 0-based CSR indices)
     function amgcl_amd_gpu_solver_create(n, ptr, col, val, config) &
          bind(c, name="amgcl_amd_gpu_solver_create")
       import c_ptr, c_int, c_double, c_char
       type(c_ptr) :: amgcl_amd_gpu_solver_create
       integer(c_int), value :: n
       integer(c_int) :: ptr(*), col(*)
       real(c_double) :: val(*)
       character(kind=c_char) :: config(*)
     end function

     function amgcl_amd_gpu_solver_solve(solver, rhs, x, iters, resid) &
          bind(c, name="amgcl_amd_gpu_solver_solve")
       import c_ptr, c_int, c_double
       integer(c_int) :: amgcl_amd_gpu_solver_solve
       type(c_ptr), value :: solver
       real(c_double) :: rhs(*), x(*)
       integer(c_int) :: iters
       real(c_double) :: resid
     end function

     subroutine amgcl_amd_gpu_solver_destroy(solver) &
          bind(c, name="amgcl_amd_gpu_solver_destroy")
       import c_ptr
       type(c_ptr), value :: solver
     end subroutine
  end interface
end module amgcl_amd
