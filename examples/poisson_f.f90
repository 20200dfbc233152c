! 3D Poisson solve from Fortran through the C API (1-based entry points)
! — parity with the reference's fortran/poisson.f90.
program poisson_f
  use amgcl_amd
  use iso_c_binding
  implicit none
  integer, parameter :: m = 16
  integer :: n, nnz, i, j, k, row, idx
  integer, allocatable :: ptr(:), col(:)
  real(c_double), allocatable :: val(:), rhs(:), x(:)
  type(c_ptr) :: solver, prm
  integer(c_int) :: iters, rc
  real(c_double) :: resid

  n = m*m*m
  allocate(ptr(n+1), col(7*n), val(7*n), rhs(n), x(n))
  ptr(1) = 1
  idx = 0
  row = 0
  do k = 1, m
    do j = 1, m
      do i = 1, m
        row = row + 1
        if (k > 1) then
          idx = idx + 1; col(idx) = row - m*m; val(idx) = -1d0
        end if
        if (j > 1) then
          idx = idx + 1; col(idx) = row - m; val(idx) = -1d0
        end if
        if (i > 1) then
          idx = idx + 1; col(idx) = row - 1; val(idx) = -1d0
        end if
        idx = idx + 1; col(idx) = row; val(idx) = 6d0
        if (i < m) then
          idx = idx + 1; col(idx) = row + 1; val(idx) = -1d0
        end if
        if (j < m) then
          idx = idx + 1; col(idx) = row + m; val(idx) = -1d0
        end if
        if (k < m) then
          idx = idx + 1; col(idx) = row + m*m; val(idx) = -1d0
        end if
        ptr(row+1) = idx + 1
      end do
    end do
  end do
  nnz = idx
  rhs = 1d0
  x = 0d0

  prm = amgcl_amd_params_create()
  call amgcl_amd_params_setf(prm, "solver.tol"//c_null_char, 1d-8)
  solver = amgcl_amd_solver_create_f(n, ptr, col, val, prm)
  rc = amgcl_amd_solver_solve(solver, rhs, x, iters, resid)
  write(*,'(a,i4,a,es10.3)') "iters=", iters, "  resid=", resid
  call amgcl_amd_solver_destroy(solver)
  call amgcl_amd_params_destroy(prm)
  if (resid < 1d-8 .and. iters < 40) then
    write(*,'(a)') "FORTRAN_OK"
  else
    write(*,'(a)') "FORTRAN_FAIL"
    stop 1
  end if
end program
