! quda_amd Fortran module (role of the reference's lib/quda_fortran.F90):
! iso_c_binding interfaces to the C ABI of include/quda_amd.h, plus the
! param derived types with matching layout. Link against libquda_amd_c.so.
!
! Classic pass-by-reference entry points (init_quda_, invert_quda_, ...)
! are also exported by the library for implicit-interface code; this
! module is the typed modern-Fortran surface.
module quda_amd
  use iso_c_binding
  implicit none

  integer(c_int), parameter :: QUDA_AMD_DOUBLE = 8
  integer(c_int), parameter :: QUDA_AMD_SINGLE = 4
  integer(c_int), parameter :: QUDA_AMD_HALF = 2
  integer(c_int), parameter :: QUDA_AMD_RECON_NO = 18
  integer(c_int), parameter :: QUDA_AMD_RECON_12 = 12
  integer(c_int), parameter :: QUDA_AMD_WILSON_DSLASH = 0
  integer(c_int), parameter :: QUDA_AMD_CLOVER_WILSON_DSLASH = 1
  integer(c_int), parameter :: QUDA_AMD_CG_INVERTER = 0
  integer(c_int), parameter :: QUDA_AMD_CGNR_INVERTER = 8
  integer(c_int), parameter :: QUDA_AMD_MAT_SOLUTION = 0
  integer(c_int), parameter :: QUDA_AMD_MATPC_SOLUTION = 1

  type, bind(c) :: quda_amd_gauge_param
    integer(c_int) :: x(4)
    integer(c_int) :: cpu_prec, cuda_prec, cuda_prec_sloppy
    integer(c_int) :: reconstruct, reconstruct_sloppy
    real(c_double) :: anisotropy
    integer(c_int) :: t_boundary
    integer(c_int) :: staggered_phase_applied
  end type

  type, bind(c) :: quda_amd_invert_param
    integer(c_int) :: dslash_type, inv_type, solution_type
    real(c_double) :: kappa, mass, mu, clover_csw, tol
    integer(c_int) :: maxiter
    real(c_double) :: reliable_delta
    integer(c_int) :: cpu_prec, cuda_prec, cuda_prec_sloppy
    integer(c_int) :: ls
    real(c_double) :: m5, b5, c5
    integer(c_int) :: iter
    real(c_double) :: true_res, secs, gflops
    type(c_ptr) :: preconditioner
  end type

  interface
    function initQuda(device) bind(c, name="initQuda") result(ierr)
      import :: c_int
      integer(c_int), value :: device
      integer(c_int) :: ierr
    end function

    function endQuda() bind(c, name="endQuda") result(ierr)
      import :: c_int
      integer(c_int) :: ierr
    end function

    function loadGaugeQuda(h_gauge, param) bind(c, name="loadGaugeQuda") &
        result(ierr)
      import :: c_ptr, c_int, quda_amd_gauge_param
      type(c_ptr), value :: h_gauge
      type(quda_amd_gauge_param) :: param
      integer(c_int) :: ierr
    end function

    function loadCloverQuda(h_clover, h_clovinv, param) &
        bind(c, name="loadCloverQuda") result(ierr)
      import :: c_ptr, c_int, quda_amd_invert_param
      type(c_ptr), value :: h_clover, h_clovinv
      type(quda_amd_invert_param) :: param
      integer(c_int) :: ierr
    end function

    function invertQuda(h_x, h_b, param) bind(c, name="invertQuda") &
        result(ierr)
      import :: c_ptr, c_int, quda_amd_invert_param
      type(c_ptr), value :: h_x, h_b
      type(quda_amd_invert_param) :: param
      integer(c_int) :: ierr
    end function

    function MatQuda(h_out, h_in, param) bind(c, name="MatQuda") &
        result(ierr)
      import :: c_ptr, c_int, quda_amd_invert_param
      type(c_ptr), value :: h_out, h_in
      type(quda_amd_invert_param) :: param
      integer(c_int) :: ierr
    end function

    function plaqQuda(plaq) bind(c, name="plaqQuda") result(ierr)
      import :: c_double, c_int
      real(c_double) :: plaq(3)
      integer(c_int) :: ierr
    end function

    function gaugeObservablesQuda(plaq, qcharge, energy) &
        bind(c, name="gaugeObservablesQuda") result(ierr)
      import :: c_double, c_int
      real(c_double) :: plaq(3), qcharge, energy(2)
      integer(c_int) :: ierr
    end function

    function performGaugeSmearQuda(stype, n_steps, coeff) &
        bind(c, name="performGaugeSmearQuda") result(ierr)
      import :: c_double, c_int
      integer(c_int), value :: stype, n_steps
      real(c_double), value :: coeff
      integer(c_int) :: ierr
    end function

    function computeGaugeForceQuda(h_mom, beta) &
        bind(c, name="computeGaugeForceQuda") result(ierr)
      import :: c_ptr, c_double, c_int
      type(c_ptr), value :: h_mom
      real(c_double), value :: beta
      integer(c_int) :: ierr
    end function

    function updateGaugeFieldQuda(h_mom, dt) &
        bind(c, name="updateGaugeFieldQuda") result(ierr)
      import :: c_ptr, c_double, c_int
      type(c_ptr), value :: h_mom
      real(c_double), value :: dt
      integer(c_int) :: ierr
    end function

    function momActionQuda(action, h_mom) &
        bind(c, name="momActionQuda") result(ierr)
      import :: c_ptr, c_double, c_int
      real(c_double) :: action
      type(c_ptr), value :: h_mom
      integer(c_int) :: ierr
    end function

    function computeGaugeFixingOVRQuda(gauge_dir, max_iter, tol) &
        bind(c, name="computeGaugeFixingOVRQuda") result(ierr)
      import :: c_double, c_int
      integer(c_int), value :: gauge_dir, max_iter
      real(c_double), value :: tol
      integer(c_int) :: ierr
    end function
  end interface
end module quda_amd
