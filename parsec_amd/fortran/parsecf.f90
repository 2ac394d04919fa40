! parsec_amd Fortran bindings (reference parity: parsec/fortran/parsecf.F90
! ISO-C wrappers over the public C API, runtime.h:156-710).
!
! ISO_C_BINDING interfaces over the pa_* C ABI exported by _core.so /
! libparsec_amd.so (src/capi.cpp). Task bodies are BIND(C) subroutines
! taking the opaque task handle; flow buffers and packed arguments come
! back through pa_task_host_ptr / pa_task_args.
!
! Build (ROCm ships amdflang):
!   amdflang -c parsec_amd/fortran/parsecf.f90
!   amdflang your_app.f90 parsecf.o -Lparsec_amd -l:_core.so \
!       -Wl,-rpath,$PWD/parsec_amd
module parsec_amd_f
  use iso_c_binding
  implicit none

  integer(c_int), parameter :: PA_ACCESS_IN = 1
  integer(c_int), parameter :: PA_ACCESS_OUT = 2
  integer(c_int), parameter :: PA_ACCESS_INOUT = 3

  interface
    function pa_context_new(nworkers, rank, world, comm, gpu) &
        bind(c, name="pa_context_new") result(ctx)
      import :: c_ptr, c_int, c_char
      integer(c_int), value :: nworkers, rank, world, gpu
      character(kind=c_char), dimension(*) :: comm
      type(c_ptr) :: ctx
    end function

    subroutine pa_context_free(ctx) bind(c, name="pa_context_free")
      import :: c_ptr
      type(c_ptr), value :: ctx
    end subroutine

    subroutine pa_context_barrier(ctx) bind(c, name="pa_context_barrier")
      import :: c_ptr
      type(c_ptr), value :: ctx
    end subroutine

    function pa_ctx_rank(ctx) bind(c, name="pa_ctx_rank") result(r)
      import :: c_ptr, c_int
      type(c_ptr), value :: ctx
      integer(c_int) :: r
    end function

    function pa_dtd_new(ctx, name) bind(c, name="pa_dtd_new") result(tp)
      import :: c_ptr, c_char
      type(c_ptr), value :: ctx
      character(kind=c_char), dimension(*) :: name
      type(c_ptr) :: tp
    end function

    subroutine pa_dtd_wait(tp) bind(c, name="pa_dtd_wait")
      import :: c_ptr
      type(c_ptr), value :: tp
    end subroutine

    subroutine pa_dtd_free(tp) bind(c, name="pa_dtd_free")
      import :: c_ptr
      type(c_ptr), value :: tp
    end subroutine

    function pa_tm_new(ctx, m, n, mb, nb, p, q, elem_size, sym) &
        bind(c, name="pa_tm_new") result(tm)
      import :: c_ptr, c_int, c_long
      type(c_ptr), value :: ctx
      integer(c_long), value :: m, n, elem_size
      integer(c_int), value :: mb, nb, p, q, sym
      type(c_ptr) :: tm
    end function

    subroutine pa_tm_free(tm) bind(c, name="pa_tm_free")
      import :: c_ptr
      type(c_ptr), value :: tm
    end subroutine

    function pa_tm_tile(tm, i, j) bind(c, name="pa_tm_tile") result(d)
      import :: c_ptr, c_int
      type(c_ptr), value :: tm
      integer(c_int), value :: i, j
      type(c_ptr) :: d
    end function

    function pa_tm_tile_host(tm, i, j) bind(c, name="pa_tm_tile_host") &
        result(p)
      import :: c_ptr, c_int
      type(c_ptr), value :: tm
      integer(c_int), value :: i, j
      type(c_ptr) :: p
    end function

    function pa_taskclass_new(name, flags, cpu, gpu) &
        bind(c, name="pa_taskclass_new") result(tc)
      import :: c_ptr, c_int, c_char, c_funptr
      character(kind=c_char), dimension(*) :: name
      integer(c_int), value :: flags
      type(c_funptr), value :: cpu, gpu
      type(c_ptr) :: tc
    end function

    subroutine pa_dtd_insert(tp, tc, args, nargs, datas, modes, nflows, &
                             prio, rank) bind(c, name="pa_dtd_insert")
      import :: c_ptr, c_int
      type(c_ptr), value :: tp, tc
      type(c_ptr), value :: args
      integer(c_int), value :: nargs, nflows, prio, rank
      type(c_ptr), dimension(*) :: datas
      integer(c_int), dimension(*) :: modes
    end subroutine

    function pa_task_args(t) bind(c, name="pa_task_args") result(p)
      import :: c_ptr
      type(c_ptr), value :: t
      type(c_ptr) :: p
    end function

    function pa_task_host_ptr(t, flow) bind(c, name="pa_task_host_ptr") &
        result(p)
      import :: c_ptr, c_int
      type(c_ptr), value :: t
      integer(c_int), value :: flow
      type(c_ptr) :: p
    end function

    subroutine pa_param_set(name, value) bind(c, name="pa_param_set")
      import :: c_char
      character(kind=c_char), dimension(*) :: name, value
    end subroutine
  end interface
end module parsec_amd_f
