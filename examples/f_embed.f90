! Fortran program embedding the parsec_amd runtime (parsecf analog).
!
! Build:
!   amdflang -c parsec_amd/fortran/parsecf.f90 -o /tmp/parsecf.o -module-dir /tmp
!   amdflang examples/f_embed.f90 /tmp/parsecf.o -module-dir /tmp \
!       -Lparsec_amd -l:libparsec_amd.so -Wl,-rpath,$PWD/parsec_amd \
!       -o /tmp/f_embed
!
! Inserts a chain of INOUT tasks per tile (Fortran task bodies through the
! dataflow engine) and verifies the result.
module f_bodies
  use iso_c_binding
  use parsec_amd_f
  implicit none
contains
  subroutine body_init(t) bind(c)
    type(c_ptr), value :: t
    real(c_double), pointer :: v(:)
    integer(c_long), pointer :: k
    call c_f_pointer(pa_task_host_ptr(t, 0), v, [8])
    call c_f_pointer(pa_task_args(t), k)
    v(1) = real(k, c_double)
  end subroutine

  subroutine body_chain(t) bind(c)
    type(c_ptr), value :: t
    real(c_double), pointer :: v(:)
    call c_f_pointer(pa_task_host_ptr(t, 0), v, [8])
    v(1) = v(1) * 2.0d0 + 1.0d0
  end subroutine
end module f_bodies

program f_embed
  use iso_c_binding
  use parsec_amd_f
  use f_bodies
  implicit none
  type(c_ptr) :: ctx, tp, tm, tc_init, tc_chain
  type(c_ptr) :: datas(1)
  integer(c_int) :: modes(1)
  integer(c_long), target :: karg
  integer :: i, s
  real(c_double), pointer :: out(:)
  real(c_double) :: expect

  ctx = pa_context_new(2, 0, 1, ""//c_null_char, -2)
  tp = pa_dtd_new(ctx, "f_embed"//c_null_char)
  tm = pa_tm_new(ctx, 8_c_long, 8_c_long, 1, 8, 1, 1, 8_c_long, 0)
  tc_init = pa_taskclass_new("f_init"//c_null_char, 0, &
                             c_funloc(body_init), c_null_funptr)
  tc_chain = pa_taskclass_new("f_chain"//c_null_char, 0, &
                              c_funloc(body_chain), c_null_funptr)

  do i = 0, 7
    karg = int(100 + i, c_long)
    datas(1) = pa_tm_tile(tm, i, 0)
    modes(1) = PA_ACCESS_OUT
    call pa_dtd_insert(tp, tc_init, c_loc(karg), 8, datas, modes, 1, 0, -1)
    modes(1) = PA_ACCESS_INOUT
    do s = 1, 5
      call pa_dtd_insert(tp, tc_chain, c_null_ptr, 0, datas, modes, 1, 0, -1)
    end do
  end do
  call pa_dtd_wait(tp)

  do i = 0, 7
    call c_f_pointer(pa_tm_tile_host(tm, i, 0), out, [8])
    expect = real(100 + i, c_double)
    do s = 1, 5
      expect = expect * 2.0d0 + 1.0d0
    end do
    if (abs(out(1) - expect) > 1.0d-12) then
      print *, "F_EMBED_FAIL tile", i, out(1), expect
      stop 1
    end if
  end do
  print *, "F_EMBED_OK"
  call pa_dtd_free(tp)
  call pa_tm_free(tm)
  call pa_context_free(ctx)
end program f_embed
