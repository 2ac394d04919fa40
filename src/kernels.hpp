// Tile task-classes for the dense linear-algebra headline apps.
//
// The reference runtime ships no compute kernels — JDF bodies call cuBLAS
// via dyld (tests/runtime/cuda/stress.jdf:133-137) and dense drivers live in
// DPLASMA. Here the tile kernel set IS part of the framework (SURVEY.md §6
// north star): fp64 POTRF/TRSM/SYRK/GEMM tile bodies with two chore
// incarnations each — rocBLAS/rocSOLVER ("library" chore) and hand-written
// CDNA4 MFMA HIP kernels (kernels_hip.cpp) — selected by the `chore_gemm`
// etc. params, mirroring the reference's chore/incarnation machinery
// (parsec_internal.h:411-459).
#pragma once

#include "dtd.hpp"

namespace pa {

enum class Reshape : uint8_t;
TaskClass& tc_reshape();
size_t reshape_bytes(size_t src_bytes, Reshape kind, size_t src_elem);
void fill_reshape_args(void* argbuf, Reshape kind, int m, int n, int ld);
size_t reshape_args_bytes();

struct TileArgs {
  int m = 0, n = 0, k = 0, ld = 0;
  int64_t i0 = 0, j0 = 0, N = 0;
  uint32_t seed = 0;
};

// Registered once; ids stable.
TaskClass& tc_spd_fill();
TaskClass& tc_potrf();
TaskClass& tc_trsm();
TaskClass& tc_syrk();
TaskClass& tc_gemm();
TaskClass& tc_gemm_nn();

TaskClass& tc_trtri();
TaskClass& tc_trsm_inv();
TaskClass& tc_geqrt();
TaskClass& tc_unmqr();
TaskClass& tc_tsqrt();
TaskClass& tc_tsmqr();

// Build the DAGs (DTD insertion; SPMD-safe: call on every rank).
void insert_spd_fill(Dtd& tp, TiledMatrix& A, uint32_t seed);
void insert_full_fill(Dtd& tp, TiledMatrix& A, uint32_t seed);
void insert_potrf(Dtd& tp, TiledMatrix& A);
void insert_geqrf(Dtd& tp, TiledMatrix& A);
// QR by block Gram-Schmidt with CholeskyQR2 panels (matrix-core rates;
// cond(A) <~ 1e7): A becomes the explicit orthonormal Q, R the upper
// tiles (kernels_qr_bcgs.cpp).
void insert_geqrf_bcgs(Dtd& tp, TiledMatrix& A, TiledMatrix& R);
void insert_getrf_nopiv(Dtd& tp, TiledMatrix& A);
void insert_fill_bf16(Dtd& tp, TiledMatrix& A, uint32_t seed);
void insert_gemm_bf16(Dtd& tp, TiledMatrix& At, TiledMatrix& B, TiledMatrix& C);
void insert_redistribute(Dtd& tp, TiledMatrix& Src, TiledMatrix& Dst);
void insert_reduce_axis(Dtd& tp, TiledMatrix& A, TiledMatrix& R, int axis);
void insert_band_to_rect(Dtd& tp, TiledMatrix& S, TiledMatrix& D);
void insert_subtile_extract(Dtd& tp, TiledMatrix& A, int tm, int tn,
                            TiledMatrix& S);
void insert_subtile_insert(Dtd& tp, TiledMatrix& S, TiledMatrix& A, int tm,
                           int tn);
void insert_apply_scale(Dtd& tp, TiledMatrix& A, double alpha, double beta);
// Tiled C = A*B, fp64 NN (idempotent DAG: k==0 overwrites C — the hipGraph
// capture/replay demo workload, and a plain library-GEMM DAG generally).
void insert_gemm_fp64(Dtd& tp, TiledMatrix& A, TiledMatrix& B, TiledMatrix& C);
// data_advise analog: prefetch one tile onto the device (no-op body).
void insert_advise_prefetch(Dtd& tp, Data* d);
// Cholesky solve / factor+solve (dplasma dpotrs/dposv analogs).
void insert_potrs(Dtd& tp, TiledMatrix& A, TiledMatrix& B);
void insert_posv(Dtd& tp, TiledMatrix& A, TiledMatrix& B);
// LU solve / factor+solve, no pivoting (dgetrs/dgesv nopiv analogs).
void insert_getrs_nopiv(Dtd& tp, TiledMatrix& A, TiledMatrix& B);
void insert_gesv_nopiv(Dtd& tp, TiledMatrix& A, TiledMatrix& B);
// Least squares via BCGS QR (dgels analog): X = R^-1 Q^T B.
void insert_gels_bcgs(Dtd& tp, TiledMatrix& A, TiledMatrix& R,
                      TiledMatrix& B, TiledMatrix& X);
void insert_reduce_sum(Dtd& tp, TiledMatrix& A, TiledMatrix& R);
// log-depth binary-tree variant (BT_reduction.jdf analog).
void insert_reduce_sum_tree(Dtd& tp, TiledMatrix& A, TiledMatrix& R);
void insert_stencil_1d(Dtd& tp, TiledMatrix& Src, TiledMatrix& Dst);
void insert_panel_fill(Dtd& tp, TiledMatrix& A, uint32_t seed);
void insert_potrf_panel(Dtd& tp, TiledMatrix& A);

}  // namespace pa
