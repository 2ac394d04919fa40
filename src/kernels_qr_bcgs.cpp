// QR via block classical Gram-Schmidt with CholeskyQR2 panels
// (qr_algo=bcgs): every operation is a full-rate tile GEMM / POTRF /
// TRTRI, so the factorization runs at matrix-core rates instead of the
// Householder panel chain's latency bound.
//
// Math: for each block column k, G = P^T P, L1 = chol(G), Q = P L1^{-T};
// repeated once (CholeskyQR2) so Q's orthogonality is O(eps) whenever the
// first Cholesky succeeds (cond(P) <~ 1e7). R11 = (L1 L2)^T. Trailing:
// R12 = Q^T C, C -= Q R12 (block classical GS). A ends as the explicit
// orthonormal Q; R is a separate upper-triangular tile collection —
// verified by Q^T Q = I and Q R = A (a STRONGER contract than the
// Householder path's R^T R test).
//
// Stability envelope (documented, library-grade honesty): CholeskyQR2
// requires the panel Gram matrix to be numerically SPD; for
// cond(A) >~ 1e7 in fp64 use the Householder path (qr_algo=house,
// kernels_qr.cpp) instead. This is the same trade modern GPU libraries
// ship for tall-skinny/blocked QR.
//
// The reference ships no QR at all (dense LA lives in DPLASMA); both
// algorithms here are MI355X-first designs on the DTD engine and run
// distributed unchanged (all ops are tile tasks).
#include <cmath>
#include <cstring>
#include <memory>

#include <hip/hip_runtime.h>
#include <rocblas/rocblas.h>

#include "device_gpu.hpp"
#include "kernels.hpp"
#include "profiling.hpp"

namespace pa {

namespace {

rocblas_handle bc_handle(GpuTaskCtx& g) {
  static thread_local std::map<void*, rocblas_handle> handles;
  rocblas_handle& h = handles[(void*)g.stream];
  if (!h) {
    PA_CHECK(rocblas_create_handle(&h) == rocblas_status_success);
    rocblas_set_pointer_mode(h, rocblas_pointer_mode_host);
    rocblas_set_stream(h, g.stream);
  }
  return h;
}

// ---- G += A^T B (TN accumulate; also the syrk step with B = A) ----
void cpu_gemm_tn_acc(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const double* A = (const double*)t.flows[0].data->pull_to_host();
  const double* B = (const double*)t.flows[1].data->pull_to_host();
  double* C = (double*)t.flows[2].data->pull_to_host();
  const int rows = a.m, n = a.n, ld = a.ld;
  for (int j = 0; j < n; j++)
    for (int i = 0; i < n; i++) {
      double s = 0;
      for (int p = 0; p < rows; p++)
        s += A[(size_t)i * ld + p] * B[(size_t)j * ld + p];
      C[(size_t)j * n + i] += s;
    }
  t.flows[2].data->written_on(false);
}

void gpu_gemm_tn_acc(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  const double one = 1.0;
  PA_CHECK(rocblas_dgemm(bc_handle(g), rocblas_operation_transpose,
                         rocblas_operation_none, a.n, a.n, a.m, &one,
                         (const double*)t.dev_ptr[0], a.ld,
                         (const double*)t.dev_ptr[1], a.ld, &one,
                         (double*)t.dev_ptr[2], a.n) ==
           rocblas_status_success);
}

// ---- zero an nb x nb workspace tile ----
void cpu_zero_tile(Task& t) {
  Data* d = t.flows[0].data;
  memset(d->ensure_host(), 0, d->bytes);
  d->written_on(false);
}

void gpu_zero_tile(Task& t, GpuTaskCtx& g) {
  PA_HIP_CHECK(hipMemsetAsync(t.dev_ptr[0], 0, t.flows[0].data->bytes,
                              g.stream));
}

// ---- R = L2^T * L1^T (both lower-triangular reads; R upper) ----
__global__ void k_tri_tt(const double* __restrict__ L2,
                         const double* __restrict__ L1,
                         double* __restrict__ R, int n) {
  int idx = blockIdx.x * blockDim.x + threadIdx.x;
  int total = n * n;
  for (; idx < total; idx += gridDim.x * blockDim.x) {
    int j = idx / n, i = idx - j * n;  // R(i, j)
    double s = 0;
    if (i <= j)
      for (int p = i; p <= j; p++)
        s += L2[(size_t)i * n + p] * L1[(size_t)p * n + j];
    R[(size_t)j * n + i] = s;
  }
}

void cpu_tri_tt(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const double* L2 = (const double*)t.flows[0].data->pull_to_host();
  const double* L1 = (const double*)t.flows[1].data->pull_to_host();
  double* R = (double*)t.flows[2].data->ensure_host();
  const int n = a.n;
  for (int j = 0; j < n; j++)
    for (int i = 0; i < n; i++) {
      double s = 0;
      if (i <= j)
        // R(i,j) = sum_p L2(p,i) L1(j,p), i <= p <= j
        for (int p = i; p <= j; p++)
          s += L2[(size_t)i * n + p] * L1[(size_t)p * n + j];
      R[(size_t)j * n + i] = s;
    }
  t.flows[2].data->written_on(false);
}

void gpu_tri_tt(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  int grid = std::min((a.n * a.n + 255) / 256, 2048);
  hipLaunchKernelGGL(k_tri_tt, dim3(grid), dim3(256), 0, g.stream,
                     (const double*)t.dev_ptr[0],
                     (const double*)t.dev_ptr[1], (double*)t.dev_ptr[2],
                     a.n);
}

// ---- C -= A * B (plain NN; C is nb-ld, A is tile-ld) ----
void cpu_gemm_nn_sub(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const double* A = (const double*)t.flows[0].data->pull_to_host();
  const double* B = (const double*)t.flows[1].data->pull_to_host();
  double* C = (double*)t.flows[2].data->pull_to_host();
  const int rows = a.m, n = a.n, kk = a.k, ld = a.ld;
  for (int j = 0; j < n; j++)
    for (int i = 0; i < rows; i++) {
      double s = 0;
      for (int p = 0; p < kk; p++)
        s += A[(size_t)p * ld + i] * B[(size_t)j * kk + p];
      C[(size_t)j * ld + i] -= s;
    }
  t.flows[2].data->written_on(false);
}

void gpu_gemm_nn_sub(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  const double mone = -1.0, one = 1.0;
  PA_CHECK(rocblas_dgemm(bc_handle(g), rocblas_operation_none,
                         rocblas_operation_none, a.m, a.n, a.k, &mone,
                         (const double*)t.dev_ptr[0], a.ld,
                         (const double*)t.dev_ptr[1], a.k, &one,
                         (double*)t.dev_ptr[2], a.ld) ==
           rocblas_status_success);
}

TaskClass make_bc_tc(const char* name, void (*cpu)(Task&),
                     void (*gpu)(Task&, GpuTaskCtx&), int id) {
  Profiler::inst().register_class(id, name);
  TaskClass tc;
  tc.name = name;
  tc.kind = TaskKind::GPU;
  tc.cpu_hook = cpu;
  tc.gpu_hook = gpu;
  tc.id = id;
  return tc;
}

TaskClass& tc_gemm_tn_acc() {
  static TaskClass tc =
      make_bc_tc("qr_gemm_tn", cpu_gemm_tn_acc, gpu_gemm_tn_acc, 50);
  return tc;
}
TaskClass& tc_zero_tile() {
  static TaskClass tc = make_bc_tc("qr_zero", cpu_zero_tile, gpu_zero_tile, 51);
  return tc;
}
TaskClass& tc_tri_tt() {
  static TaskClass tc = make_bc_tc("qr_tri_tt", cpu_tri_tt, gpu_tri_tt, 52);
  return tc;
}
TaskClass& tc_gemm_nn_sub() {
  static TaskClass tc =
      make_bc_tc("qr_gemm_nn", cpu_gemm_nn_sub, gpu_gemm_nn_sub, 53);
  return tc;
}

// One CholeskyQR pass over block column k of A: G = P^T P, L = chol(G),
// P := P L^{-T}. Leaves L in the G workspace tile (lower).
void cholqr_pass(Dtd& tp, TiledMatrix& A, int k, Data* G, Data* W,
                 int prio) {
  const int T = A.mt();
  const int nb = A.nb(), ld = A.mb();
  TileArgs za;
  za.n = nb;
  {
    Dtd::FlowSpec f[] = {{G, ACCESS_OUT}};
    tp.insert(&tc_zero_tile(), &za, sizeof(za), f, 1, prio,
              A.rank_of(k, k));
  }
  for (int m = 0; m < T; m++) {
    TileArgs a;
    a.m = A.tile_rows(m);
    a.n = nb;
    a.ld = ld;
    Dtd::FlowSpec f[] = {{A.tile(m, k), ACCESS_IN},
                         {A.tile(m, k), ACCESS_IN},
                         {G, ACCESS_INOUT}};
    tp.insert(&tc_gemm_tn_acc(), &a, sizeof(a), f, 3, prio,
              A.rank_of(k, k));
  }
  {
    TileArgs a;
    a.m = nb;
    a.n = nb;
    a.ld = nb;
    Dtd::FlowSpec f[] = {{G, ACCESS_INOUT}};
    tp.insert(&tc_potrf(), &a, sizeof(a), f, 1, prio + 1, A.rank_of(k, k));
    Dtd::FlowSpec f2[] = {{G, ACCESS_IN}, {W, ACCESS_OUT}};
    tp.insert(&tc_trtri(), &a, sizeof(a), f2, 2, prio + 1,
              A.rank_of(k, k));
  }
  for (int m = 0; m < T; m++) {
    TileArgs a;
    a.m = A.tile_rows(m);
    a.n = nb;
    a.ld = ld;
    Dtd::FlowSpec f[] = {{W, ACCESS_IN}, {A.tile(m, k), ACCESS_INOUT}};
    tp.insert(&tc_trsm_inv(), &a, sizeof(a), f, 2, prio,
              A.rank_of(m, k));
  }
}

}  // namespace

// Block classical Gram-Schmidt QR with CholeskyQR2 panels. On return
// (after wait), A holds the explicit orthonormal Q and R the upper
// tiles (R(k,n) for k <= n); R's strictly-lower tiles are untouched.
void insert_geqrf_bcgs(Dtd& tp, TiledMatrix& A, TiledMatrix& R) {
  const int T = A.nt();
  const int nb = A.nb(), ld = A.mb();
  PA_CHECK(A.m() % nb == 0 && A.mb() == A.nb(),
           "geqrf_bcgs: square tiles, M %% nb == 0");
  PA_CHECK(R.mt() >= T && R.nt() >= T && R.mb() == nb && R.nb() == nb,
           "geqrf_bcgs: R must be at least nt x nt tiles of nb x nb");
  constexpr int PANEL = 1 << 20;
  auto* ctx = A.ctx();
  // per-column workspaces: G1/W1/G2/W2 as a 4-column irregular strip
  auto Wk = std::make_shared<TiledMatrix>(ctx, (int64_t)T * nb, 4 * nb, nb,
                                          nb, A.grid_p(), A.grid_q());
  {
    std::vector<int> ranks((size_t)T * 4);
    for (int k = 0; k < T; k++)
      for (int c = 0; c < 4; c++) ranks[(size_t)k * 4 + c] = A.rank_of(k, k);
    Wk->set_rank_table(std::move(ranks));
  }
  tp.own(Wk);
  const int MT = A.mt();
  for (int k = 0; k < T; k++) {
    Data* G1 = Wk->tile(k, 0);
    Data* W1 = Wk->tile(k, 1);
    Data* G2 = Wk->tile(k, 2);
    Data* W2 = Wk->tile(k, 3);
    cholqr_pass(tp, A, k, G1, W1, PANEL);
    cholqr_pass(tp, A, k, G2, W2, PANEL);  // CholeskyQR2
    {
      // R(k,k) = (L1 L2)^T = L2^T * L1^T
      TileArgs a;
      a.n = nb;
      Dtd::FlowSpec f[] = {{G2, ACCESS_IN},
                          {G1, ACCESS_IN},
                          {R.tile(k, k), ACCESS_OUT}};
      tp.insert(&tc_tri_tt(), &a, sizeof(a), f, 3, PANEL, A.rank_of(k, k));
    }
    for (int n = k + 1; n < T; n++) {
      {
        TileArgs za;
        za.n = nb;
        Dtd::FlowSpec f[] = {{R.tile(k, n), ACCESS_OUT}};
        tp.insert(&tc_zero_tile(), &za, sizeof(za), f, 1, 1 << 18,
                  A.rank_of(k, n));
      }
      for (int m = 0; m < MT; m++) {
        TileArgs a;
        a.m = A.tile_rows(m);
        a.n = nb;
        a.ld = ld;
        Dtd::FlowSpec f[] = {{A.tile(m, k), ACCESS_IN},
                             {A.tile(m, n), ACCESS_IN},
                             {R.tile(k, n), ACCESS_INOUT}};
        tp.insert(&tc_gemm_tn_acc(), &a, sizeof(a), f, 3,
                  (1 << 18) - (n - k), A.rank_of(k, n));
      }
      for (int m = 0; m < MT; m++) {
        TileArgs a;
        a.m = A.tile_rows(m);
        a.n = nb;
        a.k = nb;
        a.ld = ld;
        Dtd::FlowSpec f[] = {{A.tile(m, k), ACCESS_IN},
                             {R.tile(k, n), ACCESS_IN},
                             {A.tile(m, n), ACCESS_INOUT}};
        tp.insert(&tc_gemm_nn_sub(), &a, sizeof(a), f, 3, -(n - k) * 4,
                  A.rank_of(m, n));
      }
    }
  }
}

}  // namespace pa
