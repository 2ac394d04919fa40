// Tile LU without pivoting (getrf_nopiv) — third dense factorization
// family (DPLASMA dgetrf_nopiv analog; the synthetic operator is
// diagonally dominant, where pivot-free LU is stable).
//
// DAG per step k: GETRF(k,k) -> TRSM_L(k,n) row panel (L^-1 from the
// left, unit-lower), TRSM_U(m,k) column panel (U^-1 from the right),
// GEMM_NN(m,n) trailing update A(m,n) -= A(m,k) A(k,n).
// GPU chores: rocSOLVER getrf_npvt (on a blocking worker, QR-panel
// pattern) + rocBLAS trsm/gemm. CPU chores: reference loops for the
// no-GPU test path.
#include <cstring>
#include <map>

#include <hip/hip_runtime.h>
#include <rocblas/rocblas.h>
#include <rocsolver/rocsolver.h>

#include "device_gpu.hpp"
#include "kernels.hpp"
#include "profiling.hpp"

namespace pa {

namespace {

rocblas_handle lu_handle(GpuTaskCtx& g) {
  static thread_local std::map<void*, rocblas_handle> handles;
  rocblas_handle& h = handles[(void*)g.stream];
  if (!h) {
    PA_CHECK(rocblas_create_handle(&h) == rocblas_status_success);
    rocblas_set_pointer_mode(h, rocblas_pointer_mode_host);
    rocblas_set_stream(h, g.stream);
  }
  return h;
}

// ---- CPU reference chores (column-major, ld = tile rows) ----
void cpu_getrf(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  double* A = (double*)t.flows[0].data->pull_to_host();
  const int n = a.n, ld = a.ld;
  for (int k = 0; k < n; k++) {
    const double piv = A[(size_t)k * ld + k];
    PA_CHECK(piv != 0.0, "getrf_nopiv: zero pivot at %d", k);
    for (int i = k + 1; i < n; i++) A[(size_t)k * ld + i] /= piv;
    for (int j = k + 1; j < n; j++) {
      const double u = A[(size_t)j * ld + k];
      for (int i = k + 1; i < n; i++)
        A[(size_t)j * ld + i] -= A[(size_t)k * ld + i] * u;
    }
  }
  t.flows[0].data->written_on(false);
}

// B := L^{-1} B  (left, unit-lower L = strict lower of diag tile)
void cpu_trsm_l(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const double* L = (const double*)t.flows[0].data->pull_to_host();
  double* B = (double*)t.flows[1].data->pull_to_host();
  const int n = a.m, cols = a.n, ld = a.ld;
  for (int j = 0; j < cols; j++)
    for (int i = 0; i < n; i++) {
      double s = B[(size_t)j * ld + i];
      for (int p = 0; p < i; p++)
        s -= L[(size_t)p * ld + i] * B[(size_t)j * ld + p];
      B[(size_t)j * ld + i] = s;
    }
  t.flows[1].data->written_on(false);
}

// B := B U^{-1}  (right, upper non-unit U = upper of diag tile)
void cpu_trsm_u(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const double* U = (const double*)t.flows[0].data->pull_to_host();
  double* B = (double*)t.flows[1].data->pull_to_host();
  const int rows = a.m, n = a.n, ld = a.ld;
  for (int j = 0; j < n; j++) {
    const double inv = 1.0 / U[(size_t)j * ld + j];
    for (int i = 0; i < rows; i++) {
      double s = B[(size_t)j * ld + i];
      for (int p = 0; p < j; p++)
        s -= B[(size_t)p * ld + i] * U[(size_t)j * ld + p];
      B[(size_t)j * ld + i] = s * inv;
    }
  }
  t.flows[1].data->written_on(false);
}

// C -= A * B  (no transpose)
void cpu_gemm_nn(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const double* A = (const double*)t.flows[0].data->pull_to_host();
  const double* B = (const double*)t.flows[1].data->pull_to_host();
  double* C = (double*)t.flows[2].data->pull_to_host();
  const int m = a.m, n = a.n, k = a.k, ld = a.ld;
  for (int j = 0; j < n; j++)
    for (int i = 0; i < m; i++) {
      double s = 0;
      for (int p = 0; p < k; p++)
        s += A[(size_t)p * ld + i] * B[(size_t)j * ld + p];
      C[(size_t)j * ld + i] -= s;
    }
  t.flows[2].data->written_on(false);
}

// ---- inverse-based TRSM chores (Cholesky's TRTRI+GEMM trick applied to
// LU: rocBLAS dtrsm decomposes into ~44 sub-rate kernels; one per-step
// inverse turns every panel solve into a single full-rate dgemm.
// W tile layout (nb x 2nb, ld=nb): cols [0,nb) = U^{-1} (upper,
// non-unit), cols [nb,2nb) = L^{-1} (unit lower). ----
void cpu_lu_trtri(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const double* A = (const double*)t.flows[0].data->pull_to_host();
  Data* wd = t.flows[1].data;
  double* W = (double*)wd->ensure_host();
  const int n = a.n, ld = a.ld;
  memset(W, 0, wd->bytes);
  double* Wu = W;
  double* Wl = W + (size_t)n * n;
  for (int c = 0; c < n; c++) {
    Wu[(size_t)c * n + c] = 1.0 / A[(size_t)c * ld + c];
    for (int r = c - 1; r >= 0; r--) {
      double sum = 0;
      for (int k = r + 1; k <= c; k++)
        sum += A[(size_t)k * ld + r] * Wu[(size_t)c * n + k];
      Wu[(size_t)c * n + r] = -sum / A[(size_t)r * ld + r];
    }
  }
  for (int c = 0; c < n; c++) {
    Wl[(size_t)c * n + c] = 1.0;
    for (int r = c + 1; r < n; r++) {
      double sum = 0;
      for (int k = c; k < r; k++)
        sum += A[(size_t)k * ld + r] * Wl[(size_t)c * n + k];
      Wl[(size_t)c * n + r] = -sum;  // unit diagonal
    }
  }
  wd->written_on(false);
}

// column panel: B <- B * U^{-1}
void cpu_trsm_u_inv(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const double* W = (const double*)t.flows[0].data->pull_to_host();
  double* B = (double*)t.flows[1].data->pull_to_host();
  const int m = a.m, n = a.n, ld = a.ld;
  const double* Wu = W;
  std::vector<double> tmp((size_t)m * n);
  for (int j = 0; j < n; j++)
    for (int i = 0; i < m; i++) tmp[(size_t)j * m + i] = B[(size_t)j * ld + i];
  for (int j = 0; j < n; j++)
    for (int i = 0; i < m; i++) {
      double sum = 0;
      for (int k = 0; k <= j; k++)
        sum += tmp[(size_t)k * m + i] * Wu[(size_t)j * n + k];
      B[(size_t)j * ld + i] = sum;
    }
  t.flows[1].data->written_on(false);
}

// row panel: B <- L^{-1} * B
void cpu_trsm_l_inv(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const double* W = (const double*)t.flows[0].data->pull_to_host();
  double* B = (double*)t.flows[1].data->pull_to_host();
  const int n = a.m, cols = a.n, ld = a.ld;
  const double* Wl = W + (size_t)n * n;
  std::vector<double> tmp((size_t)n * cols);
  for (int j = 0; j < cols; j++)
    for (int i = 0; i < n; i++) tmp[(size_t)j * n + i] = B[(size_t)j * ld + i];
  for (int j = 0; j < cols; j++)
    for (int i = 0; i < n; i++) {
      double sum = 0;
      for (int k = 0; k <= i; k++)
        sum += Wl[(size_t)k * n + i] * tmp[(size_t)j * n + k];
      B[(size_t)j * ld + i] = sum;
    }
  t.flows[1].data->written_on(false);
}

// ---- GPU chores ----
rocblas_int* lu_dev_info(GpuTaskCtx& g) {
  static thread_local std::map<void*, rocblas_int*> infos;
  rocblas_int*& p = infos[(void*)g.stream];
  if (!p) PA_HIP_CHECK(hipMalloc(&p, sizeof(rocblas_int)));
  return p;
}

void gpu_getrf(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  rocblas_int* dinfo = lu_dev_info(g);
  PA_CHECK(rocsolver_dgetrf_npvt(lu_handle(g), a.n, a.n,
                                 (double*)t.dev_ptr[0], a.ld,
                                 dinfo) == rocblas_status_success);
  // Blocking chore (worker thread): read the info scalar back so a
  // zero/tiny pivot fails as loudly as the CPU reference path does,
  // instead of silently poisoning the trailing matrix with Inf/NaN.
  rocblas_int info = 0;
  PA_HIP_CHECK(hipMemcpyAsync(&info, dinfo, sizeof(info),
                              hipMemcpyDeviceToHost, g.stream));
  PA_HIP_CHECK(hipStreamSynchronize(g.stream));
  PA_CHECK(info == 0, "getrf_nopiv: singular pivot at column %d", (int)info);
}

void gpu_trsm_l(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  const double one = 1.0;
  PA_CHECK(rocblas_dtrsm(lu_handle(g), rocblas_side_left,
                         rocblas_fill_lower, rocblas_operation_none,
                         rocblas_diagonal_unit, a.m, a.n, &one,
                         (const double*)t.dev_ptr[0], a.ld,
                         (double*)t.dev_ptr[1], a.ld) ==
           rocblas_status_success);
}

void gpu_trsm_u(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  const double one = 1.0;
  PA_CHECK(rocblas_dtrsm(lu_handle(g), rocblas_side_right,
                         rocblas_fill_upper, rocblas_operation_none,
                         rocblas_diagonal_non_unit, a.m, a.n, &one,
                         (const double*)t.dev_ptr[0], a.ld,
                         (double*)t.dev_ptr[1], a.ld) ==
           rocblas_status_success);
}

void gpu_gemm_nn(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  const double mone = -1.0, one = 1.0;
  PA_CHECK(rocblas_dgemm(lu_handle(g), rocblas_operation_none,
                         rocblas_operation_none, a.m, a.n, a.k, &mone,
                         (const double*)t.dev_ptr[0], a.ld,
                         (const double*)t.dev_ptr[1], a.ld, &one,
                         (double*)t.dev_ptr[2], a.ld) ==
           rocblas_status_success);
}

void gpu_lu_trtri(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  Data* wd = t.flows[1].data;
  double* W = (double*)t.dev_ptr[1];
  PA_HIP_CHECK(hipMemsetAsync(W, 0, wd->bytes, g.stream));
  rocblas_handle h = lu_handle(g);
  PA_CHECK(rocblas_dtrtri(h, rocblas_fill_upper, rocblas_diagonal_non_unit,
                          a.n, (const double*)t.dev_ptr[0], a.ld, W,
                          a.n) == rocblas_status_success);
  PA_CHECK(rocblas_dtrtri(h, rocblas_fill_lower, rocblas_diagonal_unit, a.n,
                          (const double*)t.dev_ptr[0], a.ld,
                          W + (size_t)a.n * a.n,
                          a.n) == rocblas_status_success);
}

// X = B * Wu into a fresh pool buffer, swapped in as the tile's device
// copy (no D2D copy — the Cholesky buffer-swap pattern).
void lu_swap_gemm(Task& t, GpuTaskCtx& g, const double* A0, int lda,
                  const double* B0, int ldb, int m, int n, int k) {
  Data* bd = t.flows[1].data;
  double* X = (double*)g.engine->dev_alloc(bd->bytes);
  const double one = 1.0, zero = 0.0;
  PA_CHECK(rocblas_dgemm(lu_handle(g), rocblas_operation_none,
                         rocblas_operation_none, m, n, k, &one, A0, lda, B0,
                         ldb, &zero, X, m) == rocblas_status_success);
  {
    SpinGuard gd(bd->lock);
    g.deferred_frees->emplace_back(bd->dev_ptr, bd->bytes);
    bd->dev_ptr = X;
  }
  t.dev_ptr[1] = X;
}

void gpu_trsm_u_inv(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  const double* Wu = (const double*)t.dev_ptr[0];
  lu_swap_gemm(t, g, (const double*)t.dev_ptr[1], a.ld, Wu, a.n, a.m, a.n,
               a.n);
}

void gpu_trsm_l_inv(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  const double* Wl = (const double*)t.dev_ptr[0] + (size_t)a.m * a.m;
  lu_swap_gemm(t, g, Wl, a.m, (const double*)t.dev_ptr[1], a.ld, a.m, a.n,
               a.m);
}

TaskClass make_lu_tc(const char* name, void (*cpu)(Task&),
                     void (*gpu)(Task&, GpuTaskCtx&), int id,
                     bool blocking = false) {
  Profiler::inst().register_class(id, name);
  TaskClass tc;
  tc.name = name;
  tc.kind = TaskKind::GPU;
  tc.cpu_hook = cpu;
  tc.gpu_hook = gpu;
  tc.gpu_blocking = blocking;
  tc.id = id;
  return tc;
}

TaskClass& tc_getrf() {
  static TaskClass tc = make_lu_tc("getrf_nopiv", cpu_getrf, gpu_getrf, 40,
                                   /*blocking=*/true);
  return tc;
}
TaskClass& tc_lutrsml() {
  static TaskClass tc = make_lu_tc("lu_trsm_l", cpu_trsm_l, gpu_trsm_l, 41);
  return tc;
}
TaskClass& tc_lutrsmu() {
  static TaskClass tc = make_lu_tc("lu_trsm_u", cpu_trsm_u, gpu_trsm_u, 42);
  return tc;
}
TaskClass& tc_lugemm() {
  static TaskClass tc = make_lu_tc("lu_gemm", cpu_gemm_nn, gpu_gemm_nn, 43);
  return tc;
}
TaskClass& tc_lu_trtri() {
  static TaskClass tc =
      make_lu_tc("lu_trtri", cpu_lu_trtri, gpu_lu_trtri, 44);
  return tc;
}
TaskClass& tc_lutrsml_inv() {
  static TaskClass tc =
      make_lu_tc("lu_trsm_l_inv", cpu_trsm_l_inv, gpu_trsm_l_inv, 45);
  return tc;
}
TaskClass& tc_lutrsmu_inv() {
  static TaskClass tc =
      make_lu_tc("lu_trsm_u_inv", cpu_trsm_u_inv, gpu_trsm_u_inv, 46);
  return tc;
}

}  // namespace

void insert_getrf_nopiv(Dtd& tp, TiledMatrix& A) {
  const int T = A.mt();
  const int nb = A.nb(), ld = A.mb();
  PA_CHECK(A.m() == A.n() && A.m() % nb == 0 && A.mb() == A.nb(),
           "getrf_nopiv: square matrix, square tiles, N %% nb == 0");
  constexpr int PANEL = 1 << 20;
  // Default: per-step TRTRI of the diag tile's two factors, then every
  // panel solve is ONE full-rate dgemm (chore_lu_trsm=rocblas restores
  // the library dtrsm chores).
  const bool inv = param_str("chore_lu_trsm", "inv") == "inv";
  std::shared_ptr<TiledMatrix> Wc;
  if (inv) {
    std::vector<int> wranks(T);
    for (int k = 0; k < T; k++) wranks[k] = A.rank_of(k, k);
    Wc = std::make_shared<TiledMatrix>(A.ctx(), (int64_t)T * nb, 2 * nb,
                                       nb, 2 * nb, A.grid_p(), A.grid_q());
    Wc->set_rank_table(std::move(wranks));
    tp.own(Wc);
  }
  for (int k = 0; k < T; k++) {
    TileArgs a;
    a.n = nb;
    a.ld = ld;
    {
      TileArgs d = a;
      d.m = nb;
      Dtd::FlowSpec f[] = {{A.tile(k, k), ACCESS_INOUT}};
      tp.insert(&tc_getrf(), &d, sizeof(d), f, 1, PANEL + 1,
                A.rank_of(k, k));
    }
    if (inv && k + 1 < T) {
      TileArgs d = a;
      d.m = nb;
      Dtd::FlowSpec f[] = {{A.tile(k, k), ACCESS_IN},
                           {Wc->tile(k, 0), ACCESS_OUT}};
      tp.insert(&tc_lu_trtri(), &d, sizeof(d), f, 2, PANEL,
                A.rank_of(k, k));
    }
    for (int n = k + 1; n < T; n++) {
      TileArgs d = a;
      d.m = nb;
      if (inv) {
        Dtd::FlowSpec f[] = {{Wc->tile(k, 0), ACCESS_IN},
                             {A.tile(k, n), ACCESS_INOUT}};
        tp.insert(&tc_lutrsml_inv(), &d, sizeof(d), f, 2,
                  (1 << 18) - (n - k), A.rank_of(k, n));
      } else {
        Dtd::FlowSpec f[] = {{A.tile(k, k), ACCESS_IN},
                             {A.tile(k, n), ACCESS_INOUT}};
        tp.insert(&tc_lutrsml(), &d, sizeof(d), f, 2, (1 << 18) - (n - k),
                  A.rank_of(k, n));
      }
    }
    for (int m = k + 1; m < T; m++) {
      TileArgs d = a;
      d.m = nb;
      if (inv) {
        Dtd::FlowSpec f[] = {{Wc->tile(k, 0), ACCESS_IN},
                             {A.tile(m, k), ACCESS_INOUT}};
        tp.insert(&tc_lutrsmu_inv(), &d, sizeof(d), f, 2, PANEL,
                  A.rank_of(m, k));
      } else {
        Dtd::FlowSpec f[] = {{A.tile(k, k), ACCESS_IN},
                             {A.tile(m, k), ACCESS_INOUT}};
        tp.insert(&tc_lutrsmu(), &d, sizeof(d), f, 2, PANEL,
                  A.rank_of(m, k));
      }
    }
    for (int m = k + 1; m < T; m++)
      for (int n = k + 1; n < T; n++) {
        TileArgs d = a;
        d.m = nb;
        d.k = nb;
        Dtd::FlowSpec f[] = {{A.tile(m, k), ACCESS_IN},
                             {A.tile(k, n), ACCESS_IN},
                             {A.tile(m, n), ACCESS_INOUT}};
        tp.insert(&tc_lugemm(), &d, sizeof(d), f, 3, -(n - k) * 4,
                  A.rank_of(m, n));
      }
  }
}

}  // namespace pa
