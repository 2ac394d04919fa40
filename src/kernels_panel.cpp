// Panel-granularity Cholesky (right-looking, lower).
// (Granularity study for the headline app; see SURVEY.md §6 and
// profiles/RESULTS.md — measured slower than tile granularity, kept as
// --algo panel.)
//
// Same factorization as insert_potrf but with column panels as the data
// granule: one Data = one N x nb column panel, UPDATE(k,n) is a single
// tall dgemm (M = N - n*nb rows) and PANEL(k) factors the diagonal block
// (hand MFMA potf2 pipeline) + one big TRSM. Coarser granules trade DAG
// parallelism for per-kernel efficiency: on ONE GPU the tall dgemms run at
// rocBLAS's solo rate (~75 TF fp64 at these shapes) instead of the ~63 TF
// effective rate of co-scheduled tile kernels; with many ranks the tile
// variant exposes more parallelism. bench.py picks per world size
// (--algo auto).
#include <cmath>
#include <map>
#include <vector>

#include <hip/hip_runtime.h>
#include <rocblas/rocblas.h>
#include <rocsolver/rocsolver.h>

#include "device_gpu.hpp"
#include "kernels.hpp"
#include "profiling.hpp"

namespace pa {

void launch_potf2(double* A, int n, int ld, hipStream_t stream);

// ------------------------------------------------------------------ fill
__global__ void k_panel_fill(double* p, int64_t rows, int cols, int64_t ld,
                             int64_t j0, int64_t N, uint32_t seed) {
  int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t total = rows * cols;
  for (; idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    int64_t c = idx / rows, r = idx - c * rows;
    int64_t i = r, j = j0 + c;
    uint64_t a = (uint64_t)(i < j ? i : j), b = (uint64_t)(i < j ? j : i);
    uint64_t h = (a * 2654435761ull) ^ (b * 40503ull) ^
                 ((uint64_t)seed * 2246822519ull);
    h ^= h >> 13;
    h *= 0x9E3779B97F4A7C15ull;
    h ^= h >> 32;
    double v = (double)(h & 0xFFFFFF) / (double)0x1000000 - 0.5;
    p[c * ld + r] = (i == j) ? v + (double)N : v;
  }
}

static void gpu_panel_fill(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  hipLaunchKernelGGL(k_panel_fill, dim3(4096), dim3(256), 0, g.stream,
                     (double*)t.dev_ptr[0], (int64_t)a.ld, a.n, (int64_t)a.ld,
                     a.j0, a.N, a.seed);
}

static void cpu_panel_fill(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  Data* d = t.flows[0].data;
  double* p = (double*)d->ensure_host();
  int64_t ld = a.ld;
  for (int c = 0; c < a.n; c++)
    for (int64_t r = 0; r < ld; r++) {
      int64_t i = r, j = a.j0 + c;
      uint64_t aa = (uint64_t)(i < j ? i : j), bb = (uint64_t)(i < j ? j : i);
      uint64_t h = (aa * 2654435761ull) ^ (bb * 40503ull) ^
                   ((uint64_t)a.seed * 2246822519ull);
      h ^= h >> 13;
      h *= 0x9E3779B97F4A7C15ull;
      h ^= h >> 32;
      double v = (double)(h & 0xFFFFFF) / (double)0x1000000 - 0.5;
      p[(size_t)c * ld + r] = (i == j) ? v + (double)a.N : v;
    }
  d->written_on(false);
}

// ------------------------------------------------------------------ chores
namespace {
rocblas_handle pan_handle(GpuTaskCtx& g) {
  static thread_local std::map<void*, rocblas_handle> handles;
  rocblas_handle& h = handles[(void*)g.stream];
  if (!h) {
    PA_CHECK(rocblas_create_handle(&h) == rocblas_status_success);
    rocblas_set_pointer_mode(h, rocblas_pointer_mode_host);
    rocblas_set_stream(h, g.stream);
  }
  return h;
}
}  // namespace

// PANEL(k): potrf of the nb x nb diagonal block (128-wide hand MFMA potf2
// pipeline + rocBLAS trsm/syrk, as the tile chore) then the big TRSM of
// the rows below.
static void gpu_panel_factor(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  const int64_t ld = a.ld;
  const int nb = a.n;
  double* diag = (double*)t.dev_ptr[0] + a.i0;  // rows k*nb.., col 0
  rocblas_handle h = pan_handle(g);
  const double one = 1.0, mone = -1.0;
  for (int j = 0; j < nb; j += 128) {
    int jb = std::min(128, nb - j);
    double* Ajj = diag + (size_t)j * ld + j;
    launch_potf2(Ajj, jb, (int)ld, g.stream);
    int rest = nb - j - jb;
    if (rest > 0) {
      double* Aij = diag + (size_t)j * ld + j + jb;
      PA_CHECK(rocblas_dtrsm(h, rocblas_side_right, rocblas_fill_lower,
                             rocblas_operation_transpose,
                             rocblas_diagonal_non_unit, rest, jb, &one, Ajj,
                             (int)ld, Aij, (int)ld) == rocblas_status_success);
      double* Att = diag + (size_t)(j + jb) * ld + j + jb;
      PA_CHECK(rocblas_dsyrk(h, rocblas_fill_lower, rocblas_operation_none,
                             rest, jb, &mone, Aij, (int)ld, &one, Att,
                             (int)ld) == rocblas_status_success);
    }
  }
  const int64_t below = a.m - nb;  // rows under the diagonal block
  if (below > 0) {
    // TRSM via the inverse (as the tile variant): W = L^{-1} once, then
    // one full-rate dgemm into scratch + a strided copy back — rocBLAS
    // dtrsm at this shape decomposes into ~9 TF-effective kernels.
    double* B = diag + nb;
    double* W = (double*)g.engine->dev_alloc((size_t)nb * nb * 8);
    double* X = (double*)g.engine->dev_alloc((size_t)below * nb * 8);
    g.deferred_frees->emplace_back(W, (size_t)nb * nb * 8);
    g.deferred_frees->emplace_back(X, (size_t)below * nb * 8);
    PA_HIP_CHECK(hipMemsetAsync(W, 0, (size_t)nb * nb * 8, g.stream));
    PA_CHECK(rocblas_dtrtri(h, rocblas_fill_lower, rocblas_diagonal_non_unit,
                            nb, diag, (int)ld, W, nb) == rocblas_status_success);
    const double zero = 0.0;
    PA_CHECK(rocblas_dgemm(h, rocblas_operation_none,
                           rocblas_operation_transpose, (int)below, nb, nb,
                           &one, B, (int)ld, W, nb, &zero, X,
                           (int)below) == rocblas_status_success);
    PA_HIP_CHECK(hipMemcpy2DAsync(B, (size_t)ld * 8, X, (size_t)below * 8,
                                  (size_t)below * 8, nb,
                                  hipMemcpyDeviceToDevice, g.stream));
  }
}

static void cpu_panel_factor(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  double* p = (double*)t.flows[0].data->pull_to_host();
  const int64_t ld = a.ld;
  const int nb = a.n;
  double* diag = p + a.i0;
  // unblocked potrf of diag + solve below, reference code
  for (int j = 0; j < nb; j++) {
    double d = diag[(size_t)j * ld + j];
    for (int q = 0; q < j; q++) d -= diag[(size_t)q * ld + j] * diag[(size_t)q * ld + j];
    d = sqrt(d);
    diag[(size_t)j * ld + j] = d;
    for (int64_t i = j + 1; i < a.m; i++) {
      double s = diag[(size_t)j * ld + i];
      for (int q = 0; q < j; q++)
        s -= diag[(size_t)q * ld + i] * diag[(size_t)q * ld + j];
      diag[(size_t)j * ld + i] = s / d;
    }
  }
  t.flows[0].data->written_on(false);
}

// UPDATE(k,n): panel_n rows n*nb.. -= A1 * A2^T with both operands from
// panel_k (one tall dgemm; the diagonal block's upper half is symmetric-
// valid garbage exactly as in the tile SYRK-as-dgemm route).
static void gpu_panel_update(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  const int64_t ld = a.ld;
  const double mone = -1.0, one = 1.0;
  const double* A1 = (const double*)t.dev_ptr[0] + a.i0;  // rows n*nb..
  const double* A2 = (const double*)t.dev_ptr[0] + a.i0;  // top nb_n rows
  double* C = (double*)t.dev_ptr[1] + a.i0;
  PA_CHECK(rocblas_dgemm(pan_handle(g), rocblas_operation_none,
                         rocblas_operation_transpose, (int)a.m, a.n, a.k,
                         &mone, A1, (int)ld, A2, (int)ld, &one, C,
                         (int)ld) == rocblas_status_success);
}

static void cpu_panel_update(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const double* pk = (const double*)t.flows[0].data->pull_to_host();
  double* pn = (double*)t.flows[1].data->pull_to_host();
  const int64_t ld = a.ld;
  const double* A1 = pk + a.i0;
  double* C = pn + a.i0;
  for (int j = 0; j < a.n; j++)
    for (int64_t i = 0; i < a.m; i++) {
      double s = 0;
      for (int q = 0; q < a.k; q++)
        s += A1[(size_t)q * ld + i] * A1[(size_t)q * ld + j];
      C[(size_t)j * ld + i] -= s;
    }
  t.flows[1].data->written_on(false);
}

static TaskClass make_pan_tc(const char* name, void (*cpu)(Task&),
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

TaskClass& tc_panel_fill() {
  static TaskClass tc =
      make_pan_tc("panel_fill", cpu_panel_fill, gpu_panel_fill, 40);
  return tc;
}
TaskClass& tc_panel_factor() {
  static TaskClass tc =
      make_pan_tc("panel_factor", cpu_panel_factor, gpu_panel_factor, 41);
  return tc;
}
TaskClass& tc_panel_update() {
  static TaskClass tc =
      make_pan_tc("panel_update", cpu_panel_update, gpu_panel_update, 42);
  return tc;
}

// A is a 1 x T collection of N x nb column panels: TiledMatrix(N, N, N, nb).
void insert_panel_fill(Dtd& tp, TiledMatrix& A, uint32_t seed) {
  PA_CHECK(A.mt() == 1, "panel collection must have mb == N");
  for (int k = 0; k < A.nt(); k++) {
    TileArgs a;
    a.n = A.tile_cols(k);
    a.ld = (int)A.m();
    a.j0 = (int64_t)k * A.nb();
    a.N = A.m();
    a.seed = seed;
    Dtd::FlowSpec f[] = {{A.tile(0, k), ACCESS_OUT}};
    tp.insert(&tc_panel_fill(), &a, sizeof(a), f, 1, 0, A.rank_of(0, k));
  }
}

void insert_potrf_panel(Dtd& tp, TiledMatrix& A) {
  PA_CHECK(A.mt() == 1, "panel collection must have mb == N");
  const int T = A.nt();
  const int64_t N = A.m();
  const int nb = A.nb();
  constexpr int PANEL = 1 << 20;
  for (int k = 0; k < T; k++) {
    {
      TileArgs a;
      a.i0 = (int64_t)k * nb;
      a.m = (int)(N - (int64_t)k * nb);
      a.n = A.tile_cols(k);
      a.ld = (int)N;
      Dtd::FlowSpec f[] = {{A.tile(0, k), ACCESS_INOUT}};
      tp.insert(&tc_panel_factor(), &a, sizeof(a), f, 1, PANEL + 1,
                A.rank_of(0, k));
    }
    for (int n = k + 1; n < T; n++) {
      TileArgs a;
      a.i0 = (int64_t)n * nb;
      a.m = (int)(N - (int64_t)n * nb);
      a.n = A.tile_cols(n);
      a.k = A.tile_cols(k);
      a.ld = (int)N;
      Dtd::FlowSpec f[] = {{A.tile(0, k), ACCESS_IN},
                           {A.tile(0, n), ACCESS_INOUT}};
      tp.insert(&tc_panel_update(), &a, sizeof(a), f, 2, -(n - k) * 4,
                A.rank_of(0, n));
    }
  }
}

}  // namespace pa
