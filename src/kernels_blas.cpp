// Library-chore tile kernels (rocBLAS / rocSOLVER) + CPU reference chores +
// synthetic SPD fill, and the Cholesky DAG builder.
//
// CPU chores are straightforward reference implementations used by the
// no-GPU test path (numerics tests compare the HIP path against these / a
// NumPy fp64 reference); the GPU path is the production path.
#include "kernels.hpp"

#include <cmath>
#include <map>
#include <string>

#include <hip/hip_runtime.h>
#include <rocblas/rocblas.h>
#include <rocsolver/rocsolver.h>

#include "device_gpu.hpp"
#include "profiling.hpp"

namespace pa {

// ------------------------------------------------------------------ fill
// Deterministic synthetic SPD matrix: symmetric hash values in [-0.5, 0.5),
// + N on the diagonal (diagonally dominant => positive definite).
__host__ __device__ inline double spd_val(int64_t i, int64_t j, int64_t N,
                                          uint32_t seed) {
  uint64_t a = (uint64_t)(i < j ? i : j), b = (uint64_t)(i < j ? j : i);
  uint64_t h = (a * 2654435761ull) ^ (b * 40503ull) ^
               ((uint64_t)seed * 2246822519ull);
  h ^= h >> 13;
  h *= 0x9E3779B97F4A7C15ull;
  h ^= h >> 32;
  double v = (double)(h & 0xFFFFFF) / (double)0x1000000 - 0.5;
  return i == j ? v + (double)N : v;
}

__global__ void k_spd_fill(double* t, int rows, int cols, int ld, int64_t i0,
                           int64_t j0, int64_t N, uint32_t seed) {
  int idx = blockIdx.x * blockDim.x + threadIdx.x;
  int total = rows * cols;
  for (; idx < total; idx += gridDim.x * blockDim.x) {
    int c = idx / rows, r = idx - c * rows;
    t[(size_t)c * ld + r] = spd_val(i0 + r, j0 + c, N, seed);
  }
}

static void cpu_spd_fill(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  Data* d = t.flows[0].data;
  double* p = (double*)d->ensure_host();
  for (int c = 0; c < a.n; c++)
    for (int r = 0; r < a.m; r++)
      p[(size_t)c * a.ld + r] = spd_val(a.i0 + r, a.j0 + c, a.N, a.seed);
  d->written_on(false);
}

static void gpu_spd_fill(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  int total = a.m * a.n;
  int block = 256;
  int grid = std::min((total + block - 1) / block, 4096);
  hipLaunchKernelGGL(k_spd_fill, dim3(grid), dim3(block), 0, g.stream,
                     (double*)t.dev_ptr[0], a.m, a.n, a.ld, a.i0, a.j0, a.N,
                     a.seed);
}

// ------------------------------------------------------------------ CPU chores
// Naive fp64 reference bodies (column-major, ld = tile rows).
static void cpu_potrf(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  double* A = (double*)t.flows[0].data->pull_to_host();
  const int n = a.n, ld = a.ld;
  for (int j = 0; j < n; j++) {
    double d = A[(size_t)j * ld + j];
    for (int k = 0; k < j; k++) d -= A[(size_t)k * ld + j] * A[(size_t)k * ld + j];
    PA_CHECK(d > 0, "potrf: matrix not SPD at column %d", j);
    d = std::sqrt(d);
    A[(size_t)j * ld + j] = d;
    for (int i = j + 1; i < n; i++) {
      double s = A[(size_t)j * ld + i];
      for (int k = 0; k < j; k++) s -= A[(size_t)k * ld + i] * A[(size_t)k * ld + j];
      A[(size_t)j * ld + i] = s / d;
    }
  }
  t.flows[0].data->written_on(false);
}

// B = B * L^{-T}  (right, lower, transposed, non-unit) — TRSM of the tile panel.
static void cpu_trsm(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const double* L = (const double*)t.flows[0].data->pull_to_host();
  double* B = (double*)t.flows[1].data->pull_to_host();
  const int m = a.m, n = a.n, ld = a.ld;
  for (int j = 0; j < n; j++) {
    double inv = 1.0 / L[(size_t)j * ld + j];
    for (int i = 0; i < m; i++) {
      double s = B[(size_t)j * ld + i];
      for (int k = 0; k < j; k++) s -= B[(size_t)k * ld + i] * L[(size_t)k * ld + j];
      B[(size_t)j * ld + i] = s * inv;
    }
  }
  t.flows[1].data->written_on(false);
}

// C = C - A*A^T (lower part only guaranteed; full tile computed is fine)
static void cpu_syrk(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const double* A = (const double*)t.flows[0].data->pull_to_host();
  double* C = (double*)t.flows[1].data->pull_to_host();
  const int n = a.n, k = a.k, ld = a.ld;
  for (int j = 0; j < n; j++)
    for (int i = j; i < n; i++) {
      double s = 0;
      for (int p = 0; p < k; p++) s += A[(size_t)p * ld + i] * A[(size_t)p * ld + j];
      C[(size_t)j * ld + i] -= s;
    }
  t.flows[1].data->written_on(false);
}

// W = L^{-1} (lower triangular inverse; upper of W zeroed)
static void cpu_trtri(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const double* L = (const double*)t.flows[0].data->pull_to_host();
  Data* wd = t.flows[1].data;
  double* W = (double*)wd->ensure_host();
  const int n = a.n, ld = a.ld;
  memset(W, 0, wd->bytes);
  for (int c = 0; c < n; c++) {
    W[(size_t)c * ld + c] = 1.0 / L[(size_t)c * ld + c];
    for (int i = c + 1; i < n; i++) {
      double s = 0;
      for (int k = c; k < i; k++) s += L[(size_t)k * ld + i] * W[(size_t)c * ld + k];
      W[(size_t)c * ld + i] = -s / L[(size_t)i * ld + i];
    }
  }
  wd->written_on(false);
}

// B = B * W^T where W = L^{-1}  (the TRSM-as-GEMM variant)
static void cpu_trsm_inv(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const double* W = (const double*)t.flows[0].data->pull_to_host();
  double* B = (double*)t.flows[1].data->pull_to_host();
  const int m = a.m, n = a.n, ld = a.ld;
  std::vector<double> tmp((size_t)n * m);
  for (int j = 0; j < n; j++)
    for (int i = 0; i < m; i++) tmp[(size_t)j * m + i] = B[(size_t)j * ld + i];
  for (int j = 0; j < n; j++)
    for (int i = 0; i < m; i++) {
      double s = 0;
      for (int k = 0; k < n; k++)
        s += tmp[(size_t)k * m + i] * W[(size_t)k * ld + j];
      B[(size_t)j * ld + i] = s;
    }
  t.flows[1].data->written_on(false);
}

// C = C - A*B^T
static void cpu_gemm(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const double* A = (const double*)t.flows[0].data->pull_to_host();
  const double* B = (const double*)t.flows[1].data->pull_to_host();
  double* C = (double*)t.flows[2].data->pull_to_host();
  const int m = a.m, n = a.n, k = a.k, ld = a.ld;
  for (int j = 0; j < n; j++)
    for (int i = 0; i < m; i++) {
      double s = 0;
      for (int p = 0; p < k; p++) s += A[(size_t)p * ld + i] * B[(size_t)p * ld + j];
      C[(size_t)j * ld + i] -= s;
    }
  t.flows[2].data->written_on(false);
}

void launch_dsyrk_v2(int n, int k, const double* A, int lda, double* C,
                     int ldc, hipStream_t stream);  // kernels_hip.cpp

// ------------------------------------------------------------------ GPU chores
// One rocBLAS handle PER EXEC STREAM: rocBLAS/rocSOLVER keep device
// workspace on the handle (e.g. dtrsm's invA), so concurrent kernels on
// different streams must not share one. Hooks run on the manager thread
// only, so a thread_local map is race-free.
static rocblas_handle blas_handle(GpuTaskCtx& g) {
  static thread_local std::map<void*, rocblas_handle> handles;
  rocblas_handle& h = handles[(void*)g.stream];
  if (!h) {
    PA_CHECK(rocblas_create_handle(&h) == rocblas_status_success);
    rocblas_set_pointer_mode(h, rocblas_pointer_mode_host);
    rocblas_set_stream(h, g.stream);
  }
  return h;
}

static rocblas_int* dev_info(GpuTaskCtx& g) {
  static thread_local std::map<void*, rocblas_int*> infos;
  rocblas_int*& p = infos[(void*)g.stream];
  if (!p) PA_HIP_CHECK(hipMalloc(&p, sizeof(rocblas_int)));
  return p;
}

void launch_potf2(double* A, int n, int ld, hipStream_t stream);  // kernels_hip

// Blocked tile POTRF: 128-wide LDS panel factorization (hand kernel) +
// rocBLAS panel-TRSM + trailing SYRK, all in-order on the task's stream.
// Critical-path op: measured ~3x faster than rocsolver_dpotrf at nb=2048.
static void gpu_potrf_blocked(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  double* A = (double*)t.dev_ptr[0];
  const int n = a.n, ld = a.ld;
  const double one = 1.0, mone = -1.0;
  rocblas_handle h = blas_handle(g);
  for (int j = 0; j < n; j += 128) {
    int jb = std::min(128, n - j);
    double* Ajj = A + (size_t)j * ld + j;
    launch_potf2(Ajj, jb, ld, g.stream);
    int rest = n - j - jb;
    if (rest > 0) {
      double* Aij = A + (size_t)j * ld + j + jb;
      PA_CHECK(rocblas_dtrsm(h, rocblas_side_right, rocblas_fill_lower,
                             rocblas_operation_transpose,
                             rocblas_diagonal_non_unit, rest, jb, &one, Ajj,
                             ld, Aij, ld) == rocblas_status_success);
      double* Att = A + (size_t)(j + jb) * ld + j + jb;
      PA_CHECK(rocblas_dsyrk(h, rocblas_fill_lower, rocblas_operation_none,
                             rest, jb, &mone, Aij, ld, &one, Att,
                             ld) == rocblas_status_success);
    }
  }
}

static void gpu_potrf(Task& t, GpuTaskCtx& g) {
  static const bool use_rocsolver =
      param_str("chore_potrf", "hip") == "rocsolver";
  if (!use_rocsolver) { gpu_potrf_blocked(t, g); return; }
  const TileArgs& a = t.arg<TileArgs>();
  rocblas_status s = rocsolver_dpotrf(blas_handle(g), rocblas_fill_lower, a.n,
                                      (double*)t.dev_ptr[0], a.ld, dev_info(g));
  PA_CHECK(s == rocblas_status_success, "rocsolver_dpotrf failed: %d", (int)s);
}

static void gpu_trsm(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  const double one = 1.0;
  rocblas_status s = rocblas_dtrsm(
      blas_handle(g), rocblas_side_right, rocblas_fill_lower,
      rocblas_operation_transpose, rocblas_diagonal_non_unit, a.m, a.n, &one,
      (const double*)t.dev_ptr[0], a.ld, (double*)t.dev_ptr[1], a.ld);
  PA_CHECK(s == rocblas_status_success, "rocblas_dtrsm failed: %d", (int)s);
}

static void gpu_syrk(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  const double mone = -1.0, one = 1.0;
  static const std::string chore = param_str("chore_syrk", "hip");
  static const bool use_syrkx = chore == "syrkx";
  static const bool use_hip = chore == "hip";
  if (use_hip) {
    // hand MFMA kernel over the lower-triangular block set only: half the
    // FLOPs of the full-tile dgemm route.
    launch_dsyrk_v2(a.n, a.k, (const double*)t.dev_ptr[0], a.ld,
                    (double*)t.dev_ptr[1], a.ld, g.stream);
    return;
  }
  if (!use_syrkx) {
    // SYRK as full-tile DGEMM(A, A^T): 2x the FLOPs but ~4x faster wall on
    // MI355X (rocBLAS syrkx measured 7.5 TF vs dgemm 60 TF at nb=2048);
    // the upper triangle of diagonal tiles is never read by the DAG.
    rocblas_status s = rocblas_dgemm(
        blas_handle(g), rocblas_operation_none, rocblas_operation_transpose,
        a.n, a.n, a.k, &mone, (const double*)t.dev_ptr[0], a.ld,
        (const double*)t.dev_ptr[0], a.ld, &one, (double*)t.dev_ptr[1], a.ld);
    PA_CHECK(s == rocblas_status_success, "syrk-as-dgemm failed: %d", (int)s);
    return;
  }
  rocblas_status s = rocblas_dsyrk(blas_handle(g), rocblas_fill_lower,
                                   rocblas_operation_none, a.n, a.k, &mone,
                                   (const double*)t.dev_ptr[0], a.ld, &one,
                                   (double*)t.dev_ptr[1], a.ld);
  PA_CHECK(s == rocblas_status_success, "rocblas_dsyrk failed: %d", (int)s);
}

static void gpu_trtri(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  Data* wd = t.flows[1].data;
  PA_HIP_CHECK(hipMemsetAsync(t.dev_ptr[1], 0, wd->bytes, g.stream));
  rocblas_status s = rocblas_dtrtri(
      blas_handle(g), rocblas_fill_lower, rocblas_diagonal_non_unit, a.n,
      (const double*)t.dev_ptr[0], a.ld, (double*)t.dev_ptr[1], a.ld);
  PA_CHECK(s == rocblas_status_success, "rocblas_dtrtri failed: %d", (int)s);
}

// TRSM via the per-step inverse: B <- B * W^T is one full-rate DGEMM
// instead of rocBLAS dtrsm's ~44-kernel decomposition (measured ~9 TF vs
// ~60-75 TF for dgemm at these tile sizes). W is computed once per panel
// step and broadcast like the diagonal tile.
static void* stream_scratch(GpuTaskCtx& g, size_t bytes) {
  static thread_local std::map<void*, std::pair<void*, size_t>> bufs;
  auto& e = bufs[(void*)g.stream];
  if (e.second < bytes) {
    // old buffer may still be in use by earlier kernels on this stream
    if (e.first) g.deferred_frees->emplace_back(e.first, e.second);
    e.first = g.engine->dev_alloc(bytes);
    e.second = bytes;
  }
  return e.first;
}

static void gpu_trsm_inv(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  Data* bd = t.flows[1].data;
  double* B = (double*)t.dev_ptr[1];
  // Out-of-place X = B * W^T into a fresh pool buffer, then swap it in as
  // the tile's device copy — no D2D copy (was ~13% of the Cholesky step).
  // The old buffer returns to the pool only after the kernel completes
  // (deferred free through the engine's retire path).
  double* X = (double*)g.engine->dev_alloc(bd->bytes);
  const double one = 1.0, zero = 0.0;
  rocblas_status s = rocblas_dgemm(
      blas_handle(g), rocblas_operation_none, rocblas_operation_transpose,
      a.m, a.n, a.n, &one, B, a.ld, (const double*)t.dev_ptr[0], a.ld,
      &zero, X, a.ld);
  PA_CHECK(s == rocblas_status_success, "trsm-as-gemm failed: %d", (int)s);
  {
    SpinGuard gd(bd->lock);
    g.deferred_frees->emplace_back(bd->dev_ptr, bd->bytes);
    bd->dev_ptr = X;
  }
  t.dev_ptr[1] = X;
}

void gpu_gemm_hip(Task& t, GpuTaskCtx& g);  // kernels_hip.cpp

static void gpu_gemm(Task& t, GpuTaskCtx& g) {
  static const bool use_hip = param_str("chore_gemm", "rocblas") == "hip";
  if (use_hip) { gpu_gemm_hip(t, g); return; }
  const TileArgs& a = t.arg<TileArgs>();
  const double mone = -1.0, one = 1.0;
  rocblas_status s = rocblas_dgemm(
      blas_handle(g), rocblas_operation_none, rocblas_operation_transpose,
      a.m, a.n, a.k, &mone, (const double*)t.dev_ptr[0], a.ld,
      (const double*)t.dev_ptr[1], a.ld, &one, (double*)t.dev_ptr[2], a.ld);
  PA_CHECK(s == rocblas_status_success, "rocblas_dgemm failed: %d", (int)s);
}

// Timed rocBLAS dgemm NT loop for kernel-level A/B against the hand kernel.
double bench_dgemm_rocblas(int m, int n, int k, int iters) {
  double *dA, *dB, *dC;
  PA_HIP_CHECK(hipMalloc(&dA, (size_t)m * k * 8));
  PA_HIP_CHECK(hipMalloc(&dB, (size_t)n * k * 8));
  PA_HIP_CHECK(hipMalloc(&dC, (size_t)m * n * 8));
  hipLaunchKernelGGL(k_spd_fill, dim3(2048), dim3(256), 0, 0, dA, m, k, m, 0,
                     0, 0, 11u);
  hipLaunchKernelGGL(k_spd_fill, dim3(2048), dim3(256), 0, 0, dB, n, k, n, 7,
                     3, 0, 12u);
  hipLaunchKernelGGL(k_spd_fill, dim3(2048), dim3(256), 0, 0, dC, m, n, m, 1,
                     9, 0, 13u);
  rocblas_handle h;
  PA_CHECK(rocblas_create_handle(&h) == rocblas_status_success);
  const double mone = -1.0, one = 1.0;
  auto run = [&] {
    rocblas_dgemm(h, rocblas_operation_none, rocblas_operation_transpose, m,
                  n, k, &mone, dA, m, dB, n, &one, dC, m);
  };
  run();
  PA_HIP_CHECK(hipDeviceSynchronize());
  double t0 = now_s();
  for (int i = 0; i < iters; i++) run();
  PA_HIP_CHECK(hipDeviceSynchronize());
  double dt = now_s() - t0;
  rocblas_destroy_handle(h);
  PA_HIP_CHECK(hipFree(dA));
  PA_HIP_CHECK(hipFree(dB));
  PA_HIP_CHECK(hipFree(dC));
  return dt;
}

// ------------------------------------------------------------------ classes
static TaskClass make_tc(const char* name, TaskKind kind,
                         void (*cpu)(Task&), void (*gpu)(Task&, GpuTaskCtx&),
                         int id) {
  Profiler::inst().register_class(id, name);
  TaskClass tc;
  tc.name = name;
  tc.kind = kind;
  tc.cpu_hook = cpu;
  tc.gpu_hook = gpu;
  tc.id = id;
  return tc;
}

TaskClass& tc_spd_fill() {
  static TaskClass tc = make_tc("spd_fill", TaskKind::GPU, cpu_spd_fill, gpu_spd_fill, 1);
  return tc;
}
TaskClass& tc_potrf() {
  static TaskClass tc = make_tc("potrf", TaskKind::GPU, cpu_potrf, gpu_potrf, 2);
  return tc;
}
TaskClass& tc_trsm() {
  static TaskClass tc = make_tc("trsm", TaskKind::GPU, cpu_trsm, gpu_trsm, 3);
  return tc;
}
TaskClass& tc_trtri() {
  static TaskClass tc = make_tc("trtri", TaskKind::GPU, cpu_trtri, gpu_trtri, 6);
  return tc;
}
TaskClass& tc_trsm_inv() {
  static TaskClass tc =
      make_tc("trsm_inv", TaskKind::GPU, cpu_trsm_inv, gpu_trsm_inv, 7);
  return tc;
}
TaskClass& tc_syrk() {
  static TaskClass tc = make_tc("syrk", TaskKind::GPU, cpu_syrk, gpu_syrk, 4);
  return tc;
}
TaskClass& tc_gemm() {
  static TaskClass tc = make_tc("gemm", TaskKind::GPU, cpu_gemm, gpu_gemm, 5);
  return tc;
}

// ---------------------------------------------------------- copy / apply
// Redistribution (data_dist/matrix/redistribute analog, same tile grid,
// any rank grids) and the generic elementwise operator (map_operator.c /
// apply.jdf analog; the scale op x = alpha*x + beta is the built-in).
__global__ void k_scale_tile(double* p, size_t n, double alpha, double beta) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n; i += (size_t)gridDim.x * blockDim.x)
    p[i] = p[i] * alpha + beta;
}

static void cpu_copy_tile(Task& t) {
  Data* src = t.flows[0].data;
  Data* dst = t.flows[1].data;
  memcpy(dst->ensure_host(), src->pull_to_host(), src->bytes);
  dst->written_on(false);
}

static void gpu_copy_tile(Task& t, GpuTaskCtx& g) {
  Data* src = t.flows[0].data;
  PA_HIP_CHECK(hipMemcpyAsync(t.dev_ptr[1], t.dev_ptr[0], src->bytes,
                              hipMemcpyDeviceToDevice, g.stream));
}

static void cpu_scale(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  double alpha, beta;
  memcpy(&alpha, &a.i0, 8);
  memcpy(&beta, &a.j0, 8);
  Data* d = t.flows[0].data;
  double* p = (double*)d->pull_to_host();
  for (size_t i = 0; i < d->bytes / 8; i++) p[i] = p[i] * alpha + beta;
  d->written_on(false);
}

static void gpu_scale(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  double alpha, beta;
  memcpy(&alpha, &a.i0, 8);
  memcpy(&beta, &a.j0, 8);
  Data* d = t.flows[0].data;
  hipLaunchKernelGGL(k_scale_tile, dim3(2048), dim3(256), 0, g.stream,
                     (double*)t.dev_ptr[0], d->bytes / 8, alpha, beta);
}

TaskClass& tc_copy_tile() {
  static TaskClass tc = make_tc("copy_tile", TaskKind::GPU, cpu_copy_tile,
                                gpu_copy_tile, 30);
  return tc;
}
TaskClass& tc_scale() {
  static TaskClass tc = make_tc("scale", TaskKind::GPU, cpu_scale, gpu_scale, 31);
  return tc;
}

__global__ void k_add_tile(double* dst, const double* src, size_t n) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n; i += (size_t)gridDim.x * blockDim.x) dst[i] += src[i];
}

static void cpu_add_tile(Task& t) {
  Data* a = t.flows[0].data;
  Data* c = t.flows[1].data;
  const double* p = (const double*)a->pull_to_host();
  double* q = (double*)c->pull_to_host();
  for (size_t i = 0; i < c->bytes / 8; i++) q[i] += p[i];
  c->written_on(false);
}

static void gpu_add_tile(Task& t, GpuTaskCtx& g) {
  Data* c = t.flows[1].data;
  hipLaunchKernelGGL(k_add_tile, dim3(2048), dim3(256), 0, g.stream,
                     (double*)t.dev_ptr[1], (const double*)t.dev_ptr[0],
                     c->bytes / 8);
}

TaskClass& tc_add_tile() {
  static TaskClass tc = make_tc("add_tile", TaskKind::GPU, cpu_add_tile,
                                gpu_add_tile, 32);
  return tc;
}

// out = a + b (pure OUTPUT third flow — the tree-reduction combiner)
__global__ void k_add2(double* out, const double* a, const double* b,
                       size_t n) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n; i += (size_t)gridDim.x * blockDim.x) out[i] = a[i] + b[i];
}

static void cpu_add2(Task& t) {
  const double* a = (const double*)t.flows[0].data->pull_to_host();
  const double* b = (const double*)t.flows[1].data->pull_to_host();
  Data* o = t.flows[2].data;
  double* q = (double*)o->ensure_host();
  for (size_t i = 0; i < o->bytes / 8; i++) q[i] = a[i] + b[i];
  o->written_on(false);
}

static void gpu_add2(Task& t, GpuTaskCtx& g) {
  Data* o = t.flows[2].data;
  hipLaunchKernelGGL(k_add2, dim3(2048), dim3(256), 0, g.stream,
                     (double*)t.dev_ptr[2], (const double*)t.dev_ptr[0],
                     (const double*)t.dev_ptr[1], o->bytes / 8);
}

TaskClass& tc_add2() {
  static TaskClass tc = make_tc("add2", TaskKind::GPU, cpu_add2, gpu_add2, 33);
  return tc;
}

// result += sum of all tiles of A (reduce.jdf / DTD reduce analog).
// The accumulation chains on `result`; contributions from each rank's
// tiles flow through the comm engine automatically.
// Binary-tree reduction (BT_reduction.jdf analog): log2(N) critical path
// instead of the flat chain's N. Internal workspace tiles are owned by
// the taskpool (the bcgs-style pattern); pair placement follows the left
// operand so cross-rank combines ride the normal dataflow.
void insert_reduce_sum_tree(Dtd& tp, TiledMatrix& A, TiledMatrix& R) {
  std::vector<Data*> level;
  std::vector<int> lrank;
  for (int m = 0; m < A.mt(); m++)
    for (int n = 0; n < (A.sym() ? m + 1 : A.nt()); n++) {
      level.push_back(A.tile(m, n));
      lrank.push_back(A.rank_of(m, n));
    }
  PA_CHECK(!level.empty(), "reduce_sum_tree: empty collection");
  auto* ctx = A.ctx();
  // one workspace strip holds every internal node (N-1 combines)
  int nw = (int)level.size();  // >= combines + final copy target
  auto W = std::make_shared<TiledMatrix>(ctx, (int64_t)A.mb() * nw, A.nb(),
                                         A.mb(), A.nb(), 1, 1);
  {
    std::vector<int> ranks((size_t)nw);
    for (int i = 0; i < nw; i++) ranks[i] = lrank[i % lrank.size()];
    W->set_rank_table(std::move(ranks));
  }
  tp.own(W);
  int wi = 0;
  while (level.size() > 1) {
    std::vector<Data*> next;
    std::vector<int> nrank;
    for (size_t i = 0; i + 1 < level.size(); i += 2) {
      bool last = level.size() == 2;
      Data* out = last ? R.tile(0, 0) : W->tile(wi, 0);
      int orank = last ? R.rank_of(0, 0) : lrank[i];
      if (!last) wi++;
      Dtd::FlowSpec f[] = {{level[i], ACCESS_IN},
                           {level[i + 1], ACCESS_IN},
                           {out, ACCESS_OUT}};
      tp.insert(&tc_add2(), nullptr, 0, f, 3, 0, orank);
      next.push_back(out);
      nrank.push_back(orank);
    }
    if (level.size() % 2) {  // odd leftover promotes unchanged
      next.push_back(level.back());
      nrank.push_back(lrank.back());
    }
    level.swap(next);
    lrank.swap(nrank);
  }
  if (level[0] != R.tile(0, 0)) {
    // single-tile collection: copy the lone input into R
    Dtd::FlowSpec f[] = {{level[0], ACCESS_IN}, {R.tile(0, 0), ACCESS_OUT}};
    tp.insert(&tc_copy_tile(), nullptr, 0, f, 2, 0, R.rank_of(0, 0));
  }
}

void insert_reduce_sum(Dtd& tp, TiledMatrix& A, TiledMatrix& R) {
  Data* r = R.tile(0, 0);
  for (int m = 0; m < A.mt(); m++)
    for (int n = 0; n < (A.sym() ? m + 1 : A.nt()); n++) {
      Dtd::FlowSpec f[] = {{A.tile(m, n), ACCESS_IN}, {r, ACCESS_INOUT}};
      tp.insert(&tc_add_tile(), nullptr, 0, f, 2, 0, R.rank_of(0, 0));
    }
}

// Reduce along one axis (reduce_col/reduce_row.jdf analogs): R's single
// tile row (axis=0: R is 1 x nt tiles, each = sum over the tile column)
// or tile column (axis=1) accumulates elementwise tile sums: the first
// addend copies (defining R's content), the rest add.
void insert_reduce_axis(Dtd& tp, TiledMatrix& A, TiledMatrix& R, int axis) {
  PA_CHECK(axis == 0 || axis == 1, "axis must be 0 (columns) or 1 (rows)");
  PA_CHECK(!A.sym(), "reduce_axis: dense collections only");
  const int outer = axis == 0 ? A.nt() : A.mt();
  const int inner = axis == 0 ? A.mt() : A.nt();
  PA_CHECK((axis == 0 ? R.nt() : R.mt()) == outer &&
               (axis == 0 ? R.mt() : R.nt()) == 1 &&
               R.tile_bytes() == A.tile_bytes(),
           "reduce_axis: R must be a single tile row/column matching A");
  for (int o = 0; o < outer; o++) {
    Data* r = axis == 0 ? R.tile(0, o) : R.tile(o, 0);
    const int rrank = axis == 0 ? R.rank_of(0, o) : R.rank_of(o, 0);
    for (int i = 0; i < inner; i++) {
      Data* a = axis == 0 ? A.tile(i, o) : A.tile(o, i);
      if (i == 0) {
        Dtd::FlowSpec f[] = {{a, ACCESS_IN}, {r, ACCESS_OUT}};
        tp.insert(&tc_copy_tile(), nullptr, 0, f, 2, 0, rrank);
      } else {
        Dtd::FlowSpec f[] = {{a, ACCESS_IN}, {r, ACCESS_INOUT}};
        tp.insert(&tc_add_tile(), nullptr, 0, f, 2, 0, rrank);
      }
    }
  }
}

// ---- general regridding (redistribute.jdf incl. non-matching tile grids,
// data_dist/matrix/redistribute/redistribute.jdf analog) ----
// One CPU piece-task per (dst tile, overlapping src tile): copies the
// intersection rectangle. Pieces of one dst tile serialize through INOUT
// chaining; cross-rank movement falls out of the normal protocol. A
// leading zero-task defines regions no src tile covers.
namespace {
struct RegridArgs {
  int r, c;        // rectangle extent (rows, cols)
  int si, sj;      // offset in src tile
  int di, dj;      // offset in dst tile
  int lds, ldd;    // column strides (elements)
  int elem;        // element size in bytes
};

void cpu_regrid_zero(Task& t) {
  Data* d = t.flows[0].data;
  memset(d->ensure_host(), 0, d->bytes);
  d->written_on(false);
}

void cpu_regrid_piece(Task& t) {
  const RegridArgs& a = t.arg<RegridArgs>();
  const char* s = (const char*)t.flows[0].data->pull_to_host();
  Data* dd = t.flows[1].data;
  char* d;
  if (t.flows[1].mode & ACCESS_IN) {
    d = (char*)dd->pull_to_host();
  } else {
    // OUT-only piece fully covers its destination (subtile extract)
    dd->begin_host_overwrite();
    d = (char*)dd->ensure_host();
  }
  const size_t rb = (size_t)a.r * a.elem;
  for (int j = 0; j < a.c; j++)
    memcpy(d + ((size_t)(a.dj + j) * a.ldd + a.di) * a.elem,
           s + ((size_t)(a.sj + j) * a.lds + a.si) * a.elem, rb);
  t.flows[1].data->written_on(false);
}

TaskClass& tc_regrid_zero() {
  static TaskClass tc = make_tc("regrid_zero", TaskKind::CPU, cpu_regrid_zero,
                                nullptr, 36);
  return tc;
}
TaskClass& tc_regrid_piece() {
  static TaskClass tc = make_tc("regrid_piece", TaskKind::CPU,
                                cpu_regrid_piece, nullptr, 37);
  return tc;
}
}  // namespace

void insert_redistribute(Dtd& tp, TiledMatrix& Src, TiledMatrix& Dst) {
  if (Src.mt() == Dst.mt() && Src.nt() == Dst.nt() &&
      Src.tile_bytes() == Dst.tile_bytes()) {
    // fast path: same tile grid — one (GPU-capable) whole-tile copy each
    for (int m = 0; m < Src.mt(); m++)
      for (int n = 0; n < (Src.sym() ? m + 1 : Src.nt()); n++) {
        Dtd::FlowSpec f[] = {{Src.tile(m, n), ACCESS_IN},
                             {Dst.tile(m, n), ACCESS_OUT}};
        tp.insert(&tc_copy_tile(), nullptr, 0, f, 2, 0, Dst.rank_of(m, n));
      }
    return;
  }
  PA_CHECK(Src.m() == Dst.m() && Src.n() == Dst.n() &&
           Src.elem_size() == Dst.elem_size() && !Src.sym() && !Dst.sym(),
           "redistribute: matrices must have equal global shape/dtype "
           "(sym storage regridding unsupported)");
  for (int dm = 0; dm < Dst.mt(); dm++)
    for (int dn = 0; dn < Dst.nt(); dn++) {
      const int rank = Dst.rank_of(dm, dn);
      {
        Dtd::FlowSpec f[] = {{Dst.tile(dm, dn), ACCESS_OUT}};
        tp.insert(&tc_regrid_zero(), nullptr, 0, f, 1, 0, rank);
      }
      const int64_t r0 = (int64_t)dm * Dst.mb(), r1 = r0 + Dst.tile_rows(dm);
      const int64_t c0 = (int64_t)dn * Dst.nb(), c1 = c0 + Dst.tile_cols(dn);
      for (int sm = (int)(r0 / Src.mb()); (int64_t)sm * Src.mb() < r1; sm++)
        for (int sn = (int)(c0 / Src.nb()); (int64_t)sn * Src.nb() < c1;
             sn++) {
          const int64_t sr0 = (int64_t)sm * Src.mb();
          const int64_t sc0 = (int64_t)sn * Src.nb();
          const int64_t ir0 = std::max(r0, sr0);
          const int64_t ir1 = std::min(r1, sr0 + Src.tile_rows(sm));
          const int64_t ic0 = std::max(c0, sc0);
          const int64_t ic1 = std::min(c1, sc0 + Src.tile_cols(sn));
          if (ir1 <= ir0 || ic1 <= ic0) continue;
          RegridArgs a;
          a.r = (int)(ir1 - ir0);
          a.c = (int)(ic1 - ic0);
          a.si = (int)(ir0 - sr0);
          a.sj = (int)(ic0 - sc0);
          a.di = (int)(ir0 - r0);
          a.dj = (int)(ic0 - c0);
          a.lds = Src.mb();
          a.ldd = Dst.mb();
          a.elem = (int)Src.elem_size();
          Dtd::FlowSpec f[] = {{Src.tile(sm, sn), ACCESS_IN},
                               {Dst.tile(dm, dn), ACCESS_INOUT}};
          tp.insert(&tc_regrid_piece(), &a, sizeof(a), f, 2, 0, rank);
        }
    }
}

// ---- band -> rectangular conversion (diag_band_to_rect.jdf analog) ----
// Copies a band-stored collection into a dense one; out-of-band tiles of
// the destination are zero-filled (they have no source).
void insert_band_to_rect(Dtd& tp, TiledMatrix& S, TiledMatrix& D) {
  PA_CHECK(S.mt() == D.mt() && S.nt() == D.nt() &&
           S.tile_bytes() == D.tile_bytes() && !D.sym(),
           "band_to_rect: matching tile grids required");
  for (int m = 0; m < S.mt(); m++)
    for (int n = 0; n < S.nt(); n++) {
      const int rank = D.rank_of(m, n);
      if (S.in_band(m, n)) {
        Dtd::FlowSpec f[] = {{S.tile(m, n), ACCESS_IN},
                             {D.tile(m, n), ACCESS_OUT}};
        tp.insert(&tc_copy_tile(), nullptr, 0, f, 2, 0, rank);
      } else {
        Dtd::FlowSpec f[] = {{D.tile(m, n), ACCESS_OUT}};
        tp.insert(&tc_regrid_zero(), nullptr, 0, f, 1, 0, rank);
      }
    }
}

// ---- recursive subtiling (subtile.c analog) ----
// View ONE tile of A as its own tiled collection S (same global shape as
// the tile, finer tiles), by copy: extract pieces tile->S, run any DAG on
// S (e.g. a recursive factorization of a diagonal block), insert back.
// The copies ride the normal dataflow, so extraction chains after the
// tile's last writer and write-back chains before its next reader —
// including across ranks.
void insert_subtile_extract(Dtd& tp, TiledMatrix& A, int tm, int tn,
                            TiledMatrix& S) {
  PA_CHECK(S.m() == A.tile_rows(tm) && S.n() == A.tile_cols(tn) &&
           S.elem_size() == A.elem_size() && !S.sym(),
           "subtile: S must have the tile's global shape");
  Data* src = A.tile(tm, tn);
  const int rank = A.rank_of(tm, tn);
  for (int i = 0; i < S.mt(); i++)
    for (int j = 0; j < S.nt(); j++) {
      RegridArgs a;
      a.r = S.tile_rows(i);
      a.c = S.tile_cols(j);
      a.si = i * S.mb();
      a.sj = j * S.nb();
      a.di = 0;
      a.dj = 0;
      a.lds = A.mb();
      a.ldd = S.mb();
      a.elem = (int)A.elem_size();
      Dtd::FlowSpec f[] = {{src, ACCESS_IN}, {S.tile(i, j), ACCESS_OUT}};
      tp.insert(&tc_regrid_piece(), &a, sizeof(a), f, 2, 0, rank);
    }
}

void insert_subtile_insert(Dtd& tp, TiledMatrix& S, TiledMatrix& A, int tm,
                           int tn) {
  PA_CHECK(S.m() == A.tile_rows(tm) && S.n() == A.tile_cols(tn) &&
           S.elem_size() == A.elem_size() && !S.sym(),
           "subtile: S must have the tile's global shape");
  Data* dst = A.tile(tm, tn);
  const int rank = A.rank_of(tm, tn);
  for (int i = 0; i < S.mt(); i++)
    for (int j = 0; j < S.nt(); j++) {
      RegridArgs a;
      a.r = S.tile_rows(i);
      a.c = S.tile_cols(j);
      a.si = 0;
      a.sj = 0;
      a.di = i * S.mb();
      a.dj = j * S.nb();
      a.lds = S.mb();
      a.ldd = A.mb();
      a.elem = (int)A.elem_size();
      Dtd::FlowSpec f[] = {{S.tile(i, j), ACCESS_IN}, {dst, ACCESS_INOUT}};
      tp.insert(&tc_regrid_piece(), &a, sizeof(a), f, 2, 0, rank);
    }
}

void insert_apply_scale(Dtd& tp, TiledMatrix& A, double alpha, double beta) {
  for (int m = 0; m < A.mt(); m++)
    for (int n = 0; n < (A.sym() ? m + 1 : A.nt()); n++) {
      TileArgs a;
      memcpy(&a.i0, &alpha, 8);
      memcpy(&a.j0, &beta, 8);
      Dtd::FlowSpec f[] = {{A.tile(m, n), ACCESS_INOUT}};
      tp.insert(&tc_scale(), &a, sizeof(a), f, 1, 0, A.rank_of(m, n));
    }
}

// ---------------------------------------------------------------- stencil
// 1-D 3-point stencil over a vector of tiles (tests/apps/stencil analog):
// dst[i] = (src[i-1] + src[i] + src[i+1]) / 3 with halos crossing tile
// boundaries through neighbor-tile flows (zero at domain edges).
__global__ void k_stencil3(double* dst, const double* left,
                           const double* mid, const double* right, int nbe) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < nbe; i += gridDim.x * blockDim.x) {
    double l = i > 0 ? mid[i - 1] : (left ? left[nbe - 1] : 0.0);
    double r = i < nbe - 1 ? mid[i + 1] : (right ? right[0] : 0.0);
    dst[i] = (l + mid[i] + r) / 3.0;
  }
}

static void cpu_stencil3(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const int nbe = a.m;
  const double* left =
      t.flows[1].data ? (const double*)t.flows[1].data->pull_to_host() : nullptr;
  const double* mid = (const double*)t.flows[0].data->pull_to_host();
  const double* right =
      t.flows[2].data ? (const double*)t.flows[2].data->pull_to_host() : nullptr;
  double* dst = (double*)t.flows[3].data->ensure_host();
  for (int i = 0; i < nbe; i++) {
    double l = i > 0 ? mid[i - 1] : (left ? left[nbe - 1] : 0.0);
    double r = i < nbe - 1 ? mid[i + 1] : (right ? right[0] : 0.0);
    dst[i] = (l + mid[i] + r) / 3.0;
  }
  t.flows[3].data->written_on(false);
}

static void gpu_stencil3(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  hipLaunchKernelGGL(k_stencil3, dim3(64), dim3(256), 0, g.stream,
                     (double*)t.dev_ptr[3], (const double*)t.dev_ptr[1],
                     (const double*)t.dev_ptr[0], (const double*)t.dev_ptr[2],
                     a.m);
}

TaskClass& tc_stencil3() {
  static TaskClass tc = make_tc("stencil3", TaskKind::GPU, cpu_stencil3,
                                gpu_stencil3, 33);
  return tc;
}

// One stencil sweep Src -> Dst (tile vectors: mt x 1 tiles of mb x 1).
void insert_stencil_1d(Dtd& tp, TiledMatrix& Src, TiledMatrix& Dst) {
  PA_CHECK(Src.nt() == 1 && Dst.nt() == 1 && Src.mt() == Dst.mt());
  const int T = Src.mt();
  for (int t = 0; t < T; t++) {
    TileArgs a;
    a.m = Src.tile_rows(t) * Src.tile_cols(0);
    Dtd::FlowSpec f[4];
    f[0] = {Src.tile(t, 0), ACCESS_IN};
    f[1] = {t > 0 ? Src.tile(t - 1, 0) : nullptr, ACCESS_IN};
    f[2] = {t < T - 1 ? Src.tile(t + 1, 0) : nullptr, ACCESS_IN};
    f[3] = {Dst.tile(t, 0), ACCESS_OUT};
    tp.insert(&tc_stencil3(), &a, sizeof(a), f, 4, 0, Dst.rank_of(t, 0));
  }
}

// ------------------------------------------------------------------ DAG builders
void insert_spd_fill(Dtd& tp, TiledMatrix& A, uint32_t seed) {
  for (int tm = 0; tm < A.mt(); tm++)
    for (int tn = 0; tn <= tm && tn < A.nt(); tn++) {
      TileArgs a;
      a.m = A.tile_rows(tm);
      a.n = A.tile_cols(tn);
      a.ld = A.mb();
      a.i0 = (int64_t)tm * A.mb();
      a.j0 = (int64_t)tn * A.nb();
      a.N = A.m();
      a.seed = seed;
      Dtd::FlowSpec f[] = {{A.tile(tm, tn), ACCESS_OUT}};
      tp.insert(&tc_spd_fill(), &a, sizeof(a), f, 1, 0, A.rank_of(tm, tn));
    }
}

// Full-matrix fill (QR and GEMM inputs read above the diagonal too).
void insert_full_fill(Dtd& tp, TiledMatrix& A, uint32_t seed) {
  for (int tm = 0; tm < A.mt(); tm++)
    for (int tn = 0; tn < A.nt(); tn++) {
      TileArgs a;
      a.m = A.tile_rows(tm);
      a.n = A.tile_cols(tn);
      a.ld = A.mb();
      a.i0 = (int64_t)tm * A.mb();
      a.j0 = (int64_t)tn * A.nb();
      a.N = A.m();
      a.seed = seed;
      Dtd::FlowSpec f[] = {{A.tile(tm, tn), ACCESS_OUT}};
      tp.insert(&tc_spd_fill(), &a, sizeof(a), f, 1, 0, A.rank_of(tm, tn));
    }
}

// Right-looking tiled Cholesky, lower triangular (the reference's headline
// dpotrf DAG shape; DPLASMA dpotrf_L equivalent).
void insert_potrf(Dtd& tp, TiledMatrix& A) {
  const int T = A.mt();
  const int ld = A.mb();
  // Critical-path priorities (DPLASMA dpotrf style): panel ops (POTRF/TRSM)
  // always outrank trailing updates, and an update targeting column n at
  // step k is more urgent the sooner column n becomes the panel (n-k small).
  // This is what creates lookahead: panel k+1 preempts the bulk of step-k
  // GEMMs in the GPU engine's priority queue.
  constexpr int PANEL = 1 << 20;
  // TRSM variant: "invgemm" (default) computes W_k = L(k,k)^{-1} once per
  // step (TRTRI task on the panel stream) and does each TRSM as a single
  // full-rate DGEMM; "rocblas" calls dtrsm directly. Must be uniform across
  // ranks (the DAG shape depends on it).
  const bool invgemm = param_str("trsm_variant", "invgemm") == "invgemm";
  std::shared_ptr<TiledMatrix> W;
  if (invgemm) {
    W = std::make_shared<TiledMatrix>(A.ctx(), A.m(), A.n(), A.mb(), A.nb(),
                                      A.grid_p(), A.grid_q());
    tp.own(W);
  }
  for (int k = 0; k < T; k++) {
    TileArgs pa_args;
    pa_args.n = A.tile_cols(k);
    pa_args.ld = ld;
    {
      Dtd::FlowSpec f[] = {{A.tile(k, k), ACCESS_INOUT}};
      tp.insert(&tc_potrf(), &pa_args, sizeof(pa_args), f, 1, PANEL + 1,
                A.rank_of(k, k));
    }
    if (invgemm && k + 1 < T) {
      Dtd::FlowSpec f[] = {{A.tile(k, k), ACCESS_IN},
                           {W->tile(k, k), ACCESS_OUT}};
      tp.insert(&tc_trtri(), &pa_args, sizeof(pa_args), f, 2, PANEL,
                A.rank_of(k, k));
    }
    for (int m = k + 1; m < T; m++) {
      TileArgs a;
      a.m = A.tile_rows(m);
      a.n = A.tile_cols(k);
      a.ld = ld;
      // Below the express-stream threshold (1<<19): TRSMs are urgent in the
      // queue but must spread across bulk streams, not serialize on the
      // panel stream.
      if (invgemm) {
        Dtd::FlowSpec f[] = {{W->tile(k, k), ACCESS_IN},
                             {A.tile(m, k), ACCESS_INOUT}};
        tp.insert(&tc_trsm_inv(), &a, sizeof(a), f, 2, (1 << 18) - (m - k),
                  A.rank_of(m, k));
      } else {
        Dtd::FlowSpec f[] = {{A.tile(k, k), ACCESS_IN},
                             {A.tile(m, k), ACCESS_INOUT}};
        tp.insert(&tc_trsm(), &a, sizeof(a), f, 2, (1 << 18) - (m - k),
                  A.rank_of(m, k));
      }
    }
    for (int n = k + 1; n < T; n++) {
      {
        TileArgs a;
        a.n = A.tile_rows(n);
        a.k = A.tile_cols(k);
        a.ld = ld;
        Dtd::FlowSpec f[] = {{A.tile(n, k), ACCESS_IN},
                             {A.tile(n, n), ACCESS_INOUT}};
        tp.insert(&tc_syrk(), &a, sizeof(a), f, 2, -(n - k) * 4 + 1,
                  A.rank_of(n, n));
      }
      for (int m = n + 1; m < T; m++) {
        TileArgs a;
        a.m = A.tile_rows(m);
        a.n = A.tile_rows(n);
        a.k = A.tile_cols(k);
        a.ld = ld;
        Dtd::FlowSpec f[] = {{A.tile(m, k), ACCESS_IN},
                             {A.tile(n, k), ACCESS_IN},
                             {A.tile(m, n), ACCESS_INOUT}};
        tp.insert(&tc_gemm(), &a, sizeof(a), f, 3, -(n - k) * 4,
                  A.rank_of(m, n));
      }
    }
  }
}

// ------------------------------------------------------- plain NN GEMM DAG
// Tiled C = A * B (fp64, rocBLAS NN per tile). The k==0 task overwrites C
// (beta = 0, OUTPUT flow), later k accumulate — which makes the whole DAG
// IDEMPOTENT: replaying it (e.g. from a captured hipGraph, gpu_graph.hpp)
// reproduces the same C from the same A/B.
struct GemmNNArgs {
  int m, n, k, lda, ldb, ldc;
  double alpha, beta;
  int transA = 0;  // 1: C = alpha*A^T*B + beta*C
};

static void cpu_gemm_nn(Task& t) {
  const GemmNNArgs& a = t.arg<GemmNNArgs>();
  const double* A = (const double*)t.flows[0].data->pull_to_host();
  const double* B = (const double*)t.flows[1].data->pull_to_host();
  // beta == 0 rides an OUTPUT-only flow: nothing valid to pull yet
  double* C = a.beta == 0.0 ? (double*)t.flows[2].data->ensure_host()
                            : (double*)t.flows[2].data->pull_to_host();
  for (int j = 0; j < a.n; j++)
    for (int i = 0; i < a.m; i++) {
      double s = 0;
      for (int p = 0; p < a.k; p++)
        s += (a.transA ? A[(size_t)i * a.lda + p] : A[(size_t)p * a.lda + i]) *
             B[(size_t)j * a.ldb + p];
      double c0 = a.beta == 0.0 ? 0.0 : a.beta * C[(size_t)j * a.ldc + i];
      C[(size_t)j * a.ldc + i] = c0 + a.alpha * s;
    }
  t.flows[2].data->written_on(false);
}

// Thin-C tiles (vector iterations, skinny updates): a library GEMM is all
// launch overhead at n<=4, and a plain kernel is trivially capture-safe.
__global__ void k_gemm_nn_thin(int m, int n, int k, double alpha,
                               const double* A, int lda, const double* B,
                               int ldb, double beta, double* C, int ldc) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  int j = blockIdx.y;
  if (i >= m || j >= n) return;
  double s = 0;
  for (int p = 0; p < k; p++)
    s += A[(size_t)p * lda + i] * B[(size_t)j * ldb + p];
  double c0 = beta == 0.0 ? 0.0 : beta * C[(size_t)j * ldc + i];
  C[(size_t)j * ldc + i] = c0 + alpha * s;
}

static void gpu_gemm_nn(Task& t, GpuTaskCtx& g) {
  const GemmNNArgs& a = t.arg<GemmNNArgs>();
  if (a.n <= 4 && !a.transA) {
    hipLaunchKernelGGL(k_gemm_nn_thin, dim3((a.m + 63) / 64, a.n), dim3(64),
                       0, g.stream, a.m, a.n, a.k, a.alpha,
                       (const double*)t.dev_ptr[0], a.lda,
                       (const double*)t.dev_ptr[1], a.ldb, a.beta,
                       (double*)t.dev_ptr[2], a.ldc);
    return;
  }
  rocblas_status s = rocblas_dgemm(
      blas_handle(g),
      a.transA ? rocblas_operation_transpose : rocblas_operation_none,
      rocblas_operation_none, a.m, a.n, a.k, &a.alpha,
      (const double*)t.dev_ptr[0], a.lda, (const double*)t.dev_ptr[1],
      a.ldb, &a.beta, (double*)t.dev_ptr[2], a.ldc);
  PA_CHECK(s == rocblas_status_success, "rocblas_dgemm NN failed: %d", (int)s);
}

TaskClass& tc_gemm_nn() {
  static TaskClass tc =
      make_tc("gemm_nn", TaskKind::GPU, cpu_gemm_nn, gpu_gemm_nn, 70);
  return tc;
}

void insert_gemm_fp64(Dtd& tp, TiledMatrix& A, TiledMatrix& B,
                      TiledMatrix& C) {
  PA_CHECK(A.nt() == B.mt() && A.mt() == C.mt() && B.nt() == C.nt(),
           "insert_gemm_fp64: tile-grid mismatch");
  for (int i = 0; i < C.mt(); i++)
    for (int j = 0; j < C.nt(); j++)
      for (int p = 0; p < A.nt(); p++) {
        // tiles are stored at the collection's full ld (= mb), partial
        // edge tiles included (see tile_numpy / TiledMatrix layout)
        GemmNNArgs a{C.tile_rows(i), C.tile_cols(j), A.tile_cols(p),
                     A.mb(),         B.mb(),         C.mb(),
                     1.0,            p == 0 ? 0.0 : 1.0};
        Dtd::FlowSpec f[] = {{A.tile(i, p), ACCESS_IN},
                             {B.tile(p, j), ACCESS_IN},
                             {C.tile(i, j), p == 0 ? ACCESS_OUT
                                                   : ACCESS_INOUT}};
        tp.insert(&tc_gemm_nn(), &a, sizeof(a), f, 3, 0, C.rank_of(i, j));
      }
}

// --------------------------------------------------------------- advise
// parsec_advise_data_on_device analog (device.h data_advise vtable row):
// a no-op GPU task with one READ flow — the engine's normal stage-in
// pulls the tile onto the device ahead of its first real consumer, on
// the h2d stream, overlapped with whatever is executing. On CPU-only
// contexts it degenerates to a no-op CPU task.
static void cpu_advise(Task&) {}
static void gpu_advise(Task&, GpuTaskCtx&) {}
TaskClass& tc_advise() {
  static TaskClass tc =
      make_tc("advise_prefetch", TaskKind::GPU, cpu_advise, gpu_advise, 71);
  return tc;
}

void insert_advise_prefetch(Dtd& tp, Data* d) {
  Dtd::FlowSpec f[] = {{d, ACCESS_IN}};
  TileArgs a{};
  tp.insert(&tc_advise(), &a, sizeof(a), f, 1, 0, -1);
}

// ----------------------------------------------------------- POTRS / POSV
// Cholesky SOLVE (dplasma dpotrs/dposv analog): after insert_potrf left
// A's lower tiles holding L, solve A X = B for a block of right-hand
// sides: forward L Y = B (tile forward substitution), then backward
// L^T X = Y. Diagonal solves are rocblas_dtrsm (left, lower, N/T);
// off-diagonal updates reuse the gemm_nn chore (alpha=-1, beta=1, with
// transA for the backward sweep).
struct TrsmSolveArgs {
  int m, n, lda, ldb;
  int trans;  // 0: solve T Z = B; 1: solve T^T Z = B
  int upper = 0;  // triangle of the diagonal tile holding the factor
  int unit = 0;   // unit diagonal (LU's L)
};

static void cpu_trsm_solve(Task& t) {
  const TrsmSolveArgs& a = t.arg<TrsmSolveArgs>();
  const double* L = (const double*)t.flows[0].data->pull_to_host();
  double* B = (double*)t.flows[1].data->pull_to_host();
  // forward order for {lower, N} and {upper, T}; backward otherwise
  const bool fwd = (a.upper == 0) != (a.trans != 0);
  auto elem = [&](int i, int p) {
    return a.trans ? L[(size_t)i * a.lda + p] : L[(size_t)p * a.lda + i];
  };
  for (int j = 0; j < a.n; j++) {
    double* b = B + (size_t)j * a.ldb;
    if (fwd) {
      for (int i = 0; i < a.m; i++) {
        double s = b[i];
        for (int p = 0; p < i; p++) s -= elem(i, p) * b[p];
        b[i] = a.unit ? s : s / L[(size_t)i * a.lda + i];
      }
    } else {
      for (int i = a.m - 1; i >= 0; i--) {
        double s = b[i];
        for (int p = i + 1; p < a.m; p++) s -= elem(i, p) * b[p];
        b[i] = a.unit ? s : s / L[(size_t)i * a.lda + i];
      }
    }
  }
  t.flows[1].data->written_on(false);
}

static void gpu_trsm_solve(Task& t, GpuTaskCtx& g) {
  const TrsmSolveArgs& a = t.arg<TrsmSolveArgs>();
  const double one = 1.0;
  rocblas_status s = rocblas_dtrsm(
      blas_handle(g), rocblas_side_left,
      a.upper ? rocblas_fill_upper : rocblas_fill_lower,
      a.trans ? rocblas_operation_transpose : rocblas_operation_none,
      a.unit ? rocblas_diagonal_unit : rocblas_diagonal_non_unit, a.m, a.n,
      &one, (const double*)t.dev_ptr[0], a.lda, (double*)t.dev_ptr[1],
      a.ldb);
  PA_CHECK(s == rocblas_status_success, "rocblas_dtrsm solve failed: %d",
           (int)s);
}

TaskClass& tc_trsm_solve() {
  static TaskClass tc = make_tc("trsm_solve", TaskKind::GPU, cpu_trsm_solve,
                                gpu_trsm_solve, 72);
  return tc;
}

void insert_potrs(Dtd& tp, TiledMatrix& A, TiledMatrix& B) {
  PA_CHECK(A.mt() == A.nt() && A.mt() == B.mt(),
           "insert_potrs: A must be square with B.mt == A.mt");
  const int mt = A.mt();
  auto trsm = [&](int k, int j, int trans) {
    TrsmSolveArgs a{A.tile_rows(k), B.tile_cols(j), A.mb(), B.mb(), trans};
    Dtd::FlowSpec f[] = {{A.tile(k, k), ACCESS_IN},
                         {B.tile(k, j), ACCESS_INOUT}};
    tp.insert(&tc_trsm_solve(), &a, sizeof(a), f, 2, (1 << 20),
              B.rank_of(k, j));
  };
  auto update = [&](int i, int k, int j, int trans) {
    // forward: B[i] -= L[i,k]   * B[k]   (i > k)
    // backward: B[i] -= L[k,i]^T * B[k]  (i < k; lower tile (k,i))
    GemmNNArgs a{B.tile_rows(i), B.tile_cols(j), A.tile_rows(k),
                 A.mb(),         B.mb(),         B.mb(),
                 -1.0,           1.0};
    a.transA = trans;
    Data* l = trans ? A.tile(k, i) : A.tile(i, k);
    Dtd::FlowSpec f[] = {{l, ACCESS_IN},
                         {B.tile(k, j), ACCESS_IN},
                         {B.tile(i, j), ACCESS_INOUT}};
    tp.insert(&tc_gemm_nn(), &a, sizeof(a), f, 3, 0, B.rank_of(i, j));
  };
  for (int j = 0; j < B.nt(); j++) {
    for (int k = 0; k < mt; k++) {  // forward: L Y = B
      trsm(k, j, 0);
      for (int i = k + 1; i < mt; i++) update(i, k, j, 0);
    }
    for (int k = mt - 1; k >= 0; k--) {  // backward: L^T X = Y
      trsm(k, j, 1);
      for (int i = k - 1; i >= 0; i--) update(i, k, j, 1);
    }
  }
}

void insert_posv(Dtd& tp, TiledMatrix& A, TiledMatrix& B) {
  insert_potrf(tp, A);
  insert_potrs(tp, A, B);
}

// LU solve (dplasma dgetrs/dgesv nopiv analogs): after insert_getrf_nopiv
// left L (unit lower) and U (upper) packed in A's tiles, solve A X = B:
// forward L Y = B (unit diagonal), then backward U X = Y. Off-diagonal
// updates use A's tiles directly: forward uses the strictly-lower tiles
// (which hold L), backward the strictly-upper tiles (which hold U).
void insert_getrs_nopiv(Dtd& tp, TiledMatrix& A, TiledMatrix& B) {
  PA_CHECK(A.mt() == A.nt() && A.mt() == B.mt(),
           "insert_getrs_nopiv: A must be square with B.mt == A.mt");
  const int mt = A.mt();
  auto trsm = [&](int k, int j, int upper) {
    TrsmSolveArgs a{A.tile_rows(k), B.tile_cols(j), A.mb(), B.mb(), 0};
    a.upper = upper;
    a.unit = upper ? 0 : 1;  // L is unit-lower, U non-unit upper
    Dtd::FlowSpec f[] = {{A.tile(k, k), ACCESS_IN},
                         {B.tile(k, j), ACCESS_INOUT}};
    tp.insert(&tc_trsm_solve(), &a, sizeof(a), f, 2, (1 << 20),
              B.rank_of(k, j));
  };
  auto update = [&](int i, int k, int j) {
    // B[i] -= A[i,k] * B[k]  (forward: i > k uses L tiles;
    //                         backward: i < k uses U tiles)
    GemmNNArgs a{B.tile_rows(i), B.tile_cols(j), A.tile_rows(k),
                 A.mb(),         B.mb(),         B.mb(),
                 -1.0,           1.0};
    Dtd::FlowSpec f[] = {{A.tile(i, k), ACCESS_IN},
                         {B.tile(k, j), ACCESS_IN},
                         {B.tile(i, j), ACCESS_INOUT}};
    tp.insert(&tc_gemm_nn(), &a, sizeof(a), f, 3, 0, B.rank_of(i, j));
  };
  for (int j = 0; j < B.nt(); j++) {
    for (int k = 0; k < mt; k++) {  // forward: L Y = B
      trsm(k, j, 0);
      for (int i = k + 1; i < mt; i++) update(i, k, j);
    }
    for (int k = mt - 1; k >= 0; k--) {  // backward: U X = Y
      trsm(k, j, 1);
      for (int i = k - 1; i >= 0; i--) update(i, k, j);
    }
  }
}

void insert_gesv_nopiv(Dtd& tp, TiledMatrix& A, TiledMatrix& B) {
  insert_getrf_nopiv(tp, A);
  insert_getrs_nopiv(tp, A, B);
}

// Least squares min ||A X - B|| via the BCGS QR (dgels analog, QR route):
// insert_geqrf_bcgs leaves Q explicit in A (m x n) and R (n x n upper) —
// X = R^{-1} Q^T B. Q^T B is a tiled TN GEMM chain; the triangular solve
// is a backward sweep with the upper-triangular trsm_solve. X must be
// A.nt x B.nt tiles. Inherits BCGS's cond(A) <~ 1e7 envelope.
void insert_gels_bcgs(Dtd& tp, TiledMatrix& A, TiledMatrix& R,
                      TiledMatrix& B, TiledMatrix& X) {
  insert_geqrf_bcgs(tp, A, R);
  PA_CHECK(X.mt() == A.nt() && X.nt() == B.nt() && B.mt() == A.mt(),
           "insert_gels_bcgs: X must be A.nt x B.nt tiles, B.mt == A.mt");
  // X[k,j] = sum_i Q[i,k]^T B[i,j]
  for (int k = 0; k < A.nt(); k++)
    for (int j = 0; j < B.nt(); j++)
      for (int i = 0; i < A.mt(); i++) {
        GemmNNArgs a{X.tile_rows(k), B.tile_cols(j), A.tile_rows(i),
                     A.mb(),         B.mb(),         X.mb(),
                     1.0,            i == 0 ? 0.0 : 1.0};
        a.transA = 1;
        Dtd::FlowSpec f[] = {{A.tile(i, k), ACCESS_IN},
                             {B.tile(i, j), ACCESS_IN},
                             {X.tile(k, j), i == 0 ? ACCESS_OUT
                                                   : ACCESS_INOUT}};
        tp.insert(&tc_gemm_nn(), &a, sizeof(a), f, 3, 0, X.rank_of(k, j));
      }
  // backward: R X = (Q^T B), R upper non-unit
  for (int j = 0; j < X.nt(); j++)
    for (int k = X.mt() - 1; k >= 0; k--) {
      TrsmSolveArgs a{X.tile_rows(k), X.tile_cols(j), R.mb(), X.mb(), 0};
      a.upper = 1;
      Dtd::FlowSpec f[] = {{R.tile(k, k), ACCESS_IN},
                           {X.tile(k, j), ACCESS_INOUT}};
      tp.insert(&tc_trsm_solve(), &a, sizeof(a), f, 2, (1 << 20),
                X.rank_of(k, j));
      for (int i = k - 1; i >= 0; i--) {
        GemmNNArgs a2{X.tile_rows(i), X.tile_cols(j), X.tile_rows(k),
                      R.mb(),         X.mb(),         X.mb(),
                      -1.0,           1.0};
        Dtd::FlowSpec f2[] = {{R.tile(i, k), ACCESS_IN},
                              {X.tile(k, j), ACCESS_IN},
                              {X.tile(i, j), ACCESS_INOUT}};
        tp.insert(&tc_gemm_nn(), &a2, sizeof(a2), f2, 3, 0,
                  X.rank_of(i, j));
      }
    }
}

// Pre-create the per-stream rocBLAS handle and give it a fixed device
// workspace so no allocation can happen inside a hipStream capture
// (gpu_graph.cpp calls this for each capture stream before BeginCapture;
// same thread, so the thread_local handle map matches the replay).
void blas_warm_stream_for_capture(hipStream_t s) {
  GpuTaskCtx g{s, 0, nullptr, nullptr};
  rocblas_handle h = blas_handle(g);
  static thread_local std::map<void*, void*> ws;
  void*& w = ws[(void*)s];
  if (!w) {
    const size_t bytes = 1u << 26;
    PA_HIP_CHECK(hipMalloc(&w, bytes));
    PA_CHECK(rocblas_set_workspace(h, w, bytes) == rocblas_status_success);
  }
}

}  // namespace pa
