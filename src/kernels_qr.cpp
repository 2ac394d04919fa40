// Tile QR (dgeqrf) — DTD headline app #2 (BASELINE.json config 4).
//
// Flat-tree tile QR in the PLASMA style the reference's users run through
// DPLASMA: GEQRT (panel QR + T factor), UNMQR (apply panel Q^T across the
// row), TSQRT (triangular-on-top-of-square QR combining R_kk with A_mk),
// TSMQR (apply the TS reflectors to row pairs). GPU chores compose
// rocSOLVER dgeqrf/dlarft/dlarfb with D2D stacking copies on the task
// stream; CPU chores are self-contained Householder reference code used by
// the no-GPU tests.
//
// Storage note (deviation from PLASMA, documented): the TS reflector block
// V has a non-identity top block here (generic dgeqrf on the stacked 2nb x
// nb matrix), so V is stored in a dedicated workspace collection V2(m,k)
// of 2nb x nb tiles instead of overwriting A(m,k); R lands in the upper
// triangles of A as usual. The factorization is verified via R^T R = A^T A.
#include <cmath>
#include <cstring>
#include <map>
#include <vector>

#include <hip/hip_runtime.h>
#include <rocblas/rocblas.h>
#include <rocsolver/rocsolver.h>

#include "device_gpu.hpp"
#include "kernels.hpp"
#include "profiling.hpp"

namespace pa {

// ============================================================ CPU reference
namespace {

inline double vval(const double* V, int ld, int r, int q) {
  // column q of a unit-lower reflector block: 0 above diag, 1 on diag
  return r < q ? 0.0 : (r == q ? 1.0 : V[(size_t)q * ld + r]);
}

void cpu_geqrf_kern(int m, int n, double* A, int ld, double* tau) {
  int kmax = m < n ? m : n;
  for (int j = 0; j < kmax; j++) {
    double alpha = A[(size_t)j * ld + j];
    double xnorm2 = 0;
    for (int i = j + 1; i < m; i++) {
      double v = A[(size_t)j * ld + i];
      xnorm2 += v * v;
    }
    if (xnorm2 == 0) {
      tau[j] = 0;
      continue;
    }
    double beta = -copysign(sqrt(alpha * alpha + xnorm2), alpha);
    tau[j] = (beta - alpha) / beta;
    double scal = 1.0 / (alpha - beta);
    for (int i = j + 1; i < m; i++) A[(size_t)j * ld + i] *= scal;
    A[(size_t)j * ld + j] = beta;
    for (int c = j + 1; c < n; c++) {
      double w = A[(size_t)c * ld + j];
      for (int i = j + 1; i < m; i++)
        w += A[(size_t)j * ld + i] * A[(size_t)c * ld + i];
      w *= tau[j];
      A[(size_t)c * ld + j] -= w;
      for (int i = j + 1; i < m; i++)
        A[(size_t)c * ld + i] -= w * A[(size_t)j * ld + i];
    }
  }
}

// forward/columnwise T factor: H_0 H_1 ... H_{k-1} = I - V T V^T
void cpu_larft_kern(int n, int k, const double* V, int ld, const double* tau,
                    double* T, int ldt) {
  std::vector<double> w(k);
  for (int i = 0; i < k; i++) {
    T[(size_t)i * ldt + i] = tau[i];
    for (int c = 0; c < i; c++) {
      double s = 0;
      for (int r = i; r < n; r++) s += vval(V, ld, r, c) * vval(V, ld, r, i);
      w[c] = s;
    }
    for (int r = 0; r < i; r++) {
      double s = 0;
      for (int c = r; c < i; c++) s += T[(size_t)c * ldt + r] * w[c];
      T[(size_t)i * ldt + r] = -tau[i] * s;
    }
  }
}

// C = (I - V T V^T)^T C = C - V T^T (V^T C)   (side=left, trans=T)
void cpu_larfb_kern(int m, int n, int k, const double* V, int ldv,
                    const double* T, int ldt, double* C, int ldc) {
  std::vector<double> W((size_t)k * n), W2((size_t)k * n);
  for (int c = 0; c < n; c++)
    for (int q = 0; q < k; q++) {
      double s = 0;
      for (int r = q; r < m; r++)
        s += vval(V, ldv, r, q) * C[(size_t)c * ldc + r];
      W[(size_t)c * k + q] = s;
    }
  for (int c = 0; c < n; c++)
    for (int q = 0; q < k; q++) {
      double s = 0;
      for (int sdx = 0; sdx <= q; sdx++)
        s += T[(size_t)q * ldt + sdx] * W[(size_t)c * k + sdx];
      W2[(size_t)c * k + q] = s;
    }
  for (int c = 0; c < n; c++)
    for (int r = 0; r < m; r++) {
      double s = 0;
      int qmax = r < k - 1 ? r : k - 1;
      for (int q = 0; q <= qmax; q++)
        s += vval(V, ldv, r, q) * W2[(size_t)c * k + q];
      C[(size_t)c * ldc + r] -= s;
    }
}

void stack_tiles(double* S, const double* top, bool triu_top,
                 const double* bot, bool triu_bot, int nb, int ld) {
  for (int c = 0; c < nb; c++) {
    for (int r = 0; r < nb; r++) {
      double v = top[(size_t)c * ld + r];
      S[(size_t)c * 2 * nb + r] = (triu_top && r > c) ? 0.0 : v;
    }
    for (int r = 0; r < nb; r++) {
      double v = bot[(size_t)c * ld + r];
      // tree combines: the bottom tile holds its level-0 V below the
      // diagonal — only its R triangle participates
      S[(size_t)c * 2 * nb + nb + r] = (triu_bot && r > c) ? 0.0 : v;
    }
  }
}

}  // namespace

// ------------------------------------------------------------ CPU chores
static void cpu_geqrt(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  double* A = (double*)t.flows[0].data->pull_to_host();
  Data* td = t.flows[1].data;
  double* T = (double*)td->ensure_host();
  memset(T, 0, td->bytes);
  std::vector<double> tau(a.n);
  cpu_geqrf_kern(a.n, a.n, A, a.ld, tau.data());
  cpu_larft_kern(a.n, a.n, A, a.ld, tau.data(), T, a.ld);
  t.flows[0].data->written_on(false);
  td->written_on(false);
}

static void cpu_unmqr(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const double* V = (const double*)t.flows[0].data->pull_to_host();
  const double* T = (const double*)t.flows[1].data->pull_to_host();
  double* C = (double*)t.flows[2].data->pull_to_host();
  cpu_larfb_kern(a.m, a.n, a.m, V, a.ld, T, a.ld, C, a.ld);
  t.flows[2].data->written_on(false);
}

static void cpu_tsqrt(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const int nb = a.n, ld = a.ld;
  double* Akk = (double*)t.flows[0].data->pull_to_host();
  double* Amk = (double*)t.flows[1].data->pull_to_host();
  Data* vd = t.flows[2].data;
  Data* td = t.flows[3].data;
  double* V2 = (double*)vd->ensure_host();
  double* T1 = (double*)td->ensure_host();
  memset(T1, 0, td->bytes);
  stack_tiles(V2, Akk, true, Amk, a.k != 0, nb, ld);
  std::vector<double> tau(nb);
  cpu_geqrf_kern(2 * nb, nb, V2, 2 * nb, tau.data());
  cpu_larft_kern(2 * nb, nb, V2, 2 * nb, tau.data(), T1, ld);
  // R update: upper triangle back into Akk
  for (int c = 0; c < nb; c++)
    for (int r = 0; r <= c; r++)
      Akk[(size_t)c * ld + r] = V2[(size_t)c * 2 * nb + r];
  t.flows[0].data->written_on(false);
  t.flows[1].data->written_on(false);
  vd->written_on(false);
  td->written_on(false);
}

static void cpu_tsmqr(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const int nb = a.m, cols = a.n, ld = a.ld;
  const double* V2 = (const double*)t.flows[0].data->pull_to_host();
  const double* T1 = (const double*)t.flows[1].data->pull_to_host();
  double* Ckn = (double*)t.flows[2].data->pull_to_host();
  double* Cmn = (double*)t.flows[3].data->pull_to_host();
  std::vector<double> S((size_t)2 * nb * cols);
  for (int c = 0; c < cols; c++) {
    memcpy(&S[(size_t)c * 2 * nb], &Ckn[(size_t)c * ld], nb * 8);
    memcpy(&S[(size_t)c * 2 * nb + nb], &Cmn[(size_t)c * ld], nb * 8);
  }
  cpu_larfb_kern(2 * nb, cols, nb, V2, 2 * nb, T1, ld, S.data(), 2 * nb);
  for (int c = 0; c < cols; c++) {
    memcpy(&Ckn[(size_t)c * ld], &S[(size_t)c * 2 * nb], nb * 8);
    memcpy(&Cmn[(size_t)c * ld], &S[(size_t)c * 2 * nb + nb], nb * 8);
  }
  t.flows[2].data->written_on(false);
  t.flows[3].data->written_on(false);
}

// ------------------------------------------------------------ GPU chores
namespace {

__global__ void k_qr_fill(double* p, size_t n, uint32_t seed) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n; i += (size_t)gridDim.x * blockDim.x) {
    uint64_t h = i * 0x9E3779B97F4A7C15ull + seed;
    h ^= h >> 13;
    h *= 0x9E3779B97F4A7C15ull;
    h ^= h >> 32;
    p[i] = (double)(h & 0xFFFFFF) / (double)0x1000000 - 0.5;
  }
}

rocblas_handle qr_handle(GpuTaskCtx& g) {
  static thread_local std::map<void*, rocblas_handle> handles;
  rocblas_handle& h = handles[(void*)g.stream];
  if (!h) {
    PA_CHECK(rocblas_create_handle(&h) == rocblas_status_success);
    rocblas_set_pointer_mode(h, rocblas_pointer_mode_host);
    rocblas_set_stream(h, g.stream);
  }
  return h;
}

double* qr_scratch(GpuTaskCtx& g, int slot, size_t bytes) {
  static thread_local std::map<std::pair<void*, int>,
                               std::pair<void*, size_t>> bufs;
  auto& e = bufs[{(void*)g.stream, slot}];
  if (e.second < bytes) {
    // the old buffer may still be referenced by earlier kernels on this
    // stream: return it to the pool only after this task retires
    if (e.first) g.deferred_frees->emplace_back(e.first, e.second);
    if (g.engine) {
      e.first = g.engine->dev_alloc(bytes);
    } else {
      PA_HIP_CHECK(hipMalloc(&e.first, bytes));  // standalone micro-bench
    }
    e.second = bytes;
  }
  return (double*)e.first;
}

__global__ void k_stack_triu(double* S, const double* top, const double* bot,
                             int nb, int ld, int triu_bot) {
  int idx = blockIdx.x * blockDim.x + threadIdx.x;
  int total = nb * nb;
  for (; idx < total; idx += gridDim.x * blockDim.x) {
    int c = idx / nb, r = idx - c * nb;
    double v = top[(size_t)c * ld + r];
    S[(size_t)c * 2 * nb + r] = (r > c) ? 0.0 : v;
    double w = bot[(size_t)c * ld + r];
    S[(size_t)c * 2 * nb + nb + r] = (triu_bot && r > c) ? 0.0 : w;
  }
}

__global__ void k_stack(double* S, const double* top, const double* bot,
                        int nb, int cols, int ld) {
  int idx = blockIdx.x * blockDim.x + threadIdx.x;
  int total = nb * cols;
  for (; idx < total; idx += gridDim.x * blockDim.x) {
    int c = idx / nb, r = idx - c * nb;
    S[(size_t)c * 2 * nb + r] = top[(size_t)c * ld + r];
    S[(size_t)c * 2 * nb + nb + r] = bot[(size_t)c * ld + r];
  }
}

__global__ void k_unstack(const double* S, double* top, double* bot, int nb,
                          int cols, int ld) {
  int idx = blockIdx.x * blockDim.x + threadIdx.x;
  int total = nb * cols;
  for (; idx < total; idx += gridDim.x * blockDim.x) {
    int c = idx / nb, r = idx - c * nb;
    top[(size_t)c * ld + r] = S[(size_t)c * 2 * nb + r];
    bot[(size_t)c * ld + r] = S[(size_t)c * 2 * nb + nb + r];
  }
}

__global__ void k_copy_triu(double* dst, const double* src, int nb, int lds,
                            int ldd) {
  int idx = blockIdx.x * blockDim.x + threadIdx.x;
  int total = nb * nb;
  for (; idx < total; idx += gridDim.x * blockDim.x) {
    int c = idx / nb, r = idx - c * nb;
    if (r <= c) dst[(size_t)c * ldd + r] = src[(size_t)c * lds + r];
  }
}

inline dim3 grid1d(int total) {
  int g = (total + 255) / 256;
  return dim3(g > 2048 ? 2048 : g);
}

}  // namespace

__global__ void k_unitlow(double* V, const double* A, int m, int k, int lda,
                          int ldv);
static void larfb_gemm(GpuTaskCtx& g, int m, int n, int k, const double* A,
                       int lda, const double* T, int ldt, double* C, int ldc,
                       int scratch_slot);

// ---------------------------------------------------------- hand panel QR
// rocSOLVER's dgeqrf at tile sizes runs an unblocked host-synced column
// loop (~350 us of idle per column; profiles/RESULTS.md). These kernels do
// the panel factorization device-side:
//  - k_geqr2: one workgroup factors an (m x 128) panel column-by-column
//    (norm reduce + scale + wave-parallel trailing update, all in-kernel);
//  - T factors come from G = V^T V (one full-rate dgemm) via k_larft_diag
//    (per-128-block upper-triangular recurrence) and a blocked combine
//    (two dgemms per block column).
__global__ void __launch_bounds__(1024) k_geqr2(double* A, int m, int ncols,
                                                int ld, double* tau) {
  __shared__ double red[1024];
  __shared__ double bc[3];  // beta, tau_j, scal
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  for (int j = 0; j < ncols; j++) {
    // 1) norm^2 of the sub-diagonal part of column j
    double acc = 0;
    for (int i = j + 1 + tid; i < m; i += 1024) {
      double v = A[(size_t)j * ld + i];
      acc += v * v;
    }
    red[tid] = acc;
    __syncthreads();
    for (int s = 512; s > 0; s >>= 1) {
      if (tid < s) red[tid] += red[tid + s];
      __syncthreads();
    }
    if (tid == 0) {
      double alpha = A[(size_t)j * ld + j];
      double nrm2 = red[0];
      if (nrm2 == 0.0) {
        bc[0] = alpha;
        bc[1] = 0.0;
        bc[2] = 0.0;
        tau[j] = 0.0;
      } else {
        double beta = -copysign(sqrt(alpha * alpha + nrm2), alpha);
        bc[0] = beta;
        bc[1] = (beta - alpha) / beta;
        bc[2] = 1.0 / (alpha - beta);
        tau[j] = bc[1];
      }
    }
    __syncthreads();
    const double tau_j = bc[1], scal = bc[2];
    if (tau_j != 0.0) {
      for (int i = j + 1 + tid; i < m; i += 1024)
        A[(size_t)j * ld + i] *= scal;
      if (tid == 0) A[(size_t)j * ld + j] = bc[0];
    }
    __syncthreads();
    if (tau_j != 0.0) {
      // 2) apply H_j to trailing columns, one wave per column round-robin
      for (int c = j + 1 + wave; c < ncols; c += 16) {
        double dot = (lane == 0) ? A[(size_t)c * ld + j] : 0.0;
        for (int i = j + 1 + lane; i < m; i += 64)
          dot += A[(size_t)j * ld + i] * A[(size_t)c * ld + i];
        for (int s = 32; s > 0; s >>= 1) dot += __shfl_down(dot, s);
        dot = __shfl(dot, 0);
        double w = tau_j * dot;
        if (lane == 0) A[(size_t)c * ld + j] -= w;
        for (int i = j + 1 + lane; i < m; i += 64)
          A[(size_t)c * ld + i] -= w * A[(size_t)j * ld + i];
      }
    }
    __syncthreads();
  }
}

// ---------------------------------------------------- multi-WG panel QR
// One LAUNCH factors a whole 128-column panel (vs k_geqr2's one-WG column
// loop at 0.7 TF, profiles/RESULTS.md): WG 0 factors 16-column sub-panels
// held in LDS (fast column loop: norm reduce + rank-1 updates never leave
// the CU), builds the 16x16 T block in LDS, then ALL workgroups apply the
// block reflector to the remaining panel columns between agent-scope grid
// barriers (MI355X_MICROARCH.md "Workgroup dispatch" release/acquire
// forms). A two-segment row map skips the structurally-zero block of
// stacked [R (upper-tri); B] TSQRT tiles, capping active rows at
// top_block + B regardless of the panel offset.
//
// Row map: local row r < len0 -> global base0 + r ; else base1 + (r-len0).
// LDS budget: rows * W * 8 <= 147456 (W=16 up to 1152 rows, W=8 to 2304).
namespace {

constexpr int QR_LDS_DOUBLES = 18432;  // 144 KiB panel + ~8 KiB reduction
constexpr int QR_MAX_ROWS_W16 = 1152;
constexpr int QR_MAX_ROWS_W8 = 2304;

__device__ inline void qr_grid_barrier(int* cnt, int nwg, int bar_no) {
  __syncthreads();
  if (threadIdx.x == 0) {
    __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __hip_atomic_fetch_add(cnt, 1, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    while (__hip_atomic_load(cnt, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_AGENT) < bar_no * nwg)
      __builtin_amdgcn_s_sleep(8);
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  }
  __syncthreads();
}

// A panel column's reflector value at local row r (unit-lower, in-place).
__device__ inline double qr_vval(const double* A, int ld, int base0,
                                 int len0, int base1, int r, int cloc,
                                 int gcol) {
  if (r < cloc) return 0.0;
  if (r == cloc) return 1.0;
  int gr = r < len0 ? base0 + r : base1 + (r - len0);
  return A[(size_t)gcol * ld + gr];
}

typedef double f64x4q __attribute__((ext_vector_type(4)));

// MFMA block-reflector apply used by the panel kernel's helper pool:
// processes 16 columns at once through v_mfma_f64_16x16x4f64, staging
// 256-row chunks of the (unit-lower) V image and of C in LDS carved from
// `sp` — W = V^T C and C -= V Y become matrix-core work instead of
// latency-bound 16-lane dot units (profiles/qr_round2.md headroom note).
// Layout notes: operands stored [k][m] in LDS ([row*16 + q] for V), MFMA
// lane map a = A[m=r16][k=ksub], D(row=ksub+4e, col=r16) — the repo's
// fp64 GEMM conventions (kernels_hip.cpp).
__device__ void qr_apply_mfma16(
    double* A, int ld, int pcol0, int base0, int len0, int base1, int rows,
    int c0, int w, const double* T16, const int* cg, int nc, double* sp,
    double* wy /*unused*/, int tid, int wave, int lane) {
  double* Vc = sp;               // [256][16]
  double* Cc = sp + 256 * 16;    // [256][16]
  double* Wr = sp + 2 * 256 * 16;  // [16 waves][16][16] partial W
  double* Ys = sp + 2 * 256 * 16 + 16 * 256;  // [16][16] col-major [c][q]
  const int r16 = lane & 15, ksub = lane >> 4;

  // ---------------- step A: W = V^T C ----------------
  f64x4q accW = {0, 0, 0, 0};
  for (int ck = 0; ck < rows; ck += 256) {
    const int clen = min(256, rows - ck);
    // stage V chunk (transposed [r][q], unit-lower semantics, zero pad)
    for (int x = tid; x < 256 * 16; x += 1024) {
      const int q = x >> 8, rr = x & 255;
      const int r = ck + rr;
      double v = 0;
      if (rr < clen && q < w) {
        const int cloc = c0 + q;
        if (r == cloc) {
          v = 1.0;
        } else if (r > cloc) {
          int gr = r < len0 ? base0 + r : base1 + (r - len0);
          v = A[(size_t)(pcol0 + cloc) * ld + gr];
        }
      }
      Vc[rr * 16 + q] = v;
    }
    // stage C chunk ([r][c], zero pad)
    for (int x = tid; x < 256 * 16; x += 1024) {
      const int c = x >> 8, rr = x & 255;
      const int r = ck + rr;
      double v = 0;
      if (rr < clen && c < nc) {
        int gr = r < len0 ? base0 + r : base1 + (r - len0);
        v = A[(size_t)(pcol0 + cg[c]) * ld + gr];
      }
      Cc[rr * 16 + c] = v;
    }
    __syncthreads();
    // wave `wave` covers k-rows [wave*16, wave*16+16) of this chunk
    {
      const int k0 = wave * 16;
#pragma unroll
      for (int kk = 0; kk < 16; kk += 4) {
        double a = Vc[(k0 + kk + ksub) * 16 + r16];
        double b = Cc[(k0 + kk + ksub) * 16 + r16];
        accW = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, accW, 0, 0, 0);
      }
    }
    __syncthreads();
  }
  // reduce the 16 per-wave W partials, then Y = T^T W
#pragma unroll
  for (int e = 0; e < 4; e++)
    Wr[wave * 256 + (ksub + 4 * e) * 16 + r16] = accW[e];
  __syncthreads();
  if (tid < 256) {
    const int q = tid & 15, c = tid >> 4;
    double wsum = 0;
    for (int wv = 0; wv < 16; wv++) wsum += Wr[wv * 256 + q * 16 + c];
    // stash W(q, c) back (reuse Wr row 0 region is unsafe; use Ys then
    // overwrite with Y below via a second barrier)
    Wr[q * 16 + c] = wsum;  // W in [q][c] at the front of Wr
  }
  __syncthreads();
  if (tid < 256) {
    const int q = tid & 15, c = tid >> 4;
    double y = 0;
    for (int p2 = 0; p2 <= q; p2++) y += T16[q * 16 + p2] * Wr[p2 * 16 + c];
    Ys[(q)*16 + c] = y;  // Y stored [k=q][c]
  }
  __syncthreads();

  // ---------------- step B: C -= V * Y ----------------
  for (int ck = 0; ck < rows; ck += 256) {
    const int clen = min(256, rows - ck);
    for (int x = tid; x < 256 * 16; x += 1024) {
      const int q = x >> 8, rr = x & 255;
      const int r = ck + rr;
      double v = 0;
      if (rr < clen && q < w) {
        const int cloc = c0 + q;
        if (r == cloc) {
          v = 1.0;
        } else if (r > cloc) {
          int gr = r < len0 ? base0 + r : base1 + (r - len0);
          v = A[(size_t)(pcol0 + cloc) * ld + gr];
        }
      }
      Vc[rr * 16 + q] = v;
    }
    __syncthreads();
    // wave owns the 16-row m-tile [wave*16, wave*16+16) of the chunk
    f64x4q accU = {0, 0, 0, 0};
    {
#pragma unroll
      for (int kk = 0; kk < 16; kk += 4) {
        double a = Vc[(wave * 16 + r16) * 16 + kk + ksub];
        double b = Ys[(kk + ksub) * 16 + r16];
        accU = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, accU, 0, 0, 0);
      }
    }
    // D(row=ksub+4e, col=r16) -> Cc[(wave*16+row)][col]
#pragma unroll
    for (int e = 0; e < 4; e++)
      Cc[(wave * 16 + ksub + 4 * e) * 16 + r16] = accU[e];
    __syncthreads();
    // coalesced global RMW per column
    for (int x = tid; x < 256 * 16; x += 1024) {
      const int c = x >> 8, rr = x & 255;
      const int r = ck + rr;
      if (rr < clen && c < nc && r >= c0) {
        int gr = r < len0 ? base0 + r : base1 + (r - len0);
        A[(size_t)(pcol0 + cg[c]) * ld + gr] -= Cc[rr * 16 + c];
      }
    }
    __syncthreads();
  }
}

__global__ void __launch_bounds__(1024) k_qr_panel_mw(
    double* A, int ld, int pcol0,  // panel base column (global)
    int base0, int len0, int base1, int len1,  // reflector row segments
    int pc,     // total columns from pcol0 (factor + apply targets)
    int fcols,  // columns to FACTOR (<=128)
    int W,      // sub-panel width (8|16)
    double* tau, double* T16s,  // T16s: 16x16 per sub-panel scratch
    int* cnt, int nwg, int nA, int amode, int look,
    unsigned long long* dbg) {
  // Producer/consumer panel pipeline (no grid barriers):
  //  - WG 0 factors 16-column sub-panels in LDS (one barrier per column,
  //    piggybacked norms, deferred scaling), builds T16 with a parallel
  //    G = V^T V, then PUBLISHES the sub-panel (agent release + counter).
  //  - helper pool A (<=7 WGs) applies each published block reflector to
  //    the remaining FACTOR columns and acks; WG 0 only stages sub-panel
  //    s+1 after ack(s), so the factor chain never waits on the trailing
  //    matrix.
  //  - pool B (the rest) streams the same applies over the trailing
  //    columns asynchronously, in publication order (each column has a
  //    fixed owner, so per-column apply order is the program order).
  // Memory protocol per MI355X_MICROARCH.md "Workgroup dispatch": plain
  // stores -> release fence + vmcnt(0) asm -> relaxed counter; consumers
  // poll relaxed, acquire-fence once, then plain loads.
  __shared__ double sp[QR_LDS_DOUBLES];  // sub-panel / V image
  __shared__ double red[16];             // norm partials (staging)
  __shared__ double bcs[3 * 16];         // per-col {vd, beta, tfac}
  __shared__ double tl[16];              // per-col tau (LDS copy)
  __shared__ double wy[2 * 64];          // dots + y for 4 columns
  __shared__ double gt[512];             // G (16x16) + local T (16x16)
  const int rows = len0 + len1;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int wg = blockIdx.x;
  const int ns = (fcols + W - 1) / W;
  const int nB = nwg - 1 - nA;

  if (wg == 0) {
    for (int si = 0; si < ns; si++) {
      const int c0 = si * W;
      const int w = min(W, fcols - c0);
      double* T16 = T16s + (size_t)si * 256;
      unsigned long long t0 = dbg ? __builtin_amdgcn_s_memrealtime() : 0;
      if (dbg && tid == 0) {
        unsigned long long t1 = __builtin_amdgcn_s_memrealtime();
        dbg[0] += t1 - t0;
        t0 = t1;
      }
      // ---- stage sub-panel columns into LDS (+ first column's norm) ----
      double acc0 = 0;
      for (int q = 0; q < w; q++) {
        const double* A0 = A + (size_t)(pcol0 + c0 + q) * ld + base0;
        const double* A1 = A + (size_t)(pcol0 + c0 + q) * ld + base1 - len0;
        double* spq = sp + (size_t)q * rows;
        for (int r = tid; r < len0; r += 1024) {
          double v = A0[r];
          spq[r] = v;
          if (q == 0 && r > c0) acc0 += v * v;
        }
        for (int r = len0 + tid; r < rows; r += 1024) {
          double v = A1[r];
          spq[r] = v;
          if (q == 0) acc0 += v * v;
        }
      }
      for (int sh = 32; sh > 0; sh >>= 1) acc0 += __shfl_down(acc0, sh);
      if (lane == 0) red[wave] = acc0;
      __syncthreads();
      if (tid == 0) {
        double nrm2 = 0;
        for (int v = 0; v < 16; v++) nrm2 += red[v];
        double alpha = sp[c0];
        double beta = alpha, vd = 1.0, tfac = 0.0, tj = 0.0;
        if (nrm2 != 0.0) {
          beta = -copysign(sqrt(alpha * alpha + nrm2), alpha);
          vd = alpha - beta;
          tj = (beta - alpha) / beta;
          tfac = tj / (vd * vd);
        }
        bcs[0] = vd;
        bcs[1] = beta;
        bcs[2] = tfac;
        tl[0] = tj;
        tau[c0] = tj;
      }
      __syncthreads();
      if (dbg && tid == 0) {
        unsigned long long t1 = __builtin_amdgcn_s_memrealtime();
        dbg[1] += t1 - t0;  // stage
        t0 = t1;
      }
      // ---- factor: one barrier per column ----
      for (int j = 0; j < w; j++) {
        const int d = c0 + j;
        const double vd = bcs[3 * j], tfac = bcs[3 * j + 2];
        if (wave > j && wave < w) {
          const double* col = sp + (size_t)j * rows;
          double* cc = sp + (size_t)wave * rows;
          double dot = (lane == 0) ? vd * cc[d] : 0.0;
          double dot1 = 0.0;
          for (int i = d + 1 + lane; i < rows; i += 128) dot += col[i] * cc[i];
          for (int i = d + 65 + lane; i < rows; i += 128)
            dot1 += col[i] * cc[i];
          dot += dot1;
          for (int sh = 32; sh > 0; sh >>= 1) dot += __shfl_down(dot, sh);
          dot = __shfl(dot, 0);
          const double wj = tfac * dot;
          if (lane == 0) cc[d] -= wj * vd;
          double nacc = 0;
          const bool mine = (wave == j + 1);
          for (int i = d + 1 + lane; i < rows; i += 64) {
            double nv = cc[i] - wj * col[i];
            cc[i] = nv;
            if (mine && i > d + 1) nacc += nv * nv;
          }
          if (mine) {
            for (int sh = 32; sh > 0; sh >>= 1) nacc += __shfl_down(nacc, sh);
            if (lane == 0) {
              double alpha = cc[d + 1];
              double beta = alpha, nvd = 1.0, tfac2 = 0.0, tj = 0.0;
              if (nacc != 0.0) {
                beta = -copysign(sqrt(alpha * alpha + nacc), alpha);
                nvd = alpha - beta;
                tj = (beta - alpha) / beta;
                tfac2 = tj / (nvd * nvd);
              }
              bcs[3 * (j + 1)] = nvd;
              bcs[3 * (j + 1) + 1] = beta;
              bcs[3 * (j + 1) + 2] = tfac2;
              tl[j + 1] = tj;
              tau[c0 + j + 1] = tj;
            }
          }
        }
        __syncthreads();
      }
      if (dbg && tid == 0) {
        unsigned long long t1 = __builtin_amdgcn_s_memrealtime();
        dbg[2] += t1 - t0;  // factor
        t0 = t1;
      }
      // ---- scale pass ----
      if (wave < w) {
        const int d = c0 + wave;
        double* col = sp + (size_t)wave * rows;
        const double inv = 1.0 / bcs[3 * wave];
        for (int i = d + 1 + lane; i < rows; i += 64) col[i] *= inv;
        if (lane == 0) col[d] = bcs[3 * wave + 1];
      }
      __syncthreads();
      // ---- write back ----
      for (int q = 0; q < w; q++) {
        double* A0 = A + (size_t)(pcol0 + c0 + q) * ld + base0;
        double* A1 = A + (size_t)(pcol0 + c0 + q) * ld + base1 - len0;
        const double* spq = sp + (size_t)q * rows;
        for (int r = tid; r < len0; r += 1024) A0[r] = spq[r];
        for (int r = len0 + tid; r < rows; r += 1024) A1[r] = spq[r];
      }
      __syncthreads();
      if (dbg && tid == 0) {
        unsigned long long t1 = __builtin_amdgcn_s_memrealtime();
        dbg[3] += t1 - t0;  // scale + writeback
        t0 = t1;
      }
      // ---- G = V^T V: 64 units of 16 lanes over the (s<q) pairs ----
      {
        const int unit = wave * 4 + (lane >> 4), l16 = lane & 15;
        const int npairs = w * (w - 1) / 2;
        for (int idx = unit; idx < npairs; idx += 64) {
          // decode idx -> (sq pair): q = row in triangle
          int q = 1;
          int rem = idx;
          while (rem >= q) {
            rem -= q;
            q++;
          }
          int sc = rem;  // 0 <= sc < q
          const int dq = c0 + q;
          const double* cs = sp + (size_t)sc * rows;
          const double* cq = sp + (size_t)q * rows;
          // i >= dq > ds always: v_s = cs[i]; v_q = 1 at dq, cq[i] below
          double dot = (l16 == 0) ? cs[dq] : 0.0;
          double d1 = 0.0;
          for (int i = dq + 1 + l16; i < rows; i += 32) dot += cs[i] * cq[i];
          for (int i = dq + 17 + l16; i < rows; i += 32) d1 += cs[i] * cq[i];
          dot += d1;
          for (int sh = 8; sh > 0; sh >>= 1)
            dot += __shfl_down(dot, sh, 16);
          if (l16 == 0) gt[sc * 16 + q] = dot;  // G(sc, q)
        }
      }
      __syncthreads();
      if (wave == 0) {
        double* Tl = gt + 256;
        const double* G = gt;
        for (int j = 0; j < w; j++) {
          double tj = tl[j];
          double sacc = 0;
          if (lane < j) {
            for (int q = lane; q < j; q++)
              sacc += Tl[q * 16 + lane] * G[q * 16 + j];
            sacc *= -tj;
          }
          if (lane < j) Tl[j * 16 + lane] = sacc;
          if (lane == j) Tl[j * 16 + j] = tj;
        }
        if (lane < 16)
          for (int j = 0; j < 16; j++)
            T16[j * 16 + lane] =
                (j < w && lane <= j) ? Tl[j * 16 + lane] : 0.0;
      }
      __syncthreads();
      if (dbg && tid == 0) {
        unsigned long long t1 = __builtin_amdgcn_s_memrealtime();
        dbg[4] += t1 - t0;  // G16 + T16
        t0 = t1;
      }
      // ---- publish sub-panel si ----
      if (tid == 0) {
        __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __hip_atomic_store(&cnt[0], si + 1, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
      }
      // ---- wait for pool A to finish sub-panel si-1 everywhere: its
      // range [c0-W+2W, fcols) includes OUR self-apply window, and the
      // si-1 reflectors must land on those columns before si's do. The
      // factor of si overlapped pool A's si-1 pass, so this wait is the
      // pipeline's only rendezvous. With qr_lookahead=0 the self-apply is
      // off and the wait covers pool A through si instead. ----
      if (tid == 0) {
        while (__hip_atomic_load(&cnt[1], __ATOMIC_RELAXED,
                                 __HIP_MEMORY_SCOPE_AGENT) <
               (si + (look ? 0 : 1)) * nA)
          __builtin_amdgcn_s_sleep(2);
        __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
      }
      __syncthreads();
      // ---- self-apply this block to the NEXT sub-panel's columns (the
      // V image is already in sp, minus the unit-lower masking) ----
      for (int c = c0 + w; look && c < min(c0 + 2 * W, fcols); c += 4) {
        const int nc = min(4, min(c0 + 2 * W, fcols) - c);
        const int ci = lane >> 4, l16 = lane & 15;
        if (ci < nc && wave < w) {
          const int gcol = pcol0 + c + ci;
          const int cloc = c0 + wave;
          const double* v = sp + (size_t)wave * rows;
          const double* A0 = A + (size_t)gcol * ld + base0;
          const double* A1 = A + (size_t)gcol * ld + base1 - len0;
          // raw sp has R above the diagonal: start at cloc, inject the
          // unit diagonal by hand
          double dot = (l16 == 0) ? A0[cloc] : 0.0;
          double d1 = 0;
          for (int r = cloc + 1 + l16; r < len0; r += 32)
            dot += v[r] * A0[r];
          for (int r = cloc + 17 + l16; r < len0; r += 32)
            d1 += v[r] * A0[r];
          for (int r = len0 + l16; r < rows; r += 32) dot += v[r] * A1[r];
          for (int r = len0 + 16 + l16; r < rows; r += 32)
            d1 += v[r] * A1[r];
          dot += d1;
          for (int sh = 8; sh > 0; sh >>= 1) dot += __shfl_down(dot, sh, 16);
          if (l16 == 0) wy[ci * 16 + wave] = dot;
        }
        __syncthreads();
        if (tid < 16 * nc) {
          const int q = tid & 15, cc2 = tid >> 4;
          double sacc = 0;
          for (int p2 = 0; p2 <= q; p2++)
            sacc += T16[q * 16 + p2] * wy[cc2 * 16 + p2];
          wy[64 + cc2 * 16 + q] = sacc;
        }
        __syncthreads();
        {
          const int cc2 = tid >> 8, t2 = tid & 255;
          if (cc2 < nc) {
            const int gcol = pcol0 + c + cc2;
            double* A0 = A + (size_t)gcol * ld + base0;
            double* A1 = A + (size_t)gcol * ld + base1 - len0;
            const double* y = wy + 64 + cc2 * 16;
            for (int r = c0 + t2; r < len0; r += 256) {
              double sacc = 0;
              for (int q = 0; q < w; q++) {
                const int cloc = c0 + q;
                double v = r < cloc ? 0.0
                                    : (r == cloc ? 1.0
                                                 : sp[(size_t)q * rows + r]);
                sacc += v * y[q];
              }
              A0[r] -= sacc;
            }
            for (int r = len0 + t2; r < rows; r += 256) {
              double sacc = 0;
              for (int q = 0; q < w; q++)
                sacc += sp[(size_t)q * rows + r] * y[q];
              A1[r] -= sacc;
            }
          }
        }
        __syncthreads();
      }
    }
    return;
  }

  // ---------------- helper workgroups ----------------
  const bool poolA = wg <= nA;
  for (int si = 0; si < ns; si++) {
    const int c0 = si * W;
    const int w = min(W, fcols - c0);
    const double* T16 = T16s + (size_t)si * 256;
    unsigned long long t0 =
        (dbg && (wg == 1 || wg == nA + 1)) ? __builtin_amdgcn_s_memrealtime()
                                           : 0;
    if (tid == 0) {
      while (__hip_atomic_load(&cnt[0], __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_AGENT) < si + 1)
        __builtin_amdgcn_s_sleep(2);
      __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
    }
    __syncthreads();
    const int dslot = (wg == 1) ? 5 : (wg == nA + 1 ? 8 : -1);
    if (dbg && dslot >= 0 && tid == 0) {
      unsigned long long t1 = __builtin_amdgcn_s_memrealtime();
      dbg[dslot] += t1 - t0;  // publish poll wait
      t0 = t1;
    }
    if (!amode || poolA) {
      // cache the scaled V image (unit-lower) in LDS (VALU apply path;
      // the MFMA path streams V chunks itself)
      for (int q = 0; q < w; q++) {
        const int gcol = pcol0 + c0 + q, cloc = c0 + q;
        const double* A0 = A + (size_t)gcol * ld + base0;
        const double* A1 = A + (size_t)gcol * ld + base1 - len0;
        double* spq = sp + (size_t)q * rows;
        for (int r = tid; r < len0; r += 1024)
          spq[r] = r < cloc ? 0.0 : (r == cloc ? 1.0 : A0[r]);
        for (int r = len0 + tid; r < rows; r += 1024) spq[r] = A1[r];
      }
    }
    __syncthreads();
    if (dbg && dslot >= 0 && tid == 0) {
      unsigned long long t1 = __builtin_amdgcn_s_memrealtime();
      dbg[dslot + 1] += t1 - t0;  // V image build
      t0 = t1;
    }
    // apply (I - V T V^T)^T to 4-column groups of my column set
    auto apply4 = [&](int cg[4], int nc) {
      const int ci = lane >> 4, l16 = lane & 15;  // unit = (q=wave, ci)
      if (ci < nc) {
        const int gcol = pcol0 + cg[ci];
        const int cloc = c0 + wave;
        const double* v = sp + (size_t)wave * rows;
        const double* A0 = A + (size_t)gcol * ld + base0;
        const double* A1 = A + (size_t)gcol * ld + base1 - len0;
        double d0 = 0, d1 = 0, d2 = 0, d3 = 0;
        for (int r = cloc + l16; r < len0; r += 64) d0 += v[r] * A0[r];
        for (int r = cloc + 16 + l16; r < len0; r += 64) d1 += v[r] * A0[r];
        for (int r = cloc + 32 + l16; r < len0; r += 64) d2 += v[r] * A0[r];
        for (int r = cloc + 48 + l16; r < len0; r += 64) d3 += v[r] * A0[r];
        for (int r = len0 + l16; r < rows; r += 64) d0 += v[r] * A1[r];
        for (int r = len0 + 16 + l16; r < rows; r += 64) d1 += v[r] * A1[r];
        for (int r = len0 + 32 + l16; r < rows; r += 64) d2 += v[r] * A1[r];
        for (int r = len0 + 48 + l16; r < rows; r += 64) d3 += v[r] * A1[r];
        double dot = (d0 + d1) + (d2 + d3);
        for (int sh = 8; sh > 0; sh >>= 1) dot += __shfl_down(dot, sh, 16);
        if (l16 == 0) wy[ci * 16 + wave] = dot;
      }
      __syncthreads();
      if (tid < 16 * nc) {
        const int q = tid & 15, c = tid >> 4;
        double sacc = 0;
        for (int p2 = 0; p2 <= q; p2++)
          sacc += T16[q * 16 + p2] * wy[c * 16 + p2];
        wy[64 + c * 16 + q] = sacc;
      }
      __syncthreads();
      {
        const int c = tid >> 8, t2 = tid & 255;  // 4 cols x 256 threads
        if (c < nc) {
          const int gcol = pcol0 + cg[c];
          double* A0 = A + (size_t)gcol * ld + base0;
          double* A1 = A + (size_t)gcol * ld + base1 - len0;
          const double* y = wy + 64 + c * 16;
          for (int r = c0 + t2; r < len0; r += 256) {
            double sacc = 0;
            for (int q = 0; q < w; q++) sacc += sp[(size_t)q * rows + r] * y[q];
            A0[r] -= sacc;
          }
          for (int r = len0 + t2; r < rows; r += 256) {
            double sacc = 0;
            for (int q = 0; q < w; q++) sacc += sp[(size_t)q * rows + r] * y[q];
            A1[r] -= sacc;
          }
        }
      }
      __syncthreads();
    };
    // Column ownership must be ABSOLUTE (c mod stride == phase), not
    // relative to cbeg: cbeg advances with the sub-panel, and a column
    // whose owner changed between sub-panels would receive reflector
    // applies from two helpers with NO mutual ordering (measured: rare
    // mid-panel corruption, first bad column inside pool A's range).
    auto apply_range = [&](int cbeg, int cend, int stride, int phase,
                           bool mfma) {
      int first = cbeg + (((phase - cbeg) % stride) + stride) % stride;
      if (amode && mfma) {
        int cg[16];
        int nc = 0;
        for (int c = first; c < cend; c += stride) {
          cg[nc++] = c;
          if (nc == 16) {
            qr_apply_mfma16(A, ld, pcol0, base0, len0, base1, rows, c0, w,
                            T16, cg, 16, sp, wy, tid, wave, lane);
            nc = 0;
          }
        }
        if (nc)
          qr_apply_mfma16(A, ld, pcol0, base0, len0, base1, rows, c0, w,
                          T16, cg, nc, sp, wy, tid, wave, lane);
        return;
      }
      int cg[4];
      int nc = 0;
      for (int c = first; c < cend; c += stride) {
        cg[nc++] = c;
        if (nc == 4) {
          apply4(cg, 4);
          nc = 0;
        }
      }
      if (nc) apply4(cg, nc);
    };
    if (poolA) {
      // small within-panel groups: the latency-optimized VALU units win;
      // the MFMA path pays per-256-row chunk staging regardless of nc
      apply_range(look ? min(c0 + 2 * W, fcols) : c0 + w, fcols, nA,
                  wg - 1, false);
      __syncthreads();
      if (dbg && dslot >= 0 && tid == 0)
        dbg[dslot + 2] += __builtin_amdgcn_s_memrealtime() - t0;  // apply
      if (tid == 0) {
        __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __hip_atomic_fetch_add(&cnt[1], 1, __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_AGENT);
      }
      if (nB == 0)  // no pool B: pool A also covers the trailing columns
        apply_range(fcols, pc, nA, wg - 1, true);
    } else {
      apply_range(fcols, pc, nB, wg - 1 - nA, true);
      if (dbg && dslot >= 0 && tid == 0)
        dbg[dslot + 2] += __builtin_amdgcn_s_memrealtime() - t0;  // apply
    }
  }
}

}  // namespace

// T diag blocks: for block b, columns j: T(0:j,j) = -tau_j T(0:j,0:j) g
// where g = G(blk rows, j) restricted to the block. G is V^T V of the
// unit-lower V; T upper-triangular, T(j,j) = tau_j.
__global__ void __launch_bounds__(128) k_larft_diag(const double* G, int ldg,
                                                    const double* tau,
                                                    double* T, int ldt,
                                                    int k) {
  // The whole 128x128 T block lives in LDS during the column recurrence
  // (global-memory round trips per column made this kernel 1.5 ms and 9%
  // of QR GPU time — gpurun_out/qr_kernel_stats.csv); one barrier per
  // column, write-back once at the end.
  __shared__ double Ts[128 * 128];
  const int b0 = blockIdx.x * 128;
  const int tid = threadIdx.x;  // 128 threads
  const int nb = min(128, k - b0);
  for (int j = 0; j < nb; j++) {
    int gj = b0 + j;
    // col = -tau_j * T(0:j,0:j) * G(b0..b0+j, gj)
    double s = 0;
    if (tid < j) {
      const double* Gc = G + (size_t)gj * ldg + b0;
      for (int q = tid; q < j; q++) s += Ts[q * 128 + tid] * Gc[q];
      s *= -tau[gj];
    }
    if (tid < j) Ts[j * 128 + tid] = s;
    if (tid == j) Ts[j * 128 + j] = tau[gj];
    __syncthreads();
  }
  for (int j = 0; j < nb; j++)
    if (tid <= j) T[(size_t)(b0 + j) * ldt + b0 + tid] = Ts[j * 128 + tid];
}

// Full tile/stacked-panel QR: factor 128-wide panels with the multi-WG
// panel kernel (k_geqr2 single-WG fallback for oversized rows), apply to
// the trailing columns with gemm-larfb, then build the full k x k T.
// ts_split > 0 marks a stacked [R (upper-tri, ts_split rows); B] tile:
// the panel kernel then skips the structurally-zero block below R's
// diagonal band, capping active rows at 128 + (m - ts_split).
static void qr_factor_hand(GpuTaskCtx& g, double* A, int m, int k, int ld,
                           double* T, int ldt, int slot0, int ts_split = 0) {
  rocblas_handle h = qr_handle(g);
  double* tau = qr_scratch(g, slot0, (size_t)k * 8);
  double* T16s = qr_scratch(g, slot0 + 9, (size_t)8 * 256 * 8);
  int* cnt = (int*)qr_scratch(g, slot0 + 10, 256);
  const int nwg = std::max(2, (int)param_int("qr_panel_wgs", 48));
  const int nA = std::min((int64_t)nwg - 1, param_int("qr_panel_poolA", 15));
  const double one = 1.0, zero = 0.0, mone = -1.0;
  PA_HIP_CHECK(hipMemsetAsync(T, 0, (size_t)ldt * k * 8, g.stream));
  for (int p = 0; p < k; p += 128) {
    int pc = std::min(128, k - p);
    double* panel = A + (size_t)p * ld + p;
    int prows = m - p;
    // row segments: dense panel, or triangular-top + dense-bottom
    int base0 = p, len0 = prows, base1 = 0, len1 = 0;
    if (ts_split > 0) {
      len0 = std::min(pc, ts_split - p);
      base1 = ts_split;
      len1 = m - ts_split;
    }
    int rows = len0 + len1;
    int W = rows <= QR_MAX_ROWS_W16 ? 16 : (rows <= QR_MAX_ROWS_W8 ? 8 : 0);
    if (W) {
      // qr_apply=kernel (default): the panel kernel both factors and
      // applies to ALL remaining tile columns (no per-panel dgemms).
      // qr_apply=gemm: the kernel applies within the 128 panel columns
      // only; the trailing matrix goes through T128 + larfb dgemms at
      // Tensile rates (A/B: the in-kernel apply units are latency-bound,
      // profiles/qr_round2.md).
      static const std::string apply_kind = param_str("qr_apply", "kernel");
      static const bool apply_gemm = apply_kind == "gemm";
      static const int amode = apply_kind == "valu" ? 0 : 1;
      int rest = k - p - pc;
      int apply_cols = apply_gemm ? pc : k - p;
      PA_HIP_CHECK(hipMemsetAsync(cnt, 0, 2 * sizeof(int), g.stream));
      hipLaunchKernelGGL(k_qr_panel_mw, dim3(nwg), dim3(1024), 0, g.stream,
                         A, ld, p, base0, len0, base1, len1, apply_cols, pc,
                         W, tau + p, T16s, cnt, nwg, nA, amode,
                         (int)param_int("qr_lookahead", 1),
                         (unsigned long long*)nullptr);
      if (apply_gemm && rest > 0) {
        double* V = qr_scratch(g, slot0 + 1, (size_t)prows * 128 * 8);
        double* G = qr_scratch(g, slot0 + 2, (size_t)128 * 128 * 8);
        hipLaunchKernelGGL(k_unitlow, grid1d(prows * pc), dim3(256), 0,
                           g.stream, V, panel, prows, pc, ld, prows);
        PA_CHECK(rocblas_dgemm(h, rocblas_operation_transpose,
                               rocblas_operation_none, pc, pc, prows, &one,
                               V, prows, V, prows, &zero, G,
                               pc) == rocblas_status_success);
        double* T128 = qr_scratch(g, slot0 + 3, (size_t)128 * 128 * 8);
        PA_HIP_CHECK(hipMemsetAsync(T128, 0, (size_t)pc * pc * 8, g.stream));
        hipLaunchKernelGGL(k_larft_diag, dim3(1), dim3(128), 0, g.stream, G,
                           pc, tau + p, T128, pc, pc);
        larfb_gemm(g, prows, rest, pc, panel, ld, T128, pc,
                   A + (size_t)(p + pc) * ld + p, ld, slot0 + 4);
      }
      continue;
    }
    hipLaunchKernelGGL(k_geqr2, dim3(1), dim3(1024), 0, g.stream, panel,
                       prows, pc, ld, tau + p);
    int rest = k - p - pc;
    if (rest > 0) {
      // T128 for this panel from G128 = V^T V
      double* V = qr_scratch(g, slot0 + 1, (size_t)prows * 128 * 8);
      double* G = qr_scratch(g, slot0 + 2, (size_t)128 * 128 * 8);
      hipLaunchKernelGGL(k_unitlow, grid1d(prows * pc), dim3(256), 0,
                         g.stream, V, panel, prows, pc, ld, prows);
      PA_CHECK(rocblas_dgemm(h, rocblas_operation_transpose,
                             rocblas_operation_none, pc, pc, prows, &one, V,
                             prows, V, prows, &zero, G,
                             pc) == rocblas_status_success);
      double* T128 = qr_scratch(g, slot0 + 3, (size_t)128 * 128 * 8);
      PA_HIP_CHECK(hipMemsetAsync(T128, 0, (size_t)pc * pc * 8, g.stream));
      hipLaunchKernelGGL(k_larft_diag, dim3(1), dim3(128), 0, g.stream, G,
                         pc, tau + p, T128, pc, pc);
      larfb_gemm(g, prows, rest, pc, panel, ld, T128, pc,
                 A + (size_t)(p + pc) * ld + p, ld, slot0 + 4);
    }
  }
  // Full T: G = V^T V over all k columns, diag blocks in one launch,
  // off-diagonal blocks by blocked combine (T[0:J,blk] = -T·G·T_blk).
  double* V = qr_scratch(g, slot0 + 1, (size_t)m * k * 8);
  double* G = qr_scratch(g, slot0 + 7, (size_t)k * k * 8);
  hipLaunchKernelGGL(k_unitlow, grid1d(m * k), dim3(256), 0, g.stream, V, A,
                     m, k, ld, m);
  PA_CHECK(rocblas_dgemm(h, rocblas_operation_transpose,
                         rocblas_operation_none, k, k, m, &one, V, m, V, m,
                         &zero, G, k) == rocblas_status_success);
  hipLaunchKernelGGL(k_larft_diag, dim3((k + 127) / 128), dim3(128), 0,
                     g.stream, G, k, tau, T, ldt, k);
  double* X = qr_scratch(g, slot0 + 8, (size_t)k * 128 * 8);
  for (int b = 128; b < k; b += 128) {
    int bc2 = std::min(128, k - b);
    // X = T(0:b,0:b) * G(0:b, b:b+bc)
    PA_CHECK(rocblas_dgemm(h, rocblas_operation_none, rocblas_operation_none,
                           b, bc2, b, &one, T, ldt, G + (size_t)b * k, k,
                           &zero, X, b) == rocblas_status_success);
    // T(0:b, blk) = -X * T_blk
    PA_CHECK(rocblas_dgemm(h, rocblas_operation_none, rocblas_operation_none,
                           b, bc2, bc2, &mone, X, b,
                           T + (size_t)b * ldt + b, ldt, &zero,
                           T + (size_t)b * ldt, ldt) == rocblas_status_success);
  }
}

static void gpu_geqrt(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  double* A = (double*)t.dev_ptr[0];
  double* T = (double*)t.dev_ptr[1];
  static const bool use_rocsolver =
      param_str("chore_qr", "hand") == "rocsolver";
  if (!use_rocsolver) {
    qr_factor_hand(g, A, a.n, a.n, a.ld, T, a.ld, 0);
    return;
  }
  double* tau = qr_scratch(g, 0, (size_t)a.n * 8);
  rocblas_handle h = qr_handle(g);
  PA_HIP_CHECK(hipMemsetAsync(T, 0, t.flows[1].data->bytes, g.stream));
  PA_CHECK(rocsolver_dgeqrf(h, a.n, a.n, A, a.ld, tau) ==
           rocblas_status_success);
  PA_CHECK(rocsolver_dlarft(h, rocblas_forward_direction,
                            rocblas_column_wise, a.n, a.n, A, a.ld, tau, T,
                            a.ld) == rocblas_status_success);
}

__global__ void k_unitlow(double* V, const double* A, int m, int k, int lda,
                          int ldv) {
  // V = unit-lower copy of A's reflector block (zeros above, 1 on diag)
  int idx = blockIdx.x * blockDim.x + threadIdx.x;
  int total = m * k;
  for (; idx < total; idx += gridDim.x * blockDim.x) {
    int c = idx / m, r = idx - c * m;
    double v = r < c ? 0.0 : (r == c ? 1.0 : A[(size_t)c * lda + r]);
    V[(size_t)c * ldv + r] = v;
  }
}

// C = (I - V T V^T)^T C via three full-rate dgemms (rocSOLVER's dlarfb
// decomposes into micro-kernels at these sizes: profiles/RESULTS.md).
// V: m x k reflectors (unit-lower inside A-storage), T: k x k upper
// (zeros elsewhere), C: m x n. Scratch: V expanded + W (k x n).
static void larfb_gemm(GpuTaskCtx& g, int m, int n, int k, const double* A,
                       int lda, const double* T, int ldt, double* C, int ldc,
                       int scratch_slot) {
  rocblas_handle h = qr_handle(g);
  double* V = qr_scratch(g, scratch_slot, (size_t)m * k * 8);
  double* W = qr_scratch(g, scratch_slot + 1, (size_t)k * n * 8);
  hipLaunchKernelGGL(k_unitlow, grid1d(m * k), dim3(256), 0, g.stream, V, A,
                     m, k, lda, m);
  const double one = 1.0, zero = 0.0, mone = -1.0;
  // W = V^T C
  PA_CHECK(rocblas_dgemm(h, rocblas_operation_transpose,
                         rocblas_operation_none, k, n, m, &one, V, m, C, ldc,
                         &zero, W, k) == rocblas_status_success);
  // W = T^T W  (T upper-triangular, stored dense with zero lower)
  double* W2 = qr_scratch(g, scratch_slot + 2, (size_t)k * n * 8);
  PA_CHECK(rocblas_dgemm(h, rocblas_operation_transpose,
                         rocblas_operation_none, k, n, k, &one, T, ldt, W, k,
                         &zero, W2, k) == rocblas_status_success);
  // C -= V W2
  PA_CHECK(rocblas_dgemm(h, rocblas_operation_none, rocblas_operation_none,
                         m, n, k, &mone, V, m, W2, k, &one, C,
                         ldc) == rocblas_status_success);
}

static void gpu_unmqr(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  larfb_gemm(g, a.m, a.n, a.m, (const double*)t.dev_ptr[0], a.ld,
             (const double*)t.dev_ptr[1], a.ld, (double*)t.dev_ptr[2], a.ld,
             4);
}

static void gpu_tsqrt(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  const int nb = a.n, ld = a.ld;
  double* Akk = (double*)t.dev_ptr[0];
  double* Amk = (double*)t.dev_ptr[1];
  double* V2 = (double*)t.dev_ptr[2];
  double* T1 = (double*)t.dev_ptr[3];
  rocblas_handle h = qr_handle(g);
  hipLaunchKernelGGL(k_stack_triu, grid1d(nb * nb), dim3(256), 0, g.stream,
                     V2, Akk, Amk, nb, ld, a.k);
  static const bool use_rocsolver =
      param_str("chore_qr", "hand") == "rocsolver";
  if (!use_rocsolver) {
    qr_factor_hand(g, V2, 2 * nb, nb, 2 * nb, T1, ld, 0, /*ts_split=*/nb);
  } else {
    double* tau = qr_scratch(g, 0, (size_t)nb * 8);
    PA_HIP_CHECK(hipMemsetAsync(T1, 0, t.flows[3].data->bytes, g.stream));
    PA_CHECK(rocsolver_dgeqrf(h, 2 * nb, nb, V2, 2 * nb, tau) ==
             rocblas_status_success);
    PA_CHECK(rocsolver_dlarft(h, rocblas_forward_direction,
                              rocblas_column_wise, 2 * nb, nb, V2, 2 * nb,
                              tau, T1, ld) == rocblas_status_success);
  }
  hipLaunchKernelGGL(k_copy_triu, grid1d(nb * nb), dim3(256), 0, g.stream,
                     Akk, V2, nb, 2 * nb, ld);
}

static void gpu_tsmqr(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  const int nb = a.m, cols = a.n, ld = a.ld;
  double* V2 = (double*)t.dev_ptr[0];
  double* T1 = (double*)t.dev_ptr[1];
  double* Ckn = (double*)t.dev_ptr[2];
  double* Cmn = (double*)t.dev_ptr[3];
  double* S = qr_scratch(g, 1, (size_t)2 * nb * cols * 8);
  hipLaunchKernelGGL(k_stack, grid1d(nb * cols), dim3(256), 0, g.stream, S,
                     Ckn, Cmn, nb, cols, ld);
  larfb_gemm(g, 2 * nb, cols, nb, V2, 2 * nb, T1, ld, S, 2 * nb, 8);
  hipLaunchKernelGGL(k_unstack, grid1d(nb * cols), dim3(256), 0, g.stream, S,
                     Ckn, Cmn, nb, cols, ld);
}

// ------------------------------------------------------------ task classes
static TaskClass make_qr_tc(const char* name, void (*cpu)(Task&),
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

TaskClass& tc_geqrt() {
  // rocSOLVER panel factorizations host-sync internally: run them on
  // worker threads (gpu_blocking) so the manager keeps the trailing
  // updates flowing. The hand panel path is fully device-side (multi-WG
  // panel kernel), so it pipelines through the manager like any chore.
  // PARSEC_MCA_qr_blocking_panels=0 restores manager execution.
  static TaskClass tc = [] {
    TaskClass c = make_qr_tc("geqrt", cpu_geqrt, gpu_geqrt, 10);
    bool rocs = param_str("chore_qr", "hand") == "rocsolver";
    c.gpu_blocking = rocs && param_int("qr_blocking_panels", 1) != 0;
    return c;
  }();
  return tc;
}
TaskClass& tc_unmqr() {
  static TaskClass tc = make_qr_tc("unmqr", cpu_unmqr, gpu_unmqr, 11);
  return tc;
}
TaskClass& tc_tsqrt() {
  static TaskClass tc = [] {
    TaskClass c = make_qr_tc("tsqrt", cpu_tsqrt, gpu_tsqrt, 12);
    bool rocs = param_str("chore_qr", "hand") == "rocsolver";
    c.gpu_blocking = rocs && param_int("qr_blocking_panels", 1) != 0;
    return c;
  }();
  return tc;
}
TaskClass& tc_tsmqr() {
  static TaskClass tc = make_qr_tc("tsmqr", cpu_tsmqr, gpu_tsmqr, 13);
  return tc;
}

// ------------------------------------------------------------ DAG builder
// Binary TS-reduction tree per panel column (the reference ecosystem's
// HQR trees): every row tile is GEQRT'd independently (level 0), then
// pairs combine through TSQRT/TSMQR with stride-doubling — panel depth
// O(log M) instead of O(M). More total flops than the flat chain (the
// per-row UNMQR applies), but the host-synced panel kernels run
// CONCURRENTLY on blocking workers, which is what bounds QR here.
static void insert_geqrf_tree(Dtd& tp, TiledMatrix& A, TiledMatrix& WT,
                              TiledMatrix& T1, TiledMatrix& V2) {
  const int T = A.mt();
  const int nb = A.nb(), ld = A.mb();
  constexpr int PANEL = 1 << 20;
  TileArgs pa_args;
  pa_args.m = nb;
  pa_args.n = nb;
  pa_args.ld = ld;
  for (int k = 0; k < T; k++) {
    // level 0: factor every row tile of the column, apply across its row
    for (int m = k; m < T; m++) {
      Dtd::FlowSpec f[] = {{A.tile(m, k), ACCESS_INOUT},
                           {WT.tile(m, k), ACCESS_OUT}};
      tp.insert(&tc_geqrt(), &pa_args, sizeof(pa_args), f, 2, PANEL + 1,
                A.rank_of(m, k));
      for (int n = k + 1; n < T; n++) {
        Dtd::FlowSpec fu[] = {{A.tile(m, k), ACCESS_IN},
                              {WT.tile(m, k), ACCESS_IN},
                              {A.tile(m, n), ACCESS_INOUT}};
        tp.insert(&tc_unmqr(), &pa_args, sizeof(pa_args), fu, 3,
                  (1 << 18) - (n - k), A.rank_of(m, n));
      }
    }
    // reduction tree: combine (a, a+step) pairs, doubling the stride
    for (int step = 1; k + step < T; step *= 2) {
      for (int a = k; a + step < T; a += 2 * step) {
        const int b = a + step;
        {
          TileArgs ta = pa_args;
          ta.k = 1;  // bottom tile participates by its R triangle only
          Dtd::FlowSpec f[] = {{A.tile(a, k), ACCESS_INOUT},
                               {A.tile(b, k), ACCESS_INOUT},
                               {V2.tile(b, k), ACCESS_OUT},
                               {T1.tile(b, k), ACCESS_OUT}};
          tp.insert(&tc_tsqrt(), &ta, sizeof(ta), f, 4, PANEL,
                    A.rank_of(b, k));
        }
        for (int n = k + 1; n < T; n++) {
          Dtd::FlowSpec f[] = {{V2.tile(b, k), ACCESS_IN},
                               {T1.tile(b, k), ACCESS_IN},
                               {A.tile(a, n), ACCESS_INOUT},
                               {A.tile(b, n), ACCESS_INOUT}};
          tp.insert(&tc_tsmqr(), &pa_args, sizeof(pa_args), f, 4,
                    -(n - k) * 4, A.rank_of(b, n));
        }
      }
    }
  }
}

void insert_geqrf(Dtd& tp, TiledMatrix& A) {
  const int T = A.mt();
  const int nb = A.nb(), ld = A.mb();
  PA_CHECK(A.m() % nb == 0 && A.mb() == A.nb(),
           "insert_geqrf: N must be a multiple of the (square) tile size");
  constexpr int PANEL = 1 << 20;
  auto* ctx = A.ctx();
  auto WT = std::make_shared<TiledMatrix>(ctx, A.m(), A.n(), nb, nb,
                                          A.grid_p(), A.grid_q());
  auto T1 = std::make_shared<TiledMatrix>(ctx, A.m(), A.n(), nb, nb,
                                          A.grid_p(), A.grid_q());
  auto V2 = std::make_shared<TiledMatrix>(ctx, (int64_t)2 * A.m(), A.n(),
                                          2 * nb, nb, A.grid_p(), A.grid_q());
  tp.own(WT);
  tp.own(T1);
  tp.own(V2);
  if (param_str("qr_tree", "flat") == "binary") {
    insert_geqrf_tree(tp, A, *WT, *T1, *V2);
    return;
  }
  for (int k = 0; k < T; k++) {
    TileArgs pa_args;
    pa_args.m = nb;
    pa_args.n = nb;
    pa_args.ld = ld;
    {
      Dtd::FlowSpec f[] = {{A.tile(k, k), ACCESS_INOUT},
                           {WT->tile(k, k), ACCESS_OUT}};
      tp.insert(&tc_geqrt(), &pa_args, sizeof(pa_args), f, 2, PANEL + 1,
                A.rank_of(k, k));
    }
    for (int n = k + 1; n < T; n++) {
      Dtd::FlowSpec f[] = {{A.tile(k, k), ACCESS_IN},
                           {WT->tile(k, k), ACCESS_IN},
                           {A.tile(k, n), ACCESS_INOUT}};
      tp.insert(&tc_unmqr(), &pa_args, sizeof(pa_args), f, 3,
                (1 << 18) - (n - k), A.rank_of(k, n));
    }
    for (int m = k + 1; m < T; m++) {
      {
        Dtd::FlowSpec f[] = {{A.tile(k, k), ACCESS_INOUT},
                             {A.tile(m, k), ACCESS_INOUT},
                             {V2->tile(m, k), ACCESS_OUT},
                             {T1->tile(m, k), ACCESS_OUT}};
        tp.insert(&tc_tsqrt(), &pa_args, sizeof(pa_args), f, 4, PANEL,
                  A.rank_of(m, k));
      }
      for (int n = k + 1; n < T; n++) {
        Dtd::FlowSpec f[] = {{V2->tile(m, k), ACCESS_IN},
                             {T1->tile(m, k), ACCESS_IN},
                             {A.tile(k, n), ACCESS_INOUT},
                             {A.tile(m, n), ACCESS_INOUT}};
        tp.insert(&tc_tsmqr(), &pa_args, sizeof(pa_args), f, 4,
                  -(n - k) * 4, A.rank_of(m, n));
      }
    }
  }
}

// Standalone micro-benchmark of the TS/tile panel factorization path:
// mode 0 = full qr_factor_hand, 1 = panel kernels only, 2 = rocsolver
// dgeqrf. Returns seconds for `iters` factorizations of an m x k tile.
double bench_qr_factor(int m, int k, int ts_split, int iters, int mode) {
  double *dA, *dT;
  PA_HIP_CHECK(hipMalloc(&dA, (size_t)m * k * 8));
  PA_HIP_CHECK(hipMalloc(&dT, (size_t)k * k * 8));
  hipStream_t s;
  PA_HIP_CHECK(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
  std::vector<std::pair<void*, size_t>> defer;
  GpuTaskCtx g{s, 0, nullptr, &defer};
  double* tau = qr_scratch(g, 0, (size_t)k * 8);
  double* T16s = qr_scratch(g, 9, (size_t)8 * 256 * 8);
  int* cnt = (int*)qr_scratch(g, 10, 256);
  unsigned long long* dbg = nullptr;
  if (mode == 5) {
    PA_HIP_CHECK(hipMalloc(&dbg, 16 * 8));
    PA_HIP_CHECK(hipMemset(dbg, 0, 16 * 8));
    mode = 1;
  }
  const int nwg = std::max(2, (int)param_int("qr_panel_wgs", 48));
  const int nA = std::min((int64_t)nwg - 1, param_int("qr_panel_poolA", 15));
  auto run = [&] {
    hipLaunchKernelGGL(k_qr_fill, dim3(2048), dim3(256), 0, s, dA,
                       (size_t)m * k, 7);
    if (mode == 0) {
      qr_factor_hand(g, dA, m, k, m, dT, k, 0, ts_split);
    } else if (mode == 1 || mode >= 16) {
      for (int p = 0; p < k; p += 128) {
        int pc = std::min(128, k - p);
        int base0 = p, len0 = m - p, base1 = 0, len1 = 0;
        if (ts_split > 0) {
          len0 = std::min(pc, ts_split - p);
          base1 = ts_split;
          len1 = m - ts_split;
        }
        int rows = len0 + len1;
        int W = rows <= QR_MAX_ROWS_W16 ? 16
                                        : (rows <= QR_MAX_ROWS_W8 ? 8 : 0);
        PA_CHECK(W, "bench: rows too large");
        PA_HIP_CHECK(hipMemsetAsync(cnt, 0, 2 * sizeof(int), s));
        hipLaunchKernelGGL(k_qr_panel_mw, dim3(nwg), dim3(1024), 0, s, dA,
                           m, p, base0, len0, base1, len1, k - p, pc, W,
                           tau + p, T16s, cnt, nwg, nA,
                           param_str("qr_apply", "kernel") == "valu" ? 0 : 1,
                           (int)param_int("qr_lookahead", 1), dbg);
      }
    } else {
      PA_CHECK(rocsolver_dgeqrf(qr_handle(g), m, k, dA, m, tau) ==
               rocblas_status_success);
    }
  };
  run();
  PA_HIP_CHECK(hipStreamSynchronize(s));
  double t0 = now_s();
  for (int i = 0; i < iters; i++) run();
  PA_HIP_CHECK(hipStreamSynchronize(s));
  double dt = now_s() - t0;
  if (dbg) {
    unsigned long long h[16];
    PA_HIP_CHECK(hipMemcpy(h, dbg, 16 * 8, hipMemcpyDeviceToHost));
    const char* names[] = {"wg0 ack-wait", "wg0 stage", "wg0 factor",
                           "wg0 scale+wb", "wg0 G16+T16", "A poll",
                           "A vbuild", "A apply", "B poll", "B vbuild",
                           "B apply"};
    for (int i = 0; i < 11; i++)
      fprintf(stderr, "[qrdbg] %-12s %8.1f us/tile\n", names[i],
              h[i] / 0.1 / (iters + 1));  // 100 MHz ticks
    PA_HIP_CHECK(hipFree(dbg));
  }
  PA_HIP_CHECK(hipFree(dA));
  PA_HIP_CHECK(hipFree(dT));
  return dt;
}

}  // namespace pa
