// Hand-written CDNA4 (gfx950) fp64 MFMA tile kernels.
//
// The reference ships no compute kernels (JDF bodies call
// cuBLAS); these are the hand-optimized chores for the headline dense
// apps, written for the gfx950 execution model directly.
//
// v_mfma_f64_16x16x4f64: one wave computes a 16x16 fp64 tile with K-step 4,
// 2048 FLOP per instruction at ~64 cycles/SIMD (the fp64 matrix rate equals
// the fp64 vector rate on MI355X, ~78.6 TF/s chip peak). At that cadence a
// 4-wave 128x128 LDS-staged block is compute-bound: 8 ds_read_b64 feed
// 16 MFMAs (1024 SIMD-cycles), so staging and even bank conflicts hide
// entirely under the matrix pipe — the structure below aims for clean
// global coalescing and 16 independent accumulators per wave, which is what
// the fp64 pipe actually needs (MI355X_MICROARCH.md: per-instruction
// constants; fragment maps per cdna4 ISA: A[i=l&15][k=l>>4],
// B[k=l>>4][j=l&15], C row=(l>>4)*4+e, col=l&15).
//
// Kernel set: dgemm_nt (C -= A*B^T — the Cholesky trailing update, also
// reused by SYRK on the full tile), with bounds-checked edges. rocBLAS
// remains the alternate chore; select with PARSEC_MCA_chore_gemm=hip|rocblas.
#include <hip/hip_runtime.h>

#include "device_gpu.hpp"
#include "kernels.hpp"

namespace pa {

typedef double f64x4 __attribute__((ext_vector_type(4)));

#define BM 128
#define BN 128
#define BKD 16

__launch_bounds__(256)
__global__ void k_dgemm_nt(int m, int n, int k, const double* __restrict__ A,
                           int lda, const double* __restrict__ B, int ldb,
                           double* __restrict__ C, int ldc) {
  // C[m x n] -= A[m x k] * B[n x k]^T, column-major.
  __shared__ double As[BKD][BM + 1];
  __shared__ double Bs[BKD][BN + 1];
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wr = wave >> 1, wc = wave & 1;  // 2x2 waves of 64x64 output
  const int bm0 = blockIdx.x * BM, bn0 = blockIdx.y * BN;
  const int ksub = lane >> 4;   // k offset within the 4-deep MFMA step
  const int r16 = lane & 15;

  f64x4 acc[4][4] = {};

  for (int k0 = 0; k0 < k; k0 += BKD) {
    // Stage A-panel [BM][BKD] and B-panel [BN][BKD] k-major in LDS.
    // 256 threads x 8 elements; global access is column-contiguous.
    for (int x = threadIdx.x; x < BM * BKD; x += 256) {
      int i = x & (BM - 1), kk = x >> 7;
      int gi = bm0 + i, gk = k0 + kk;
      As[kk][i] = (gi < m && gk < k) ? A[(size_t)gk * lda + gi] : 0.0;
    }
    for (int x = threadIdx.x; x < BN * BKD; x += 256) {
      int j = x & (BN - 1), kk = x >> 7;
      int gj = bn0 + j, gk = k0 + kk;
      Bs[kk][j] = (gj < n && gk < k) ? B[(size_t)gk * ldb + gj] : 0.0;
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < BKD; kk += 4) {
      double a[4], b[4];
#pragma unroll
      for (int f = 0; f < 4; f++) a[f] = As[kk + ksub][wr * 64 + f * 16 + r16];
#pragma unroll
      for (int f = 0; f < 4; f++) b[f] = Bs[kk + ksub][wc * 64 + f * 16 + r16];
#pragma unroll
      for (int i = 0; i < 4; i++)
#pragma unroll
        for (int j = 0; j < 4; j++)
          acc[i][j] =
              __builtin_amdgcn_mfma_f64_16x16x4f64(a[i], b[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  // f64 16x16x4 C/D lane map (verified on gfx950 hardware): element e of
  // lane l holds D[4*e + (l>>4)][l&15] — register index strides ROWS by 4,
  // unlike the f32/bf16 16x16 map where (l>>4) selects the 4-row group.
#pragma unroll
  for (int j = 0; j < 4; j++) {
    int col = bn0 + wc * 64 + j * 16 + r16;
    if (col >= n) continue;
    double* cp = C + (size_t)col * ldc;
#pragma unroll
    for (int i = 0; i < 4; i++) {
      int row0 = bm0 + wr * 64 + i * 16 + ksub;
#pragma unroll
      for (int e = 0; e < 4; e++) {
        int row = row0 + e * 4;
        if (row < m) cp[row] -= acc[i][j][e];
      }
    }
  }
}

// ------------------------------------------------------------------ v2
// 8-wave 128x128 double-buffered variant tuned for occupancy: acc is 2x4
// fragments per wave (64 VGPR), LDS 66 KB -> 2 blocks/CU co-resident, which
// hides each block's staging under the other's MFMA phase (fp64 MFMA issue
// is 64 cyc/SIMD, so 4 co-resident waves/SIMD keep the matrix pipe fed
// without hand pipelining). 1-D grid with a bijective XCD-aware remap so
// consecutive blocks share B panels within one XCD's L2 (guide T1).
__device__ __forceinline__ int xcd_swizzle(int id, int nwg) {
  int q = nwg >> 3, r = nwg & 7;
  int xcd = id & 7, pos = id >> 3;
  // blocks per XCD: first r XCDs get q+1
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
}

template <bool TRI>
__launch_bounds__(512)
__global__ void k_dgemm_nt_v2(int m, int n, int k, const double* __restrict__ A,
                              int lda, const double* __restrict__ B, int ldb,
                              double* __restrict__ C, int ldc, int nbx) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr int LDT = 129;
  double* As = (double*)smem;                  // [2][BKD][LDT]
  double* Bs = As + 2 * BKD * LDT;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wr = wave >> 1, wc = wave & 1;  // 4(M) x 2(N) waves: 32x64 each
  const int nwg = gridDim.x;
  int id = xcd_swizzle(blockIdx.x, nwg);
  int bx, by;
  if (TRI) {
    // lower-triangular block set: id -> (bx >= by); SYRK writes only the
    // lower blocks (diag blocks compute their full tile: the upper half of
    // a diagonal tile is symmetric-valid and never read by the DAG).
    bx = 0;
    int rem = id;
    while (rem > bx) { bx++; rem -= bx; }
    by = rem;
  } else {
    bx = id % nbx;
    by = id / nbx;
  }
  const int bm0 = bx * BM, bn0 = by * BN;
  const int ksub = lane >> 4;
  const int r16 = lane & 15;

  f64x4 acc[2][4] = {};

  auto stage = [&](int buf, int k0) {
    double* as = As + buf * BKD * LDT;
    double* bs = Bs + buf * BKD * LDT;
#pragma unroll
    for (int x = 0; x < BM * BKD; x += 512) {
      int xi = x + tid;
      int i = xi & (BM - 1), kk = xi >> 7;
      int gi = bm0 + i, gk = k0 + kk;
      as[kk * LDT + i] = (gi < m && gk < k) ? A[(size_t)gk * lda + gi] : 0.0;
    }
#pragma unroll
    for (int x = 0; x < BN * BKD; x += 512) {
      int xi = x + tid;
      int j = xi & (BN - 1), kk = xi >> 7;
      int gj = bn0 + j, gk = k0 + kk;
      bs[kk * LDT + j] = (gj < n && gk < k) ? B[(size_t)gk * ldb + gj] : 0.0;
    }
  };

  stage(0, 0);
  __syncthreads();
  const int ntiles = (k + BKD - 1) / BKD;
  for (int t = 0; t < ntiles; t++) {
    if (t + 1 < ntiles) stage((t + 1) & 1, (t + 1) * BKD);
    const double* as = As + (t & 1) * BKD * LDT;
    const double* bs = Bs + (t & 1) * BKD * LDT;
#pragma unroll
    for (int kk = 0; kk < BKD; kk += 4) {
      double a[2], b[4];
#pragma unroll
      for (int f = 0; f < 2; f++)
        a[f] = as[(kk + ksub) * LDT + wr * 32 + f * 16 + r16];
#pragma unroll
      for (int f = 0; f < 4; f++)
        b[f] = bs[(kk + ksub) * LDT + wc * 64 + f * 16 + r16];
#pragma unroll
      for (int i = 0; i < 2; i++)
#pragma unroll
        for (int j = 0; j < 4; j++)
          acc[i][j] =
              __builtin_amdgcn_mfma_f64_16x16x4f64(a[i], b[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

#pragma unroll
  for (int j = 0; j < 4; j++) {
    int col = bn0 + wc * 64 + j * 16 + r16;
    if (col >= n) continue;
    double* cp = C + (size_t)col * ldc;
#pragma unroll
    for (int i = 0; i < 2; i++) {
      int row0 = bm0 + wr * 32 + i * 16 + ksub;
#pragma unroll
      for (int e = 0; e < 4; e++) {
        int row = row0 + e * 4;
        if (row < m) cp[row] -= acc[i][j][e];
      }
    }
  }
}

// ------------------------------------------------------------------ v3
// glds-staged variant (the structure that lifted the bf16 kernel 2x):
// async global->LDS (16 B/lane), [k][m] lane-linear LDS image (one K-row =
// 1 KB per glds), double-buffered BK=16, one drain barrier per K-tile.
// Full tiles only; edges fall back to v2.
template <bool TRI>
__launch_bounds__(512)
__global__ void k_dgemm_nt_v3(int m, int n, int k, const double* __restrict__ A,
                              int lda, const double* __restrict__ B, int ldb,
                              double* __restrict__ C, int ldc, int nbx) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  double* As = (double*)smem;                  // [2][BKD][BM]
  double* Bs = As + 2 * BKD * BM;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int wr = wave >> 1, wc = wave & 1;
  int id = xcd_swizzle(blockIdx.x, gridDim.x);
  int bx, by;
  if (TRI) {
    bx = 0;
    int rem = id;
    while (rem > bx) { bx++; rem -= bx; }
    by = rem;
  } else {
    bx = id % nbx;
    by = id / nbx;
  }
  const int bm0 = bx * BM, bn0 = by * BN;
  const int ksub = lane >> 4, r16 = lane & 15;

  f64x4 acc[2][4] = {};

  auto stage = [&](int buf, int k0) {
#pragma unroll
    for (int p = 0; p < 2; p++) {
      int krow = wave * 2 + p;
      const double* srcA = A + (size_t)(k0 + krow) * lda + bm0 + 2 * lane;
      double* dstA = As + buf * BKD * BM + krow * BM;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcA,
          (__attribute__((address_space(3))) unsigned int*)dstA, 16, 0, 0);
      const double* srcB = B + (size_t)(k0 + krow) * ldb + bn0 + 2 * lane;
      double* dstB = Bs + buf * BKD * BN + krow * BN;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcB,
          (__attribute__((address_space(3))) unsigned int*)dstB, 16, 0, 0);
    }
  };

  stage(0, 0);
  __syncthreads();
  const int ntiles = k / BKD;
  for (int t = 0; t < ntiles; t++) {
    if (t + 1 < ntiles) stage((t + 1) & 1, (t + 1) * BKD);
    const double* as = As + (t & 1) * BKD * BM;
    const double* bs = Bs + (t & 1) * BKD * BN;
#pragma unroll
    for (int kk = 0; kk < BKD; kk += 4) {
      double a[2], b[4];
#pragma unroll
      for (int f = 0; f < 2; f++)
        a[f] = as[(kk + ksub) * BM + wr * 32 + f * 16 + r16];
#pragma unroll
      for (int f = 0; f < 4; f++)
        b[f] = bs[(kk + ksub) * BN + wc * 64 + f * 16 + r16];
#pragma unroll
      for (int i = 0; i < 2; i++)
#pragma unroll
        for (int j = 0; j < 4; j++)
          acc[i][j] =
              __builtin_amdgcn_mfma_f64_16x16x4f64(a[i], b[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

#pragma unroll
  for (int j = 0; j < 4; j++) {
    int col = bn0 + wc * 64 + j * 16 + r16;
    double* cp = C + (size_t)col * ldc;
#pragma unroll
    for (int i = 0; i < 2; i++) {
      int row0 = bm0 + wr * 32 + i * 16 + ksub;
#pragma unroll
      for (int e = 0; e < 4; e++) cp[row0 + e * 4] -= acc[i][j][e];
    }
  }
}

static void launch_dgemm_v2(int m, int n, int k, const double* A, int lda,
                            const double* B, int ldb, double* C, int ldc,
                            hipStream_t stream) {
  if (m % BM == 0 && n % BN == 0 && k % BKD == 0) {
    constexpr size_t lds = 2 * BKD * (BM + BN) * 8;
    static bool attr3 = false;
    if (!attr3) {
      hipFuncSetAttribute((const void*)k_dgemm_nt_v3<false>,
                          hipFuncAttributeMaxDynamicSharedMemorySize, lds);
      hipFuncSetAttribute((const void*)k_dgemm_nt_v3<true>,
                          hipFuncAttributeMaxDynamicSharedMemorySize, lds);
      attr3 = true;
    }
    int nbx = (m + BM - 1) / BM, nby = (n + BN - 1) / BN;
    hipLaunchKernelGGL(k_dgemm_nt_v3<false>, dim3(nbx * nby), dim3(512), lds,
                       stream, m, n, k, A, lda, B, ldb, C, ldc, nbx);
    return;
  }
  static bool attr_set = false;
  constexpr size_t lds = 4 * BKD * 129 * 8;
  if (!attr_set) {
    hipFuncSetAttribute((const void*)k_dgemm_nt_v2<false>,
                        hipFuncAttributeMaxDynamicSharedMemorySize, lds);
    attr_set = true;
  }
  int nbx = (m + BM - 1) / BM, nby = (n + BN - 1) / BN;
  hipLaunchKernelGGL(k_dgemm_nt_v2<false>, dim3(nbx * nby), dim3(512), lds,
                     stream, m, n, k, A, lda, B, ldb, C, ldc, nbx);
}

// C(lower) -= A A^T : triangular block grid on the same structure.
void launch_dsyrk_v2(int n, int k, const double* A, int lda, double* C,
                     int ldc, hipStream_t stream) {
  static bool attr_set = false;
  constexpr size_t lds = 4 * BKD * 129 * 8;
  if (!attr_set) {
    hipFuncSetAttribute((const void*)k_dgemm_nt_v2<true>,
                        hipFuncAttributeMaxDynamicSharedMemorySize, lds);
    attr_set = true;
  }
  int ntb = (n + BM - 1) / BM;
  int ntiles = ntb * (ntb + 1) / 2;
  if (n % BM == 0 && k % BKD == 0) {
    constexpr size_t lds3 = 2 * BKD * (BM + BN) * 8;
    static bool attr3t = false;
    if (!attr3t) {
      hipFuncSetAttribute((const void*)k_dgemm_nt_v3<true>,
                          hipFuncAttributeMaxDynamicSharedMemorySize, lds3);
      attr3t = true;
    }
    hipLaunchKernelGGL(k_dgemm_nt_v3<true>, dim3(ntiles), dim3(512), lds3,
                       stream, n, n, k, A, lda, A, lda, C, ldc, 0);
    return;
  }
  hipLaunchKernelGGL(k_dgemm_nt_v2<true>, dim3(ntiles), dim3(512), lds,
                     stream, n, n, k, A, lda, A, lda, C, ldc, 0);
}

void gpu_gemm_hip(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  launch_dgemm_v2(a.m, a.n, a.k, (const double*)t.dev_ptr[0], a.ld,
                  (const double*)t.dev_ptr[1], a.ld, (double*)t.dev_ptr[2],
                  a.ld, g.stream);
}

// ---------------------------------------------------------------- potf2 v2
// Unblocked-in-LDS Cholesky of a <=128 fp64 panel with MFMA trailing
// updates: 16-column micro-panels factor scalar-sequentially (cheap), the
// rank-16 trailing update runs on v_mfma_f64_16x16x4f64 over 16x16 LDS
// tiles. One workgroup (4 waves); replaces both rocSOLVER's potf2 (~233 us)
// and the scalar LDS version (~830 us) — the tile-POTRF panel is the
// critical path of the whole Cholesky DAG.
__global__ void __launch_bounds__(256) k_potf2_mfma(double* A, int n, int ld) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr int LDP = 129;
  double* S = (double*)smem;     // [<=128][LDP] column-major
  double* dinv = S + 128 * LDP;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  for (int x = tid; x < n * n; x += 256) {
    int j = x / n, i = x - j * n;
    if (i >= j) S[j * LDP + i] = A[(size_t)j * ld + i];
  }
  __syncthreads();
  for (int jb = 0; jb < n; jb += 16) {
    const int jbe = jb + 16 < n ? jb + 16 : n;
    // ---- factor the 16-column micro-panel (rows jb..n) ----
    for (int j = jb; j < jbe; j++) {
      if (tid == 0) {
        double d = S[j * LDP + j];
        S[j * LDP + j] = d = sqrt(d);
        *dinv = 1.0 / d;
      }
      __syncthreads();
      for (int i = j + 1 + tid; i < n; i += 256) S[j * LDP + i] *= *dinv;
      __syncthreads();
      // independent trailing columns of the micro-panel: one wave each
      for (int c = j + 1 + wave; c < jbe; c += 4) {
        double ljc = S[j * LDP + c];
        for (int i = c + lane; i < n; i += 64)
          S[c * LDP + i] -= S[j * LDP + i] * ljc;
      }
      __syncthreads();
    }
    // ---- rank-16 MFMA trailing update: S[t,t] -= P P^T, t >= jb+16 ----
    const int t0 = jbe;
    const int nt = (n - t0 + 15) >> 4;  // trailing 16-blocks
    if (nt > 0) {
      const int ntiles = nt * (nt + 1) / 2;
      const int r16 = lane & 15, g4 = lane >> 4;
      for (int idx = wave; idx < ntiles; idx += 4) {
        // idx -> (ti, tj) over the lower-triangular block set
        int ti = 0, rem = idx;
        while (rem > ti) { ti++; rem -= ti; }
        int tj = rem;
        const int ro = t0 + ti * 16, co = t0 + tj * 16;
        const int arow = ro + r16, brow = co + r16;
        f64x4 acc;
#pragma unroll
        for (int e = 0; e < 4; e++) {
          int r = ro + 4 * e + g4, c = co + r16;
          acc[e] = (r < n && c < n) ? S[c * LDP + r] : 0.0;
        }
#pragma unroll
        for (int kk = 0; kk < 16; kk += 4) {
          int kcol = jb + kk + g4;
          double a = (arow < n) ? -S[kcol * LDP + arow] : 0.0;
          double b = (brow < n) ? S[kcol * LDP + brow] : 0.0;
          acc = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc, 0, 0, 0);
        }
#pragma unroll
        for (int e = 0; e < 4; e++) {
          int r = ro + 4 * e + g4, c = co + r16;
          if (r < n && c < n && r >= c) S[c * LDP + r] = acc[e];
        }
      }
    }
    __syncthreads();
  }
  for (int x = tid; x < n * n; x += 256) {
    int j = x / n, i = x - j * n;
    if (i >= j) A[(size_t)j * ld + i] = S[j * LDP + i];
  }
}

void launch_potf2(double* A, int n, int ld, hipStream_t stream) {
  PA_CHECK(n <= 128);
  constexpr size_t lds = 128 * 129 * 8 + 16;
  static bool attr_set = false;
  if (!attr_set) {
    hipFuncSetAttribute((const void*)k_potf2_mfma,
                        hipFuncAttributeMaxDynamicSharedMemorySize, lds);
    attr_set = true;
  }
  hipLaunchKernelGGL(k_potf2_mfma, dim3(1), dim3(256), lds, stream, A, n, ld);
}

// ------------------------------------------------------------ test harness
// Standalone host entry for numerics tests: C -= A*B^T on device, host I/O.
void test_dgemm_nt_hip(int m, int n, int k, const double* A, int lda,
                       const double* B, int ldb, double* C, int ldc) {
  double *dA, *dB, *dC;
  PA_HIP_CHECK(hipMalloc(&dA, (size_t)lda * k * 8));
  PA_HIP_CHECK(hipMalloc(&dB, (size_t)ldb * k * 8));
  PA_HIP_CHECK(hipMalloc(&dC, (size_t)ldc * n * 8));
  PA_HIP_CHECK(hipMemcpy(dA, A, (size_t)lda * k * 8, hipMemcpyHostToDevice));
  PA_HIP_CHECK(hipMemcpy(dB, B, (size_t)ldb * k * 8, hipMemcpyHostToDevice));
  PA_HIP_CHECK(hipMemcpy(dC, C, (size_t)ldc * n * 8, hipMemcpyHostToDevice));
  launch_dgemm_v2(m, n, k, dA, lda, dB, ldb, dC, ldc, 0);
  PA_HIP_CHECK(hipGetLastError());
  PA_HIP_CHECK(hipMemcpy(C, dC, (size_t)ldc * n * 8, hipMemcpyDeviceToHost));
  PA_HIP_CHECK(hipFree(dA));
  PA_HIP_CHECK(hipFree(dB));
  PA_HIP_CHECK(hipFree(dC));
}

__global__ void k_bench_fill(double* p, size_t n, uint32_t seed) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n; i += (size_t)gridDim.x * blockDim.x) {
    uint64_t h = (i * 2654435761ull) ^ ((uint64_t)seed * 2246822519ull);
    h ^= h >> 13;
    h *= 0x9E3779B97F4A7C15ull;
    h ^= h >> 32;
    p[i] = (double)(h & 0xFFFFFF) / (double)0x1000000 - 0.5;
  }
}

// Timed device-resident GEMM loop: impl 0 = hand v2, 1 = hand v1.
// rocBLAS comparison lives in kernels_blas.cpp. Returns seconds.
double bench_dgemm_hip(int m, int n, int k, int iters, int impl) {
  double *dA, *dB, *dC;
  PA_HIP_CHECK(hipMalloc(&dA, (size_t)m * k * 8));
  PA_HIP_CHECK(hipMalloc(&dB, (size_t)n * k * 8));
  PA_HIP_CHECK(hipMalloc(&dC, (size_t)m * n * 8));
  // random-ish fill (never bench on zero-filled operands — DVFS inflates)
  auto fill = [](double* p, size_t nelem, uint32_t seed) {
    hipLaunchKernelGGL(k_bench_fill, dim3(2048), dim3(256), 0, 0, p, nelem,
                       seed);
  };
  fill(dA, (size_t)m * k, 11);
  fill(dB, (size_t)n * k, 12);
  fill(dC, (size_t)m * n, 13);
  auto run = [&] {
    if (impl == 0) launch_dgemm_v2(m, n, k, dA, m, dB, n, dC, m, 0);
    else {
      dim3 grid((m + BM - 1) / BM, (n + BN - 1) / BN);
      hipLaunchKernelGGL(k_dgemm_nt, grid, dim3(256), 0, 0, m, n, k, dA, m,
                         dB, n, dC, m);
    }
  };
  run();
  PA_HIP_CHECK(hipDeviceSynchronize());
  double t0 = now_s();
  for (int i = 0; i < iters; i++) run();
  PA_HIP_CHECK(hipDeviceSynchronize());
  double dt = now_s() - t0;
  PA_HIP_CHECK(hipFree(dA));
  PA_HIP_CHECK(hipFree(dB));
  PA_HIP_CHECK(hipFree(dC));
  return dt;
}

void test_potf2_hip(double* A, int n) {
  double* dA;
  PA_HIP_CHECK(hipMalloc(&dA, (size_t)n * n * 8));
  PA_HIP_CHECK(hipMemcpy(dA, A, (size_t)n * n * 8, hipMemcpyHostToDevice));
  launch_potf2(dA, n, n, 0);
  PA_HIP_CHECK(hipGetLastError());
  PA_HIP_CHECK(hipMemcpy(A, dA, (size_t)n * n * 8, hipMemcpyDeviceToHost));
  PA_HIP_CHECK(hipFree(dA));
}

}  // namespace pa
