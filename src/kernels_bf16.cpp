// bf16 MFMA tile-GEMM DAG (BASELINE.json config 5).
//
// Hand-written CDNA4 kernel on v_mfma_f32_16x16x32_bf16 (gfx950 2xK form,
// ~2.5 PF dense chip peak). Convention: C[m][n] (fp32, col-major) +=
// sum_k At(k,m)^T * B(k,n) with A and B tiles stored K-major (kb x mb /
// kb x nb, col-major) — a TN GEMM, which makes BOTH operand fragments
// K-contiguous so LDS reads are single ds_read_b128 per fragment
// (cdna_hip_programming.md §5: the B^T-input convention of the ladder).
//
// Structure (correctness-first step of the guide's ladder): 128x128 block,
// 4 waves (2x2 of 64x64), BK=32, reg-staged LDS with +16B row pad, 4x4
// fp32x4 accumulators per wave, XCD-aware block swizzle. glds/8-phase
// pipelining are the next rungs (tracked in docs/DESIGN.md).
#include <cmath>

#include <hip/hip_runtime.h>

#include "device_gpu.hpp"
#include "kernels.hpp"
#include "profiling.hpp"

namespace pa {

// The reference ships no compute kernels (GPU bodies come
// from applications/cuBLAS); this bf16 MFMA tile-GEMM is the
// BASELINE.json config-5 headline app, written CDNA4-first.

typedef __bf16 bf16;
typedef short bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define GB_BM 128
#define GB_BN 128
#define GB_BK 32
#define GB_PAD 8  // bf16 elements of row padding (16 B): bank de-phasing

__device__ __forceinline__ int bf_swz(int id, int nwg) {
  int q = nwg >> 3, r = nwg & 7;
  int xcd = id & 7, pos = id >> 3;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
}

__launch_bounds__(256)
__global__ void k_gemm_bf16_tn(int m, int n, int k,
                               const bf16* __restrict__ A, int lda,
                               const bf16* __restrict__ B, int ldb,
                               float* __restrict__ C, int ldc, int nbx,
                               int accum) {
  // A: k x m col-major (lda >= k), B: k x n col-major, C: m x n col-major.
  constexpr int LDS_K = GB_BK + GB_PAD;
  __shared__ bf16 As[GB_BM * LDS_K];
  __shared__ bf16 Bs[GB_BN * LDS_K];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int wr = wave >> 1, wc = wave & 1;
  int id = bf_swz(blockIdx.x, gridDim.x);
  const int bm0 = (id % nbx) * GB_BM, bn0 = (id / nbx) * GB_BN;
  const int g16 = lane >> 4, r16 = lane & 15;

  f32x4 acc[4][4] = {};

  for (int k0 = 0; k0 < k; k0 += GB_BK) {
    // stage: each thread copies 16 B (8 bf16 along K) per row chunk.
    // 128 rows x (32/8=4) chunks = 512 slots for A, same for B.
    for (int x = tid; x < GB_BM * (GB_BK / 8); x += 256) {
      int row = x >> 2, ck = (x & 3) * 8;
      int gm = bm0 + row, gk = k0 + ck;
      bf16x8 v = {};
      if (gm < m && gk + 7 < k) {
        v = *(const bf16x8*)&A[(size_t)gm * lda + gk];
      } else if (gm < m) {
        for (int e = 0; e < 8 && gk + e < k; e++)
          ((short*)&v)[e] = ((const short*)A)[(size_t)gm * lda + gk + e];
      }
      *(bf16x8*)&As[row * LDS_K + ck] = v;
    }
    for (int x = tid; x < GB_BN * (GB_BK / 8); x += 256) {
      int row = x >> 2, ck = (x & 3) * 8;
      int gn = bn0 + row, gk = k0 + ck;
      bf16x8 v = {};
      if (gn < n && gk + 7 < k) {
        v = *(const bf16x8*)&B[(size_t)gn * ldb + gk];
      } else if (gn < n) {
        for (int e = 0; e < 8 && gk + e < k; e++)
          ((short*)&v)[e] = ((const short*)B)[(size_t)gn * ldb + gk + e];
      }
      *(bf16x8*)&Bs[row * LDS_K + ck] = v;
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < GB_BK; kk += 32) {
      // fragment: lane holds 8 bf16 at K offset kk + g16*8
      bf16x8 a[4], b[4];
#pragma unroll
      for (int f = 0; f < 4; f++)
        a[f] = *(const bf16x8*)&As[(wr * 64 + f * 16 + r16) * LDS_K + kk + g16 * 8];
#pragma unroll
      for (int f = 0; f < 4; f++)
        b[f] = *(const bf16x8*)&Bs[(wc * 64 + f * 16 + r16) * LDS_K + kk + g16 * 8];
#pragma unroll
      for (int i = 0; i < 4; i++)
#pragma unroll
        for (int j = 0; j < 4; j++)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[i], b[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  // C/D map (guide §3): col = lane&15, row = (lane>>4)*4 + e
#pragma unroll
  for (int j = 0; j < 4; j++) {
    int col = bn0 + wc * 64 + j * 16 + r16;
    if (col >= n) continue;
    float* cp = C + (size_t)col * ldc;
#pragma unroll
    for (int i = 0; i < 4; i++) {
      int row0 = bm0 + wr * 64 + i * 16 + g16 * 4;
#pragma unroll
      for (int e = 0; e < 4; e++) {
        int row = row0 + e;
        if (row < m) cp[row] = accum ? cp[row] + acc[i][j][e] : acc[i][j][e];
      }
    }
  }
}

// ---------------------------------------------------------------- v2: glds
// Step-3 of the guide's optimization ladder: async global->LDS staging via
// `global_load_lds` (16 B per lane, wave-uniform LDS base), double-buffered
// K-tiles (BK=64), one vmcnt-drain barrier per tile. The LDS image is
// lane-linear ([row][K] bf16, K fastest), which the TN tile layout feeds
// with contiguous 16 B per lane. Full tiles only; edge tiles take the
// reg-staged kernel above.
#define GB2_BK 64

__launch_bounds__(256)
__global__ void k_gemm_bf16_tn_v2(int m, int n, int k,
                                  const bf16* __restrict__ A, int lda,
                                  const bf16* __restrict__ B, int ldb,
                                  float* __restrict__ C, int ldc, int nbx,
                                  int accum) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* As = (bf16*)smem;                      // [2][128][GB2_BK]
  bf16* Bs = As + 2 * GB_BM * GB2_BK;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int wr = wave >> 1, wc = wave & 1;
  int id = bf_swz(blockIdx.x, gridDim.x);
  const int bm0 = (id % nbx) * GB_BM, bn0 = (id / nbx) * GB_BN;
  const int g16 = lane >> 4, r16 = lane & 15;
  const int srow = lane >> 3;  // staging lane map (8 rows x 8 K-chunks)

  f32x4 acc[4][4] = {};

  auto stage = [&](int buf, int k0) {
    // 4 pieces of 8 rows per wave for each operand: 64 lanes x 16 B = 1 KB
    // per glds instruction, lane-linear into [row][K].
#pragma unroll
    for (int p = 0; p < 4; p++) {
      int row0 = (wave * 4 + p) * 8;
      // T2 bank swizzle via the SOURCE address (glds writes lane-linear,
      // rule 21): element k-offset for this lane = 8*((lane&7) ^ (row&7)),
      // matched by the XOR on the read side.
      int skc = (((lane & 7) ^ ((row0 + srow) & 7)) * 8);
      const bf16* srcA = A + (size_t)(bm0 + row0 + srow) * lda + k0 + skc;
      bf16* dstA = As + buf * GB_BM * GB2_BK + row0 * GB2_BK;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcA,
          (__attribute__((address_space(3))) unsigned int*)dstA, 16, 0, 0);
      const bf16* srcB = B + (size_t)(bn0 + row0 + srow) * ldb + k0 + skc;
      bf16* dstB = Bs + buf * GB_BN * GB2_BK + row0 * GB2_BK;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcB,
          (__attribute__((address_space(3))) unsigned int*)dstB, 16, 0, 0);
    }
  };

  stage(0, 0);
  __syncthreads();
  const int ntiles = k / GB2_BK;
  for (int t = 0; t < ntiles; t++) {
    if (t + 1 < ntiles) stage((t + 1) & 1, (t + 1) * GB2_BK);
    const bf16* as = As + (t & 1) * GB_BM * GB2_BK;
    const bf16* bs = Bs + (t & 1) * GB_BN * GB2_BK;
#pragma unroll
    for (int kk = 0; kk < GB2_BK; kk += 32) {
      bf16x8 a[4], b[4];
#pragma unroll
      for (int f = 0; f < 4; f++) {
        int row = wr * 64 + f * 16 + r16;
        a[f] = *(const bf16x8*)&as[row * GB2_BK +
                                   ((kk + g16 * 8) ^ ((row & 7) << 3))];
      }
#pragma unroll
      for (int f = 0; f < 4; f++) {
        int row = wc * 64 + f * 16 + r16;
        b[f] = *(const bf16x8*)&bs[row * GB2_BK +
                                   ((kk + g16 * 8) ^ ((row & 7) << 3))];
      }
#pragma unroll
      for (int i = 0; i < 4; i++)
#pragma unroll
        for (int j = 0; j < 4; j++)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[i], b[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

#pragma unroll
  for (int j = 0; j < 4; j++) {
    int col = bn0 + wc * 64 + j * 16 + r16;
    float* cp = C + (size_t)col * ldc;
#pragma unroll
    for (int i = 0; i < 4; i++) {
      int row0 = bm0 + wr * 64 + i * 16 + g16 * 4;
#pragma unroll
      for (int e = 0; e < 4; e++) {
        int row = row0 + e;
        cp[row] = accum ? cp[row] + acc[i][j][e] : acc[i][j][e];
      }
    }
  }
}

// ------------------------------------------------------------- v3: 256^2
// Deep-pipelined schedule (T3+T4 of the guide): 256x256 tile, 8 waves
// (2M x 4N, 128x64 per wave), K processed in 32-wide "K-half" phases over a
// 4-slot LDS ring (A+B 32 KB per slot, 128 KB total -> 1 block/CU, 2
// waves/SIMD). Each phase: counted `s_waitcnt vmcnt(8)` (the K-half staged
// 3 phases ago has landed; 2 newer phases x 4 glds stay IN FLIGHT across
// the barrier), one raw `s_barrier` (never __syncthreads: it would drain
// the glds queue — guide §5 'Pipelining across barriers'), 12 ds_read_b128
// fragment loads, 4 glds staging the phase+3 K-half, 32 MFMAs. The LDS
// image is XOR-swizzled through the glds source address (key
// ((row>>2)&3)*8 elements) so the b128 fragment reads are conflict-free.
#define GB3_BM 256
#define GB3_BN 256
#define GB3_KH 32  // K per phase

__launch_bounds__(512)
__global__ void k_gemm_bf16_tn_v3(int m, int n, int k,
                                  const bf16* __restrict__ A, int lda,
                                  const bf16* __restrict__ B, int ldb,
                                  float* __restrict__ C, int ldc, int nbx,
                                  int accum) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* ring = (bf16*)smem;  // 4 slots x [A 256x32 | B 256x32]
  constexpr int SLOT = (GB3_BM + GB3_BN) * GB3_KH;  // bf16 elements
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int wr = wave >> 2, wc = wave & 3;  // 2(M) x 4(N): 128x64 per wave
  int id = bf_swz(blockIdx.x, gridDim.x);
  const int bm0 = (id % nbx) * GB3_BM, bn0 = (id / nbx) * GB3_BN;
  const int g16 = lane >> 4, r16 = lane & 15;
  const int srow = lane >> 2;           // staging: 16 rows x 4 chunks
  const int schunk = lane & 3;

  f32x4 acc[8][4] = {};

  // stage K-half `ph` (globally k0 = ph*32) into ring slot ph&3
  auto stage = [&](int ph) {
    const int k0 = ph * GB3_KH;
    bf16* slot = ring + (ph & 3) * SLOT;
#pragma unroll
    for (int p = 0; p < 2; p++) {
      int row0 = (wave * 2 + p) * 16;
      int row = row0 + srow;
      int kc = 8 * (schunk ^ ((row >> 2) & 3));  // source pre-swizzle
      const bf16* srcA = A + (size_t)(bm0 + row) * lda + k0 + kc;
      bf16* dstA = slot + row0 * GB3_KH;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcA,
          (__attribute__((address_space(3))) unsigned int*)dstA, 16, 0, 0);
      const bf16* srcB = B + (size_t)(bn0 + row) * ldb + k0 + kc;
      bf16* dstB = slot + GB3_BM * GB3_KH + row0 * GB3_KH;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcB,
          (__attribute__((address_space(3))) unsigned int*)dstB, 16, 0, 0);
    }
  };

  const int P = k / GB3_KH;
  stage(0);
  if (P > 1) stage(1);
  if (P > 2) stage(2);
  for (int ph = 0; ph < P; ph++) {
    // counted wait: the K-half for THIS phase (own glds) has landed;
    // newer phases' loads stay in flight across the barrier
    if (ph + 3 <= P - 1) {
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    } else if (ph + 2 == P - 1) {
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    } else if (ph + 1 == P - 1) {
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    const bf16* as = ring + (ph & 3) * SLOT;
    const bf16* bs = as + GB3_BM * GB3_KH;
    bf16x8 a[8], b[4];
#pragma unroll
    for (int f = 0; f < 8; f++) {
      int row = wr * 128 + f * 16 + r16;
      a[f] = *(const bf16x8*)&as[row * GB3_KH +
                                 ((g16 * 8) ^ (((row >> 2) & 3) * 8))];
    }
#pragma unroll
    for (int f = 0; f < 4; f++) {
      int row = wc * 64 + f * 16 + r16;
      b[f] = *(const bf16x8*)&bs[row * GB3_KH +
                                 ((g16 * 8) ^ (((row >> 2) & 3) * 8))];
    }
    if (ph + 3 < P) stage(ph + 3);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 8; i++)
#pragma unroll
      for (int j = 0; j < 4; j++)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[i], b[j],
                                                            acc[i][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    // no trailing barrier: the next phase's counted-wait + barrier is the
    // only synchronization needed (slot reuse is 3 phases away)
  }

#pragma unroll
  for (int j = 0; j < 4; j++) {
    int col = bn0 + wc * 64 + j * 16 + r16;
    float* cp = C + (size_t)col * ldc;
#pragma unroll
    for (int i = 0; i < 8; i++) {
      int row0 = bm0 + wr * 128 + i * 16 + g16 * 4;
#pragma unroll
      for (int e = 0; e < 4; e++) {
        int row = row0 + e;
        cp[row] = accum ? cp[row] + acc[i][j][e] : acc[i][j][e];
      }
    }
  }
}

// ------------------------------------------------------------- v4: 8-phase
// The guide's 256^2 8-phase sub-phase interleave
// (cdna_hip_programming.md §5 "The 256² 8-phase template", quoted
// 1320-1340 TF on random data there): same 4-slot K-half ring, glds and
// XOR swizzle as v3, but each K-half's 32-MFMA burst is split into two
// 16-MFMA sub-phases bracketed by raw barriers — [ds-loads (+glds) |
// s_barrier | lgkmcnt(0) | setprio(1) 16xMFMA setprio(0) | s_barrier] —
// so each SIMD pairs one wave's MFMA segment with its partner's LDS/DMA
// load segment instead of stalling in the ds_read->MFMA latency window
// (round-1 PMC: MFMA-pipe 39% busy, SQ_WAIT_ANY 50%,
// profiles/bf16_gemm_v3_pmc.md).
// MEASURED (this repo, round 2): v4 709 TF @4096^3 / 787 @8192^3 vs v3
// 751/784 solo, and 875 vs 943 TF in the whole-job DAG — the sub-phase
// barriers alone do NOT reproduce the template's number without its exact
// hand schedule, so v3 stays the default (PARSEC_MCA_bf16_kernel=4 opts
// in). Kept as the measured A/B for the round-2 plan item.
__launch_bounds__(512)
__global__ void k_gemm_bf16_tn_v4(int m, int n, int k,
                                  const bf16* __restrict__ A, int lda,
                                  const bf16* __restrict__ B, int ldb,
                                  float* __restrict__ C, int ldc, int nbx,
                                  int accum) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* ring = (bf16*)smem;  // 4 slots x [A 256x32 | B 256x32]
  constexpr int SLOT = (GB3_BM + GB3_BN) * GB3_KH;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int wr = wave >> 2, wc = wave & 3;
  int id = bf_swz(blockIdx.x, gridDim.x);
  const int bm0 = (id % nbx) * GB3_BM, bn0 = (id / nbx) * GB3_BN;
  const int g16 = lane >> 4, r16 = lane & 15;
  const int srow = lane >> 2;
  const int schunk = lane & 3;

  f32x4 acc[8][4] = {};

  auto stage = [&](int ph) {
    const int k0 = ph * GB3_KH;
    bf16* slot = ring + (ph & 3) * SLOT;
#pragma unroll
    for (int p = 0; p < 2; p++) {
      int row0 = (wave * 2 + p) * 16;
      int row = row0 + srow;
      int kc = 8 * (schunk ^ ((row >> 2) & 3));
      const bf16* srcA = A + (size_t)(bm0 + row) * lda + k0 + kc;
      bf16* dstA = slot + row0 * GB3_KH;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcA,
          (__attribute__((address_space(3))) unsigned int*)dstA, 16, 0, 0);
      const bf16* srcB = B + (size_t)(bn0 + row) * ldb + k0 + kc;
      bf16* dstB = slot + GB3_BM * GB3_KH + row0 * GB3_KH;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcB,
          (__attribute__((address_space(3))) unsigned int*)dstB, 16, 0, 0);
    }
  };

  const int P = k / GB3_KH;
  stage(0);
  if (P > 1) stage(1);
  if (P > 2) stage(2);
  for (int ph = 0; ph < P; ph++) {
    // the K-half for THIS phase has landed; newer loads stay in flight
    if (ph + 3 <= P - 1) {
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    } else if (ph + 2 == P - 1) {
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    } else if (ph + 1 == P - 1) {
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    const bf16* as = ring + (ph & 3) * SLOT;
    const bf16* bs = as + GB3_BM * GB3_KH;
    // ---- sub-phase 0: load a[0..3] + b[0..3], stage ph+3, 16 MFMA ----
    bf16x8 a[8], b[4];
#pragma unroll
    for (int f = 0; f < 4; f++) {
      int row = wr * 128 + f * 16 + r16;
      a[f] = *(const bf16x8*)&as[row * GB3_KH +
                                 ((g16 * 8) ^ (((row >> 2) & 3) * 8))];
    }
#pragma unroll
    for (int f = 0; f < 4; f++) {
      int row = wc * 64 + f * 16 + r16;
      b[f] = *(const bf16x8*)&bs[row * GB3_KH +
                                 ((g16 * 8) ^ (((row >> 2) & 3) * 8))];
    }
    if (ph + 3 < P) stage(ph + 3);
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; i++)
#pragma unroll
      for (int j = 0; j < 4; j++)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[i], b[j],
                                                            acc[i][j], 0, 0,
                                                            0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
    // ---- sub-phase 1: load a[4..7], 16 MFMA ----
#pragma unroll
    for (int f = 4; f < 8; f++) {
      int row = wr * 128 + f * 16 + r16;
      a[f] = *(const bf16x8*)&as[row * GB3_KH +
                                 ((g16 * 8) ^ (((row >> 2) & 3) * 8))];
    }
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 4; i < 8; i++)
#pragma unroll
      for (int j = 0; j < 4; j++)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[i], b[j],
                                                            acc[i][j], 0, 0,
                                                            0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
  }

#pragma unroll
  for (int j = 0; j < 4; j++) {
    int col = bn0 + wc * 64 + j * 16 + r16;
    float* cp = C + (size_t)col * ldc;
#pragma unroll
    for (int i = 0; i < 8; i++) {
      int row0 = bm0 + wr * 128 + i * 16 + g16 * 4;
#pragma unroll
      for (int e = 0; e < 4; e++) {
        int row = row0 + e;
        cp[row] = accum ? cp[row] + acc[i][j][e] : acc[i][j][e];
      }
    }
  }
}

static void launch_gemm_bf16(int m, int n, int k, const void* A, int lda,
                             const void* B, int ldb, float* C, int ldc,
                             hipStream_t stream, int accum = 1) {
  if (m % GB3_BM == 0 && n % GB3_BN == 0 && k % GB3_KH == 0 && k >= 4 * GB3_KH) {
    constexpr size_t lds = 4 * (GB3_BM + GB3_BN) * GB3_KH * 2;
    int variant = (int)param_int("bf16_kernel", 3);
    const void* kf = variant >= 4 ? (const void*)k_gemm_bf16_tn_v4
                                  : (const void*)k_gemm_bf16_tn_v3;
    static bool attr3 = false;
    if (!attr3) {
      hipFuncSetAttribute((const void*)k_gemm_bf16_tn_v3,
                          hipFuncAttributeMaxDynamicSharedMemorySize, lds);
      hipFuncSetAttribute((const void*)k_gemm_bf16_tn_v4,
                          hipFuncAttributeMaxDynamicSharedMemorySize, lds);
      attr3 = true;
    }
    int nbx = m / GB3_BM, nby = n / GB3_BN;
    if (variant >= 4)
      hipLaunchKernelGGL(k_gemm_bf16_tn_v4, dim3(nbx * nby), dim3(512), lds,
                         stream, m, n, k, (const bf16*)A, lda,
                         (const bf16*)B, ldb, C, ldc, nbx, accum);
    else
      hipLaunchKernelGGL(k_gemm_bf16_tn_v3, dim3(nbx * nby), dim3(512), lds,
                         stream, m, n, k, (const bf16*)A, lda,
                         (const bf16*)B, ldb, C, ldc, nbx, accum);
    (void)kf;
    return;
  }
  int nbx = (m + GB_BM - 1) / GB_BM, nby = (n + GB_BN - 1) / GB_BN;
  if (m % GB_BM == 0 && n % GB_BN == 0 && k % GB2_BK == 0) {
    constexpr size_t lds = 2 * (GB_BM + GB_BN) * GB2_BK * 2;
    static bool attr_set = false;
    if (!attr_set) {
      hipFuncSetAttribute((const void*)k_gemm_bf16_tn_v2,
                          hipFuncAttributeMaxDynamicSharedMemorySize, lds);
      attr_set = true;
    }
    hipLaunchKernelGGL(k_gemm_bf16_tn_v2, dim3(nbx * nby), dim3(256), lds,
                       stream, m, n, k, (const bf16*)A, lda, (const bf16*)B,
                       ldb, C, ldc, nbx, accum);
    return;
  }
  hipLaunchKernelGGL(k_gemm_bf16_tn, dim3(nbx * nby), dim3(256), 0, stream,
                     m, n, k, (const bf16*)A, lda, (const bf16*)B, ldb, C,
                     ldc, nbx, accum);
}

// ------------------------------------------------------------------ fill
__global__ void k_fill_bf16(bf16* p, size_t nelem, uint32_t seed) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < nelem; i += (size_t)gridDim.x * blockDim.x) {
    uint64_t h = (i * 2654435761ull) ^ ((uint64_t)seed * 2246822519ull);
    h ^= h >> 13;
    h *= 0x9E3779B97F4A7C15ull;
    h ^= h >> 32;
    p[i] = (bf16)((float)(h & 0xFFFF) / 65536.0f - 0.5f);
  }
}

// host-side bf16 helpers (CPU chores / tests)
static inline float bf2f(uint16_t b) {
  uint32_t u = (uint32_t)b << 16;
  float f;
  memcpy(&f, &u, 4);
  return f;
}
static inline uint16_t f2bf(float f) {
  uint32_t u;
  memcpy(&u, &f, 4);
  uint32_t r = (u + 0x7FFF + ((u >> 16) & 1)) >> 16;
  return (uint16_t)r;
}

static void cpu_fill_bf16(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  Data* d = t.flows[0].data;
  uint16_t* p = (uint16_t*)d->ensure_host();
  size_t nelem = d->bytes / 2;
  uint32_t seed = a.seed;
  size_t base = (size_t)a.i0;
  for (size_t i = 0; i < nelem; i++) {
    uint64_t h = ((base + i) * 2654435761ull) ^
                 ((uint64_t)seed * 2246822519ull);
    h ^= h >> 13;
    h *= 0x9E3779B97F4A7C15ull;
    h ^= h >> 32;
    p[i] = f2bf((float)(h & 0xFFFF) / 65536.0f - 0.5f);
  }
  d->written_on(false);
}

static void gpu_fill_bf16(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  Data* d = t.flows[0].data;
  hipLaunchKernelGGL(k_fill_bf16, dim3(2048), dim3(256), 0, g.stream,
                     (bf16*)((char*)t.dev_ptr[0] + 0), d->bytes / 2, a.seed);
}

// Note: GPU fill hashes the tile-local index; CPU fill must match for
// cross-checks, so both hash tile-local index + per-tile seed.

// ------------------------------------------------------------------ chores
static void cpu_gemm_bf16(Task& t) {
  const TileArgs& a = t.arg<TileArgs>();
  const int accum = (int)a.j0;
  const uint16_t* A = (const uint16_t*)t.flows[0].data->pull_to_host();
  const uint16_t* B = (const uint16_t*)t.flows[1].data->pull_to_host();
  Data* cd = t.flows[2].data;
  float* C = accum ? (float*)cd->pull_to_host() : (float*)cd->ensure_host();
  const int m = a.m, n = a.n, kk = a.k, lda = a.ld;
  const int ldc = cd->coll->mb();
  for (int j = 0; j < n; j++)
    for (int i = 0; i < m; i++) {
      float s = 0;
      for (int p = 0; p < kk; p++)
        s += bf2f(A[(size_t)i * lda + p]) * bf2f(B[(size_t)j * lda + p]);
      C[(size_t)j * ldc + i] = accum ? C[(size_t)j * ldc + i] + s : s;
    }
  cd->written_on(false);
}

static void gpu_gemm_bf16(Task& t, GpuTaskCtx& g) {
  const TileArgs& a = t.arg<TileArgs>();
  const int ldc = t.flows[2].data->coll->mb();
  launch_gemm_bf16(a.m, a.n, a.k, t.dev_ptr[0], a.ld, t.dev_ptr[1], a.ld,
                   (float*)t.dev_ptr[2], ldc, g.stream, (int)a.j0);
}

static TaskClass make_bf_tc(const char* name, void (*cpu)(Task&),
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

TaskClass& tc_fill_bf16() {
  static TaskClass tc = make_bf_tc("fill_bf16", cpu_fill_bf16, gpu_fill_bf16, 20);
  return tc;
}
TaskClass& tc_gemm_bf16() {
  static TaskClass tc = make_bf_tc("gemm_bf16", cpu_gemm_bf16, gpu_gemm_bf16, 21);
  return tc;
}

// ------------------------------------------------------------ DAG builders
void insert_fill_bf16(Dtd& tp, TiledMatrix& A, uint32_t seed) {
  PA_CHECK(A.elem_size() == 2);
  for (int tm = 0; tm < A.mt(); tm++)
    for (int tn = 0; tn < A.nt(); tn++) {
      TileArgs a;
      a.seed = seed ^ (uint32_t)(tm * 9973 + tn);
      a.i0 = 0;
      Dtd::FlowSpec f[] = {{A.tile(tm, tn), ACCESS_OUT}};
      tp.insert(&tc_fill_bf16(), &a, sizeof(a), f, 1, 0, A.rank_of(tm, tn));
    }
}

// C (fp32 mb x nb tiles) += At^T B over the shared K tiling. At: K x M
// bf16 tiles (kb x mb each); B: K x N bf16 tiles (kb x nb).
void insert_gemm_bf16(Dtd& tp, TiledMatrix& At, TiledMatrix& B,
                      TiledMatrix& C) {
  PA_CHECK(At.elem_size() == 2 && B.elem_size() == 2 && C.elem_size() == 4);
  PA_CHECK(At.mt() == B.mt(), "K tilings must match");
  PA_CHECK(At.nt() == C.mt() && B.nt() == C.nt());
  const int KT = At.mt();
  for (int m = 0; m < C.mt(); m++)
    for (int n = 0; n < C.nt(); n++)
      for (int k = 0; k < KT; k++) {
        TileArgs a;
        a.m = At.tile_cols(m);
        a.n = B.tile_cols(n);
        a.k = At.tile_rows(k);
        a.ld = At.mb();
        a.j0 = (k != 0);  // accumulate flag; k==0 overwrites C
        Dtd::FlowSpec f[] = {{At.tile(k, m), ACCESS_IN},
                             {B.tile(k, n), ACCESS_IN},
                             {C.tile(m, n),
                              k == 0 ? ACCESS_OUT : ACCESS_INOUT}};
        tp.insert(&tc_gemm_bf16(), &a, sizeof(a), f, 3, -(k),
                  C.rank_of(m, n));
      }
}

// kernel-level microbench (TFLOP/s)
double bench_gemm_bf16(int m, int n, int k, int iters) {
  bf16 *dA, *dB;
  float* dC;
  PA_HIP_CHECK(hipMalloc(&dA, (size_t)m * k * 2));
  PA_HIP_CHECK(hipMalloc(&dB, (size_t)n * k * 2));
  PA_HIP_CHECK(hipMalloc(&dC, (size_t)m * n * 4));
  hipLaunchKernelGGL(k_fill_bf16, dim3(2048), dim3(256), 0, 0, dA,
                     (size_t)m * k, 1u);
  hipLaunchKernelGGL(k_fill_bf16, dim3(2048), dim3(256), 0, 0, dB,
                     (size_t)n * k, 2u);
  PA_HIP_CHECK(hipMemset(dC, 0, (size_t)m * n * 4));
  launch_gemm_bf16(m, n, k, dA, k, dB, k, dC, m, 0, 1);
  PA_HIP_CHECK(hipDeviceSynchronize());
  double t0 = now_s();
  for (int i = 0; i < iters; i++)
    launch_gemm_bf16(m, n, k, dA, k, dB, k, dC, m, 0, 1);
  PA_HIP_CHECK(hipDeviceSynchronize());
  double dt = now_s() - t0;
  PA_HIP_CHECK(hipFree(dA));
  PA_HIP_CHECK(hipFree(dB));
  PA_HIP_CHECK(hipFree(dC));
  return 2.0 * m * n * k * iters / dt / 1e12;
}

// host-I/O numerics harness
void test_gemm_bf16_hip(int m, int n, int k, const uint16_t* A,
                        const uint16_t* B, float* C) {
  bf16 *dA, *dB;
  float* dC;
  PA_HIP_CHECK(hipMalloc(&dA, (size_t)m * k * 2));
  PA_HIP_CHECK(hipMalloc(&dB, (size_t)n * k * 2));
  PA_HIP_CHECK(hipMalloc(&dC, (size_t)m * n * 4));
  PA_HIP_CHECK(hipMemcpy(dA, A, (size_t)m * k * 2, hipMemcpyHostToDevice));
  PA_HIP_CHECK(hipMemcpy(dB, B, (size_t)n * k * 2, hipMemcpyHostToDevice));
  PA_HIP_CHECK(hipMemcpy(dC, C, (size_t)m * n * 4, hipMemcpyHostToDevice));
  launch_gemm_bf16(m, n, k, dA, k, dB, k, dC, m, 0, 1);
  PA_HIP_CHECK(hipGetLastError());
  PA_HIP_CHECK(hipMemcpy(C, dC, (size_t)m * n * 4, hipMemcpyDeviceToHost));
  PA_HIP_CHECK(hipFree(dA));
  PA_HIP_CHECK(hipFree(dB));
  PA_HIP_CHECK(hipFree(dC));
}

}  // namespace pa
