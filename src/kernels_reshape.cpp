// Reshape promises: per-flow data conversion between producer and
// consumer shape/type.
//
// Reference parity (SURVEY.md §2.1 "Reshape engine",
// parsec/parsec_reshape.c:1-786 + datacopy futures): a consumer flow may
// declare a reshape; the engine then materializes a CONVERTED copy of the
// producer's version — lazily, once per (version, kind, consumer rank),
// shared by every consumer that asks for the same conversion (the
// "promise"). MI355X-native design: the converted copy is a standalone
// Data produced by an ordinary conversion task (HIP kernel on-device, CPU
// fallback), so fetch/renaming/eviction all treat it like any tile; the
// MPI-datatype machinery of the reference collapses to explicit
// conversion kernels, which is what a single-node HBM-resident runtime
// actually wants (no wire repacking exists to piggyback on).
#include <cstring>

#include <hip/hip_runtime.h>

#include "device_gpu.hpp"
#include "dtd.hpp"
#include "kernels.hpp"
#include "profiling.hpp"

namespace pa {

namespace {

struct ReshapeArgs {
  int kind;  // Reshape enum
  int m, n, ld;  // source tile dims (col-major)
};

// ---- GPU kernels ----
__global__ void k_rs_transpose(const double* __restrict__ S,
                               double* __restrict__ D, int m, int n, int ld) {
  // D (n x m, col-major, ldd = n) = S^T; LDS-tiled 32x32 for coalesced
  // loads AND stores.
  __shared__ double t[32][33];
  int bx = blockIdx.x * 32, by = blockIdx.y * 32;
  int x = threadIdx.x, y = threadIdx.y;
  for (int yy = y; yy < 32; yy += 8) {
    int r = bx + x, c = by + yy;
    t[yy][x] = (r < m && c < n) ? S[(size_t)c * ld + r] : 0.0;
  }
  __syncthreads();
  for (int yy = y; yy < 32; yy += 8) {
    int r = by + x, c = bx + yy;  // D row = S col
    if (r < n && c < m) D[(size_t)c * n + r] = t[x][yy];
  }
}

__global__ void k_rs_to_bf16(const double* __restrict__ S,
                             __bf16* __restrict__ D, size_t nelem) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < nelem; i += (size_t)gridDim.x * blockDim.x)
    D[i] = (__bf16)(float)S[i];
}

__global__ void k_rs_from_bf16(const __bf16* __restrict__ S,
                               double* __restrict__ D, size_t nelem) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < nelem; i += (size_t)gridDim.x * blockDim.x)
    D[i] = (double)(float)S[i];
}

__global__ void k_rs_tri(const double* __restrict__ S,
                         double* __restrict__ D, int m, int n, int ld,
                         int lower) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t total = (size_t)m * n;
  for (; i < total; i += (size_t)gridDim.x * blockDim.x) {
    int c = (int)(i / m), r = (int)(i - (size_t)c * m);
    double v = S[(size_t)c * ld + r];
    bool keep = lower ? (r >= c) : (r <= c);
    D[(size_t)c * m + r] = keep ? v : 0.0;
  }
}

void cpu_reshape(Task& t) {
  const ReshapeArgs& a = t.arg<ReshapeArgs>();
  const void* s = t.flows[0].data->pull_to_host();
  Data* dd = t.flows[1].data;
  void* d = dd->ensure_host();
  const int m = a.m, n = a.n, ld = a.ld;
  switch ((Reshape)a.kind) {
    case Reshape::TRANSPOSE: {
      const double* S = (const double*)s;
      double* D = (double*)d;
      for (int c = 0; c < n; c++)
        for (int r = 0; r < m; r++) D[(size_t)r * n + c] = S[(size_t)c * ld + r];
      break;
    }
    case Reshape::TO_BF16: {
      const double* S = (const double*)s;
      uint16_t* D = (uint16_t*)d;
      for (int c = 0; c < n; c++)
        for (int r = 0; r < m; r++) {
          float f = (float)S[(size_t)c * ld + r];
          uint32_t u;
          memcpy(&u, &f, 4);
          D[(size_t)c * m + r] =
              (uint16_t)((u + 0x7FFF + ((u >> 16) & 1)) >> 16);
        }
      break;
    }
    case Reshape::FROM_BF16: {
      const uint16_t* S = (const uint16_t*)s;
      double* D = (double*)d;
      for (int c = 0; c < n; c++)
        for (int r = 0; r < m; r++) {
          uint32_t u = (uint32_t)S[(size_t)c * ld + r] << 16;
          float f;
          memcpy(&f, &u, 4);
          D[(size_t)c * m + r] = (double)f;
        }
      break;
    }
    case Reshape::TRIL:
    case Reshape::TRIU: {
      const double* S = (const double*)s;
      double* D = (double*)d;
      bool lower = (Reshape)a.kind == Reshape::TRIL;
      for (int c = 0; c < n; c++)
        for (int r = 0; r < m; r++) {
          bool keep = lower ? (r >= c) : (r <= c);
          D[(size_t)c * m + r] = keep ? S[(size_t)c * ld + r] : 0.0;
        }
      break;
    }
    default:
      fatal("cpu_reshape: bad kind %d", a.kind);
  }
  dd->written_on(false);
}

void gpu_reshape(Task& t, GpuTaskCtx& g) {
  const ReshapeArgs& a = t.arg<ReshapeArgs>();
  const void* s = t.dev_ptr[0];
  void* d = t.dev_ptr[1];
  const int m = a.m, n = a.n, ld = a.ld;
  size_t total = (size_t)m * n;
  dim3 g1((unsigned)std::min<size_t>((total + 255) / 256, 2048));
  switch ((Reshape)a.kind) {
    case Reshape::TRANSPOSE:
      hipLaunchKernelGGL(k_rs_transpose,
                         dim3((m + 31) / 32, (n + 31) / 32), dim3(32, 8), 0,
                         g.stream, (const double*)s, (double*)d, m, n, ld);
      break;
    case Reshape::TO_BF16:
      hipLaunchKernelGGL(k_rs_to_bf16, g1, dim3(256), 0, g.stream,
                         (const double*)s, (__bf16*)d, total);
      break;
    case Reshape::FROM_BF16:
      hipLaunchKernelGGL(k_rs_from_bf16, g1, dim3(256), 0, g.stream,
                         (const __bf16*)s, (double*)d, total);
      break;
    case Reshape::TRIL:
    case Reshape::TRIU:
      hipLaunchKernelGGL(k_rs_tri, g1, dim3(256), 0, g.stream,
                         (const double*)s, (double*)d, m, n, ld,
                         (Reshape)a.kind == Reshape::TRIL ? 1 : 0);
      break;
    default:
      fatal("gpu_reshape: bad kind %d", a.kind);
  }
}

}  // namespace

TaskClass& tc_reshape() {
  static TaskClass tc = [] {
    Profiler::inst().register_class(60, "reshape");
    TaskClass c;
    c.name = "reshape";
    c.kind = TaskKind::GPU;
    c.cpu_hook = cpu_reshape;
    c.gpu_hook = gpu_reshape;
    c.id = 60;
    return c;
  }();
  return tc;
}

size_t reshape_bytes(size_t src_bytes, Reshape kind, size_t src_elem) {
  switch (kind) {
    case Reshape::TO_BF16:
      PA_CHECK(src_elem == 8, "TO_BF16 expects fp64 source tiles");
      return src_bytes / 4;
    case Reshape::FROM_BF16:
      PA_CHECK(src_elem == 2, "FROM_BF16 expects bf16 source tiles");
      return src_bytes * 4;
    default:
      return src_bytes;
  }
}

void fill_reshape_args(void* argbuf, Reshape kind, int m, int n, int ld) {
  ReshapeArgs a{(int)kind, m, n, ld};
  memcpy(argbuf, &a, sizeof(a));
}

size_t reshape_args_bytes() { return sizeof(ReshapeArgs); }

}  // namespace pa
