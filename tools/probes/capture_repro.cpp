// Minimal stream-capture topology probe: which cross-stream event pattern
// breaks hipStreamEndCapture on ROCm 7.2?
#include <hip/hip_runtime.h>
#include <cstdio>
#define CK(x) do { hipError_t _ck = (x); if (_ck != hipSuccess) { \
  printf("ERR %s at %d: %s\n", hipGetErrorString(_ck), __LINE__, #x); \
  fflush(stdout); return 1; } } while (0)

__global__ void knop(double* p) { if (p) p[threadIdx.x] += 1.0; }

int run_case(int pattern) {
  printf("case %d...\n", pattern); fflush(stdout);
  hipStream_t s[4];
  for (auto& x : s) CK(hipStreamCreateWithFlags(&x, hipStreamNonBlocking));
  double* buf; CK(hipMalloc(&buf, 4096));
  hipEvent_t evs[64]; int ne = 0;
  auto mkev = [&]() { hipEvent_t e; hipEventCreateWithFlags(&e, hipEventDisableTiming); evs[ne++] = e; return e; };
  CK(hipStreamBeginCapture(s[0], hipStreamCaptureModeThreadLocal));
  hipEvent_t fork = mkev();
  CK(hipEventRecord(fork, s[0]));
  for (int i = 1; i < 4; i++) CK(hipStreamWaitEvent(s[i], fork, 0));
  auto K = [&](int si) { hipLaunchKernelGGL(knop, dim3(1), dim3(64), 0, s[si], buf); };
  if (pattern == 0) {
    // simple: chain s1 -> s2 (single event, single wait)
    K(1); hipEvent_t e = mkev(); CK(hipEventRecord(e, s[1]));
    CK(hipStreamWaitEvent(s[2], e, 0)); K(2);
  } else if (pattern == 1) {
    // consumer waits TWO events from two streams before launching
    K(1); hipEvent_t e1 = mkev(); CK(hipEventRecord(e1, s[1]));
    K(2); hipEvent_t e2 = mkev(); CK(hipEventRecord(e2, s[2]));
    CK(hipStreamWaitEvent(s[3], e1, 0));
    CK(hipStreamWaitEvent(s[3], e2, 0)); K(3);
  } else if (pattern == 2) {
    // producer records TWO events back-to-back, two consumers
    K(1); hipEvent_t e1 = mkev(), e2 = mkev();
    CK(hipEventRecord(e1, s[1])); CK(hipEventRecord(e2, s[1]));
    CK(hipStreamWaitEvent(s[2], e1, 0)); K(2);
    CK(hipStreamWaitEvent(s[3], e2, 0)); K(3);
  } else if (pattern == 3) {
    // one event waited by TWO streams
    K(1); hipEvent_t e1 = mkev(); CK(hipEventRecord(e1, s[1]));
    CK(hipStreamWaitEvent(s[2], e1, 0)); K(2);
    CK(hipStreamWaitEvent(s[3], e1, 0)); K(3);
  } else if (pattern == 4) {
    // mesh: s1<->s2 cross in both directions over time
    K(1); hipEvent_t e1 = mkev(); CK(hipEventRecord(e1, s[1]));
    CK(hipStreamWaitEvent(s[2], e1, 0)); K(2);
    hipEvent_t e2 = mkev(); CK(hipEventRecord(e2, s[2]));
    CK(hipStreamWaitEvent(s[1], e2, 0)); K(1);
  }
  for (int i = 1; i < 4; i++) {
    hipEvent_t je = mkev();
    CK(hipEventRecord(je, s[i]));
    CK(hipStreamWaitEvent(s[0], je, 0));
  }
  printf("  ending capture\n"); fflush(stdout);
  hipGraph_t g; CK(hipStreamEndCapture(s[0], &g));
  hipGraphExec_t ge; CK(hipGraphInstantiate(&ge, g, nullptr, nullptr, 0));
  CK(hipGraphLaunch(ge, s[0]));
  CK(hipStreamSynchronize(s[0]));
  printf("  case %d OK\n", pattern); fflush(stdout);
  hipGraphExecDestroy(ge); hipGraphDestroy(g);
  for (int i = 0; i < ne; i++) hipEventDestroy(evs[i]);
  for (auto& x : s) hipStreamDestroy(x);
  hipFree(buf);
  return 0;
}

int main(int argc, char** argv) {
  int only = argc > 1 ? atoi(argv[1]) : -1;
  for (int p = 0; p < 5; p++)
    if (only < 0 || p == only) run_case(p);
  printf("ALL DONE\n");
  return 0;
}
