// Is rocblas_dgemm capture-safe for degenerate shapes (n=1)?
#include <hip/hip_runtime.h>
#include <rocblas/rocblas.h>
#include <cstdio>
#define CK(x) do { hipError_t _ck = (x); if (_ck != hipSuccess) { \
  printf("ERR %s at %d\n", hipGetErrorString(_ck), __LINE__); fflush(stdout); return 1; } } while (0)
#define RB(x) do { rocblas_status _rb = (x); if (_rb != rocblas_status_success) { \
  printf("RBERR %d at %d\n", (int)_rb, __LINE__); fflush(stdout); return 1; } } while (0)

int run_case(int n) {
  printf("case n=%d...\n", n); fflush(stdout);
  hipStream_t s; CK(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
  rocblas_handle h; RB(rocblas_create_handle(&h));
  RB(rocblas_set_pointer_mode(h, rocblas_pointer_mode_host));
  RB(rocblas_set_stream(h, s));
  void* ws; CK(hipMalloc(&ws, 1u<<26));
  RB(rocblas_set_workspace(h, ws, 1u<<26));
  double *A, *B, *C;
  CK(hipMalloc(&A, 64*64*8)); CK(hipMalloc(&B, 64*64*8)); CK(hipMalloc(&C, 64*64*8));
  const double one = 1.0, zero = 0.0;
  // warm the exact shape OUTSIDE capture first
  RB(rocblas_dgemm(h, rocblas_operation_none, rocblas_operation_none,
                   64, n, 64, &one, A, 64, B, 64, &zero, C, 64));
  CK(hipStreamSynchronize(s));
  CK(hipStreamBeginCapture(s, hipStreamCaptureModeThreadLocal));
  RB(rocblas_dgemm(h, rocblas_operation_none, rocblas_operation_none,
                   64, n, 64, &one, A, 64, B, 64, &zero, C, 64));
  printf("  ending capture\n"); fflush(stdout);
  hipGraph_t g; CK(hipStreamEndCapture(s, &g));
  hipGraphExec_t ge; CK(hipGraphInstantiate(&ge, g, nullptr, nullptr, 0));
  CK(hipGraphLaunch(ge, s)); CK(hipStreamSynchronize(s));
  printf("  n=%d OK\n", n); fflush(stdout);
  hipGraphExecDestroy(ge); hipGraphDestroy(g);
  hipFree(A); hipFree(B); hipFree(C); hipFree(ws);
  rocblas_destroy_handle(h); hipStreamDestroy(s);
  return 0;
}
int main() { run_case(64); run_case(8); run_case(1); printf("ALL DONE\n"); return 0; }
