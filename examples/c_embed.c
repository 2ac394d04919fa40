/* Standalone C program embedding the parsec_amd runtime through its C ABI
 * (the reference is consumed exactly this way as a C library).
 *
 * Build (from the repo root; _core.so carries the whole runtime):
 *   gcc -O2 examples/c_embed.c -o /tmp/c_embed \
 *       -Lparsec_amd -l:_core.so -Wl,-rpath,$PWD/parsec_amd
 * Run:
 *   /tmp/c_embed
 *
 * Inserts a chain of N INOUT tasks on one tile plus an independent task
 * per other tile, waits, and verifies the dataflow ordering end-to-end.
 */
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

/* C ABI of parsec_amd/_core.so (see src/capi.cpp) */
extern void* pa_context_new(int nworkers, int rank, int world,
                            const char* comm, int gpu);
extern void pa_context_free(void* ctx);
extern void* pa_dtd_new(void* ctx, const char* name);
extern void pa_dtd_wait(void* dtd);
extern void pa_dtd_free(void* dtd);
extern void* pa_tm_new(void* ctx, long m, long n, int mb, int nb, int p,
                       int q, long elem_size, int sym);
extern void pa_tm_free(void* tm);
extern void* pa_tm_tile(void* tm, int i, int j);
extern void* pa_taskclass_new(const char* name, int flags, void (*cpu)(void*),
                              void (*gpu)(void*, void*));
extern void pa_dtd_insert(void* dtd, void* tc, const void* args, int nargs,
                          void** datas, const int* modes, int nflows,
                          int prio, int rank);
extern void* pa_task_args(void* t);
extern void* pa_task_host_ptr(void* t, int flow);
extern void* pa_tm_tile_host(void* tm, int i, int j);

enum { ACCESS_IN = 1, ACCESS_OUT = 2, ACCESS_INOUT = 3 };

static void body_init(void* t) {
  double* v = (double*)pa_task_host_ptr(t, 0);
  long* k = (long*)pa_task_args(t);
  v[0] = (double)*k;
}

static void body_chain(void* t) {
  double* v = (double*)pa_task_host_ptr(t, 0);
  v[0] = v[0] * 2.0 + 1.0;
}

int main(void) {
  void* ctx = pa_context_new(2, 0, 1, "", -2 /* CPU only */);
  void* tp = pa_dtd_new(ctx, "c_embed");
  void* A = pa_tm_new(ctx, 8, 8, 1, 8, 1, 1, 8, 0); /* 8 tiles of 8 dbl */

  void* tc_init = pa_taskclass_new("c_init", 0, body_init, NULL);
  void* tc_chain = pa_taskclass_new("c_chain", 0, body_chain, NULL);

  for (int i = 0; i < 8; i++) {
    long k = 100 + i;
    void* d = pa_tm_tile(A, i, 0);
    int mode = ACCESS_OUT;
    pa_dtd_insert(tp, tc_init, &k, sizeof(k), &d, &mode, 1, 0, 0);
  }
  /* 10-deep chain on tile 3: value = ((103*2+1)*2+1)... */
  for (int s = 0; s < 10; s++) {
    void* d = pa_tm_tile(A, 3, 0);
    int mode = ACCESS_INOUT;
    pa_dtd_insert(tp, tc_chain, NULL, 0, &d, &mode, 1, 0, 0);
  }
  pa_dtd_wait(tp);

  double want = 103.0;
  for (int s = 0; s < 10; s++) want = want * 2.0 + 1.0;
  const double* v3 = (const double*)pa_tm_tile_host(A, 3, 0);
  const double* v5 = (const double*)pa_tm_tile_host(A, 5, 0);
  if (v3[0] != want || v5[0] != 105.0) {
    fprintf(stderr, "FAIL: got %f want %f (v5 %f)\n", v3[0], want, v5[0]);
    return 1;
  }
  pa_dtd_free(tp);
  pa_tm_free(A);
  pa_context_free(ctx);
  printf("C_EMBED_OK\n");
  return 0;
}
