// C ABI for PTG-generated code (and any external front-end).
//
// The reference's generated JDF code links against the public parsec C API
// (runtime.h:156-710); here the generated C++ from parsec_amd.ptg uses this
// minimal exported surface (the rest of _core.so is visibility-hidden).
#include <hip/hip_runtime.h>

#include <cstring>

#include "data.hpp"
#include "device_gpu.hpp"
#include "dtd.hpp"
#include "pins.hpp"
#include "runtime.hpp"

#define PA_EXPORT extern "C" __attribute__((visibility("default")))

using namespace pa;

namespace {
void cpu_trampoline(Task& t) {
  auto fn = (void (*)(void*))t.tc->user_cpu;
  fn(&t);
  // CPU bodies write host memory: mark OUT flows.
  for (int i = 0; i < t.nflows; i++)
    if (t.flows[i].data && (t.flows[i].mode & ACCESS_OUT))
      t.flows[i].data->written_on(false);
}

void gpu_trampoline(Task& t, GpuTaskCtx& g) {
  auto fn = (void (*)(void*, void*))t.tc->user_gpu;
  fn(&t, (void*)g.stream);
}
}  // namespace

PA_EXPORT int pa_ctx_rank(void* ctx) { return ((Context*)ctx)->rank(); }
PA_EXPORT int pa_ctx_world(void* ctx) { return ((Context*)ctx)->world(); }
PA_EXPORT int pa_ctx_has_gpu(void* ctx) { return ((Context*)ctx)->has_gpu(); }

PA_EXPORT void* pa_tm_tile(void* tm, int i, int j) {
  return ((TiledMatrix*)tm)->tile(i, j);
}
PA_EXPORT int pa_tm_rank_of(void* tm, int i, int j) {
  return ((TiledMatrix*)tm)->rank_of(i, j);
}
PA_EXPORT int pa_data_home_rank(void* d) { return ((Data*)d)->home_rank; }

// flags: bit0 = has GPU body, bit1 = GPU body is host-blocking (runs on
// a worker thread with its own stream — BODY [type=HIP blocking=on]).
PA_EXPORT void* pa_taskclass_new(const char* name, int flags,
                                 void (*cpu)(void*),
                                 void (*gpu)(void*, void*)) {
  TaskClass* tc = new TaskClass();
  tc->name = name;
  tc->kind = ((flags & 1) && gpu) ? TaskKind::GPU : TaskKind::CPU;
  tc->gpu_blocking = (flags & 2) != 0;
  tc->user_cpu = (void*)cpu;
  tc->user_gpu = (void*)gpu;
  if (cpu) tc->cpu_hook = cpu_trampoline;
  if (gpu) tc->gpu_hook = gpu_trampoline;
  return tc;
}

PA_EXPORT void* pa_task_args(void* t) { return ((Task*)t)->args; }
PA_EXPORT void* pa_task_dev_ptr(void* t, int flow) {
  return ((Task*)t)->dev_ptr[flow];
}
PA_EXPORT void* pa_task_host_ptr(void* t, int flow) {
  Task* task = (Task*)t;
  Data* d = task->flows[flow].data;
  if (!d) return nullptr;
  // WRITE-only flows produce the content: hand out the buffer untouched.
  if (!(task->flows[flow].mode & ACCESS_IN)) return d->ensure_host();
  if (!d->host_valid && !d->dev_valid)
    fatal("task %s seq=%lu flow %d: input has no valid copy (key=%ld "
          "ver=%lu) — insertion preceded its producer?",
          task->tc->name.c_str(), (unsigned long)task->seq, flow,
          (long)d->key, (unsigned long)d->version);
  return d->pull_to_host();
}

PA_EXPORT void* pa_dtd_insert_begin(void* dtd, void* tc, const void* args,
                                    int nargs, void** datas, const int* modes,
                                    int nflows, int prio, int rank) {
  Dtd::FlowSpec fs[MAX_FLOWS];
  for (int i = 0; i < nflows; i++)
    fs[i] = {(Data*)datas[i], (AccessMode)modes[i]};
  return ((Dtd*)dtd)->insert_begin((const TaskClass*)tc, args, (size_t)nargs,
                                   fs, nflows, prio, rank);
}
PA_EXPORT void pa_dtd_insert_commit(void* dtd, void* task) {
  ((Dtd*)dtd)->insert_commit((Task*)task);
}
PA_EXPORT void pa_task_edge(void* pred, void* succ) {
  task_add_edge((Task*)pred, (Task*)succ);
}
PA_EXPORT void pa_task_retain(void* t) { ((Task*)t)->retain(); }

// Fresh scratch datum owned by the taskpool (JDF `<- NEW [size=...]`
// arena-tile analog). home_rank = the creating instance's rank.
PA_EXPORT void* pa_dtd_scratch(void* dtd, long bytes, int home_rank) {
  auto* tp = (Dtd*)dtd;
  auto holder = std::make_shared<Data>();
  holder->ctx_direct = tp->context();
  holder->home_rank = home_rank;
  holder->owner_rank = home_rank;
  holder->bytes = (size_t)bytes;
  // A fresh scratch tile is valid-but-uninitialized memory (the reference's
  // arena-allocated NEW tiles behave the same): zero it so the first access
  // — which may be a read (e.g. an INOUT CTL token) — is defined.
  memset(holder->ensure_host(), 0, holder->bytes);
  holder->host_valid = true;
  Data* d = holder.get();
  tp->own(std::shared_ptr<void>(holder, holder.get()));
  return d;
}
PA_EXPORT void pa_task_release(void* t) { ((Task*)t)->release(); }
PA_EXPORT void* pa_task_taskpool(void* t) { return ((Task*)t)->tp; }

// PINS COMPLETE hook + pool-completion callback + owned-resource handle:
// the seam the compact (never-materialized) PTG iterator hangs off
// (src/ptg_runtime.hpp Compact; jdf2c.c compact-iteration analog).
PA_EXPORT long pa_pins_on_complete(void (*cb)(void*, void*), void* user) {
  return Pins::inst().add(
      [cb, user](PinsEv, const Task* t, int) { cb((void*)t, user); },
      1u << (int)PinsEv::COMPLETE);
}
PA_EXPORT void pa_pins_off(long h) { Pins::inst().remove((int)h); }
PA_EXPORT void pa_dtd_on_complete(void* dtd, void (*cb)(void*), void* user) {
  ((Dtd*)dtd)->on_complete([cb, user] { cb(user); });
}
PA_EXPORT void pa_dtd_own_ptr(void* dtd, void* p, void (*deleter)(void*)) {
  ((Dtd*)dtd)->own(std::shared_ptr<void>(p, deleter));
}

// ---- standalone C embedding surface (parsec_init/parsec_fini analog for
// C programs linking _core.so directly; the reference is consumed as a C
// library — runtime.h:156-710 — and this keeps that story true here).
PA_EXPORT void* pa_context_new(int nworkers, int rank, int world,
                               const char* comm, int gpu) {
  Context::Options o;
  o.nworkers = nworkers;
  o.rank = rank;
  o.world = world;
  o.comm = comm ? comm : "";
  o.gpu_device = gpu;
  return new Context(o);
}
PA_EXPORT void pa_context_free(void* ctx) { delete (Context*)ctx; }
PA_EXPORT void pa_context_barrier(void* ctx) { ((Context*)ctx)->barrier(); }

PA_EXPORT void* pa_dtd_new(void* ctx, const char* name) {
  return new Dtd((Context*)ctx, name ? name : "dtd");
}
PA_EXPORT void pa_dtd_wait(void* dtd) { ((Dtd*)dtd)->wait(); }
PA_EXPORT void pa_dtd_free(void* dtd) { delete (Dtd*)dtd; }

PA_EXPORT void* pa_tm_new(void* ctx, long m, long n, int mb, int nb, int p,
                          int q, long elem_size, int sym) {
  return new TiledMatrix((Context*)ctx, m, n, mb, nb, p, q,
                         (size_t)elem_size, sym != 0);
}
PA_EXPORT void pa_tm_free(void* tm) { delete (TiledMatrix*)tm; }

PA_EXPORT void pa_param_set(const char* name, const char* value) {
  param_set(name, value);
}

// Convenience DTD insertion for C callers: one call, no two-phase needed.
PA_EXPORT void pa_dtd_insert(void* dtd, void* tc, const void* args,
                             int nargs, void** datas, const int* modes,
                             int nflows, int prio, int rank) {
  void* t = pa_dtd_insert_begin(dtd, tc, args, nargs, datas, modes, nflows,
                                prio, rank);
  pa_dtd_insert_commit(dtd, t);
}

// Driver-side tile readback (tile_numpy analog for C callers): valid host
// pointer to the tile's current contents. Call only after wait().
PA_EXPORT void* pa_tm_tile_host(void* tm, int i, int j) {
  return ((TiledMatrix*)tm)->tile(i, j)->pull_to_host();
}
