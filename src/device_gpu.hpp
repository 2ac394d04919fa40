// HIP device engine for one MI355X (gfx950).
//
// Reference parity: the role of mca/device/device_gpu.c
// (3613 LoC asynchronous GPU pipeline: manager thread, stage-in / exec /
// stage-out streams, per-stream event rings, zone-malloc pool, LRU) built
// natively on hipStream/hipEvent:
//  - a dedicated manager thread owns the GPU (the reference elects a manager
//    by CAS on gpu_device->mutex, device_gpu.c:3376-3424; a dedicated thread
//    is the steady-state of that protocol and removes the handoff races),
//  - stage-in H2D on a copy stream, kernels round-robin over N exec streams,
//    per-stream in-order event rings retire tasks (device_gpu.c:2593-2745),
//  - device memory from a pre-reserved HBM3E slab with size-class free lists
//    (zone_malloc analog, utils/zone_malloc.c) — on 288 GB HBM the working
//    set of the headline apps is fully resident, so LRU eviction is a
//    capacity escape hatch, not the steady state.
#pragma once

#include <hip/hip_runtime.h>

#include <map>
#include <memory>
#include <mutex>
#include <queue>
#include <vector>

#include "common.hpp"
#include "runtime.hpp"
#include "zone_alloc.hpp"

namespace pa {

#define PA_HIP_CHECK(expr)                                                   \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    if (_e != hipSuccess)                                                    \
      ::pa::fatal("HIP error %s at %s:%d: %s", hipGetErrorString(_e),        \
                  __FILE__, __LINE__, #expr);                                \
  } while (0)

struct GpuTaskCtx {
  hipStream_t stream;
  int device;
  class GpuEngine* engine;
  // buffers a hook retires: returned to the pool only after the task's
  // kernels complete on the stream (safe pool reuse).
  std::vector<std::pair<void*, size_t>>* deferred_frees;
};

class GpuEngine {
 public:
  // Returns nullptr when no GPU is visible (CPU-only container).
  static std::unique_ptr<GpuEngine> create(Context* ctx, int device);
  GpuEngine(Context* ctx, int device);
  ~GpuEngine();

  void enqueue(Task* t);  // task with satisfied deps, TaskKind::GPU
  // Host-blocking chore executed synchronously on the CALLING (worker)
  // thread with a thread-local stream (see TaskClass::gpu_blocking).
  void run_blocking(Task* t);

  // Device memory pool (slab + size-class free lists + LRU eviction when
  // capacity is exhausted: clean copies are dropped, dirty ones written
  // back to host first — zone_malloc + dual-LRU analog,
  // device_gpu.c:1210-1623).
  void* dev_alloc(size_t bytes);
  void dev_free(void* p, size_t bytes);
  void note_resident(Data* d);   // track for eviction
  void forget(Data* d);          // drop from residency (Data teardown)
  void pin(Data* d);             // dev_refs++ under d->lock
  void unpin(Data* d);

  // Synchronous copies (rare paths: pull_to_host, test readback).
  void copy_d2h(void* dst, const void* src, size_t bytes);
  void copy_h2d(void* dst, const void* src, size_t bytes);

  int device() const { return device_; }
  // GPU's NUMA locality from sysfs (vpmap/hwloc analog on one node):
  // node id (-1 unknown) and that node's CPU list for worker binding.
  int numa_node() const { return numa_node_; }
  const std::vector<int>& numa_cpus() const { return numa_cpus_; }
  // publish device facts into the context info registry (info.c analog)
  void publish_info(Context* ctx);
  hipStream_t comm_stream() const { return comm_stream_; }
  // Quiesce all engine streams (used before external readback).
  void sync_all();

  // stats (device.c:461-658 statistics analog)
  struct Stats {
    std::atomic<uint64_t> tasks{0};
    std::atomic<uint64_t> bytes_h2d{0};
    std::atomic<uint64_t> bytes_d2h{0};
    std::atomic<uint64_t> evictions{0};
    // flow bytes tasks needed on-device vs bytes actually staged H2D:
    // the required-vs-transferred accounting that catches staging waste
    // (device.c:611-658 statistics analog)
    std::atomic<uint64_t> bytes_required{0};
  } stats;

 private:
  struct InFlight {
    Task* task;
    hipEvent_t event;
    int stream_idx;
    uint64_t t0_ns;
    hipEvent_t start_ev = nullptr;  // timing pair (profiler on only)
    std::vector<std::pair<void*, size_t>> deferred_frees;
  };
  struct PQEntry {
    Task* t;
    bool operator<(const PQEntry& o) const {
      if (t->priority != o.t->priority) return t->priority < o.t->priority;
      return t->seq > o.t->seq;
    }
  };

  void manager_main();
  void stage_in_and_launch(Task* t);
  void stage_flows(Task* t, hipStream_t es);
  hipEvent_t event_get();
  void event_put(hipEvent_t e);

  Context* ctx_;
  int device_;
  int numa_node_ = -1;
  std::vector<int> numa_cpus_;
  void detect_numa();
  hipStream_t h2d_stream_{}, d2h_stream_{}, comm_stream_{};
  std::vector<hipStream_t> exec_streams_;
  int next_stream_ = 0;

  std::mutex q_mtx_;
  std::condition_variable q_cv_;
  std::priority_queue<PQEntry> queue_;
  std::atomic<bool> stop_{false};
  std::thread manager_;

  std::vector<std::deque<InFlight>> inflight_;  // per exec stream, in-order
  size_t n_inflight_ = 0;
  size_t max_inflight_per_stream_;
  std::vector<hipEvent_t> event_pool_;
  // device-timestamp calibration for per-stream GPU spans: t_gpu(ns) =
  // ref_ns_ + hipEventElapsedTime(ref_ev_, ev)
  std::vector<hipEvent_t> tev_pool_;  // timing-enabled events
  hipEvent_t ref_ev_ = nullptr;
  uint64_t ref_ns_ = 0;
  hipEvent_t tev_get();
  void gpu_span_calibrate();

  // memory pool
  bool evict_one(size_t bytes);
  bool retire_pass();  // poll event rings; true if anything retired
  // Return writeback-drained eviction buffers to the pool (mem_mtx_ held).
  bool reap_draining_locked();
  std::thread::id manager_tid_;
  std::mutex mem_mtx_;
  void* slab_ = nullptr;
  size_t slab_bytes_ = 0;
  bool hard_cap_ = false;  // gpu_mem_limit_mb set: no hipMalloc escape
  ZoneAlloc zone_;         // coalescing slab allocator (zone_malloc analog)
  // Overflow blocks from the hipMalloc escape hatch (outside the slab):
  // recycled by exact size, never coalesced (rare path).
  std::map<size_t, std::vector<void*>> extern_lists_;
  // Evicted-dirty buffers whose async D2H writeback is still in flight:
  // reusable only once the event completes (transfer_gpu.c W2R-task analog
  // — eviction never synchronizes the manager).
  struct Draining {
    void* buf;
    size_t bytes;
    hipEvent_t ev;
  };
  std::vector<Draining> draining_;
  SpinLock ev_lock_;  // event_pool_ is shared with eviction callers
  std::vector<Data*> resident_;
  std::atomic<uint64_t> lru_clock_{1};
};

}  // namespace pa
