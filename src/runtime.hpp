// Core runtime: Task / TaskClass / Taskpool / Scheduler / Context.
//
// Reference parity: this is the MI355X-native equivalent of
// parsec/parsec.c + parsec/scheduling.c + parsec/parsec_internal.h:117-563.
// Differences by design, not omission:
//  - Dependency tracking is dynamic successor-edge lists built at insertion
//    (the DTD model, insert_function.c:3027-3120); PTG-generated taskpools
//    (parsec_amd.ptg) drive the same Task/edge machinery.
//  - Devices: one CPU device + one HIP device per process ("one process per
//    GPU" is the MI355X scaling model; ranks talk RCCL over xGMI), so the
//    mca/device multi-vendor registry (device.c:100-310) collapses to a
//    device-kind dispatch at schedule time.
//  - Termination detection: local counters (mca/termdet/local) — correct for
//    deterministic SPMD insertion, where each rank's DAG closes locally.
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <deque>
#include <functional>
#include <map>
#include <memory>
#include <mutex>
#include <queue>
#include <string>
#include <thread>
#include <vector>

#include "common.hpp"

namespace pa {

class Context;
class Taskpool;
struct Task;
struct Data;
class GpuEngine;
class CommEngine;

// Max dataflow flows per task (reference MAX_PARAM_COUNT=20,
// parsec_internal.h:461-471; linear algebra task classes need <=6).
constexpr int MAX_FLOWS = 8;
constexpr int MAX_ARGS_BYTES = 64;

enum AccessMode : uint8_t {
  ACCESS_NONE = 0,
  ACCESS_IN = 1,
  ACCESS_OUT = 2,
  ACCESS_INOUT = 3,
  // Modifier (OR with IN/OUT/INOUT): exempt this flow from dependency
  // tracking — no chaining edges, no WAR/WAW/renaming, no transfers
  // (PARSEC_DONT_TRACK analog, dtd_test_flag_dont_track). The caller
  // asserts the access cannot conflict; the buffer must be locally
  // valid. Staging/dirty marking still follow the base IN/OUT bits.
  ACCESS_UNTRACKED = 8,
};

enum class TaskKind : uint8_t {
  CPU,        // run by a worker thread
  GPU,        // run by the HIP device engine (falls back to CPU hook if no GPU)
  COMM_SEND,  // run by the comm engine
  COMM_RECV,
};

struct GpuTaskCtx;  // defined in device_gpu.hpp

struct TaskClass {
  std::string name;
  TaskKind kind = TaskKind::CPU;
  void (*cpu_hook)(Task&) = nullptr;
  void (*gpu_hook)(Task&, GpuTaskCtx&) = nullptr;
  // Called when the task object is released (to free owned resources, e.g.
  // a Python callable reference).
  void (*destruct)(Task&) = nullptr;
  double flops = 0.0;  // time_estimate analog (parsec_internal.h:411-459)
  // GPU chores that synchronize the host internally (e.g. rocSOLVER's
  // unblocked panel factorizations) run on a WORKER thread with their own
  // stream instead of the engine's manager thread, so the manager keeps
  // launching bulk work while the panel blocks (PARSEC_HOOK_RETURN_ASYNC
  // decoupling analog for host-synchronous libraries).
  bool gpu_blocking = false;
  int id = -1;
  // opaque user hooks for the C ABI (PTG-generated code) trampolines
  void* user_cpu = nullptr;
  void* user_gpu = nullptr;
};

struct FlowRef {
  Data* data = nullptr;
  uint8_t mode = ACCESS_NONE;
};

// parsec_task_t analog (parsec_internal.h:524-563), sized for cache locality.
struct Task {
  Taskpool* tp = nullptr;
  const TaskClass* tc = nullptr;
  int32_t priority = 0;
  uint64_t seq = 0;  // insertion sequence, tie-break for priority queues
  FlowRef flows[MAX_FLOWS];
  int nflows = 0;
  alignas(8) uint8_t args[MAX_ARGS_BYTES];

  // --- dependency engine state ---
  std::atomic<int32_t> deps_remaining{1};  // starts at 1: insertion guard
  SpinLock lock;
  bool completed = false;
  std::vector<Task*> succs;
  std::atomic<int32_t> refcnt{1};

  // --- comm state (send/recv tasks) ---
  int peer = -1;         // src (recv) or dst (send) rank
  uint64_t comm_seq = 0; // per-channel sequence tag

  // --- device state ---
  void* dev_ptr[MAX_FLOWS] = {};  // staged device pointers, set by GPU engine

  template <typename T>
  T& arg(size_t off = 0) { return *reinterpret_cast<T*>(args + off); }
  template <typename T>
  const T& arg(size_t off = 0) const { return *reinterpret_cast<const T*>(args + off); }

  void retain() { refcnt.fetch_add(1, std::memory_order_relaxed); }
  void release();
};

Task* task_new(Taskpool* tp, const TaskClass* tc);

// Register dependency edge pred -> succ. Returns true if the edge was
// created (succ->deps_remaining incremented); false if pred already
// completed. Caller must have called succ->deps guard protocol:
// edges are only added between task_new(succ) and schedule_if_ready(succ).
bool task_add_edge(Task* pred, Task* succ);

// Decrement the insertion guard / a dependency; schedule when it hits 0.
void task_dec_deps(Task* t);

// Mark complete: resolve successor edges, notify taskpool, release.
void task_complete(Task* t);

// ------------------------------------------------------------------ taskpool
// parsec_taskpool_t analog (parsec_internal.h:117-163): tracks pending local
// tasks; wait() blocks until the local DAG has drained (termdet "local",
// mca/termdet/termdet_local_module.c semantics).
class Taskpool {
 public:
  explicit Taskpool(Context* ctx, std::string name = "tp");
  ~Taskpool();

  Context* context() const { return ctx_; }
  const std::string& name() const { return name_; }

  void task_created();             // nb_pending++
  void task_done();                // nb_pending--; wake waiters at 0
  void wait();                     // drain; main thread participates in progress
  // Distributed termination detection for DYNAMIC task graphs (tasks may
  // be activated by control messages at any time, so no rank knows its
  // final task count up front — mca/termdet fourcounter analog): a
  // coordinator runs quiescence waves over system control messages and
  // declares completion after two stable waves with zero pending tasks
  // and matched control-message counts. One taskpool at a time.
  void wait_dynamic();
  int64_t pending() const { return nb_pending_.load(std::memory_order_acquire); }

  uint64_t next_seq() { return seq_++; }

  // completion callback chaining (parsec_compose, compound.c:17-135)
  void on_complete(std::function<void()> cb);

  // Keep an auxiliary resource (e.g. a workspace collection) alive until
  // the taskpool is destroyed.
  void own(std::shared_ptr<void> p) { owned_.push_back(std::move(p)); }

 private:
  Context* ctx_;
  std::string name_;
  std::atomic<int64_t> nb_pending_{0};
  std::mutex mtx_;
  std::condition_variable cv_;
  uint64_t seq_ = 0;
  std::vector<std::function<void()>> on_complete_;
  std::vector<std::shared_ptr<void>> owned_;
};

// ------------------------------------------------------------------ scheduler
// Work-stealing scheduler: per-worker deque (LIFO pop for locality, FIFO
// steal) + shared priority queue for externally-released and prioritized
// tasks. Equivalent role to mca/sched/lfq (hbbuffer + system queue,
// sched_lfq_module.c:60-205) on one NUMA node.
class Scheduler {
 public:
  explicit Scheduler(int nworkers);
  ~Scheduler();

  void push(Task* t, int worker_hint);  // hint -1: external thread
  Task* pop(int worker);                // may return nullptr
  void wake_all();
  // Park worker until work might be available or stop_flag set.
  void park(int worker, const std::atomic<bool>& stop);
  size_t approx_pending() const { return npending_.load(std::memory_order_relaxed); }

 private:
  struct PQEntry {
    Task* t;
    int32_t key;  // priority (negated under sched=ip)
    bool operator<(const PQEntry& o) const {
      if (key != o.key) return key < o.key;
      return t->seq > o.t->seq;  // earlier insertion first
    }
  };
  struct WorkerQ {
    SpinLock lock;
    std::deque<Task*> dq;
    // per-worker priority heap (sched=pbq/ltq modes — maxheap.c analog)
    std::priority_queue<PQEntry> heap;
    // Approximate size for the steal fast path: reading dq.empty() on
    // another worker's deque without its lock is a data race; this atomic
    // mirror is the race-free heuristic.
    std::atomic<uint32_t> sz{0};
  };
  int nworkers_;
  int mode_ = 0;  // 0=ws 1=fifo 2=lifo 3=spq 4=rnd 5=pbq 6=ip (PARSEC_MCA_sched)
  bool invert_prio_ = false;  // sched=ip: inverse priority order
  std::vector<std::unique_ptr<WorkerQ>> wq_;
  SpinLock pq_lock_;
  std::priority_queue<PQEntry> pq_;
  std::atomic<size_t> npending_{0};
  std::mutex sleep_mtx_;
  std::condition_variable sleep_cv_;
};

// ------------------------------------------------------------------ context
// parsec_context_t analog (parsec_init, parsec.c:405-980): worker threads,
// devices, comm engine; one instance per process (= per GPU in multi-rank).
class Context {
 public:
  struct Options {
    int nworkers = -1;       // -1: from param sched_workers / hw
    int rank = 0;
    int world = 1;
    std::string comm = "";   // "", "tcp", "rccl"
    int gpu_device = -1;     // -1: auto (device `local_rank`), -2: disable
  };
  explicit Context(const Options& opt);
  ~Context();

  int rank() const { return rank_; }
  int world() const { return world_; }
  int nworkers() const { return nworkers_; }
  bool has_gpu() const { return gpu_ != nullptr; }
  GpuEngine* gpu() { return gpu_.get(); }
  CommEngine* comm() { return comm_.get(); }
  Scheduler& sched() { return *sched_; }

  // Route a ready task to its executor (CPU sched / GPU engine / comm).
  void dispatch(Task* t, int worker_hint = -1);

  // Per-context info registry (class/info.c analog: extensible key/value
  // slots on runtime objects; components publish facts about themselves
  // at attach time — device topology, engine kinds — consumed by tools
  // and printed in the stats table). Scoped keys: "device.name",
  // "comm.kind", ...
  void info_set(const std::string& key, const std::string& value) {
    std::lock_guard<std::mutex> g(info_mtx_);
    info_[key] = value;
  }
  std::string info_get(const std::string& key) const {
    std::lock_guard<std::mutex> g(info_mtx_);
    auto it = info_.find(key);
    return it == info_.end() ? std::string() : it->second;
  }
  std::map<std::string, std::string> info_all() const {
    std::lock_guard<std::mutex> g(info_mtx_);
    return info_;
  }

  // Execute one pending CPU task if any (used by wait() to contribute).
  bool progress_one();

  void barrier();  // inter-rank barrier through the comm engine

  static thread_local int tls_worker_id;

 private:
  void worker_main(int id);

  int rank_, world_, nworkers_;
  std::unique_ptr<Scheduler> sched_;
  std::vector<std::thread> workers_;
  std::atomic<bool> stop_{false};
  std::unique_ptr<GpuEngine> gpu_;
  std::unique_ptr<CommEngine> comm_;
  mutable std::mutex info_mtx_;
  std::map<std::string, std::string> info_;
  // Live stats publisher (properties-dictionary / aggregator_visu analog,
  // dictionary.c + tools/aggregator_visu): PARSEC_MCA_live_stats=<path>
  // snapshots the runtime counters to <path>.<rank> every
  // live_stats_interval_ms; `python -m parsec_amd.tools.live_top <path>`
  // is the live dashboard.
  std::thread live_thread_;
  std::atomic<bool> live_stop_{false};
  void live_stats_main(std::string path, int interval_ms);
};

void run_cpu_task(Task* t);  // execute hook + complete

}  // namespace pa
