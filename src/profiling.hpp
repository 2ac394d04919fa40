// Observability: binary event tracing, DOT DAG dump, runtime counters.
//
// Reference parity:
//  - Tracing: per-thread event streams with a dictionary of event classes,
//    dumped as a compact binary trace ("PBT" analog: parsec/profiling.c,
//    parsec_binary_profile.h) convertible to Chrome trace JSON
//    (parsec_amd/tools/trace2chrome.py ~ tools/profiling/h5toctf.py).
//  - DOT grapher (parsec_prof_grapher.c): every task node + dependency edge
//    when profile_dot is set.
//  - Counters (PINS/papi_sde analogs): per-context scheduler/executed/
//    steal counts, exposed to Python and printed at teardown with
//    PARSEC_MCA_stats=1 (device.c:611-658 statistics dump analog).
#pragma once

#include <atomic>
#include <cstdint>
#include <cstdio>
#include <mutex>
#include <string>
#include <vector>

namespace pa {

struct Task;

enum class Ev : uint16_t {
  EXEC = 1,       // CPU task body
  GPU_TASK = 2,   // GPU task launch..retire (host view)
  STAGE_IN = 3,
  COMM_SEND = 4,
  COMM_RECV = 5,
  SCHED = 6,
  GPU_SPAN = 7,  // device-side span on one exec stream (hipEvent-timed);
                 // tid = 1000 + stream index (its own lane in the viewer)
};

struct TraceRec {
  uint64_t t0_ns, t1_ns;
  uint32_t tid;
  uint16_t kind;
  uint16_t class_id;
  uint64_t seq;
};

class Profiler {
 public:
  static Profiler& inst();

  bool enabled() const { return enabled_.load(std::memory_order_relaxed); }
  void start(const std::string& filename);
  void stop_and_dump();

  void record(Ev kind, uint16_t class_id, uint64_t seq, uint64_t t0_ns,
              uint64_t t1_ns);
  // record with an explicit synthetic lane id (GPU stream rows)
  void record_tid(Ev kind, uint16_t class_id, uint64_t seq, uint64_t t0_ns,
                  uint64_t t1_ns, uint32_t tid);
  void register_class(int id, const std::string& name);

  static uint64_t now_ns();

  // --- DOT grapher ---
  bool dot_enabled() const { return dot_ != nullptr; }
  void dot_node(const Task* t);
  void dot_edge(const Task* pred, const Task* succ);
  void dot_open(const std::string& filename);
  void dot_close();

 private:
  struct Buf {
    std::vector<TraceRec> recs;
    uint32_t tid;
  };
  Buf* tls_buf();

  void flush_buf_locked(Buf* b);

  std::atomic<bool> enabled_{false};
  std::string filename_;
  static constexpr size_t HEADER_PAD = 8192;
  long header_off_ = 0;
  std::string header_json_locked() const;
  std::mutex mtx_;
  FILE* file_ = nullptr;  // records stream to disk per chunk: a long traced
                          // run must not grow host RAM (profiling.c:74-160
                          // file-backed buffers analog)
  std::vector<Buf*> bufs_;
  std::vector<std::pair<int, std::string>> classes_;
  FILE* dot_ = nullptr;
  std::mutex dot_mtx_;
};

// Debug history ring (PARSEC_DEBUG_HISTORY analog, debug.c history
// buffers): when PARSEC_MCA_debug_history=N, the last N runtime events
// (task begin/complete/schedule, comm posts, GPU submit/retire) are kept
// in a ring and dumped to stderr when fatal() fires — the postmortem for
// hangs/aborts without a full trace.
void debug_history_init();
bool debug_history_on();
void debug_history_add(const char* fmt, ...)
#if defined(__GNUC__)
    __attribute__((format(printf, 1, 2)))
#endif
    ;
void debug_history_dump();

// rocTX ranges (the reference's NVTX sink, profiling_nvtx.c, done the
// ROCm-native way): with PARSEC_MCA_profile_roctx=1, task bodies and GPU
// submissions get roctxRangePush/Pop so rocprofv3's marker trace shows
// task-class names interleaved with kernels. libroctx64 is dlopen'd
// lazily: zero cost when off.
void roctx_init();
bool roctx_on();
void roctx_push(const char* name);
void roctx_pop();

// global counters (papi_sde-style software counters)
struct RuntimeCounters {
  std::atomic<uint64_t> tasks_executed_cpu{0};
  std::atomic<uint64_t> tasks_executed_gpu{0};
  std::atomic<uint64_t> tasks_scheduled{0};
  std::atomic<uint64_t> steals{0};
  std::atomic<uint64_t> comm_msgs{0};
  std::atomic<uint64_t> comm_bytes{0};
  // Copy renames: incoming versions (and OUTPUT-only rewrites) that got a
  // fresh buffer instead of WAR-waiting on readers of the old version
  // (datarepo/arena copy analog).
  std::atomic<uint64_t> renames{0};
};
RuntimeCounters& counters();

}  // namespace pa
