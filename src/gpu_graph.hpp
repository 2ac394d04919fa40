// hipGraph capture/replay of a taskpool's GPU schedule (MI355X-native
// steady-state replay; the reference has no equivalent — CUDA-graph-style
// replay is the idiomatic way to amortize launch overhead for repeated
// launch-bound DAGs on this hardware, per the CDNA4 programming guide).
//
// Protocol:
//   tp.capture_begin()    — before any insert; world==1 + GPU required
//   ... inserts ...       — every task must run its (non-blocking) GPU chore
//   g = tp.capture_end()  — waits the pool (the RECORD pass: normal
//                           execution, with the launch order, stream
//                           assignment, cross-stream dependency edges,
//                           tasks and device-resident tiles retained),
//                           then re-issues the exact launch sequence into
//                           a hipStream capture with explicit event edges
//                           and instantiates a hipGraphExec_t.
//   g.launch(n)           — replay the whole DAG n times: one
//                           hipGraphLaunch each, no per-task host work.
//
// Semantics follow CUDA/HIP graphs: replays re-run the same kernels on
// the SAME device buffers (tiles are pinned on-device for the graph's
// lifetime), so the DAG should be idempotent or externally re-seeded.
// Lifetime: destroy the GpuGraph BEFORE the collections whose tiles it
// pins (dropping a buffer under a live graph is UB, as with CUDA graphs).
// Copy renaming is suspended for the captured pool during the record pass
// (replays need stable buffers): OUTPUT rewrites WAR-serialize instead.
// CPU tasks, comm tasks, and host-blocking GPU chores in the captured
// pool fail capture loudly with the offending task named.
#pragma once
#include <hip/hip_runtime.h>

#include <atomic>
#include <string>
#include <unordered_map>
#include <utility>
#include <vector>

#include "runtime.hpp"

namespace pa {

class GpuEngine;
struct Data;

struct GpuGraphRecorder {
  Taskpool* tp = nullptr;
  SpinLock lock;
  std::vector<Task*> log;       // launch order (retained)
  std::vector<int> stream_of;   // engine exec-stream index per entry
  std::unordered_map<uint64_t, int> idx_by_seq;  // task seq -> log index
  std::vector<std::pair<uint64_t, uint64_t>> edges;  // (pred, succ) seqs
  std::vector<Data*> pinned;    // one entry per pin() taken (dups ok)
  bool failed = false;
  std::string why;

  void fail(const char* fmt, const char* a);
};

// Set while a capture record pass is active (one at a time, process-wide).
extern std::atomic<GpuGraphRecorder*> g_gpu_recorder;

class GpuGraph {
 public:
  // Builds + instantiates from a completed record pass; takes ownership
  // of the recorder's retained tasks and pins. Throws std::runtime_error
  // on ineligible pools.
  GpuGraph(GpuEngine* eng, GpuGraphRecorder&& rec);
  ~GpuGraph();
  GpuGraph(const GpuGraph&) = delete;
  GpuGraph& operator=(const GpuGraph&) = delete;

  // Replay the captured DAG `iters` times back-to-back, then synchronize.
  void launch(int iters = 1);
  int nodes() const { return n_nodes_; }
  int n_tasks() const { return (int)tasks_.size(); }

 private:
  GpuEngine* eng_;
  hipGraph_t graph_ = nullptr;
  hipGraphExec_t exec_ = nullptr;
  std::vector<hipStream_t> cs_;   // capture/replay streams
  std::vector<hipEvent_t> evs_;   // fork/join + cross-stream edge events
  std::vector<Task*> tasks_;      // retained record-pass tasks
  std::vector<Data*> pinned_;
  std::vector<std::pair<void*, size_t>> deferred_;  // capture-time allocs
  int n_nodes_ = 0;
};

}  // namespace pa
