// hipGraph capture/replay — see gpu_graph.hpp for the protocol.
#include "gpu_graph.hpp"

#include <algorithm>
#include <stdexcept>

#include "data.hpp"
#include "device_gpu.hpp"
#include "dtd.hpp"

namespace pa {

// kernels_blas.cpp: pre-create the per-stream rocBLAS handle + device
// workspace so no hipMalloc happens inside the stream capture (rocBLAS
// allocates lazily, and allocation during capture is illegal).
void blas_warm_stream_for_capture(hipStream_t s);

std::atomic<GpuGraphRecorder*> g_gpu_recorder{nullptr};

void GpuGraphRecorder::fail(const char* fmt, const char* a) {
  SpinGuard g(lock);
  if (failed) return;
  failed = true;
  char buf[256];
  snprintf(buf, sizeof(buf), fmt, a);
  why = buf;
}

GpuGraph::GpuGraph(GpuEngine* eng, GpuGraphRecorder&& rec) : eng_(eng) {
  const bool dbg = param_int("graph_debug", 0) != 0;
#define GG_DBG(...) do { if (dbg) { fprintf(stderr, "[graph] " __VA_ARGS__); fputc('\n', stderr); fflush(stderr); } } while (0)
  GG_DBG("ctor: %zu tasks %zu edges", rec.log.size(), rec.edges.size());
  tasks_ = std::move(rec.log);
  pinned_ = std::move(rec.pinned);
  const int n = (int)tasks_.size();
  PA_HIP_CHECK(hipSetDevice(eng_->device()));

  // Compact the engine stream indices the record pass used into
  // capture-stream slots (typically all 8 exec streams).
  std::vector<int> smap(n);
  {
    std::vector<int> remap;  // engine idx -> capture idx
    for (int k = 0; k < n; k++) {
      int es = rec.stream_of[k];
      auto it = std::find(remap.begin(), remap.end(), es);
      if (it == remap.end()) { remap.push_back(es); smap[k] = (int)remap.size() - 1; }
      else smap[k] = (int)(it - remap.begin());
    }
    cs_.resize(remap.size());
  }
  GG_DBG("streams: %zu", cs_.size());
  for (auto& s : cs_) {
    PA_HIP_CHECK(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
    blas_warm_stream_for_capture(s);
  }
  GG_DBG("warmed");

  // Cross-stream ordering via WAVEFRONT LEVELS with a fork/join barrier
  // between consecutive levels. Arbitrary per-edge event meshes crash
  // hipStreamEndCapture on ROCm 7.2 at >2 streams (isolated with
  // tools/probes programs: fork/join and short chains are fine, the
  // general mesh is not) — the barrier pattern is the one ROCm handles,
  // costs ~(streams+1) events per level, and keeps all within-level
  // parallelism: tasks at equal depth run concurrently across streams,
  // which is the wavefront schedule of the DAG.
  std::vector<int> level(n, 0);
  {
    std::vector<std::vector<int>> preds(n);
    for (auto& [ps, ss] : rec.edges) {
      auto pi = rec.idx_by_seq.find(ps), si = rec.idx_by_seq.find(ss);
      if (pi == rec.idx_by_seq.end() || si == rec.idx_by_seq.end()) continue;
      int p = pi->second, q = si->second;
      PA_CHECK(p < q, "graph capture: edge against launch order (%d -> %d)",
               p, q);
      preds[q].push_back(p);
    }
    for (int k = 0; k < n; k++)
      for (int p : preds[k])
        if (level[p] + 1 > level[k]) level[k] = level[p] + 1;
  }
  // Stable order by (level, launch order): preserves the record pass's
  // stream spreading within each level.
  std::vector<int> order(n);
  for (int k = 0; k < n; k++) order[k] = k;
  std::stable_sort(order.begin(), order.end(),
                   [&](int a, int b) { return level[a] < level[b]; });

  auto mkevent = [&]() {
    hipEvent_t e;
    PA_HIP_CHECK(hipEventCreateWithFlags(&e, hipEventDisableTiming));
    evs_.push_back(e);
    return e;
  };
  auto barrier = [&]() {  // join every stream into cs_[0], fork back out
    for (size_t s2 = 1; s2 < cs_.size(); s2++) {
      hipEvent_t je = mkevent();
      PA_HIP_CHECK(hipEventRecord(je, cs_[s2]));
      PA_HIP_CHECK(hipStreamWaitEvent(cs_[0], je, 0));
    }
    hipEvent_t fe = mkevent();
    PA_HIP_CHECK(hipEventRecord(fe, cs_[0]));
    for (size_t s2 = 1; s2 < cs_.size(); s2++)
      PA_HIP_CHECK(hipStreamWaitEvent(cs_[s2], fe, 0));
  };

  // Record pass is done and the engine is idle; re-issue the launches into
  // a stream capture. ThreadLocal mode: the engine's idle manager thread
  // keeps polling its own (uncaptured) streams legally.
  GG_DBG("edges built");
  PA_HIP_CHECK(hipStreamBeginCapture(cs_[0], hipStreamCaptureModeThreadLocal));
  {
    hipEvent_t fork = mkevent();
    PA_HIP_CHECK(hipEventRecord(fork, cs_[0]));
    for (size_t s2 = 1; s2 < cs_.size(); s2++)
      PA_HIP_CHECK(hipStreamWaitEvent(cs_[s2], fork, 0));
  }
  int cur_level = 0;
  for (int oi = 0; oi < n; oi++) {
    int k = order[oi];
    if (level[k] != cur_level) {
      barrier();
      cur_level = level[k];
    }
    hipStream_t s = cs_[smap[k]];
    Task* t = tasks_[k];
    GG_DBG("launch %d/%d %s seq=%lu stream=%d level=%d", oi, n,
           t->tc->name.c_str(), (unsigned long)t->seq, smap[k], level[k]);
    std::vector<std::pair<void*, size_t>> deferred;
    GpuTaskCtx gctx{s, eng_->device(), eng_, &deferred};
    t->tc->gpu_hook(*t, gctx);
    for (auto& db : deferred) deferred_.push_back(db);
  }
  for (size_t s = 1; s < cs_.size(); s++) {
    hipEvent_t je = mkevent();
    PA_HIP_CHECK(hipEventRecord(je, cs_[s]));
    PA_HIP_CHECK(hipStreamWaitEvent(cs_[0], je, 0));
  }
  GG_DBG("ending capture");
  PA_HIP_CHECK(hipStreamEndCapture(cs_[0], &graph_));
  PA_HIP_CHECK(hipGraphInstantiate(&exec_, graph_, nullptr, nullptr, 0));
  size_t nn = 0;
  PA_HIP_CHECK(hipGraphGetNodes(graph_, nullptr, &nn));
  n_nodes_ = (int)nn;
  GG_DBG("instantiated: %d nodes", n_nodes_);
#undef GG_DBG
}

void GpuGraph::launch(int iters) {
  PA_HIP_CHECK(hipSetDevice(eng_->device()));
  for (int i = 0; i < iters; i++)
    PA_HIP_CHECK(hipGraphLaunch(exec_, cs_[0]));
  PA_HIP_CHECK(hipStreamSynchronize(cs_[0]));
}

GpuGraph::~GpuGraph() {
  if (exec_) hipGraphExecDestroy(exec_);
  if (graph_) hipGraphDestroy(graph_);
  for (auto e : evs_) hipEventDestroy(e);
  for (auto s : cs_) hipStreamDestroy(s);
  for (auto& [p, b] : deferred_) eng_->dev_free(p, b);
  for (Data* d : pinned_) eng_->unpin(d);
  for (Task* t : tasks_) t->release();
}

// ------------------------------------------------------------- Dtd methods
void Dtd::capture_begin() {
  Context* c = context();
  if (c->world() != 1)
    throw std::runtime_error(
        "gpu graph capture: single-process only (comm tasks are not "
        "capturable)");
  if (!c->gpu())
    throw std::runtime_error("gpu graph capture requires a visible GPU");
  PA_CHECK(g_gpu_recorder.load(std::memory_order_acquire) == nullptr,
           "another gpu graph capture is already active");
  auto* rec = new GpuGraphRecorder();
  rec->tp = this;
  g_gpu_recorder.store(rec, std::memory_order_release);
}

std::unique_ptr<GpuGraph> Dtd::capture_end() {
  GpuGraphRecorder* rec = g_gpu_recorder.load(std::memory_order_acquire);
  PA_CHECK(rec && rec->tp == this, "capture_end without capture_begin");
  wait();  // the RECORD pass: normal execution with logging
  g_gpu_recorder.store(nullptr, std::memory_order_release);
  if (param_int("graph_debug", 0))
    fprintf(stderr, "[graph] record pass done: %zu launches, failed=%d\n",
            rec->log.size(), (int)rec->failed);
  std::unique_ptr<GpuGraphRecorder> owned(rec);
  GpuEngine* eng = context()->gpu();
  if (owned->failed || owned->log.empty()) {
    for (Data* d : owned->pinned) eng->unpin(d);
    for (Task* t : owned->log) t->release();
    throw std::runtime_error(
        "gpu graph capture failed: " +
        (owned->why.empty() ? std::string("no GPU tasks were recorded")
                            : owned->why));
  }
  return std::make_unique<GpuGraph>(eng, std::move(*owned));
}

}  // namespace pa
