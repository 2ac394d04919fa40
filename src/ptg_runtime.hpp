// Runtime support header for PTG-generated C++ (parsec_amd.ptg compiler).
//
// Generated code materializes the task graph (instances + explicit RAW/CTL
// edges from the JDF arrows), topologically orders it with a deterministic
// tie-break, and inserts through the DTD chaining engine — which derives
// RAW/WAR/WAW and all inter-rank transfers from the per-tile access
// sequence. Explicit edges are still applied (they are what orders CTL
// flows and guards the topo order). This differs from the reference's
// jdf2c (compact, never-materialized dependency iteration, jdf2c.c:3047+)
// by design: on one MI355X node the instance set of the headline DAGs is
// tiny next to HBM, and materialization makes the distributed insertion
// order deterministic, which is what the RCCL comm layer keys on.
//
// Only the pa_* C ABI of _core.so is used; this header is self-contained.
#pragma once

#include <hip/hip_runtime.h>

#include <array>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <map>
#include <queue>
#include <vector>

extern "C" {
int pa_ctx_rank(void* ctx);
int pa_ctx_world(void* ctx);
int pa_ctx_has_gpu(void* ctx);
void* pa_tm_tile(void* tm, int i, int j);
int pa_tm_rank_of(void* tm, int i, int j);
int pa_data_home_rank(void* d);
void* pa_taskclass_new(const char* name, int want_gpu, void (*cpu)(void*),
                       void (*gpu)(void*, void*));
void* pa_task_args(void* t);
void* pa_task_dev_ptr(void* t, int flow);
void* pa_task_host_ptr(void* t, int flow);
void* pa_dtd_insert_begin(void* dtd, void* tc, const void* args, int nargs,
                          void** datas, const int* modes, int nflows,
                          int prio, int rank);
void pa_dtd_insert_commit(void* dtd, void* task);
void pa_task_edge(void* pred, void* succ);
void pa_task_retain(void* t);
void pa_task_release(void* t);
void* pa_dtd_scratch(void* dtd, long bytes, int home_rank);
}

namespace paptg {

constexpr int MAXP = 8;   // params per task (packed into the 64-byte args)
constexpr int MAXF = 8;

using PKey = std::pair<int, std::array<long, MAXP>>;

struct Inst {
  int cls = 0;
  std::array<long, MAXP> P{};
  int np = 0;
  int rank = 0;
  int prio = 0;
  void* datas[MAXF] = {};
  int modes[MAXF] = {};
  int nflows = 0;
  std::vector<PKey> pred_keys;
  std::vector<int> preds;
  void* task = nullptr;
  int indeg = 0;
};

class Graph {
 public:
  Graph(void* ctx, void* dtd) : ctx_(ctx), dtd_(dtd) {
    rank_ = pa_ctx_rank(ctx);
  }

  int add(Inst&& inst) {
    PKey key{inst.cls, inst.P};
    int id = (int)insts_.size();
    auto r = index_.emplace(key, id);
    if (!r.second) {
      fprintf(stderr, "[ptg] duplicate task instance (class %d)\n", inst.cls);
      abort();
    }
    insts_.push_back(std::move(inst));
    return id;
  }

  int find(const PKey& key) const {
    auto it = index_.find(key);
    return it == index_.end() ? -1 : it->second;
  }

  // Resolve pred keys to ids, topo-sort (Kahn, deterministic tie-break by
  // (class, params)), insert through the DTD engine, apply explicit edges.
  void run() {
    const int n = (int)insts_.size();
    std::vector<std::vector<int>> out(n);
    for (int i = 0; i < n; i++) {
      for (auto& k : insts_[i].pred_keys) {
        int p = find(k);
        if (p < 0) {
          fprintf(stderr,
                  "[ptg] unresolved dependency: class %d references a "
                  "non-existent predecessor instance of class %d\n",
                  insts_[i].cls, k.first);
          abort();
        }
        insts_[i].preds.push_back(p);
        out[p].push_back(i);
        insts_[i].indeg++;
      }
    }
    auto cmp = [&](int a, int b) {
      const Inst &A = insts_[a], &B = insts_[b];
      if (A.cls != B.cls) return A.cls > B.cls;
      return A.P > B.P;
    };
    std::priority_queue<int, std::vector<int>, decltype(cmp)> ready(cmp);
    for (int i = 0; i < n; i++)
      if (insts_[i].indeg == 0) ready.push(i);
    std::vector<void*> retained;
    int done = 0;
    while (!ready.empty()) {
      int i = ready.top();
      ready.pop();
      Inst& in = insts_[i];
      long args[MAXP];
      for (int p = 0; p < MAXP; p++) args[p] = in.P[p];
      in.task = pa_dtd_insert_begin(dtd_, classes_[in.cls], args,
                                    (int)sizeof(args), in.datas, in.modes,
                                    in.nflows, in.prio, in.rank);
      if (in.task) {
        pa_task_retain(in.task);
        retained.push_back(in.task);
        for (int p : in.preds)
          if (insts_[p].task) pa_task_edge(insts_[p].task, in.task);
      }
      pa_dtd_insert_commit(dtd_, in.task);
      done++;
      for (int s : out[i])
        if (--insts_[s].indeg == 0) ready.push(s);
    }
    if (done != n) {
      fprintf(stderr, "[ptg] dependency cycle: %d of %d tasks ordered\n",
              done, n);
      abort();
    }
    for (void* t : retained) pa_task_release(t);
  }

  void set_classes(std::vector<void*> cls) { classes_ = std::move(cls); }
  int rank() const { return rank_; }

 private:
  void* ctx_;
  void* dtd_;
  int rank_;
  std::vector<Inst> insts_;
  std::map<PKey, int> index_;
  std::vector<void*> classes_;
};

// ---- NEW-tile registry (JDF `<- NEW [size=...]`): one scratch datum per
// (class, params, flow), owned by the taskpool, shared by every consumer
// that resolves its binding to this instance's flow.
inline std::map<std::tuple<int, std::array<long, MAXP>, int>, void*>&
new_tile_map() {
  static std::map<std::tuple<int, std::array<long, MAXP>, int>, void*> m;
  return m;
}
inline void*& new_tile_dtd() {
  static void* d = nullptr;
  return d;
}
inline void new_tiles_reset(void* dtd) {
  new_tile_map().clear();
  new_tile_dtd() = dtd;
}
inline void* ptg_new_tile(int cls, const long* P, int flow, long bytes,
                          int rank) {
  std::array<long, MAXP> key{};
  for (int i = 0; i < MAXP; i++) key[i] = P[i];
  void*& slot = new_tile_map()[{cls, key, flow}];
  if (!slot) slot = pa_dtd_scratch(new_tile_dtd(), bytes, rank);
  return slot;
}

}  // namespace paptg
