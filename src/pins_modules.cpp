// Built-in PINS modules (mca/pins/{task_profiler,print_steals,
// iterators_checker} analogs), selected with
//   PARSEC_MCA_pins=task_profiler,print_steals,iterators_checker
// (comma list). Each hangs off the PINS callback chain (src/pins.hpp) and
// prints its report to stderr at context teardown, like the stats tables.
// State is process-wide (one context per process is the deployment model —
// one rank per GPU); concurrent contexts share the tables.
#include <atomic>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <string>
#include <vector>

#include "pins.hpp"
#include "profiling.hpp"
#include "runtime.hpp"

namespace pa {

namespace {

constexpr int MAXC = 256;  // task-class id space (ids are small ints)
constexpr int MAXW = 256;  // worker id space

struct ClassRow {
  std::atomic<const char*> name{nullptr};
  std::atomic<uint64_t> count{0};
  std::atomic<uint64_t> total_ns{0};
  std::atomic<uint64_t> max_ns{0};
};

struct State {
  bool task_profiler = false, print_steals = false, iter_checker = false;
  int handle = -1;
  ClassRow prof[MAXC];
  std::atomic<uint64_t> steals[MAXW]{};
  std::atomic<uint64_t> created[MAXC]{}, completed[MAXC]{};
};

State& st() {
  static State s;
  return s;
}

thread_local uint64_t t_exec_start_ns = 0;

void note_name(ClassRow& r, const Task* t) {
  const char* cur = r.name.load(std::memory_order_relaxed);
  if (!cur)
    r.name.compare_exchange_strong(cur, t->tc->name.c_str(),
                                   std::memory_order_relaxed);
}

void atomic_max(std::atomic<uint64_t>& m, uint64_t v) {
  uint64_t cur = m.load(std::memory_order_relaxed);
  while (v > cur && !m.compare_exchange_weak(cur, v)) {
  }
}

void pins_cb(PinsEv e, const Task* t, int worker) {
  State& s = st();
  switch (e) {
    case PinsEv::EXEC_BEGIN:
      t_exec_start_ns = Profiler::now_ns();
      break;
    case PinsEv::EXEC_END: {
      if (!s.task_profiler || !t) break;
      uint64_t dt = Profiler::now_ns() - t_exec_start_ns;
      int id = t->tc->id & (MAXC - 1);
      ClassRow& r = s.prof[id];
      note_name(r, t);
      r.count.fetch_add(1, std::memory_order_relaxed);
      r.total_ns.fetch_add(dt, std::memory_order_relaxed);
      atomic_max(r.max_ns, dt);
      break;
    }
    case PinsEv::STEAL:
      if (s.print_steals && worker >= 0 && worker < MAXW)
        s.steals[worker].fetch_add(1, std::memory_order_relaxed);
      break;
    case PinsEv::CREATE:
      if (s.iter_checker && t)
        s.created[t->tc->id & (MAXC - 1)].fetch_add(
            1, std::memory_order_relaxed);
      break;
    case PinsEv::COMPLETE:
      if (s.iter_checker && t) {
        int id = t->tc->id & (MAXC - 1);
        note_name(s.prof[id], t);
        s.completed[id].fetch_add(1, std::memory_order_relaxed);
      }
      break;
    default:
      break;
  }
}

}  // namespace

void pins_modules_install() {
  State& s = st();
  if (s.handle >= 0) return;  // already installed (second context)
  std::string mods = param_str("pins", "");
  if (mods.empty()) return;
  uint32_t mask = 0;
  size_t pos = 0;
  while (pos < mods.size()) {
    size_t e = mods.find(',', pos);
    std::string m = mods.substr(pos, e == std::string::npos ? e : e - pos);
    pos = e == std::string::npos ? mods.size() : e + 1;
    if (m == "task_profiler") {
      s.task_profiler = true;
      mask |= (1u << (int)PinsEv::EXEC_BEGIN) | (1u << (int)PinsEv::EXEC_END);
    } else if (m == "print_steals") {
      s.print_steals = true;
      mask |= (1u << (int)PinsEv::STEAL);
    } else if (m == "iterators_checker") {
      s.iter_checker = true;
      mask |= (1u << (int)PinsEv::CREATE) | (1u << (int)PinsEv::COMPLETE);
    } else if (!m.empty()) {
      fprintf(stderr, "[parsec_amd] unknown pins module '%s' (have: "
              "task_profiler, print_steals, iterators_checker)\n", m.c_str());
    }
  }
  if (mask) s.handle = Pins::inst().add(pins_cb, mask);
}

void pins_modules_finalize(int rank) {
  State& s = st();
  if (s.handle < 0) return;
  Pins::inst().remove(s.handle);
  s.handle = -1;
  if (s.task_profiler) {
    fprintf(stderr, "[pins:task_profiler] rank %d — per-class CPU body "
            "times:\n", rank);
    for (int i = 0; i < MAXC; i++) {
      ClassRow& r = s.prof[i];
      uint64_t n = r.count.load();
      if (!n) continue;
      double tot_ms = (double)r.total_ns.load() / 1e6;
      fprintf(stderr,
              "[pins:task_profiler]   %-16s n=%-8lu total=%.2f ms "
              "avg=%.1f us max=%.1f us\n",
              r.name.load() ? r.name.load() : "?", (unsigned long)n, tot_ms,
              tot_ms * 1e3 / (double)n, (double)r.max_ns.load() / 1e3);
    }
  }
  if (s.print_steals) {
    uint64_t tot = 0;
    for (auto& w : s.steals) tot += w.load();
    fprintf(stderr, "[pins:print_steals] rank %d total=%lu:", rank,
            (unsigned long)tot);
    for (int w = 0; w < MAXW; w++)
      if (s.steals[w].load())
        fprintf(stderr, " w%d=%lu", w, (unsigned long)s.steals[w].load());
    fprintf(stderr, "\n");
  }
  if (s.iter_checker) {
    // created == completed per class once the context drained: any
    // imbalance means a task leaked or double-completed (the reference's
    // iterators_checker validates successor iteration the same spirit).
    bool ok = true;
    for (int i = 0; i < MAXC; i++) {
      uint64_t cr = s.created[i].load(), co = s.completed[i].load();
      if (cr != co) {
        ok = false;
        fprintf(stderr,
                "[pins:iterators_checker] ANOMALY class %s (id %d): "
                "created=%lu completed=%lu\n",
                s.prof[i].name.load() ? s.prof[i].name.load() : "?", i,
                (unsigned long)cr, (unsigned long)co);
      }
    }
    if (ok)
      fprintf(stderr, "[pins:iterators_checker] rank %d OK: every created "
              "task completed\n", rank);
  }
}

}  // namespace pa
