// PINS — Performance INStrumentation callback chain.
//
// Reference parity: mca/pins/pins.h:26-55 registers callback chains at task
// lifecycle events, masked at runtime (parsec_pins_enable_mask) and invoked
// via PARSEC_PINS() from scheduling.c:185,300,514. Modules like
// task_profiler / print_steals / iterators_checker hang off this seam.
//
// Here: a process-wide registry of std::function callbacks with an atomic
// event mask. The hot-path cost when no callback is armed is one relaxed
// load + branch (PA_PINS macro). The built-in trace/counter sinks stay
// hard-wired for speed; PINS is the extension seam (tools, tests, user
// instrumentation — Python callbacks via pybind acquire the GIL in their
// wrapper, so they may fire from any worker thread).
#pragma once

#include <atomic>
#include <cstdint>
#include <functional>
#include <mutex>
#include <vector>

namespace pa {

struct Task;

enum class PinsEv : int {
  EXEC_BEGIN = 0,    // CPU body about to run (worker thread)
  EXEC_END = 1,      // CPU body returned
  SCHEDULE = 2,      // task became ready and was pushed to a queue
  COMPLETE = 3,      // task completed (deps released)
  GPU_SUBMIT = 4,    // GPU hook enqueued work on its exec stream
  GPU_RETIRE = 5,    // GPU task's completion event retired
  CREATE = 6,        // task object created (insertion)
  RELEASE_DEPS = 7,  // successors of a completed task were released
  STAGE_IN = 8,      // GPU stage-in for a task's flows finished issuing
  COMM_POST = 9,     // send/recv posted to the wire engine
  COMM_DONE = 10,    // wire transfer completed
  STEAL = 11,        // task obtained from another worker's queue
  COUNT = 12,
};

using PinsFn = std::function<void(PinsEv, const Task*, int worker)>;

class Pins {
 public:
  static Pins& inst() {
    static Pins p;
    return p;
  }

  // Returns a handle for unregister. mask: bitwise OR of (1<<PinsEv).
  int add(PinsFn fn, uint32_t mask) {
    std::lock_guard<std::mutex> g(mtx_);
    cbs_.push_back({next_id_, mask, std::move(fn)});
    rebuild_mask();
    return next_id_++;
  }

  void remove(int id) {
    std::lock_guard<std::mutex> g(mtx_);
    for (auto it = cbs_.begin(); it != cbs_.end(); ++it)
      if (it->id == id) {
        cbs_.erase(it);
        break;
      }
    rebuild_mask();
  }

  bool armed(PinsEv e) const {
    return mask_.load(std::memory_order_relaxed) & (1u << (int)e);
  }

  void fire(PinsEv e, const Task* t, int worker) {
    // snapshot under the lock, invoke outside it: callbacks may register/
    // remove callbacks (or block) without deadlocking the chain
    std::vector<PinsFn> snap;
    {
      std::lock_guard<std::mutex> g(mtx_);
      for (auto& c : cbs_)
        if (c.mask & (1u << (int)e)) snap.push_back(c.fn);
    }
    for (auto& fn : snap) fn(e, t, worker);
  }

 private:
  struct Cb {
    int id;
    uint32_t mask;
    PinsFn fn;
  };
  void rebuild_mask() {
    uint32_t m = 0;
    for (auto& c : cbs_) m |= c.mask;
    mask_.store(m, std::memory_order_relaxed);
  }
  std::mutex mtx_;
  std::vector<Cb> cbs_;
  std::atomic<uint32_t> mask_{0};
  int next_id_ = 1;
};

// Built-in modules (src/pins_modules.cpp): PARSEC_MCA_pins=
// task_profiler,print_steals,iterators_checker — installed at context
// construction, reports at teardown.
void pins_modules_install();
void pins_modules_finalize(int rank);

#define PA_PINS(ev, task, worker)                           \
  do {                                                      \
    auto& _p = ::pa::Pins::inst();                          \
    if (_p.armed(ev)) _p.fire(ev, task, worker);            \
  } while (0)

}  // namespace pa
