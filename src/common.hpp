// parsec_amd — MI355X-native task-dataflow runtime (PaRSEC-class).
// Core utilities: spinlock, debug output, time, config params.
//
// Design parity: reference layers 0-2 (parsec/include/parsec/sys/atomic.h,
// parsec/utils/debug.c, parsec/utils/mca_param.c) re-imagined as C++17 on
// std::atomic — no OpenMPI-style object system is needed in C++ (RAII +
// shared_ptr replace PARSEC_OBJ_NEW/RETAIN/RELEASE, parsec_object.c:1-232).
#pragma once

#include <atomic>
#include <cstdarg>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <chrono>
#include <thread>

namespace pa {

// ---------------------------------------------------------------- spinlock
// Tasks hold this for O(ns) critical sections (successor-edge registration);
// a futex-based mutex would dominate the dependency-release hot path.
class SpinLock {
  std::atomic<bool> locked_{false};
 public:
  void lock() noexcept {
    for (;;) {
      if (!locked_.exchange(true, std::memory_order_acquire)) return;
      while (locked_.load(std::memory_order_relaxed))
        std::this_thread::yield();
    }
  }
  bool try_lock() noexcept {
    return !locked_.exchange(true, std::memory_order_acquire);
  }
  void unlock() noexcept { locked_.store(false, std::memory_order_release); }
};

struct SpinGuard {
  SpinLock& l;
  explicit SpinGuard(SpinLock& lk) : l(lk) { l.lock(); }
  ~SpinGuard() { l.unlock(); }
};

// ---------------------------------------------------------------- timing
inline double now_s() {
  using namespace std::chrono;
  return duration<double>(steady_clock::now().time_since_epoch()).count();
}

// ---------------------------------------------------------------- debug
// Leveled debug output (reference: parsec/utils/debug.c).
// Level from PARSEC_DEBUG_VERBOSE (0 = errors only).
int debug_level();

void debug_out(int level, const char* fmt, ...)
#if defined(__GNUC__)
    __attribute__((format(printf, 2, 3)))
#endif
    ;

#define PA_DEBUG(lvl, ...) \
  do { if (::pa::debug_level() >= (lvl)) ::pa::debug_out(lvl, __VA_ARGS__); } while (0)

// Error callback invoked (once) with the message before the process
// aborts (runtime.h:34-38 parsec_weaksym_exit / error-callback analog):
// lets an embedding application log/flush/teardown. The handler must not
// insert tasks; returning continues into abort().
void set_fatal_handler(void (*fn)(const char* msg));

[[noreturn]] void fatal(const char* fmt, ...)
#if defined(__GNUC__)
    __attribute__((format(printf, 1, 2)))
#endif
    ;

// fatal_check overloads let PA_CHECK carry an optional printf message
// (previously the varargs were silently dropped).
[[noreturn]] void fatal_check(const char* file, int line, const char* cond);
[[noreturn]] void fatal_check(const char* file, int line, const char* cond,
                              const char* fmt, ...)
#if defined(__GNUC__)
    __attribute__((format(printf, 4, 5)))
#endif
    ;

#define PA_CHECK(cond, ...) \
  do { if (!(cond)) ::pa::fatal_check(__FILE__, __LINE__, #cond, ##__VA_ARGS__); } while (0)

// ---------------------------------------------------------------- params
// MCA-parameter-style config (reference: utils/mca_param.c, 2606 LoC).
// Sources, in priority order: explicit set (--mca equivalent via API),
// environment PARSEC_MCA_<name>, registered default.
int64_t param_int(const char* name, int64_t dflt);
std::string param_str(const char* name, const char* dflt);
void param_set(const std::string& name, const std::string& value);
// Dump all registered params (parsec_usage analog).
std::string param_dump();

}  // namespace pa
