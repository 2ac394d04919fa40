#include "common.hpp"

// Reference parity: MCA parameter sourcing (utils/mca_param.c:1-2606 —
// env/API precedence), leveled debug output (utils/debug.c), and the
// clean-abort path (parsec.c:1127 parsec_abort + runtime.h:34-38 error
// callbacks; see set_fatal_handler).

#include <atomic>

#include <map>
#include <mutex>

namespace pa {

int debug_level() {
  static int lvl = [] {
    const char* e = getenv("PARSEC_DEBUG_VERBOSE");
    return e ? atoi(e) : 0;
  }();
  return lvl;
}

void debug_out(int level, const char* fmt, ...) {
  char buf[4096];
  va_list ap;
  va_start(ap, fmt);
  vsnprintf(buf, sizeof(buf), fmt, ap);
  va_end(ap);
  fprintf(stderr, "[parsec_amd:%d] %s\n", level, buf);
}

static std::atomic<void (*)(const char*)> g_fatal_handler{nullptr};

void set_fatal_handler(void (*fn)(const char* msg)) {
  g_fatal_handler.store(fn);
}

void fatal_check(const char* file, int line, const char* cond) {
  fatal("check failed %s:%d: %s", file, line, cond);
}

void fatal_check(const char* file, int line, const char* cond,
                 const char* fmt, ...) {
  char msg[2048];
  va_list ap;
  va_start(ap, fmt);
  vsnprintf(msg, sizeof(msg), fmt, ap);
  va_end(ap);
  fatal("check failed %s:%d: %s — %s", file, line, cond, msg);
}

void fatal(const char* fmt, ...) {
  char buf[4096];
  va_list ap;
  va_start(ap, fmt);
  vsnprintf(buf, sizeof(buf), fmt, ap);
  va_end(ap);
  fprintf(stderr, "[parsec_amd FATAL] %s\n", buf);
  fflush(stderr);
  // postmortem: dump the debug-history ring (PARSEC_MCA_debug_history=N)
  void debug_history_dump();
  debug_history_dump();
  if (auto* fn = g_fatal_handler.exchange(nullptr)) fn(buf);
  abort();
}

// ------------------------------------------------------------------ params
namespace {
std::mutex g_param_mtx;
std::map<std::string, std::string>& overrides() {
  static std::map<std::string, std::string> m;
  return m;
}
std::map<std::string, std::string>& registry() {  // name -> default, for dump
  static std::map<std::string, std::string> m;
  return m;
}

bool lookup(const char* name, std::string& out) {
  {
    std::lock_guard<std::mutex> g(g_param_mtx);
    auto it = overrides().find(name);
    if (it != overrides().end()) { out = it->second; return true; }
  }
  std::string env = "PARSEC_MCA_" + std::string(name);
  if (const char* e = getenv(env.c_str())) { out = e; return true; }
  return false;
}
}  // namespace

int64_t param_int(const char* name, int64_t dflt) {
  {
    std::lock_guard<std::mutex> g(g_param_mtx);
    registry().emplace(name, std::to_string(dflt));
  }
  std::string v;
  if (lookup(name, v)) return strtoll(v.c_str(), nullptr, 0);
  return dflt;
}

std::string param_str(const char* name, const char* dflt) {
  {
    std::lock_guard<std::mutex> g(g_param_mtx);
    registry().emplace(name, dflt);
  }
  std::string v;
  if (lookup(name, v)) return v;
  return dflt;
}

void param_set(const std::string& name, const std::string& value) {
  std::lock_guard<std::mutex> g(g_param_mtx);
  overrides()[name] = value;
}

std::string param_dump() {
  std::lock_guard<std::mutex> g(g_param_mtx);
  std::string s;
  for (auto& kv : registry())
    s += kv.first + " (default: " + kv.second + ")\n";
  return s;
}

}  // namespace pa
