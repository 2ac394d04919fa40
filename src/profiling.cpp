#include "profiling.hpp"

// Reference parity: profiling core (parsec/profiling.c per-thread event
// buffers + dictionary, parsec_binary_profile.h PBT container), DOT
// grapher (parsec_prof_grapher.c), PINS-style counters
// (mca/pins/*, papi_sde.c); rocTX sink = profiling_nvtx.c done the ROCm
// way; debug-history ring = PARSEC_DEBUG_HISTORY (utils/debug.c).

#include <dlfcn.h>

#include <cstdarg>

#include <chrono>

#include "runtime.hpp"

namespace pa {

RuntimeCounters& counters() {
  static RuntimeCounters c;
  return c;
}

Profiler& Profiler::inst() {
  static Profiler p;
  return p;
}

uint64_t Profiler::now_ns() {
  using namespace std::chrono;
  return (uint64_t)duration_cast<nanoseconds>(
             steady_clock::now().time_since_epoch())
      .count();
}

Profiler::Buf* Profiler::tls_buf() {
  static thread_local Buf* b = nullptr;
  static std::atomic<uint32_t> next_tid{0};
  if (!b) {
    b = new Buf();
    b->tid = next_tid.fetch_add(1);
    b->recs.reserve(1 << 16);
    std::lock_guard<std::mutex> g(mtx_);
    bufs_.push_back(b);
  }
  return b;
}

void Profiler::start(const std::string& filename) {
  std::lock_guard<std::mutex> g(mtx_);
  filename_ = filename;
  file_ = fopen(filename_.c_str(), "wb");
  if (!file_) {
    fprintf(stderr, "[parsec_amd] cannot open trace file %s\n",
            filename_.c_str());
    return;
  }
  // Header: magic + ONE dictionary JSON line padded to a fixed width,
  // then raw records streamed in buffer-sized chunks. Task classes
  // register LAZILY (function-local statics, first use), so the final
  // dictionary is only known at stop — stop_and_dump() rewrites this
  // padded line in place with the complete class map.
  fprintf(file_, "PABT1\n");
  header_off_ = ftell(file_);
  std::string line = header_json_locked();
  line.resize(HEADER_PAD - 1, ' ');
  fprintf(file_, "%s\n", line.c_str());
  enabled_.store(true, std::memory_order_release);
}

std::string Profiler::header_json_locked() const {
  std::string s = "{\"classes\":{";
  bool first = true;
  for (auto& [id, name] : classes_) {
    s += (first ? "" : ",");
    s += "\"" + std::to_string(id) + "\":\"" + name + "\"";
    first = false;
  }
  s += "},\"kinds\":{\"1\":\"exec\",\"2\":\"gpu_task\",\"3\":"
       "\"stage_in\",\"4\":\"comm_send\",\"5\":\"comm_recv\","
       "\"6\":\"sched\",\"7\":\"gpu_span\"},"
       "\"rec_bytes\":" + std::to_string(sizeof(TraceRec)) + "}";
  PA_CHECK(s.size() < HEADER_PAD, "trace header dictionary too large");
  return s;
}

void Profiler::register_class(int id, const std::string& name) {
  std::lock_guard<std::mutex> g(mtx_);
  classes_.emplace_back(id, name);
}

void Profiler::flush_buf_locked(Buf* b) {
  if (file_ && !b->recs.empty())
    fwrite(b->recs.data(), sizeof(TraceRec), b->recs.size(), file_);
  b->recs.clear();
}

void Profiler::record(Ev kind, uint16_t class_id, uint64_t seq, uint64_t t0,
                      uint64_t t1) {
  Buf* b = tls_buf();
  b->recs.push_back(TraceRec{t0, t1, b->tid, (uint16_t)kind, class_id, seq});
  if (b->recs.size() >= (1u << 16)) {
    std::lock_guard<std::mutex> g(mtx_);
    flush_buf_locked(b);
  }
}

void Profiler::record_tid(Ev kind, uint16_t class_id, uint64_t seq,
                          uint64_t t0, uint64_t t1, uint32_t tid) {
  Buf* b = tls_buf();
  b->recs.push_back(TraceRec{t0, t1, tid, (uint16_t)kind, class_id, seq});
  if (b->recs.size() >= (1u << 16)) {
    std::lock_guard<std::mutex> g(mtx_);
    flush_buf_locked(b);
  }
}

void Profiler::stop_and_dump() {
  if (!enabled_.exchange(false)) return;
  std::lock_guard<std::mutex> g(mtx_);
  if (!file_) return;
  {
    // rewrite the padded header with the now-complete class dictionary
    long cur = ftell(file_);
    fseek(file_, header_off_, SEEK_SET);
    std::string line = header_json_locked();
    line.resize(HEADER_PAD - 1, ' ');
    fprintf(file_, "%s\n", line.c_str());
    fseek(file_, cur, SEEK_SET);
  }
  for (Buf* b : bufs_) flush_buf_locked(b);
  fclose(file_);
  file_ = nullptr;
}

void Profiler::dot_open(const std::string& filename) {
  std::lock_guard<std::mutex> g(dot_mtx_);
  dot_ = fopen(filename.c_str(), "w");
  if (dot_) fprintf(dot_, "digraph parsec_amd {\n");
}

void Profiler::dot_close() {
  std::lock_guard<std::mutex> g(dot_mtx_);
  if (dot_) {
    fprintf(dot_, "}\n");
    fclose(dot_);
    dot_ = nullptr;
  }
}

void Profiler::dot_node(const Task* t) {
  std::lock_guard<std::mutex> g(dot_mtx_);
  if (!dot_) return;
  fprintf(dot_, "  t%lu [label=\"%s(%lu)\"];\n", (unsigned long)t->seq,
          t->tc->name.c_str(), (unsigned long)t->seq);
}

void Profiler::dot_edge(const Task* pred, const Task* succ) {
  std::lock_guard<std::mutex> g(dot_mtx_);
  if (!dot_) return;
  fprintf(dot_, "  t%lu -> t%lu;\n", (unsigned long)pred->seq,
          (unsigned long)succ->seq);
}

}  // namespace pa

// ---------------------------------------------------------------- rocTX
namespace pa {
namespace {
int (*p_roctx_push)(const char*) = nullptr;
int (*p_roctx_pop)() = nullptr;
bool roctx_enabled = false;
}  // namespace

void roctx_init() {
  if (!param_int("profile_roctx", 0)) return;
  // rocprofv3 (rocprofiler-sdk) intercepts the SDK roctx, not the legacy
  // roctracer one — prefer it so markers actually land in the trace.
  void* h = dlopen("librocprofiler-sdk-roctx.so", RTLD_NOW | RTLD_LOCAL);
  if (!h) h = dlopen("librocprofiler-sdk-roctx.so.1", RTLD_NOW | RTLD_LOCAL);
  if (!h) h = dlopen("libroctx64.so", RTLD_NOW | RTLD_LOCAL);
  if (!h) h = dlopen("libroctx64.so.4", RTLD_NOW | RTLD_LOCAL);
  if (!h) {
    fprintf(stderr, "[parsec_amd] profile_roctx=1 but libroctx64 not found\n");
    return;
  }
  p_roctx_push = (int (*)(const char*))dlsym(h, "roctxRangePushA");
  p_roctx_pop = (int (*)())dlsym(h, "roctxRangePop");
  roctx_enabled = p_roctx_push && p_roctx_pop;
}
bool roctx_on() { return roctx_enabled; }
void roctx_push(const char* name) { if (roctx_enabled) p_roctx_push(name); }
void roctx_pop() { if (roctx_enabled) p_roctx_pop(); }
}  // namespace pa

// ------------------------------------------------------- debug history
namespace pa {
namespace {
struct DebugHistory {
  std::mutex mtx;
  std::vector<std::string> ring;
  size_t cap = 0, next = 0;
  uint64_t dropped = 0;
};
DebugHistory& dh() {
  static DebugHistory h;
  return h;
}
}  // namespace

void debug_history_init() {
  long n = param_int("debug_history", 0);
  if (n > 0) {
    auto& h = dh();
    std::lock_guard<std::mutex> g(h.mtx);
    h.cap = (size_t)n;
    h.ring.resize(h.cap);
  }
}
bool debug_history_on() { return dh().cap != 0; }

void debug_history_add(const char* fmt, ...) {
  auto& h = dh();
  if (!h.cap) return;
  char buf[256];
  va_list ap;
  va_start(ap, fmt);
  vsnprintf(buf, sizeof(buf), fmt, ap);
  va_end(ap);
  char line[320];
  snprintf(line, sizeof(line), "[%14.6f] %s",
           (double)Profiler::now_ns() / 1e9, buf);
  std::lock_guard<std::mutex> g(h.mtx);
  h.ring[h.next % h.cap] = line;
  h.next++;
}

void debug_history_dump() {
  auto& h = dh();
  if (!h.cap) return;
  std::lock_guard<std::mutex> g(h.mtx);
  size_t n = h.next < h.cap ? h.next : h.cap;
  size_t start = h.next < h.cap ? 0 : h.next - h.cap;
  fprintf(stderr, "[parsec_amd] debug history (last %zu events):\n", n);
  for (size_t i = 0; i < n; i++)
    fprintf(stderr, "  %s\n", h.ring[(start + i) % h.cap].c_str());
  fflush(stderr);
}
}  // namespace pa
