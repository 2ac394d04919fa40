#include "runtime.hpp"

#include <cstring>

#include "data.hpp"
#include "device_gpu.hpp"
#include "gpu_graph.hpp"
#include "comm.hpp"
#include "pins.hpp"
#include "profiling.hpp"

#include <pthread.h>
#include <unistd.h>

namespace pa {

thread_local int Context::tls_worker_id = -1;

// ------------------------------------------------------------------ Task
// Task mempool (mempool.c / private_mempool analog): tasks churn at
// hundreds of thousands per second, so allocation goes through a shared
// pool with per-thread caches instead of the global heap. Recycled tasks
// keep their succs vector capacity — the dependency engine's only
// per-task allocation amortizes away.
namespace {
constexpr size_t TASK_CACHE_MAX = 128, TASK_CACHE_BATCH = 64;
SpinLock g_task_pool_lock;
std::vector<Task*> g_task_pool;
thread_local std::vector<Task*> t_task_cache;

Task* task_alloc() {
  if (t_task_cache.empty()) {
    SpinGuard g(g_task_pool_lock);
    size_t take = std::min(TASK_CACHE_BATCH, g_task_pool.size());
    for (size_t i = 0; i < take; i++) {
      t_task_cache.push_back(g_task_pool.back());
      g_task_pool.pop_back();
    }
  }
  if (!t_task_cache.empty()) {
    Task* t = t_task_cache.back();
    t_task_cache.pop_back();
    return t;
  }
  return new Task();
}

void task_recycle(Task* t) {
  // reset to construction state (succs keeps its capacity)
  t->tp = nullptr;
  t->tc = nullptr;
  t->priority = 0;
  t->nflows = 0;
  for (auto& f : t->flows) f = FlowRef{};
  t->deps_remaining.store(1, std::memory_order_relaxed);
  t->completed = false;
  t->succs.clear();
  t->refcnt.store(1, std::memory_order_relaxed);
  t->peer = -1;
  t->comm_seq = 0;
  memset(t->dev_ptr, 0, sizeof(t->dev_ptr));
  if (t_task_cache.size() >= TASK_CACHE_MAX) {
    SpinGuard g(g_task_pool_lock);
    for (size_t i = 0; i < TASK_CACHE_BATCH; i++) {
      g_task_pool.push_back(t_task_cache.back());
      t_task_cache.pop_back();
    }
  }
  t_task_cache.push_back(t);
}
}  // namespace

Task* task_new(Taskpool* tp, const TaskClass* tc) {
  Task* t = task_alloc();
  t->tp = tp;
  t->tc = tc;
  t->seq = tp->next_seq();
  tp->task_created();
  PA_PINS(PinsEv::CREATE, t, Context::tls_worker_id);
  Profiler& pr = Profiler::inst();
  if (pr.dot_enabled()) pr.dot_node(t);
  return t;
}

void Task::release() {
  if (refcnt.fetch_sub(1, std::memory_order_acq_rel) == 1) {
    if (tc->destruct) tc->destruct(*this);
    task_recycle(this);
  }
}

bool task_add_edge(Task* pred, Task* succ) {
  Profiler& pr = Profiler::inst();
  if (pr.dot_enabled()) pr.dot_edge(pred, succ);
  // hipGraph record pass: keep the edge even when the predecessor already
  // completed — a REPLAY runs everything concurrently, so the device-side
  // ordering must be rebuilt from every edge (gpu_graph.hpp).
  if (GpuGraphRecorder* rec = g_gpu_recorder.load(std::memory_order_acquire);
      rec && pred->tp == rec->tp && succ->tp == rec->tp) {
    SpinGuard g(rec->lock);
    rec->edges.emplace_back(pred->seq, succ->seq);
  }
  // Dependency-release race protocol (SURVEY.md §7 "hard parts"): the edge
  // is registered under the predecessor's lock; if the predecessor already
  // completed, the successor does not wait on it.
  pred->lock.lock();
  if (pred->completed) {
    pred->lock.unlock();
    return false;
  }
  succ->deps_remaining.fetch_add(1, std::memory_order_relaxed);
  pred->succs.push_back(succ);
  pred->lock.unlock();
  return true;
}

void task_dec_deps(Task* t) {
  if (t->deps_remaining.fetch_sub(1, std::memory_order_acq_rel) == 1) {
    t->tp->context()->dispatch(t, Context::tls_worker_id);
  }
}

void task_complete(Task* t) {
  t->lock.lock();
  t->completed = true;
  std::vector<Task*> succs;
  succs.swap(t->succs);
  t->lock.unlock();
  for (Task* s : succs) task_dec_deps(s);
  PA_PINS(PinsEv::RELEASE_DEPS, t, Context::tls_worker_id);
  if (debug_history_on())
    debug_history_add("done %s seq=%lu", t->tc->name.c_str(),
                      (unsigned long)t->seq);
  PA_PINS(PinsEv::COMPLETE, t, Context::tls_worker_id);
  Taskpool* tp = t->tp;
  t->release();
  tp->task_done();
}

// Periodic snapshot of the live runtime properties: one JSON object per
// write, atomically replaced (write temp + rename) so readers never see a
// torn snapshot. Fields mirror the stats table plus instantaneous queue
// depth — the reference exposes the same through its shared-memory
// properties dictionary consumed by the aggregator dashboards.
void Context::live_stats_main(std::string path, int interval_ms) {
  const std::string tmp = path + ".tmp";
  const auto t0 = std::chrono::steady_clock::now();
  while (!live_stop_.load(std::memory_order_acquire)) {
    RuntimeCounters& c = counters();
    double up = std::chrono::duration<double>(
                    std::chrono::steady_clock::now() - t0).count();
    FILE* f = fopen(tmp.c_str(), "w");
    if (f) {
      fprintf(f,
              "{\"rank\": %d, \"world\": %d, \"workers\": %d, "
              "\"uptime_s\": %.3f, \"ready_queue\": %zu, "
              "\"tasks_cpu\": %lu, \"tasks_gpu\": %lu, "
              "\"scheduled\": %lu, \"steals\": %lu, "
              "\"comm_msgs\": %lu, \"comm_bytes\": %lu, "
              "\"renames\": %lu",
              rank_, world_, nworkers_, up, sched_->approx_pending(),
              (unsigned long)c.tasks_executed_cpu.load(),
              (unsigned long)c.tasks_executed_gpu.load(),
              (unsigned long)c.tasks_scheduled.load(),
              (unsigned long)c.steals.load(),
              (unsigned long)c.comm_msgs.load(),
              (unsigned long)c.comm_bytes.load(),
              (unsigned long)c.renames.load());
      if (gpu_)
        fprintf(f,
                ", \"gpu_tasks\": %lu, \"gpu_h2d\": %lu, "
                "\"gpu_d2h\": %lu, \"gpu_evictions\": %lu",
                (unsigned long)gpu_->stats.tasks.load(),
                (unsigned long)gpu_->stats.bytes_h2d.load(),
                (unsigned long)gpu_->stats.bytes_d2h.load(),
                (unsigned long)gpu_->stats.evictions.load());
      fprintf(f, "}\n");
      fclose(f);
      rename(tmp.c_str(), path.c_str());
    }
    for (int i = 0; i < interval_ms && !live_stop_.load(); i += 20)
      std::this_thread::sleep_for(std::chrono::milliseconds(20));
  }
  remove(path.c_str());
}

void run_cpu_task(Task* t) {
  if (t->tc->kind == TaskKind::GPU && t->tp->context()->gpu()) {
    // blocking GPU chore routed through the CPU scheduler (on CPU-only
    // contexts GPU-kind tasks run their cpu_hook below instead)
    t->tp->context()->gpu()->run_blocking(t);
    return;
  }
  if (GpuGraphRecorder* rec = g_gpu_recorder.load(std::memory_order_acquire);
      rec && t->tp == rec->tp)
    rec->fail("CPU task '%s' in a captured taskpool is not replayable",
              t->tc->name.c_str());
  Profiler& pr = Profiler::inst();
  // OUTPUT-only flows are about to be produced on the host: drop stale
  // device validity first so LRU writeback cannot race the body's writes.
  for (int i = 0; i < t->nflows; i++)
    if (t->flows[i].data && t->flows[i].mode == ACCESS_OUT)
      t->flows[i].data->begin_host_overwrite();
  if (t->tc->cpu_hook) {
    if (debug_history_on())
      debug_history_add("exec %s seq=%lu w=%d", t->tc->name.c_str(),
                        (unsigned long)t->seq, Context::tls_worker_id);
    PA_PINS(PinsEv::EXEC_BEGIN, t, Context::tls_worker_id);
    if (roctx_on()) roctx_push(t->tc->name.c_str());
    if (pr.enabled()) {
      uint64_t t0 = Profiler::now_ns();
      t->tc->cpu_hook(*t);
      pr.record(Ev::EXEC, (uint16_t)t->tc->id, t->seq, t0, Profiler::now_ns());
    } else {
      t->tc->cpu_hook(*t);
    }
    if (roctx_on()) roctx_pop();
    PA_PINS(PinsEv::EXEC_END, t, Context::tls_worker_id);
  }
  counters().tasks_executed_cpu.fetch_add(1, std::memory_order_relaxed);
  task_complete(t);
}

// ------------------------------------------------------------------ Taskpool
Taskpool::Taskpool(Context* ctx, std::string name)
    : ctx_(ctx), name_(std::move(name)) {}

Taskpool::~Taskpool() {
  PA_CHECK(nb_pending_.load() == 0);
}

void Taskpool::task_created() {
  nb_pending_.fetch_add(1, std::memory_order_acq_rel);
}

void Taskpool::task_done() {
  if (nb_pending_.fetch_sub(1, std::memory_order_acq_rel) == 1) {
    std::vector<std::function<void()>> cbs;
    {
      std::lock_guard<std::mutex> g(mtx_);
      cbs.swap(on_complete_);
    }
    for (auto& cb : cbs) cb();
    cv_.notify_all();
  }
}

void Taskpool::on_complete(std::function<void()> cb) {
  bool fire = false;
  {
    std::lock_guard<std::mutex> g(mtx_);
    if (nb_pending_.load(std::memory_order_acquire) == 0) fire = true;
    else on_complete_.push_back(std::move(cb));
  }
  if (fire) cb();
}

void Taskpool::wait() {
  // Main thread contributes to CPU progress while waiting
  // (__parsec_context_wait, scheduling.c:727-863).
  while (nb_pending_.load(std::memory_order_acquire) != 0) {
    if (!ctx_->progress_one()) {
      std::unique_lock<std::mutex> g(mtx_);
      if (nb_pending_.load(std::memory_order_acquire) == 0) break;
      cv_.wait_for(g, std::chrono::microseconds(200));
    }
  }
}

void Taskpool::wait_dynamic() {
  CommEngine* ce = ctx_->comm();
  if (!ce || ctx_->world() == 1) {
    wait();
    return;
  }
  constexpr uint32_t SYS = 0x80000000u;
  constexpr uint32_t TD_PROBE = SYS | 1, TD_REPLY = SYS | 2,
                     TD_DONE = SYS | 3;
  const int world = ctx_->world(), rank = ctx_->rank();
  struct TdState {
    std::mutex m;
    std::condition_variable cv;
    uint64_t wave = 0;
    int got = 0;
    uint64_t sum_p = 0, sum_s = 0, sum_r = 0;
    std::atomic<bool> done{false};
  };
  auto st = std::make_shared<TdState>();
  // epoch disambiguates replies across successive wait_dynamic calls: a
  // straggler reply stashed between calls must never count toward a
  // later call's wave of the same number
  static std::atomic<uint64_t> td_epoch{0};
  const uint64_t epoch = td_epoch.fetch_add(1) + 1;
  auto triple = [this, ce](uint64_t* v) {
    v[0] = (uint64_t)pending();
    v[1] = ce->ctl_sent();
    v[2] = ce->ctl_recvd();
  };
  ce->set_sys_handler([this, st, ce, rank, triple, epoch](
                          int src, uint32_t tag, const std::string& pl) {
    if (tag == TD_PROBE) {
      uint64_t rep[5];
      memcpy(rep, pl.data(), 16);  // {epoch, wave}
      triple(rep + 2);
      ce->send_ctl(src, SYS | 2, rep, sizeof(rep));
    } else if (tag == TD_REPLY) {
      uint64_t rep[5];
      memcpy(rep, pl.data(), sizeof(rep));
      std::lock_guard<std::mutex> g(st->m);
      if (rep[0] == epoch && rep[1] == st->wave) {
        st->got++;
        st->sum_p += rep[2];
        st->sum_s += rep[3];
        st->sum_r += rep[4];
        st->cv.notify_all();
      }
    } else if (tag == TD_DONE) {
      st->done.store(true, std::memory_order_release);
    }
  });
  if (rank == 0) {
    uint64_t prev_s = ~0ull, prev_r = ~0ull;
    bool prev_valid = false;
    for (;;) {
      while (pending() > 0)
        if (!ctx_->progress_one())
          std::this_thread::sleep_for(std::chrono::microseconds(100));
      uint64_t wave;
      {
        std::lock_guard<std::mutex> g(st->m);
        wave = ++st->wave;
        st->got = 0;
        st->sum_p = st->sum_s = st->sum_r = 0;
      }
      uint64_t probe[2] = {epoch, wave};
      for (int d = 1; d < world; d++)
        ce->send_ctl(d, TD_PROBE, probe, sizeof(probe));
      std::unique_lock<std::mutex> g(st->m);
      st->cv.wait_for(g, std::chrono::seconds(60),
                      [&] { return st->got == world - 1; });
      PA_CHECK(st->got == world - 1, "wait_dynamic: probe wave timed out");
      uint64_t own[3];
      triple(own);
      uint64_t P = st->sum_p + own[0];
      uint64_t S = st->sum_s + own[1];
      uint64_t R = st->sum_r + own[2];
      g.unlock();
      if (P == 0 && S == R && prev_valid && S == prev_s && R == prev_r) {
        for (int d = 1; d < world; d++)
          ce->send_ctl(d, TD_DONE, nullptr, 0);
        break;
      }
      prev_s = S;
      prev_r = R;
      prev_valid = true;
      std::this_thread::sleep_for(std::chrono::milliseconds(1));
    }
  } else {
    while (!st->done.load(std::memory_order_acquire))
      if (!ctx_->progress_one())
        std::this_thread::sleep_for(std::chrono::microseconds(200));
  }
  wait();  // belt-and-braces local drain
  ce->set_sys_handler(CommEngine::CtlHandler{});
}

// ------------------------------------------------------------------ Scheduler
Scheduler::Scheduler(int nworkers) : nworkers_(nworkers) {
  // MCA "sched" analog (mca/sched/*): ws = per-worker deques + steal (lfq
  // style, default); fifo/lifo = one shared queue (gd/ll styles); spq =
  // everything through the shared priority queue (sched_spq/ap analog);
  // rnd = random victim order (sched_rnd, for schedule-robustness tests).
  // The priority queue always serves prioritized and externally-released
  // tasks.
  // Module names accept the reference's MCA names as aliases: lfq -> ws,
  // gd -> fifo, ll -> lifo, ap -> spq, ltq/llp/lhq/pbq -> pbq (per-worker
  // priority heaps + nearest-first steal: on one NUMA-ish node per GPU the
  // hbbuffer hierarchy collapses to this), ip = inverse priority.
  std::string kind = param_str("sched", "ws");
  mode_ = kind == "fifo" || kind == "gd" ? 1
          : kind == "lifo" || kind == "ll" ? 2
          : kind == "spq" || kind == "ap" ? 3
          : kind == "rnd" ? 4
          : kind == "pbq" || kind == "ltq" || kind == "llp" || kind == "lhq"
                ? 5
          : kind == "ip" ? 6
          : 0;
  invert_prio_ = kind == "ip";
  for (int i = 0; i < nworkers_; i++) wq_.emplace_back(new WorkerQ());
}

Scheduler::~Scheduler() = default;

void Scheduler::push(Task* t, int worker_hint) {
  const int32_t key = invert_prio_ ? -t->priority : t->priority;
  if (mode_ == 3 || mode_ == 6) {  // spq/ip: single shared priority queue
    pq_lock_.lock();
    pq_.push(PQEntry{t, key});
    pq_lock_.unlock();
    npending_.fetch_add(1, std::memory_order_release);
    sleep_cv_.notify_one();
    return;
  }
  if (mode_ == 5) {  // pbq/ltq: per-worker priority heaps
    int w = worker_hint >= 0 && worker_hint < nworkers_ ? worker_hint : 0;
    WorkerQ& q = *wq_[w];
    q.lock.lock();
    q.heap.push(PQEntry{t, key});
    q.sz.store((uint32_t)q.heap.size(), std::memory_order_relaxed);
    q.lock.unlock();
    npending_.fetch_add(1, std::memory_order_release);
    sleep_cv_.notify_one();
    return;
  }
  if (mode_ == 1 || mode_ == 2) {
    WorkerQ& q = *wq_[0];
    q.lock.lock();
    if (mode_ == 2) q.dq.push_front(t);
    else q.dq.push_back(t);
    q.sz.store((uint32_t)q.dq.size(), std::memory_order_relaxed);
    q.lock.unlock();
    npending_.fetch_add(1, std::memory_order_release);
    sleep_cv_.notify_one();
    return;
  }
  if (worker_hint >= 0 && worker_hint < nworkers_ && t->priority == 0) {
    WorkerQ& q = *wq_[worker_hint];
    q.lock.lock();
    q.dq.push_front(t);
    q.sz.store((uint32_t)q.dq.size(), std::memory_order_relaxed);
    q.lock.unlock();
  } else {
    pq_lock_.lock();
    pq_.push(PQEntry{t, key});
    pq_lock_.unlock();
  }
  npending_.fetch_add(1, std::memory_order_release);
  sleep_cv_.notify_one();
}

Task* Scheduler::pop(int worker) {
  Task* t = nullptr;
  if (mode_ == 5) {  // pbq/ltq: local heap, then steal other heaps' tops
    int w = worker >= 0 ? worker : 0;
    for (int d = 0; d < nworkers_; d++) {
      int v = (w + d) % nworkers_;
      WorkerQ& q = *wq_[v];
      if (d > 0 && q.sz.load(std::memory_order_relaxed) == 0) continue;
      q.lock.lock();
      if (!q.heap.empty()) {
        t = q.heap.top().t;
        q.heap.pop();
        q.sz.store((uint32_t)q.heap.size(), std::memory_order_relaxed);
      }
      q.lock.unlock();
      if (t) {
        npending_.fetch_sub(1, std::memory_order_relaxed);
        if (d > 0) {
          counters().steals.fetch_add(1, std::memory_order_relaxed);
          PA_PINS(PinsEv::STEAL, t, worker);
        }
        return t;
      }
    }
    // externally-released tasks (hint -1 lands on worker 0's heap), plus
    // anything routed through the shared pq by other components
    pq_lock_.lock();
    if (!pq_.empty()) { t = pq_.top().t; pq_.pop(); }
    pq_lock_.unlock();
    if (t) npending_.fetch_sub(1, std::memory_order_relaxed);
    return t;
  }
  if (mode_ == 3 || mode_ == 6) {
    pq_lock_.lock();
    if (!pq_.empty()) { t = pq_.top().t; pq_.pop(); }
    pq_lock_.unlock();
    if (t) npending_.fetch_sub(1, std::memory_order_relaxed);
    return t;
  }
  if (mode_ == 1 || mode_ == 2) {
    WorkerQ& q = *wq_[0];
    q.lock.lock();
    if (!q.dq.empty()) {
      t = q.dq.front();
      q.dq.pop_front();
      q.sz.store((uint32_t)q.dq.size(), std::memory_order_relaxed);
    }
    q.lock.unlock();
    if (t) { npending_.fetch_sub(1, std::memory_order_relaxed); return t; }
    pq_lock_.lock();
    if (!pq_.empty()) { t = pq_.top().t; pq_.pop(); }
    pq_lock_.unlock();
    if (t) npending_.fetch_sub(1, std::memory_order_relaxed);
    return t;
  }
  if (worker >= 0) {
    WorkerQ& q = *wq_[worker];
    q.lock.lock();
    if (!q.dq.empty()) {
      t = q.dq.front();
      q.dq.pop_front();
      q.sz.store((uint32_t)q.dq.size(), std::memory_order_relaxed);
    }
    q.lock.unlock();
    if (t) { npending_.fetch_sub(1, std::memory_order_relaxed); return t; }
  }
  // shared priority queue
  pq_lock_.lock();
  if (!pq_.empty()) { t = pq_.top().t; pq_.pop(); }
  pq_lock_.unlock();
  if (t) { npending_.fetch_sub(1, std::memory_order_relaxed); return t; }
  // steal (FIFO end) from other workers: nearest-first (hwloc-distance
  // analog on one NUMA node), or a per-call random ordering (sched=rnd)
  // that shakes out order-dependent bugs in tests
  uint32_t rot = 1;
  if (mode_ == 4) {
    static thread_local uint32_t rng = 0x9E3779B9u ^ (uint32_t)(uintptr_t)&t;
    rng = rng * 1664525u + 1013904223u;
    rot = 1 + rng % (uint32_t)(nworkers_ > 1 ? nworkers_ - 1 : 1);
  }
  for (int d = 1; d < nworkers_; d++) {
    int dd = (int)((d * rot - 1) % (uint32_t)(nworkers_ > 1 ? nworkers_ : 1)) + 1;
    int v = (worker >= 0 ? (worker + dd) % nworkers_ : dd - 1);
    WorkerQ& q = *wq_[v];
    if (q.sz.load(std::memory_order_relaxed) == 0) continue;
    q.lock.lock();
    if (!q.dq.empty()) {
      t = q.dq.back();
      q.dq.pop_back();
      q.sz.store((uint32_t)q.dq.size(), std::memory_order_relaxed);
    }
    q.lock.unlock();
    if (t) {
      npending_.fetch_sub(1, std::memory_order_relaxed);
      counters().steals.fetch_add(1, std::memory_order_relaxed);
      PA_PINS(PinsEv::STEAL, t, worker);
      return t;
    }
  }
  return nullptr;
}

void Scheduler::wake_all() { sleep_cv_.notify_all(); }

void Scheduler::park(int, const std::atomic<bool>& stop) {
  std::unique_lock<std::mutex> g(sleep_mtx_);
  if (stop.load(std::memory_order_acquire)) return;
  if (npending_.load(std::memory_order_acquire) != 0) return;
  sleep_cv_.wait_for(g, std::chrono::microseconds(500));
}

// ------------------------------------------------------------------ Context
Context::Context(const Options& opt) : rank_(opt.rank), world_(opt.world) {
  int nw = opt.nworkers;
  if (nw < 0) nw = (int)param_int("sched_workers", -1);
  if (nw < 0) {
    nw = (int)std::thread::hardware_concurrency();
    // leave room for main + gpu manager + comm threads
    nw = nw > 3 ? nw - 3 : 1;
  }
  nworkers_ = nw;
  sched_.reset(new Scheduler(nworkers_));

  if (opt.gpu_device != -2) gpu_ = GpuEngine::create(this, opt.gpu_device);
  comm_ = CommEngine::create(this, opt.comm);
  // components publish their facts into the info registry (info.c analog)
  info_set("runtime.rank", std::to_string(rank_));
  info_set("runtime.world", std::to_string(world_));
  info_set("runtime.workers", std::to_string(nworkers_));
  info_set("sched.kind", param_str("sched", "ws"));
  if (comm_) info_set("comm.kind", comm_->kind());
  if (gpu_) gpu_->publish_info(this);

  pins_modules_install();
  {
    std::string lp = param_str("live_stats", "");
    if (!lp.empty()) {
      int iv = (int)param_int("live_stats_interval_ms", 500);
      live_thread_ = std::thread(&Context::live_stats_main, this,
                                 lp + "." + std::to_string(rank_), iv);
    }
  }
  roctx_init();
  debug_history_init();
  std::string prof = param_str("profile_filename", "");
  if (!prof.empty()) Profiler::inst().start(prof + "." + std::to_string(rank_));
  std::string dot = param_str("profile_dot", "");
  if (!dot.empty()) Profiler::inst().dot_open(dot + "." + std::to_string(rank_));
  for (int i = 0; i < nworkers_; i++)
    workers_.emplace_back([this, i] { worker_main(i); });
  PA_DEBUG(1, "context up: rank %d/%d, %d workers, gpu=%d", rank_, world_,
           nworkers_, has_gpu());
}

Context::~Context() {
  live_stop_.store(true, std::memory_order_release);
  if (live_thread_.joinable()) live_thread_.join();
  stop_.store(true, std::memory_order_release);
  sched_->wake_all();
  for (auto& w : workers_) w.join();
  pins_modules_finalize(rank_);
  if (param_int("stats", 0)) {
    // device-statistics table at fini (device.c:611-658 analog):
    // per-rank task counts, required-vs-transferred bytes, evictions,
    // renames, per-peer comm traffic.
    RuntimeCounters& c = counters();
    fprintf(stderr,
            "[parsec_amd stats] rank %d: cpu_tasks=%lu gpu_tasks=%lu "
            "scheduled=%lu steals=%lu comm_msgs=%lu comm_bytes=%lu "
            "renames=%lu\n",
            rank_, (unsigned long)c.tasks_executed_cpu.load(),
            (unsigned long)c.tasks_executed_gpu.load(),
            (unsigned long)c.tasks_scheduled.load(),
            (unsigned long)c.steals.load(),
            (unsigned long)c.comm_msgs.load(),
            (unsigned long)c.comm_bytes.load(),
            (unsigned long)c.renames.load());
    if (gpu_) {
      auto& g = gpu_->stats;
      fprintf(stderr,
              "[parsec_amd stats] rank %d gpu: tasks=%lu required=%lu "
              "h2d=%lu d2h=%lu evictions=%lu\n",
              rank_, (unsigned long)g.tasks.load(),
              (unsigned long)g.bytes_required.load(),
              (unsigned long)g.bytes_h2d.load(),
              (unsigned long)g.bytes_d2h.load(),
              (unsigned long)g.evictions.load());
    }
    if (comm_) {
      const auto& ps = comm_->peer_stats();
      for (size_t p = 0; p < ps.size(); p++)
        if (ps[p].sent_msgs.load() || ps[p].recv_msgs.load())
          fprintf(stderr,
                  "[parsec_amd stats] rank %d <-> peer %zu: sent=%lu msgs "
                  "%lu B, recv=%lu msgs %lu B\n",
                  rank_, p, (unsigned long)ps[p].sent_msgs.load(),
                  (unsigned long)ps[p].sent_bytes.load(),
                  (unsigned long)ps[p].recv_msgs.load(),
                  (unsigned long)ps[p].recv_bytes.load());
    }
  }
  comm_.reset();
  gpu_.reset();
  Profiler::inst().stop_and_dump();
  Profiler::inst().dot_close();
}

void Context::worker_main(int id) {
  tls_worker_id = id;
  // Optional core binding (bindthread.c/vpmap analog). Default off: with
  // one process per GPU the OS spreads ranks reasonably. sched_bind=1:
  // linear, core = (local_rank * (nworkers+2) + id) % ncores.
  // sched_bind=numa: bind this rank's workers round-robin over the CPUs
  // of the NUMA node its GPU hangs off (sysfs detection,
  // GpuEngine::detect_numa) — the binding that actually matters on a
  // dual-socket 8-GPU box; falls back to linear when detection fails.
  std::string bind = param_str("sched_bind", "0");
  if (bind != "0" && !bind.empty()) {
    const char* lr = getenv("LOCAL_RANK");
    int local = lr ? atoi(lr) : rank_;
    int core = -1;
    if (bind == "numa" && gpu_ && !gpu_->numa_cpus().empty()) {
      const auto& cpus = gpu_->numa_cpus();
      core = cpus[(size_t)(local * (nworkers_ + 2) + id) % cpus.size()];
    } else {
      long ncores = sysconf(_SC_NPROCESSORS_ONLN);
      if (ncores > 0) core = (local * (nworkers_ + 2) + id) % (int)ncores;
    }
    if (core >= 0) {
      cpu_set_t set;
      CPU_ZERO(&set);
      CPU_SET(core, &set);
      pthread_setaffinity_np(pthread_self(), sizeof(set), &set);
    }
  }
  while (!stop_.load(std::memory_order_acquire)) {
    Task* t = sched_->pop(id);
    if (t) {
      run_cpu_task(t);
    } else {
      sched_->park(id, stop_);
    }
  }
}

void Context::dispatch(Task* t, int worker_hint) {
  counters().tasks_scheduled.fetch_add(1, std::memory_order_relaxed);
  PA_PINS(PinsEv::SCHEDULE, t, Context::tls_worker_id);
  switch (t->tc->kind) {
    case TaskKind::GPU:
      if (gpu_ && t->tc->gpu_blocking) {
        // host-synchronous chore: run on a worker so the manager keeps
        // launching bulk work behind it
        sched_->push(t, worker_hint);
        return;
      }
      if (gpu_) { gpu_->enqueue(t); return; }
      [[fallthrough]];
    case TaskKind::CPU:
      sched_->push(t, worker_hint);
      return;
    case TaskKind::COMM_SEND:
    case TaskKind::COMM_RECV:
      comm_->enqueue(t);
      return;
  }
}

bool Context::progress_one() {
  Task* t = sched_->pop(-1);
  if (!t) return false;
  run_cpu_task(t);
  return true;
}

void Context::barrier() {
  if (comm_) comm_->barrier();
}

}  // namespace pa
