#include "device_gpu.hpp"

#include <algorithm>

#include "data.hpp"
#include "gpu_graph.hpp"
#include "pins.hpp"
#include "profiling.hpp"

namespace pa {

std::unique_ptr<GpuEngine> GpuEngine::create(Context* ctx, int device) {
  int count = 0;
  hipError_t e = hipGetDeviceCount(&count);
  if (e != hipSuccess || count == 0) {
    PA_DEBUG(1, "no HIP device visible; CPU-only context");
    return nullptr;
  }
  if (device < 0) {
    // one process per GPU: local rank picks the device; respects
    // HIP_VISIBLE_DEVICES/torchrun LOCAL_RANK when present.
    const char* lr = getenv("LOCAL_RANK");
    device = lr ? atoi(lr) % count : ctx->rank() % count;
  }
  return std::unique_ptr<GpuEngine>(new GpuEngine(ctx, device));
}

GpuEngine::GpuEngine(Context* ctx, int device) : ctx_(ctx), device_(device) {
  PA_HIP_CHECK(hipSetDevice(device_));
  PA_HIP_CHECK(hipStreamCreateWithFlags(&h2d_stream_, hipStreamNonBlocking));
  PA_HIP_CHECK(hipStreamCreateWithFlags(&d2h_stream_, hipStreamNonBlocking));
  PA_HIP_CHECK(hipStreamCreateWithFlags(&comm_stream_, hipStreamNonBlocking));
  int nstreams = (int)param_int("gpu_exec_streams", 8);
  exec_streams_.resize(nstreams);
  inflight_.resize(nstreams);
  for (int i = 0; i < nstreams; i++)
    PA_HIP_CHECK(hipStreamCreateWithFlags(&exec_streams_[i], hipStreamNonBlocking));
  max_inflight_per_stream_ = (size_t)param_int("gpu_max_inflight", 64);
  next_stream_ = nstreams > 1 ? 1 : 0;

  // Reserve the HBM slab (parsec_device_memory_reserve analog,
  // device_gpu.c:867-992). Default 85% of free memory; kernels and
  // libraries get the rest.
  size_t free_b = 0, total_b = 0;
  PA_HIP_CHECK(hipMemGetInfo(&free_b, &total_b));
  int pct = (int)param_int("gpu_mem_percent", 85);
  slab_bytes_ = free_b / 100 * pct;
  int64_t cap_mb = param_int("gpu_mem_limit_mb", 0);
  if (cap_mb > 0) {
    slab_bytes_ = (size_t)cap_mb << 20;
    hard_cap_ = true;
  }
  if (slab_bytes_) {
    hipError_t e = hipMalloc(&slab_, slab_bytes_);
    if (e != hipSuccess) {
      slab_bytes_ /= 2;
      PA_HIP_CHECK(hipMalloc(&slab_, slab_bytes_));
    }
  }
  zone_.init(slab_bytes_);
  detect_numa();
  PA_DEBUG(1, "GPU %d engine: %d exec streams, slab %.1f GB, numa %d",
           device_, nstreams, slab_bytes_ / 1e9, numa_node_);
  manager_ = std::thread([this] { manager_main(); });
}

GpuEngine::~GpuEngine() {
  stop_.store(true, std::memory_order_release);
  q_cv_.notify_all();
  if (manager_.joinable()) manager_.join();
  PA_HIP_CHECK(hipSetDevice(device_));
  for (auto& dr : draining_) {
    hipEventSynchronize(dr.ev);  // writebacks must land before teardown
    hipEventDestroy(dr.ev);
  }
  for (auto& e : event_pool_) hipEventDestroy(e);
  for (auto& e : tev_pool_) hipEventDestroy(e);
  if (ref_ev_) hipEventDestroy(ref_ev_);
  for (auto s : exec_streams_) hipStreamDestroy(s);
  hipStreamDestroy(h2d_stream_);
  hipStreamDestroy(d2h_stream_);
  hipStreamDestroy(comm_stream_);
  if (slab_) hipFree(slab_);
}

// Resolve the GPU's NUMA node and its CPU list via sysfs
// (/sys/bus/pci/devices/<busid>/numa_node +
// /sys/devices/system/node/node<k>/cpulist). The vpmap/hwloc analog a
// one-node deployment actually needs: bind each rank's workers to the
// socket its GPU hangs off (PARSEC_MCA_sched_bind=numa).
void GpuEngine::detect_numa() {
  char busid[64] = {0};
  if (hipDeviceGetPCIBusId(busid, sizeof(busid), device_) != hipSuccess)
    return;
  for (char* p = busid; *p; p++) *p = (char)tolower(*p);
  char path[128];
  snprintf(path, sizeof(path), "/sys/bus/pci/devices/%s/numa_node", busid);
  FILE* f = fopen(path, "r");
  if (!f) return;
  if (fscanf(f, "%d", &numa_node_) != 1) numa_node_ = -1;
  fclose(f);
  if (numa_node_ < 0) return;
  snprintf(path, sizeof(path), "/sys/devices/system/node/node%d/cpulist",
           numa_node_);
  f = fopen(path, "r");
  if (!f) return;
  char list[512] = {0};
  if (fgets(list, sizeof(list), f)) {
    // "0-31,64-95" -> expanded cpu ids
    char* save = nullptr;
    for (char* tok = strtok_r(list, ",\n", &save); tok;
         tok = strtok_r(nullptr, ",\n", &save)) {
      int lo = 0, hi = 0;
      if (sscanf(tok, "%d-%d", &lo, &hi) == 2)
        for (int c = lo; c <= hi; c++) numa_cpus_.push_back(c);
      else if (sscanf(tok, "%d", &lo) == 1)
        numa_cpus_.push_back(lo);
    }
  }
  fclose(f);
}

void GpuEngine::publish_info(Context* ctx) {
  hipDeviceProp_t prop{};
  if (hipGetDeviceProperties(&prop, device_) == hipSuccess) {
    ctx->info_set("device.name", prop.gcnArchName);
    ctx->info_set("device.cus", std::to_string(prop.multiProcessorCount));
    ctx->info_set("device.hbm_gb",
                  std::to_string(prop.totalGlobalMem >> 30));
  }
  ctx->info_set("device.id", std::to_string(device_));
  ctx->info_set("device.slab_gb",
                std::to_string((double)slab_bytes_ / (1 << 30)));
  ctx->info_set("device.exec_streams",
                std::to_string(exec_streams_.size()));
  ctx->info_set("device.numa_node", std::to_string(numa_node_));
  ctx->info_set("device.numa_cpus", std::to_string(numa_cpus_.size()));
}

void GpuEngine::enqueue(Task* t) {
  {
    std::lock_guard<std::mutex> g(q_mtx_);
    queue_.push(PQEntry{t});
  }
  q_cv_.notify_one();
}

hipEvent_t GpuEngine::event_get() {
  {
    SpinGuard g(ev_lock_);
    if (!event_pool_.empty()) {
      hipEvent_t e = event_pool_.back();
      event_pool_.pop_back();
      return e;
    }
  }
  hipEvent_t e;
  PA_HIP_CHECK(hipEventCreateWithFlags(&e, hipEventDisableTiming));
  return e;
}

void GpuEngine::event_put(hipEvent_t e) {
  SpinGuard g(ev_lock_);
  event_pool_.push_back(e);
}

hipEvent_t GpuEngine::tev_get() {
  {
    SpinGuard g(ev_lock_);
    if (!tev_pool_.empty()) {
      hipEvent_t e = tev_pool_.back();
      tev_pool_.pop_back();
      return e;
    }
  }
  hipEvent_t e;
  PA_HIP_CHECK(hipEventCreate(&e));  // timing enabled
  return e;
}

void GpuEngine::gpu_span_calibrate() {
  if (ref_ev_) return;
  PA_HIP_CHECK(hipEventCreate(&ref_ev_));
  PA_HIP_CHECK(hipEventRecord(ref_ev_, exec_streams_[0]));
  PA_HIP_CHECK(hipEventSynchronize(ref_ev_));
  ref_ns_ = Profiler::now_ns();
}

bool GpuEngine::reap_draining_locked() {
  bool any = false;
  for (size_t i = 0; i < draining_.size();) {
    hipError_t e = hipEventQuery(draining_[i].ev);
    if (e == hipErrorNotReady) {
      i++;
      continue;
    }
    PA_HIP_CHECK(e);
    Draining dr = draining_[i];
    draining_.erase(draining_.begin() + i);
    event_put(dr.ev);
    if (dr.buf >= slab_ && dr.buf < (char*)slab_ + slab_bytes_)
      zone_.free((size_t)((char*)dr.buf - (char*)slab_), dr.bytes);
    else
      extern_lists_[dr.bytes].push_back(dr.buf);
    any = true;
  }
  return any;
}

void* GpuEngine::dev_alloc(size_t bytes) {
  bytes = (bytes + 255) & ~size_t(255);
  const double deadline =
      now_s() + (double)param_int("gpu_alloc_timeout_s", 120);
  while (now_s() < deadline) {
    {
      std::lock_guard<std::mutex> g(mem_mtx_);
      reap_draining_locked();
      size_t off = zone_.alloc(bytes);
      if (off != ZoneAlloc::NPOS) return (char*)slab_ + off;
      auto it = extern_lists_.find(bytes);
      if (it != extern_lists_.end() && !it->second.empty()) {
        void* p = it->second.back();
        it->second.pop_back();
        return p;
      }
    }
    // Capacity exhausted: evict the least-recently-used unpinned copy
    // (writeback if dirty), then retry.
    if (evict_one(bytes)) continue;
    if (!hard_cap_) {
      void* p = nullptr;
      if (hipMalloc(&p, bytes) == hipSuccess) return p;
    }
    // nothing evictable right now: in-flight tasks still hold pins.
    // The manager retires its own rings here (it is the only thread that
    // can); other threads wait for it.
    if (std::this_thread::get_id() == manager_tid_) {
      if (!retire_pass()) std::this_thread::yield();
    } else {
      std::this_thread::sleep_for(std::chrono::microseconds(200));
    }
  }
  fatal("GPU %d out of memory allocating %zu bytes (slab %zu/%zu in use, "
        "largest free %zu)",
        device_, bytes, zone_.in_use(), slab_bytes_, zone_.largest_free());
}

void GpuEngine::note_resident(Data* d) {
  std::lock_guard<std::mutex> g(mem_mtx_);
  for (Data* r : resident_)
    if (r == d) return;
  resident_.push_back(d);
}

void GpuEngine::forget(Data* d) {
  std::lock_guard<std::mutex> g(mem_mtx_);
  resident_.erase(std::remove(resident_.begin(), resident_.end(), d),
                  resident_.end());
}

void GpuEngine::pin(Data* d) {
  d->dev_refs++;
  d->dev_last_use = lru_clock_.fetch_add(1);
}

void GpuEngine::unpin(Data* d) {
  SpinGuard g(d->lock);
  d->dev_refs--;
  d->dev_last_use = lru_clock_.fetch_add(1);
}

bool GpuEngine::evict_one(size_t) {
  // The whole scan holds mem_mtx_: a Data being destroyed must first pass
  // forget() (which also takes mem_mtx_), so no candidate can dangle.
  // Tile locks are only try_lock'd (their holders may take mem_mtx_ via
  // dev_free, and try_lock never blocks, so no lock-order deadlock).
  std::lock_guard<std::mutex> g(mem_mtx_);
  PA_HIP_CHECK(hipSetDevice(device_));  // callers may be worker threads
  std::vector<Data*> cand = resident_;
  std::sort(cand.begin(), cand.end(), [](Data* a, Data* b) {
    return a->dev_last_use < b->dev_last_use;
  });
  for (Data* d : cand) {
    if (!d->lock.try_lock()) continue;
    if (!d->dev_ptr || d->dev_refs > 0) {
      d->lock.unlock();
      continue;
    }
    if (d->h2d_pending) {
      // The staging fence may long have completed: query instead of
      // skipping forever (a tile once staged from host would otherwise be
      // permanently unevictable and the evictable pool would only shrink).
      if (d->h2d_event &&
          hipEventQuery((hipEvent_t)d->h2d_event) == hipSuccess) {
        d->h2d_pending = false;
      } else {
        d->lock.unlock();
        continue;
      }
    }
    void* buf = d->dev_ptr;
    size_t rb = (d->bytes + 255) & ~size_t(255);
    bool drained_free = true;
    if (d->dev_valid && !d->host_valid) {
      // Dirty: asynchronous writeback (transfer_gpu.c:1-362 analog — the
      // reference makes eviction a D2H *task*; here the buffer parks on the
      // draining list until its event completes, and the manager never
      // synchronizes inside an allocation).
      if (!d->host_ptr) {
        if (posix_memalign(&d->host_ptr, 4096, d->bytes) != 0) {
          d->lock.unlock();
          continue;
        }
      }
      PA_HIP_CHECK(hipMemcpyAsync(d->host_ptr, buf, d->bytes,
                                  hipMemcpyDeviceToHost, d2h_stream_));
      stats.bytes_d2h += d->bytes;
      if (!d->d2h_event)
        PA_HIP_CHECK(hipEventCreateWithFlags((hipEvent_t*)&d->d2h_event,
                                             hipEventDisableTiming));
      PA_HIP_CHECK(hipEventRecord((hipEvent_t)d->d2h_event, d2h_stream_));
      d->d2h_pending = true;
      d->host_valid = true;  // valid in stream order behind d2h_event
      hipEvent_t drev = event_get();
      PA_HIP_CHECK(hipEventRecord(drev, d2h_stream_));
      draining_.push_back(Draining{buf, rb, drev});
      drained_free = false;
    }
    d->dev_ptr = nullptr;
    d->dev_valid = false;
    d->lock.unlock();
    if (drained_free) {
      if (buf >= slab_ && buf < (char*)slab_ + slab_bytes_)
        zone_.free((size_t)((char*)buf - (char*)slab_), rb);
      else
        extern_lists_[rb].push_back(buf);
    }
    resident_.erase(std::remove(resident_.begin(), resident_.end(), d),
                    resident_.end());
    PA_DEBUG(2, "evicted tile %lu (%zu bytes)", (unsigned long)d->key,
             d->bytes);
    stats.evictions++;
    return true;
  }
  return false;
}

void GpuEngine::dev_free(void* p, size_t bytes) {
  bytes = (bytes + 255) & ~size_t(255);
  std::lock_guard<std::mutex> g(mem_mtx_);
  if (p >= slab_ && p < (char*)slab_ + slab_bytes_)
    zone_.free((size_t)((char*)p - (char*)slab_), bytes);
  else
    extern_lists_[bytes].push_back(p);
}

void GpuEngine::copy_d2h(void* dst, const void* src, size_t bytes) {
  PA_HIP_CHECK(hipSetDevice(device_));
  PA_HIP_CHECK(hipMemcpyAsync(dst, src, bytes, hipMemcpyDeviceToHost, d2h_stream_));
  PA_HIP_CHECK(hipStreamSynchronize(d2h_stream_));
  stats.bytes_d2h += bytes;
}

void GpuEngine::copy_h2d(void* dst, const void* src, size_t bytes) {
  PA_HIP_CHECK(hipSetDevice(device_));
  PA_HIP_CHECK(hipMemcpyAsync(dst, src, bytes, hipMemcpyHostToDevice, h2d_stream_));
  PA_HIP_CHECK(hipStreamSynchronize(h2d_stream_));
  stats.bytes_h2d += bytes;
}

void GpuEngine::sync_all() {
  PA_HIP_CHECK(hipSetDevice(device_));
  PA_HIP_CHECK(hipDeviceSynchronize());
}

// Stage-in copies for every READ flow whose valid copy is on the host
// (parsec_device_data_stage_in, device_gpu.c:1800-2168, minus the peer-GPU
// branch: peers are other processes here, reached through the comm engine).
// Per-flow pin + device allocation + H2D staging + copy-fence wait; sets
// t->dev_ptr[]. Thread-safe: callable from the manager and from worker
// threads running blocking chores (per-Data locks serialize the
// decisions; HIP stream APIs are thread-safe).
void GpuEngine::stage_flows(Task* t, hipStream_t es) {
  for (int i = 0; i < t->nflows; i++) {
    Data* d = t->flows[i].data;
    if (!d) { t->dev_ptr[i] = nullptr; continue; }
    bool need_alloc;
    {
      SpinGuard g(d->lock);
      pin(d);  // before alloc: eviction skips pinned tiles
      need_alloc = !d->dev_ptr;
    }
    if (need_alloc) {
      // alloc outside the tile lock: dev_alloc may evict (taking other
      // tiles' locks) or wait on retirements
      void* p = dev_alloc(d->bytes);
      SpinGuard g(d->lock);
      if (!d->dev_ptr) {
        d->dev_ptr = p;
        note_resident(d);
      } else {
        dev_free(p, d->bytes);
      }
    }
    SpinGuard g(d->lock);
    if (t->flows[i].mode & ACCESS_IN)
      stats.bytes_required += d->bytes;
    if ((t->flows[i].mode & ACCESS_IN) && !d->dev_valid) {
      PA_CHECK(d->host_valid, "stage-in: no valid copy for tile");
      // A host copy produced by an async eviction writeback may still be in
      // flight on the d2h stream: order the re-stage H2D behind it.
      if (d->d2h_pending)
        PA_HIP_CHECK(
            hipStreamWaitEvent(h2d_stream_, (hipEvent_t)d->d2h_event, 0));
      PA_HIP_CHECK(hipMemcpyAsync(d->dev_ptr, d->host_ptr, d->bytes,
                                  hipMemcpyHostToDevice, h2d_stream_));
      stats.bytes_h2d += d->bytes;
      d->dev_valid = true;  // valid in stream order on h2d_stream_
      if (!d->h2d_event)
        PA_HIP_CHECK(hipEventCreateWithFlags((hipEvent_t*)&d->h2d_event,
                                             hipEventDisableTiming));
      PA_HIP_CHECK(hipEventRecord((hipEvent_t)d->h2d_event, h2d_stream_));
      d->h2d_pending = true;
    }
    if (!(t->flows[i].mode & ACCESS_IN) && !d->dev_valid) {
      d->dev_valid = true;  // OUTPUT-only: content produced by this task
    }
    // Every consumer of a staged tile waits on ITS copy fence — including
    // tasks that found the copy already issued by an earlier task.
    if (d->h2d_pending && (t->flows[i].mode & ACCESS_IN))
      PA_HIP_CHECK(hipStreamWaitEvent(es, (hipEvent_t)d->h2d_event, 0));
    t->dev_ptr[i] = d->dev_ptr;
  }
}

void GpuEngine::stage_in_and_launch(Task* t) {
  // Stream 0 is reserved for critical-path (panel) tasks so they never
  // queue behind bulk updates; others round-robin over the remaining
  // streams (the reference's exec_stream[2..n] round-robin,
  // device_gpu.c:3445-3535, with an express lane added).
  int si;
  if (t->priority >= (1 << 19) || (int)exec_streams_.size() == 1) {
    si = 0;
    if (!inflight_[0].empty()) {
      // Another express task is already in flight (concurrent panel DAGs,
      // e.g. tree QR): spread to the least-loaded stream instead of
      // serializing the whole panel tier behind stream 0.
      for (int k2 = 1; k2 < (int)exec_streams_.size(); k2++)
        if (inflight_[k2].size() < inflight_[si].size()) si = k2;
    }
  } else {
    si = next_stream_;
    next_stream_ = next_stream_ + 1;
    if (next_stream_ >= (int)exec_streams_.size()) next_stream_ = 1;
  }
  hipStream_t es = exec_streams_[si];
  stage_flows(t, es);
  PA_PINS(PinsEv::STAGE_IN, t, -1);
  std::vector<std::pair<void*, size_t>> deferred;
  GpuTaskCtx gctx{es, device_, this, &deferred};
  if (debug_history_on())
    debug_history_add("gpu_submit %s seq=%lu", t->tc->name.c_str(),
                      (unsigned long)t->seq);
  const bool spans = Profiler::inst().enabled();
  hipEvent_t sev = nullptr;
  if (spans) {
    gpu_span_calibrate();
    sev = tev_get();
    PA_HIP_CHECK(hipEventRecord(sev, es));
  }
  if (roctx_on()) roctx_push(t->tc->name.c_str());
  t->tc->gpu_hook(*t, gctx);
  if (roctx_on()) roctx_pop();
  PA_PINS(PinsEv::GPU_SUBMIT, t, -1);
  if (GpuGraphRecorder* rec = g_gpu_recorder.load(std::memory_order_acquire);
      rec && t->tp == rec->tp) {
    // hipGraph record pass (gpu_graph.hpp): retain the task, keep every
    // flow pinned on-device (stable pointers for replays), log the launch.
    t->retain();
    for (int i = 0; i < t->nflows; i++)
      if (Data* d = t->flows[i].data) { pin(d); rec->pinned.push_back(d); }
    SpinGuard rg(rec->lock);
    rec->idx_by_seq[t->seq] = (int)rec->log.size();
    rec->log.push_back(t);
    rec->stream_of.push_back(si);
  }
  hipEvent_t ev = spans ? tev_get() : event_get();
  PA_HIP_CHECK(hipEventRecord(ev, es));
  inflight_[si].push_back(
      InFlight{t, ev, si, Profiler::now_ns(), sev, std::move(deferred)});
  n_inflight_++;
}

bool GpuEngine::retire_pass() {
  bool progress = false;
  for (auto& ring : inflight_) {
    while (!ring.empty()) {
      InFlight& f = ring.front();
      hipError_t e = hipEventQuery(f.event);
      if (e == hipErrorNotReady) break;
      PA_HIP_CHECK(e);
      Task* t = f.task;
      for (int i = 0; i < t->nflows; i++) {
        Data* d = t->flows[i].data;
        if (!d) continue;
        if (t->flows[i].mode & ACCESS_OUT) d->written_on(true);
        unpin(d);
      }
      Profiler& pr = Profiler::inst();
      if (pr.enabled())
        pr.record(Ev::GPU_TASK, (uint16_t)t->tc->id, t->seq, f.t0_ns,
                  Profiler::now_ns());
      if (f.start_ev) {
        // device-side span on this exec stream: its own trace lane
        float ms0 = 0, ms1 = 0;
        hipEventElapsedTime(&ms0, ref_ev_, f.start_ev);
        hipEventElapsedTime(&ms1, ref_ev_, f.event);
        if (pr.enabled())
          pr.record_tid(Ev::GPU_SPAN, (uint16_t)t->tc->id, t->seq,
                        ref_ns_ + (uint64_t)((double)ms0 * 1e6),
                        ref_ns_ + (uint64_t)((double)ms1 * 1e6),
                        1000 + (uint32_t)f.stream_idx);
        SpinGuard g2(ev_lock_);
        tev_pool_.push_back(f.start_ev);
        tev_pool_.push_back(f.event);
        f.start_ev = nullptr;
        f.event = nullptr;
      }
      counters().tasks_executed_gpu.fetch_add(1, std::memory_order_relaxed);
      PA_PINS(PinsEv::GPU_RETIRE, t, -1);
      for (auto& [p2, b2] : f.deferred_frees) dev_free(p2, b2);
      if (f.event) event_put(f.event);
      ring.pop_front();
      n_inflight_--;
      stats.tasks++;
      task_complete(t);
      progress = true;
    }
  }
  return progress;
}

void GpuEngine::run_blocking(Task* t) {
  if (GpuGraphRecorder* rec = g_gpu_recorder.load(std::memory_order_acquire);
      rec && t->tp == rec->tp)
    rec->fail("host-blocking GPU chore '%s' is not graph-capturable "
              "(it synchronizes inside the hook)",
              t->tc->name.c_str());
  PA_HIP_CHECK(hipSetDevice(device_));
  static thread_local hipStream_t bs = nullptr;
  if (!bs) PA_HIP_CHECK(hipStreamCreateWithFlags(&bs, hipStreamNonBlocking));
  stage_flows(t, bs);
  std::vector<std::pair<void*, size_t>> deferred;
  GpuTaskCtx gctx{bs, device_, this, &deferred};
  if (debug_history_on())
    debug_history_add("gpu_blocking %s seq=%lu", t->tc->name.c_str(),
                      (unsigned long)t->seq);
  if (roctx_on()) roctx_push(t->tc->name.c_str());
  t->tc->gpu_hook(*t, gctx);
  if (roctx_on()) roctx_pop();
  PA_PINS(PinsEv::GPU_SUBMIT, t, -1);
  PA_HIP_CHECK(hipStreamSynchronize(bs));
  for (int i = 0; i < t->nflows; i++) {
    Data* d = t->flows[i].data;
    if (!d) continue;
    if (t->flows[i].mode & ACCESS_OUT) d->written_on(true);
    unpin(d);
  }
  for (auto& [p2, b2] : deferred) dev_free(p2, b2);
  counters().tasks_executed_gpu.fetch_add(1, std::memory_order_relaxed);
  PA_PINS(PinsEv::GPU_RETIRE, t, -1);
  stats.tasks++;
  task_complete(t);
}

void GpuEngine::manager_main() {
  PA_HIP_CHECK(hipSetDevice(device_));
  manager_tid_ = std::this_thread::get_id();
  while (true) {
    // 1) retire completed tasks (in-order per stream)
    bool progress = retire_pass();
    // 2) launch new work while there is room (global cap; the panel
    //    stream may exceed its share — its latency is the priority)
    while (true) {
      if (n_inflight_ >= max_inflight_per_stream_ * exec_streams_.size()) break;
      Task* t = nullptr;
      {
        std::lock_guard<std::mutex> g(q_mtx_);
        if (!queue_.empty()) { t = queue_.top().t; queue_.pop(); }
      }
      if (!t) break;
      stage_in_and_launch(t);
      progress = true;
    }
    if (stop_.load(std::memory_order_acquire) && n_inflight_ == 0) {
      std::lock_guard<std::mutex> g(q_mtx_);
      if (queue_.empty()) break;
    }
    if (!progress) {
      if (n_inflight_ > 0) {
        // work in flight: poll tightly but yield the core briefly
        std::this_thread::yield();
      } else {
        std::unique_lock<std::mutex> g(q_mtx_);
        if (queue_.empty() && !stop_.load(std::memory_order_acquire))
          q_cv_.wait_for(g, std::chrono::microseconds(100));
      }
    }
  }
}

// ---------------------------------------------------------------- Data
void Data::drop_buffers() {
  // Drop DTD chaining references to completed tasks.
  if (last_local_writer) {
    last_local_writer->release();
    last_local_writer = nullptr;
  }
  for (Task* r : local_readers) r->release();
  local_readers.clear();
  Context* c = coll ? coll->ctx() : ctx_direct;
  GpuEngine* eng = c ? c->gpu() : nullptr;
  if (eng) {
    eng->forget(this);  // never leave a dangling pointer in the LRU set
    if (dev_ptr) eng->dev_free(dev_ptr, bytes);
  }
  dev_ptr = nullptr;
  dev_valid = false;
  if (d2h_pending) {  // writeback still in flight targets host_ptr
    hipEventSynchronize((hipEvent_t)d2h_event);
    d2h_pending = false;
  }
  if (h2d_event) hipEventDestroy((hipEvent_t)h2d_event);
  if (d2h_event) hipEventDestroy((hipEvent_t)d2h_event);
  h2d_event = d2h_event = nullptr;
  h2d_pending = false;
  if (host_ptr) free(host_ptr);
  host_ptr = nullptr;
  host_valid = false;
}

Data::~Data() { drop_buffers(); }

void* Data::ensure_host() {
  if (!host_ptr) {
    if (posix_memalign(&host_ptr, 4096, bytes) != 0)
      fatal("host allocation of %zu bytes failed", bytes);
  }
  return host_ptr;
}

void* Data::pull_to_host() {
  Context* c = coll ? coll->ctx() : ctx_direct;
  GpuEngine* eng = c ? c->gpu() : nullptr;
  SpinGuard g(lock);
  ensure_host();
  if (!host_valid) {
    PA_CHECK(dev_valid && eng, "pull_to_host: no valid copy");
    eng->sync_all();  // quiesce producers before readback
    eng->copy_d2h(host_ptr, dev_ptr, bytes);
    host_valid = true;
  } else if (d2h_pending) {
    // Eviction writeback in flight: the host copy is valid only behind
    // its fence.
    PA_HIP_CHECK(hipEventSynchronize((hipEvent_t)d2h_event));
    d2h_pending = false;
  }
  return host_ptr;
}

void Data::begin_host_overwrite() {
  SpinGuard g(lock);
  dev_valid = false;
  if (d2h_pending) {
    // An in-flight writeback targets the buffer we are about to overwrite.
    PA_HIP_CHECK(hipEventSynchronize((hipEvent_t)d2h_event));
    d2h_pending = false;
  }
}

void Data::written_on(bool device) {
  SpinGuard g(lock);
  if (device) {
    dev_valid = true;
    host_valid = false;
  } else {
    host_valid = true;
    dev_valid = false;
  }
}

// ---------------------------------------------------------------- TiledMatrix
TiledMatrix::TiledMatrix(Context* ctx, int64_t m, int64_t n, int mb, int nb,
                         int p, int q, size_t elem_size, bool sym)
    : ctx_(ctx), ctx_rank_(ctx->rank()), m_(m), n_(n), mb_(mb), nb_(nb),
      p_(p), q_(q), elem_(elem_size), sym_(sym) {
  mt_ = (int)((m + mb - 1) / mb);
  nt_ = (int)((n + nb - 1) / nb);
  PA_CHECK(p_ * q_ == ctx->world(), "grid p*q must equal world size");
  tiles_.resize((size_t)mt_ * nt_);
}

TiledMatrix::~TiledMatrix() = default;  // ~Data returns device buffers

IrregularCollection::~IrregularCollection() = default;  // ~Data handles it

Data* IrregularCollection::add(uint64_t key, int rank, size_t bytes) {
  auto& slot = map_[key];
  if (!slot) {
    slot = std::make_unique<Data>();
    slot->key = key;
    slot->icoll = this;
    slot->ctx_direct = ctx_;
    slot->home_rank = rank;
    slot->owner_rank = rank;
    slot->bytes = bytes;
  }
  return slot.get();
}

Data* IrregularCollection::at(uint64_t key) {
  auto it = map_.find(key);
  PA_CHECK(it != map_.end(), "IrregularCollection: unknown key %llu",
           (unsigned long long)key);
  return it->second.get();
}

void TiledMatrix::set_rank_table(std::vector<int> table) {
  PA_CHECK((int64_t)table.size() == (int64_t)mt_ * nt_,
           "rank table must have mt*nt entries");
  for (int r : table)
    PA_CHECK(r >= 0 && r < ctx_->world(), "rank table entry out of range");
  PA_CHECK(!any_tiles(), "set_rank_table: before first tile access");
  ranks_ = std::move(table);
}

Data* TiledMatrix::rename_tile(Data* old) {
  size_t idx = (size_t)old->key;
  PA_CHECK(idx < tiles_.size() && tiles_[idx].get() == old,
           "rename_tile: not the current copy");
  auto d = std::make_unique<Data>();
  d->key = old->key;
  d->coll = this;
  d->home_rank = old->home_rank;
  d->bytes = old->bytes;
  retired_.push_back(std::move(tiles_[idx]));
  tiles_[idx] = std::move(d);
  return tiles_[idx].get();
}

Data* IrregularCollection::rename(Data* old) {
  auto it = map_.find(old->key);
  PA_CHECK(it != map_.end() && it->second.get() == old,
           "rename: not the current copy");
  auto d = std::make_unique<Data>();
  d->key = old->key;
  d->icoll = this;
  d->ctx_direct = ctx_;
  d->home_rank = old->home_rank;
  d->bytes = old->bytes;
  retired_.push_back(std::move(it->second));
  it->second = std::move(d);
  return it->second.get();
}

Data* TiledMatrix::tile(int tm, int tn) {
  if (sym_ && tn > tm) std::swap(tm, tn);
  PA_CHECK(tm >= 0 && tm < mt_ && tn >= 0 && tn < nt_);
  PA_CHECK(in_band(tm, tn), "tile (%d,%d) outside the stored band", tm, tn);
  size_t idx = (size_t)tm * nt_ + tn;
  if (!tiles_[idx]) {
    auto d = std::make_unique<Data>();
    d->key = idx;
    d->coll = this;
    d->home_rank = rank_of(tm, tn);
    d->owner_rank = d->home_rank;
    d->bytes = tile_bytes();
    tiles_[idx] = std::move(d);
  }
  return tiles_[idx].get();
}

}  // namespace pa
