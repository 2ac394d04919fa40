#include "dtd.hpp"

#include "gpu_graph.hpp"

#include "kernels.hpp"
#include "profiling.hpp"

// Reference parity notes are in dtd.hpp's header (insert_function.c
// chaining, window throttling insert_function.c:75-76, data_flush).

namespace pa {

TaskClass COMM_SEND_CLASS = [] {
  TaskClass tc;
  tc.name = "comm_send";
  tc.kind = TaskKind::COMM_SEND;
  return tc;
}();

TaskClass COMM_RECV_CLASS = [] {
  TaskClass tc;
  tc.name = "comm_recv";
  tc.kind = TaskKind::COMM_RECV;
  return tc;
}();

constexpr int COMM_PRIORITY = 1 << 28;  // transfers go out as early as possible

// Internal reclaim task: depends on every live user of a renamed-out copy
// and deletes it once they drain (the datarepo retire protocol rebuilt on
// the dependency engine itself).
TaskClass RECLAIM_CLASS = [] {
  TaskClass tc;
  tc.name = "copy_reclaim";
  tc.kind = TaskKind::CPU;
  tc.cpu_hook = [](Task& t) { t.arg<Data*>()->drop_buffers(); };
  return tc;
}();

Dtd::Dtd(Context* ctx, std::string name) : Taskpool(ctx, std::move(name)) {
  me_ = ctx->rank();
  world_ = ctx->world();
  PA_CHECK(world_ <= 64, "sent_mask is a 64-bit rank bitmap");
  chan_seq_.assign((size_t)world_ * world_, 0);
  window_ = param_int("dtd_window_size", 16384);
  threshold_ = param_int("dtd_threshold_size", 8192);
  // One-to-many tile fan-out (collective propagation, remote_dep.c:322-437
  // chain/binomial trees): "binomial" re-sends through earlier recipients
  // (binary tree over discovery order, deterministic on every rank), so at
  // world 8 the owner's out-links stop being the fan-out bound. Default
  // unicast: single-node xGMI is per-link bound and direct sends already
  // use distinct links (profiles/RESULTS.md; re-decide with SCALE data).
  bcast_tree_ = param_str("bcast_tree", "unicast") == "binomial";
}

Dtd::~Dtd() = default;

void Dtd::set_local_writer(Data* d, Task* w) {
  if (d->last_local_writer) d->last_local_writer->release();
  for (Task* r : d->local_readers) r->release();
  d->local_readers.clear();
  d->last_local_writer = w;
  if (w) w->retain();
}

void Dtd::make_send(Data* d, int dst, uint64_t seq) {
  Task* st = task_new(this, &COMM_SEND_CLASS);
  st->peer = dst;
  st->comm_seq = seq;
  st->flows[0] = {d, ACCESS_IN};
  st->nflows = 1;
  st->priority = COMM_PRIORITY;
  if (d->last_local_writer) task_add_edge(d->last_local_writer, st);
  d->local_readers.push_back(st);
  st->retain();
  task_dec_deps(st);  // release insertion guard
}

// If `d`'s current buffer still has live local users, swap a fresh copy
// into the collection slot (the rename) and schedule the old one for
// reclamation behind its users. Returns the Data future work must use.
// A Data* held by the application is a persistent NAME for the tile: after
// a rename the collection slot holds a fresh copy, so insertion-side entry
// points canonicalize to the current copy first (stale handles from before
// a rename keep working, like repo keys in the reference).
static Data* current_copy(Data* d) {
  if (d->coll) return d->coll->current_by_key(d->key);
  if (d->icoll) return d->icoll->current_by_key(d->key);
  return d;
}

Data* Dtd::maybe_rename(Data* d) {
  auto live = [](Task* t) {
    if (!t) return false;
    t->lock.lock();
    bool l = !t->completed;
    t->lock.unlock();
    return l;
  };
  bool any_live = live(d->last_local_writer);
  for (Task* r : d->local_readers) {
    if (any_live) break;
    any_live = live(r);
  }
  if (!any_live) return d;  // nobody reads the old version: reuse in place
  if (!d->coll && !d->icoll)
    return d;  // standalone scratch datum (NEW tile): WAR-serialize instead
  // hipGraph record pass: replays need STABLE buffers, and the reclaim
  // task is a CPU task — WAR-serialize through the caller's edges instead
  // of renaming (gpu_graph.hpp).
  if (GpuGraphRecorder* rec = g_gpu_recorder.load(std::memory_order_acquire);
      rec && rec->tp == this)
    return d;
  Data* nd = d->coll ? d->coll->rename_tile(d) : d->icoll->rename(d);
  nd->version = d->version;
  nd->owner_rank = d->owner_rank;
  nd->sent_mask = d->sent_mask;
  nd->recip_order = d->recip_order;
  Task* rc = task_new(this, &RECLAIM_CLASS);
  rc->arg<Data*>() = d;  // shell owned by the collection
  if (d->last_local_writer) task_add_edge(d->last_local_writer, rc);
  for (Task* r : d->local_readers) task_add_edge(r, rc);
  counters().renames.fetch_add(1, std::memory_order_relaxed);
  task_dec_deps(rc);
  return nd;
}

Data* Dtd::make_recv(Data* d, int src, uint64_t seq) {
  d = maybe_rename(d);
  Task* rt = task_new(this, &COMM_RECV_CLASS);
  rt->peer = src;
  rt->comm_seq = seq;
  rt->flows[0] = {d, ACCESS_INOUT};
  rt->nflows = 1;
  rt->priority = COMM_PRIORITY;
  // Only reachable without rename when every old user already completed
  // (then these add no edges) — the recv never WAR-waits on live readers.
  if (d->last_local_writer) task_add_edge(d->last_local_writer, rt);
  for (Task* r : d->local_readers) task_add_edge(r, rt);
  set_local_writer(d, rt);
  d->local_present = true;
  d->local_present_version = d->version;
  task_dec_deps(rt);
  return d;
}

Data* Dtd::read_flow(Data* d, Task* t, int task_rank) {
  const int O = d->owner_rank, R = task_rank;
  if (O != R && !(d->sent_mask & (1ull << R))) {
    // Every rank advances the replicated channel counter; only the
    // endpoints create the actual transfer tasks. With the binomial tree,
    // the Nth recipient fetches from recipient (N-1)/2 (whose recv task is
    // already that rank's last_local_writer, so the forwarded send chains
    // behind it automatically).
    int src = O;
    if (bcast_tree_) {
      size_t idx = d->recip_order.size();
      if (idx > 0) src = d->recip_order[(idx - 1) / 2];
      d->recip_order.push_back(R);
    }
    uint64_t seq = chan_next(src, R);
    d->sent_mask |= 1ull << R;
    if (src == me_) make_send(d, R, seq);
    if (R == me_) d = make_recv(d, src, seq);
  }
  if (R == me_) {
    if (d->last_local_writer) task_add_edge(d->last_local_writer, t);
    d->local_readers.push_back(t);
    t->retain();
  }
  return d;
}

Data* Dtd::write_flow(Data* d, Task* t, int task_rank, bool output_only) {
  const int R = task_rank;
  if (R == me_) {
    if (output_only) {
      // A pure OUTPUT rewrite need not wait for readers of the previous
      // version: rename instead of WAR-serializing (datarepo copy-per-
      // version semantics).
      d = maybe_rename(d);
    }
    if (d->last_local_writer && d->last_local_writer != t)
      task_add_edge(d->last_local_writer, t);  // WAW / RAW on buffer
    for (Task* r : d->local_readers)
      if (r != t) task_add_edge(r, t);  // WAR
  }
  d->version++;
  d->owner_rank = R;
  d->sent_mask = 0;
  d->recip_order.clear();
  d->reshaped_.clear();  // promises were for the previous version
  if (R == me_) {
    set_local_writer(d, t);
    d->local_present = true;
    d->local_present_version = d->version;
  }
  // Remote writer: local copy (if any) becomes stale; local_present_version
  // keeps the old version so a later local reader triggers a fetch.
  return d;
}

bool Dtd::insert(const TaskClass* tc, const void* args, size_t args_bytes,
                 const FlowSpec* flows, int nflows, int priority, int rank) {
  Task* t = insert_begin(tc, args, args_bytes, flows, nflows, priority, rank);
  insert_commit(t);
  return t != nullptr;
}

Task* Dtd::insert_begin(const TaskClass* tc, const void* args,
                        size_t args_bytes, const FlowSpec* flows, int nflows,
                        int priority, int rank) {
  insert_mtx_.lock();  // released in insert_commit
  PA_CHECK(nflows <= MAX_FLOWS);
  PA_CHECK(args_bytes <= MAX_ARGS_BYTES);
  int task_rank = rank;
  if (task_rank < 0) {
    for (int i = 0; i < nflows && task_rank < 0; i++)
      if ((flows[i].mode & ACCESS_OUT) &&
          !(flows[i].mode & ACCESS_UNTRACKED))
        task_rank = flows[i].d->home_rank;
    if (task_rank < 0) task_rank = 0;
  }
  PA_CHECK(task_rank < world_);

  Task* t = nullptr;
  if (task_rank == me_) {
    t = task_new(this, tc);
    t->priority = priority;
    if (args_bytes) memcpy(t->args, args, args_bytes);
    t->nflows = nflows;
  }
  for (int i = 0; i < nflows; i++) {
    Data* d = flows[i].d ? current_copy(flows[i].d) : nullptr;
    if (d && flows[i].reshape != Reshape::NONE) {
      PA_CHECK(flows[i].mode == ACCESS_IN,
               "reshape: only READ flows consume converted copies");
      d = reshaped_promise(d, flows[i].reshape, task_rank);
    }
    if (d && !(flows[i].mode & ACCESS_UNTRACKED)) {
      // NULL flow (e.g. absent stencil halo) and UNTRACKED flows skip
      // the chaining protocol entirely (PARSEC_DONT_TRACK analog)
      if (flows[i].mode & ACCESS_IN) d = read_flow(d, t, task_rank);
      if (flows[i].mode & ACCESS_OUT)
        d = write_flow(d, t, task_rank,
                       /*output_only=*/!(flows[i].mode & ACCESS_IN));
    }
    // The task binds to the renamed copy, not the pointer the caller held.
    if (t) t->flows[i] = {d, flows[i].mode};
  }
  return t;
}

void Dtd::insert_commit(Task* t) {
  insert_mtx_.unlock();
  if (t) task_dec_deps(t);

  // Window throttling (insert_function.c:75-76): the inserter joins
  // progress when too far ahead of execution.
  if (pending() > window_) {
    while (pending() > threshold_) {
      if (!context()->progress_one())
        std::this_thread::sleep_for(std::chrono::microseconds(50));
    }
  }
}

// Lazily materialize (once per {version, kind, consumer rank}) the
// converted copy of `d` and return it; every consumer asking for the same
// conversion shares it — the datacopy-future/promise semantics of
// parsec_reshape.c, with an explicit HIP/CPU conversion task instead of
// MPI-datatype repacking.
Data* Dtd::reshaped_promise(Data* d, Reshape kind, int consumer_rank) {
  for (auto& [k2, r2, rd] : d->reshaped_)
    if (k2 == (uint8_t)kind && r2 == consumer_rank) return rd;
  PA_CHECK(d->coll, "reshape: flow must be backed by a tiled collection");
  TiledMatrix* A = d->coll;
  int tm = (int)(d->key / A->nt()), tn = (int)(d->key % A->nt());
  int rows = A->tile_rows(tm), cols = A->tile_cols(tn);
  auto holder = std::make_shared<Data>();
  holder->ctx_direct = context();
  holder->home_rank = consumer_rank;
  holder->owner_rank = consumer_rank;
  holder->bytes = reshape_bytes(d->bytes, kind, A->elem_size());
  Data* rd = holder.get();
  own(std::shared_ptr<void>(holder, rd));
  alignas(8) uint8_t args[MAX_ARGS_BYTES] = {};
  fill_reshape_args(args, kind, rows, cols, A->mb());
  FlowSpec f[2] = {{d, ACCESS_IN, Reshape::NONE},
                   {rd, ACCESS_OUT, Reshape::NONE}};
  insert(&tc_reshape(), args, reshape_args_bytes(), f, 2, 1 << 16,
         consumer_rank);
  // the nested insert may have renamed d (remote fetch): record the
  // promise on the CURRENT copy
  Data* cur = d->coll ? d->coll->current_by_key(d->key) : d;
  cur->reshaped_.push_back({(uint8_t)kind, consumer_rank, rd});
  return rd;
}

void Dtd::flush(Data* d) {
  std::lock_guard<std::recursive_mutex> g(insert_mtx_);
  d = current_copy(d);
  const int O = d->owner_rank, H = d->home_rank;
  if (O == H) return;
  uint64_t seq = chan_next(O, H);
  if (O == me_) make_send(d, H, seq);
  if (H == me_) d = make_recv(d, O, seq);
  d->owner_rank = H;
  d->sent_mask = 0;
  d->recip_order.clear();
}

void Dtd::flush_all(TiledMatrix& A) {
  for (int m = 0; m < A.mt(); m++)
    for (int n = 0; n < A.nt(); n++) flush(A.tile(m, n));
}

}  // namespace pa
