#include "dtd.hpp"

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

Dtd::Dtd(Context* ctx, std::string name) : Taskpool(ctx, std::move(name)) {
  me_ = ctx->rank();
  world_ = ctx->world();
  PA_CHECK(world_ <= 64, "sent_mask is a 64-bit rank bitmap");
  chan_seq_.assign((size_t)world_ * world_, 0);
  window_ = param_int("dtd_window_size", 16384);
  threshold_ = param_int("dtd_threshold_size", 8192);
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

void Dtd::make_recv(Data* d, int src, uint64_t seq) {
  Task* rt = task_new(this, &COMM_RECV_CLASS);
  rt->peer = src;
  rt->comm_seq = seq;
  rt->flows[0] = {d, ACCESS_INOUT};
  rt->nflows = 1;
  rt->priority = COMM_PRIORITY;
  // The recv overwrites the local buffer: WAR against every local user.
  if (d->last_local_writer) task_add_edge(d->last_local_writer, rt);
  for (Task* r : d->local_readers) task_add_edge(r, rt);
  set_local_writer(d, rt);
  d->local_present = true;
  d->local_present_version = d->version;
  task_dec_deps(rt);
}

void Dtd::read_flow(Data* d, Task* t, int task_rank) {
  const int O = d->owner_rank, R = task_rank;
  if (O != R && !(d->sent_mask & (1ull << R))) {
    // Every rank advances the replicated channel counter; only the
    // endpoints create the actual transfer tasks.
    uint64_t seq = chan_next(O, R);
    d->sent_mask |= 1ull << R;
    if (O == me_) make_send(d, R, seq);
    if (R == me_) make_recv(d, O, seq);
  }
  if (R == me_) {
    if (d->last_local_writer) task_add_edge(d->last_local_writer, t);
    d->local_readers.push_back(t);
    t->retain();
  }
}

void Dtd::write_flow(Data* d, Task* t, int task_rank) {
  const int R = task_rank;
  if (R == me_) {
    if (d->last_local_writer && d->last_local_writer != t)
      task_add_edge(d->last_local_writer, t);  // WAW / RAW on buffer
    for (Task* r : d->local_readers)
      if (r != t) task_add_edge(r, t);  // WAR
  }
  d->version++;
  d->owner_rank = R;
  d->sent_mask = 0;
  if (R == me_) {
    set_local_writer(d, t);
    d->local_present = true;
    d->local_present_version = d->version;
  }
  // Remote writer: local copy (if any) becomes stale; local_present_version
  // keeps the old version so a later local reader triggers a fetch.
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
      if (flows[i].mode & ACCESS_OUT) task_rank = flows[i].d->home_rank;
    if (task_rank < 0) task_rank = 0;
  }
  PA_CHECK(task_rank < world_);

  Task* t = nullptr;
  if (task_rank == me_) {
    t = task_new(this, tc);
    t->priority = priority;
    if (args_bytes) memcpy(t->args, args, args_bytes);
    for (int i = 0; i < nflows; i++)
      t->flows[i] = {flows[i].d, flows[i].mode};
    t->nflows = nflows;
  }
  for (int i = 0; i < nflows; i++) {
    if (!flows[i].d) continue;  // NULL flow (e.g. absent stencil halo)
    if (flows[i].mode & ACCESS_IN) read_flow(flows[i].d, t, task_rank);
    if (flows[i].mode & ACCESS_OUT) write_flow(flows[i].d, t, task_rank);
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

void Dtd::flush(Data* d) {
  std::lock_guard<std::recursive_mutex> g(insert_mtx_);
  const int O = d->owner_rank, H = d->home_rank;
  if (O == H) return;
  uint64_t seq = chan_next(O, H);
  if (O == me_) make_send(d, H, seq);
  if (H == me_) make_recv(d, O, seq);
  d->owner_rank = H;
  d->sent_mask = 0;
}

void Dtd::flush_all(TiledMatrix& A) {
  for (int m = 0; m < A.mt(); m++)
    for (int n = 0; n < A.nt(); n++) flush(A.tile(m, n));
}

}  // namespace pa
