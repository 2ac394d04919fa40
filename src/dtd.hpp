// DTD — Dynamic Task Discovery interface.
//
// Reference parity (SURVEY.md §2.5 DTD, interfaces/dtd/insert_function.c):
// sequential-looking insert_task calls; per-tile last-writer/readers chaining
// builds RAW/WAR/WAW edges (insert_function.c:3027-3120); window throttling
// (insert_function.c:75-76); data_flush pushes final versions home
// (parsec_dtd_data_flush.c).
//
// Distributed model, MI355X-native: every rank executes the same insertion
// stream (SPMD, as the reference requires for distributed DTD) and derives
// every inter-rank transfer deterministically — tile {version, owner_rank,
// sent_mask} are replicated state machines, and each (src,dst) channel
// carries a sequence number all ranks compute identically. This replaces the
// reference's activation-message protocol (remote_dep_mpi.c:1984-2292): on a
// single node with RCCL p2p over xGMI, matched send/recv pairs in a
// deterministic per-channel order need no wire handshake at all.
#pragma once

#include <mutex>

#include "data.hpp"
#include "runtime.hpp"

namespace pa {

extern TaskClass COMM_SEND_CLASS;
extern TaskClass COMM_RECV_CLASS;

// Per-flow reshape kinds (reshape-promise engine, parsec_reshape.c:1-786
// analog — see src/kernels_reshape.cpp).
enum class Reshape : uint8_t {
  NONE = 0,
  TRANSPOSE,  // fp64 m x n -> n x m
  TO_BF16,    // fp64 -> bf16 (same shape)
  FROM_BF16,  // bf16 -> fp64
  TRIL,       // fp64, keep lower triangle, zero above
  TRIU,       // fp64, keep upper triangle, zero below
};

class Dtd : public Taskpool {
 public:
  Dtd(Context* ctx, std::string name = "dtd");
  ~Dtd();

  struct FlowSpec {
    Data* d;
    AccessMode mode;
    // READ flows only: consume a CONVERTED copy of the producer's version
    // (lazily materialized once per {version, kind, consumer rank} and
    // shared by all consumers — the reference's reshape "promises").
    Reshape reshape = Reshape::NONE;
  };

  // Insert one task. `rank` -1 selects the home rank of the first written
  // tile (AFFINITY default). Runs/creates the task only on its rank; all
  // ranks update the replicated tile state machines. Returns true when
  // the task is LOCAL (a Task object was created on this rank).
  bool insert(const TaskClass* tc, const void* args, size_t args_bytes,
              const FlowSpec* flows, int nflows, int priority = 0,
              int rank = -1);

  // Two-phase variant for front-ends that add explicit (e.g. CTL) edges:
  // begin() performs the dataflow chaining and returns the local Task with
  // its insertion guard still held (nullptr when the task is remote);
  // commit() releases the guard and applies window throttling.
  Task* insert_begin(const TaskClass* tc, const void* args, size_t args_bytes,
                     const FlowSpec* flows, int nflows, int priority,
                     int rank);
  void insert_commit(Task* t);

  // Push the current version of d back to its home rank (DTD data_flush).
  void flush(Data* d);
  void flush_all(TiledMatrix& A);

  // hipGraph capture/replay of this pool's GPU schedule (gpu_graph.hpp):
  // capture_begin() BEFORE the inserts, capture_end() instead of wait()
  // (it waits — the record pass — then instantiates the graph).
  void capture_begin();
  std::unique_ptr<class GpuGraph> capture_end();

 private:
  // read/write_flow and make_recv return the (possibly renamed) current
  // Data: when a recv or an OUTPUT-only rewrite targets a tile whose old
  // version still has live local readers, the collection slot gets a fresh
  // copy (datarepo/arena renaming, datarepo.h:25-92 /
  // remote_dep_mpi.c:572-615) and the old one is reclaimed by an internal
  // task once its readers drain — instead of WAR-serializing the transfer
  // behind every reader.
  Data* read_flow(Data* d, Task* t, int task_rank);
  Data* reshaped_promise(Data* d, Reshape kind, int consumer_rank);
  Data* write_flow(Data* d, Task* t, int task_rank, bool output_only);
  Data* maybe_rename(Data* d);
  void make_send(Data* d, int dst, uint64_t seq);
  Data* make_recv(Data* d, int src, uint64_t seq);
  static void set_local_writer(Data* d, Task* w);
  uint64_t chan_next(int src, int dst) {
    return chan_seq_[(size_t)src * world_ + dst]++;
  }

  int me_, world_;
  // Serializes inserters: the chaining state machines are sequential by
  // construction. A recursive mutex so a task body running inline during
  // window throttling may itself insert (untied tasks, dtd_test_untie
  // analog). Deterministic ORDER across ranks remains the app's contract
  // exactly as in the reference's distributed DTD.
  std::recursive_mutex insert_mtx_;
  std::vector<uint64_t> chan_seq_;
  int64_t window_;
  int64_t threshold_;
  bool bcast_tree_ = false;
};

}  // namespace pa
