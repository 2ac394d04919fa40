// Data subsystem: versioned multi-device data (tiles) + tiled collections.
//
// Reference parity (SURVEY.md §2.1 data subsystem, §2.6 collections):
//  - pa::Data = parsec_data_t + its parsec_data_copy_t set (data.c:153-260,
//    data_internal.h:30-85) collapsed to the two locations that exist on an
//    MI355X node process: host DRAM and this process's GPU HBM. Peer-GPU
//    copies live in peer *processes* and move over RCCL (comm.hpp).
//  - Version/ownership transfer (data.c:301-360) becomes {host_valid,
//    dev_valid} + writer-invalidates semantics under Data::lock.
//  - TiledMatrix = parsec_matrix_block_cyclic_t (two_dim_rectangle_cyclic.c)
//    with storage sized for 288 GB HBM3E: tiles live on the GPU by default
//    and host copies are created only on demand.
#pragma once

#include <cstdint>
#include <tuple>
#include <unordered_map>
#include <vector>

#include "common.hpp"
#include "runtime.hpp"

namespace pa {

class Context;

// Where the authoritative copy of a tile currently is.
struct Data {
  // identity
  uint64_t key = 0;
  class TiledMatrix* coll = nullptr;   // owning tiled collection (if any)
  class IrregularCollection* icoll = nullptr;  // or owning irregular coll
  class Context* ctx_direct = nullptr; // set instead for irregular data
  int home_rank = 0;
  size_t bytes = 0;

  SpinLock lock;  // guards the copy/validity state below
  void* host_ptr = nullptr;
  void* dev_ptr = nullptr;
  bool host_valid = false;
  bool dev_valid = false;
  // H2D staging fence: recorded on the copy stream when an async H2D is
  // issued; EVERY consumer stream must wait on it before reading dev_ptr
  // (a task that did not issue the copy still races it otherwise).
  void* h2d_event = nullptr;  // hipEvent_t, lazily created, owned here
  bool h2d_pending = false;
  // D2H writeback fence (async dirty eviction): recorded on the d2h stream;
  // host readers wait on it, and a re-stage H2D must order after it.
  void* d2h_event = nullptr;  // hipEvent_t, lazily created, owned here
  bool d2h_pending = false;
  // GPU residency management (LRU eviction): pinned while any in-flight
  // GPU task or comm transfer uses the device copy.
  int dev_refs = 0;          // guarded by lock
  uint64_t dev_last_use = 0; // engine-managed LRU stamp

  // ---- DTD chaining state (single inserter thread; no lock needed) ----
  uint32_t version = 0;        // logical version, bumped per writer insertion
  int owner_rank = 0;          // rank owning `version` (SPMD-tracked)
  Task* last_local_writer = nullptr;     // last local task writing the buffer
  std::vector<Task*> local_readers;      // local users since that writer
  uint64_t sent_mask = 0;                // ranks already sent current version
  std::vector<int> recip_order;          // recipients of current version, in
                                         // discovery order (broadcast tree)
  // reshape promises for the CURRENT version: {kind, consumer_rank, copy}
  std::vector<std::tuple<uint8_t, int, Data*>> reshaped_;
  uint32_t local_present_version = 0;    // version the local buffer will hold
  bool local_present = false;

  ~Data();

  // Free host+device buffers and drop chain references, keeping the shell
  // alive (renamed-out copies: the object must survive — application code
  // may hold stale handles that insertion canonicalizes through — but its
  // buffers are reclaimed as soon as the last reader drains).
  void drop_buffers();

  // Ensure a host buffer exists (allocates page-aligned memory).
  void* ensure_host();
  // Make the host copy valid (D2H if needed). Safe from any thread.
  void* pull_to_host();
  // Invalidate all copies except the one on `device` (true=GPU).
  void written_on(bool device);
  // Call BEFORE overwriting the host buffer outside the GPU engine (CPU
  // OUTPUT-only bodies, comm recv, external setters): drops the device
  // copy's validity so a concurrent dirty-eviction writeback cannot land
  // on top of the in-progress host write.
  void begin_host_overwrite();
};

// 2D block-cyclic tiled matrix of an elementary type (fp64 for the Cholesky
// headline; elem_size parametrizes dtype).
class TiledMatrix {
 public:
  // sym=true: symmetric storage — tile(m,n) with n>m aliases tile(n,m)
  // (sym_two_dim_rectangle_cyclic analog: only the lower triangle is
  // stored/distributed).
  TiledMatrix(Context* ctx, int64_t m, int64_t n, int mb, int nb,
              int p, int q, size_t elem_size = 8, bool sym = false);
  ~TiledMatrix();

  Context* ctx() const { return ctx_; }
  int64_t m() const { return m_; }
  int64_t n() const { return n_; }
  int mb() const { return mb_; }
  int nb() const { return nb_; }
  int mt() const { return mt_; }
  int nt() const { return nt_; }
  int grid_p() const { return p_; }
  int grid_q() const { return q_; }
  size_t elem_size() const { return elem_; }
  size_t tile_bytes() const { return (size_t)mb_ * nb_ * elem_; }
  bool sym() const { return sym_; }

  // ---- distribution variants (all SPMD: identical on every rank) ----
  // k-cyclic: kp consecutive tile-rows (kq tile-cols) share a grid row
  // (col) before cycling (two_dim_rectangle_cyclic k-cyclicity analog).
  void set_kcyclic(int kp, int kq) {
    PA_CHECK(kp >= 1 && kq >= 1 && !any_tiles(),
             "set_kcyclic: before first tile access");
    kp_ = kp; kq_ = kq;
  }
  // tabular: arbitrary per-tile rank table, size mt*nt, row-major
  // (two_dim_tabular analog). Overrides the cyclic mapping.
  void set_rank_table(std::vector<int> table);
  // band storage: only tiles with tm-kl <= tn <= tm+ku exist; accessing an
  // out-of-band tile is an error (band-cyclic distribution analog).
  void set_band(int kl, int ku) {
    PA_CHECK(kl >= 0 && ku >= 0 && !any_tiles(),
             "set_band: before first tile access");
    band_ = true; kl_ = kl; ku_ = ku;
  }
  bool in_band(int tm, int tn) const {
    if (sym_ && tn > tm) { int t = tm; tm = tn; tn = t; }
    return !band_ || (tn >= tm - kl_ && tn <= tm + ku_);
  }

  int rank_of(int tm, int tn) const {
    if (sym_ && tn > tm) { int t = tm; tm = tn; tn = t; }
    if (!ranks_.empty()) return ranks_[(size_t)tm * nt_ + tn];
    return ((tm / kp_) % p_) * q_ + ((tn / kq_) % q_);
  }
  bool is_local(int tm, int tn) const {
    return rank_of(tm, tn) == ctx_rank_;
  }
  Data* tile(int tm, int tn);

  // (datarepo/arena analog, datarepo.h:25-92: an incoming version never
  // overwrites a buffer still being read.)
  // Copy renaming: park `old` on the retired list (shell stays alive for
  // the collection's lifetime; its buffers are freed by the DTD reclaim
  // task) and install a fresh Data in the slot. Returns the fresh copy.
  Data* rename_tile(Data* old);
  Data* current_by_key(uint64_t key) { return tiles_[(size_t)key].get(); }

  // rows/cols of a (possibly partial) edge tile
  int tile_rows(int tm) const {
    int64_t r = m_ - (int64_t)tm * mb_;
    return r >= mb_ ? mb_ : (int)r;
  }
  int tile_cols(int tn) const {
    int64_t c = n_ - (int64_t)tn * nb_;
    return c >= nb_ ? nb_ : (int)c;
  }

 private:
  Context* ctx_;
  int ctx_rank_;
  int64_t m_, n_;
  int mb_, nb_, mt_, nt_, p_, q_;
  size_t elem_;
  bool sym_ = false;
  int kp_ = 1, kq_ = 1;
  bool band_ = false;
  int kl_ = 0, ku_ = 0;
  std::vector<int> ranks_;  // tabular override (empty = cyclic)
  std::vector<std::unique_ptr<Data>> tiles_;  // mt*nt, metadata lazy
  std::vector<std::unique_ptr<Data>> retired_;  // renamed-out copy shells

  bool any_tiles() const {
    for (auto& t : tiles_)
      if (t) return true;
    return false;
  }
};

// Irregular key->data collection (hash_datadist analog,
// data_dist/hash_datadist.c: arbitrary keys, explicit per-key rank —
// trees/graphs/ragged structures).
class IrregularCollection {
 public:
  explicit IrregularCollection(Context* ctx) : ctx_(ctx) {}
  ~IrregularCollection();

  // Register (or return) the datum for `key`; rank/bytes fixed on first
  // registration and must be identical on every rank (SPMD).
  Data* add(uint64_t key, int rank, size_t bytes);
  Data* at(uint64_t key);
  Data* rename(Data* old);  // see TiledMatrix::rename_tile
  Data* current_by_key(uint64_t key) {
    auto it = map_.find(key);
    return it == map_.end() ? nullptr : it->second.get();
  }
  Context* ctx() const { return ctx_; }
  size_t size() const { return map_.size(); }

 private:
  Context* ctx_;
  std::unordered_map<uint64_t, std::unique_ptr<Data>> map_;
  std::vector<std::unique_ptr<Data>> retired_;  // renamed-out copy shells
};

}  // namespace pa
