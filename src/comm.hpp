// Communication engine: dataflow transfers between ranks (= GPUs).
//
// Reference parity: the comm-engine vtable
// (parsec_comm_engine.h:14-207) + funnelled comm thread
// (remote_dep_mpi.c:423-481, 1143-1271), re-designed for one MI355X node:
//  - Deterministic SPMD task insertion makes every transfer predictable on
//    both ends, so the eager-activate/GET handshake of remote_dep_mpi.c
//    collapses to matched send/recv pairs ordered per (src,dst) channel by a
//    sequence number every rank derives independently — no activation AMs,
//    no rank-bitmap reconstruction on the wire.
//  - Backends: "tcp" (host sockets; the CPU-testable engine, also the
//    control plane) and "rccl" (ncclSend/ncclRecv on xGMI, device-resident
//    payloads, one stream per peer to keep the 7 xGMI links independently
//    busy). Multi-node would layer MPI under the same interface.
#pragma once

#include <atomic>
#include <functional>
#include <memory>
#include <string>
#include <vector>

#include "common.hpp"
#include "runtime.hpp"

namespace pa {

class CommEngine {
 public:
  static std::unique_ptr<CommEngine> create(Context* ctx, const std::string& kind);
  // Always a TCP engine (control plane under the RCCL data engine).
  static std::unique_ptr<CommEngine> create_tcp(Context* ctx);
  virtual ~CommEngine() = default;

  // Send/recv task whose local dependencies are satisfied. The engine owns
  // completion (task_complete) once the wire transfer finishes.
  virtual void enqueue(Task* t) = 0;
  virtual void barrier() = 0;
  virtual const char* kind() const = 0;

  // ---- control messages (active-message seed, parsec_comm_engine.h
  // AM tags analog): small tagged payloads outside the deterministic
  // dataflow protocol — the transport for dynamic-DAG activation and
  // distributed termination detection. Handler runs on the comm thread.
  using CtlHandler =
      std::function<void(int src, uint32_t tag, const std::string&)>;
  // Tags with the high bit set are SYSTEM messages (termination detection);
  // they route to the sys handler and are excluded from traffic counts so
  // quiescence waves can stabilize.
  static constexpr uint32_t CTL_SYS_BIT = 0x80000000u;
  virtual void send_ctl(int dst, uint32_t tag, const void* p, size_t n) {
    (void)dst; (void)tag; (void)p; (void)n;
    fatal("comm engine '%s' has no control-message path", kind());
  }
  virtual void set_ctl_handler(CtlHandler h) { ctl_handler_ = std::move(h); }
  virtual void set_sys_handler(CtlHandler h) { sys_handler_ = std::move(h); }
  // user (non-system) control traffic counters, for quiescence accounting
  virtual uint64_t ctl_sent() const {
    return ctl_sent_.load(std::memory_order_relaxed);
  }
  virtual uint64_t ctl_recvd() const {
    return ctl_recvd_.load(std::memory_order_relaxed);
  }

  // Per-peer traffic accounting (device stats table analog,
  // device.c:611-658: the counters that explain a scaling curve).
  struct PeerStat {
    std::atomic<uint64_t> sent_msgs{0}, sent_bytes{0};
    std::atomic<uint64_t> recv_msgs{0}, recv_bytes{0};
  };
  const std::vector<PeerStat>& peer_stats() const { return peer_stats_; }

 protected:
  CtlHandler ctl_handler_;
  CtlHandler sys_handler_;
  std::atomic<uint64_t> ctl_sent_{0}, ctl_recvd_{0};
  void init_peer_stats(int world) {
    peer_stats_ = std::vector<PeerStat>(world);
  }
  void note_sent(int peer, uint64_t bytes) {
    if (peer >= 0 && peer < (int)peer_stats_.size()) {
      peer_stats_[peer].sent_msgs.fetch_add(1, std::memory_order_relaxed);
      peer_stats_[peer].sent_bytes.fetch_add(bytes, std::memory_order_relaxed);
    }
  }
  void note_recvd(int peer, uint64_t bytes) {
    if (peer >= 0 && peer < (int)peer_stats_.size()) {
      peer_stats_[peer].recv_msgs.fetch_add(1, std::memory_order_relaxed);
      peer_stats_[peer].recv_bytes.fetch_add(bytes, std::memory_order_relaxed);
    }
  }
  std::vector<PeerStat> peer_stats_;
};

// world==1: no-op engine.
class NullComm : public CommEngine {
 public:
  void enqueue(Task* t) override;
  void barrier() override {}
  const char* kind() const override { return "null"; }
};

}  // namespace pa
