// Controller: decides which tensors are ready on ALL ranks each cycle and
// broadcasts an ordered ResponseList so every rank issues collectives in an
// identical order (the Horovod coordination invariant — SURVEY.md §2.1
// "Controller / coordination", §3.2). Rank 0 is the coordinator; workers
// send Requests and receive ResponseLists over the unix-socket control
// plane (wire.h). Also hosts the stall inspector.
#pragma once

#include <chrono>
#include <deque>
#include <unordered_map>
#include <unordered_set>

#include "message.h"

namespace axon {

// Demuxing wrapper over the per-peer channels: control frames (REQUEST /
// RESPONSE_LIST / SHUTDOWN) and data frames (CPU_DATA / RENDEZVOUS / ACK)
// interleave on one socket; each side reads what it needs and stashes the
// rest. Single background thread per process => no locking.
class ControlPlane {
 public:
  ControlPlane() = default;
  void init_coordinator(const std::string& path, int world_size, double timeout);
  void init_worker(const std::string& path, int rank, double timeout);

  void send(int peer, MsgType t, const std::vector<uint8_t>& payload);
  // Non-blocking: pop next control frame from peer, false if none pending.
  bool try_recv_control(int peer, MsgType* t, std::vector<uint8_t>* payload);
  // Blocking: next data frame (CPU_DATA/RENDEZVOUS/ACK) from peer.
  std::vector<uint8_t> recv_data(int peer);
  int world_size() const { return static_cast<int>(chans_.size()); }
  void close_all();

 private:
  Channel& chan(int peer);
  // Reads frames from peer socket until predicate class available or
  // (non-blocking mode) socket drained.
  struct Inbox {
    std::deque<std::pair<MsgType, std::vector<uint8_t>>> control;
    std::deque<std::pair<MsgType, std::vector<uint8_t>>> data;
    bool eof = false;  // peer closed; queued frames remain consumable
  };
  void pump(int peer, bool blocking_for_data);
  std::vector<Channel> chans_;  // coordinator: by rank; worker: [0] only
  std::unordered_map<int, Inbox> inbox_;
  int my_rank_ = 0;
};

class Controller {
 public:
  Controller(int rank, int size, ControlPlane* cp);

  // One coordination step. `own` = requests newly generated on this rank
  // this cycle. Returns responses to execute, in the globally agreed order.
  // Sets *shutdown when the coordinator announced shutdown; sets
  // *new_cycle_ms > 0 when the autotuner shipped a cycle-time update.
  std::vector<Response> run_cycle(const std::vector<Request>& own, bool* shutdown,
                                  double* new_cycle_ms);

  // Autotuner hooks (rank 0): change the fusion threshold locally and
  // attach a cycle-time update to the next broadcast.
  void set_fusion_threshold(int64_t t) { fusion_threshold_ = t; }
  int64_t fusion_threshold() const { return fusion_threshold_; }
  void announce_cycle_time(double ms) { pending_cycle_ms_ = ms; }

  // Coordinator only: announce shutdown to all workers (drained on their
  // next cycle).
  void coordinator_shutdown();
  // Worker: notify coordinator we are shutting down? (not needed: launcher
  // owns process lifetime; coordinator-driven shutdown only.)

  int rank() const { return rank_; }
  int size() const { return size_; }

 private:
  struct PendingTensor {
    std::vector<Request> reqs;      // indexed by GLOBAL rank; empty name = missing
    uint32_t have = 0;              // bitmask of ranks that requested
    std::chrono::steady_clock::time_point first_seen;
    bool warned = false;
    bool queued = false;            // already pushed to ready_
    int order = 0;                  // arrival order of first request
    int pset = 0;                   // process-set id
  };

  void ingest(const Request& q);
  void maybe_ready(const std::string& name, PendingTensor& pt);
  void rescan_ready();  // after joins / late process-set registration
  void learn(const Response& r);  // adopt coordinator-assigned cache ids
  std::vector<Response> assemble();            // coordinator
  Response make_response(PendingTensor& pt);   // validate + build
  void fuse(std::vector<Response>& rs);
  void check_stalls();

  int rank_, size_;
  ControlPlane* cp_;
  // --- response cache (both roles) ---
  // Steady-state allreduces re-submit as 4-byte ids: the coordinator
  // assigns an id the first time a tensor's response is emitted; every
  // rank learns the id from the response; subsequent submissions with an
  // unchanged fingerprint go on the wire as the id alone.
  static std::string fingerprint(const Request& q);
  static bool cacheable(OpType op);
  std::unordered_map<std::string, uint32_t> cache_;       // name -> id
  std::unordered_map<std::string, std::string> cache_fp_; // name -> fp
  std::unordered_map<std::string, std::string> sent_fp_;  // awaiting id
  // Coordinator: per-rank request templates by cache id. Per-rank because
  // allgather first-dims and alltoall splits legally differ across ranks.
  std::unordered_map<uint32_t, std::unordered_map<int, Request>> templates_;
  uint32_t next_cache_id_ = 0;
  int64_t cache_capacity_;
  // --- coordinator state ---
  std::unordered_map<std::string, PendingTensor> table_;
  std::deque<std::string> ready_;   // names ready this cycle, arrival order
  uint32_t joined_mask_ = 0;
  int last_joined_ = -1;
  int order_counter_ = 0;
  int last_ps_version_ = 1;  // re-scan readiness on set register/remove
  int64_t fusion_threshold_;
  double stall_warn_sec_;
  bool pending_shutdown_ = false;
  double pending_cycle_ms_ = 0.0;
};

}  // namespace axon
