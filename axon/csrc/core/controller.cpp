#include "controller.h"

#include <algorithm>
#include <sstream>

#include "operations.h"  // ps_get/ps_count (process-set registry)

namespace axon {

// ---------------------------------------------------------------- ControlPlane

void ControlPlane::init_coordinator(const std::string& path, int world_size,
                                    double timeout) {
  chans_ = server_accept(path, world_size, timeout);
  my_rank_ = 0;
}

void ControlPlane::init_worker(const std::string& path, int rank, double timeout) {
  chans_.clear();
  chans_.push_back(client_connect(path, rank, timeout));
  my_rank_ = rank;
}

Channel& ControlPlane::chan(int peer) {
  if (my_rank_ == 0) return chans_.at(peer);
  return chans_.at(0);  // workers only talk to rank 0
}

void ControlPlane::send(int peer, MsgType t, const std::vector<uint8_t>& payload) {
  chan(peer).send(t, payload);
}

void ControlPlane::pump(int peer, bool blocking_for_data) {
  Channel& c = chan(peer);
  Inbox& ib = inbox_[peer];
  for (;;) {
    if (ib.eof) {
      // Peer closed: queued frames (e.g. the final shutdown ResponseList)
      // stay consumable; only a consumer that needs MORE data errors out.
      if (blocking_for_data && ib.data.empty())
        throw std::runtime_error("axon: control channel closed by peer " +
                                 std::to_string(peer));
      return;
    }
    if (!blocking_for_data && !c.readable(0)) return;
    if (blocking_for_data && !ib.data.empty()) return;
    MsgType t;
    std::vector<uint8_t> payload;
    if (!c.recv(&t, &payload)) {
      ib.eof = true;
      continue;
    }
    if (t == MsgType::REQUEST || t == MsgType::RESPONSE_LIST || t == MsgType::SHUTDOWN)
      ib.control.emplace_back(t, std::move(payload));
    else
      ib.data.emplace_back(t, std::move(payload));
  }
}

bool ControlPlane::try_recv_control(int peer, MsgType* t,
                                    std::vector<uint8_t>* payload) {
  Inbox& ib = inbox_[peer];
  if (ib.control.empty()) pump(peer, /*blocking_for_data=*/false);
  if (ib.control.empty()) {
    if (ib.eof)  // fail-fast: peer died without a shutdown handshake
      throw std::runtime_error("axon: control channel closed by peer " +
                               std::to_string(peer));
    return false;
  }
  *t = ib.control.front().first;
  *payload = std::move(ib.control.front().second);
  ib.control.pop_front();
  return true;
}

std::vector<uint8_t> ControlPlane::recv_data(int peer) {
  Inbox& ib = inbox_[peer];
  while (ib.data.empty()) pump(peer, /*blocking_for_data=*/true);
  auto payload = std::move(ib.data.front().second);
  ib.data.pop_front();
  return payload;
}

void ControlPlane::close_all() {
  for (auto& c : chans_) c.close_fd();
}

// ------------------------------------------------------------------ Controller

Controller::Controller(int rank, int size, ControlPlane* cp)
    : rank_(rank), size_(size), cp_(cp) {
  if (size > 32)
    throw std::runtime_error("axon: world_size > 32 unsupported (single node)");
  // Default threshold re-tuned for xGMI: fewer, larger collectives
  //  — 128 MiB vs the NVLink-era 64 MiB default.
  fusion_threshold_ = env_int("AXON_FUSION_THRESHOLD", 128 << 20);
  stall_warn_sec_ = env_double("AXON_STALL_CHECK_TIME", 60.0);
  cache_capacity_ = env_int("AXON_CACHE_CAPACITY", 2048);
}

std::string Controller::fingerprint(const Request& q) {
  std::string fp;
  fp.reserve(32 + q.shape.size() * 8);
  fp += static_cast<char>(q.op);
  fp += static_cast<char>(q.red);
  fp += static_cast<char>(q.dtype);
  fp += static_cast<char>(q.wire_dtype);
  fp += q.device < 0 ? 'c' : 'g';
  fp += std::to_string(q.root_rank);
  fp += '/';
  fp += std::to_string(q.process_set);
  for (auto d : q.shape) {
    fp += ':';
    fp += std::to_string(d);
  }
  for (auto v : q.splits) {
    fp += ',';
    fp += std::to_string(v);
  }
  return fp;
}

bool Controller::cacheable(OpType op) {
  // Every data op negotiates through the cache (SURVEY.md §2.1 "Response
  // cache"); JOIN/BARRIER are control-shaped and stay full-frame.
  return op == OpType::ALLREDUCE || op == OpType::BROADCAST ||
         op == OpType::ALLGATHER || op == OpType::REDUCESCATTER ||
         op == OpType::ALLTOALL;
}

std::vector<Response> Controller::run_cycle(const std::vector<Request>& own,
                                            bool* shutdown,
                                            double* new_cycle_ms) {
  *shutdown = false;
  *new_cycle_ms = 0.0;
  if (rank_ != 0) {
    // Worker: ship requests (cached steady-state allreduces as bare ids),
    // collect any pending response lists.
    if (!own.empty()) {
      std::vector<const Request*> full;
      std::vector<uint32_t> ids;
      for (auto& q : own) {
        if (cacheable(q.op)) {
          auto it = cache_.find(q.name);
          if (it != cache_.end() && cache_fp_[q.name] == fingerprint(q)) {
            ids.push_back(it->second);
            continue;
          }
          sent_fp_[q.name] = fingerprint(q);
        }
        full.push_back(&q);
      }
      Writer w;
      w.u32(static_cast<uint32_t>(full.size()));
      for (auto* q : full) q->serialize(w);
      w.u32(static_cast<uint32_t>(ids.size()));
      for (uint32_t id : ids) w.u32(id);
      cp_->send(0, MsgType::REQUEST, w.buf);
    }
    std::vector<Response> out;
    MsgType t;
    std::vector<uint8_t> payload;
    // Stop polling once the shutdown flag is seen: the coordinator closes
    // its end right after the final broadcast, and polling past it would
    // misread the EOF as a peer failure.
    while (!*shutdown && cp_->try_recv_control(0, &t, &payload)) {
      if (t == MsgType::RESPONSE_LIST) {
        ResponseList rl = ResponseList::deserialize(payload);
        if (rl.shutdown) *shutdown = true;
        if (rl.new_cycle_ms > 0) *new_cycle_ms = rl.new_cycle_ms;
        for (auto& r : rl.responses) {
          learn(r);
          out.push_back(std::move(r));
        }
      }
    }
    return out;
  }

  // Coordinator: ingest own requests, then drain workers'.
  for (auto& q : own) ingest(q);
  for (int peer = 1; peer < size_; ++peer) {
    MsgType t;
    std::vector<uint8_t> payload;
    while (cp_->try_recv_control(peer, &t, &payload)) {
      if (t != MsgType::REQUEST) continue;
      Reader r(payload);
      uint32_t n = r.u32();
      for (uint32_t i = 0; i < n; ++i) ingest(Request::deserialize(r));
      if (!r.done()) {
        uint32_t nc = r.u32();
        for (uint32_t i = 0; i < nc; ++i) {
          uint32_t id = r.u32();
          auto it = templates_.find(id);
          if (it == templates_.end() || !it->second.count(peer)) {
            AXON_LOG(LogLevel::ERROR, "axon: unknown cache id " +
                                          std::to_string(id) + " from rank " +
                                          std::to_string(peer));
            continue;
          }
          Request q = it->second.at(peer);
          q.rank = peer;
          ingest(q);
        }
      }
    }
  }

  // A process set registered after its members' requests arrived: re-check
  // readiness of everything deferred on an unknown set.
  if (ps_version() != last_ps_version_) {
    last_ps_version_ = ps_version();
    rescan_ready();
  }
  std::vector<Response> rs = assemble();
  check_stalls();

  if (!rs.empty() || pending_shutdown_ || pending_cycle_ms_ > 0) {
    ResponseList rl;
    rl.responses = rs;
    rl.shutdown = pending_shutdown_;
    rl.new_cycle_ms = pending_cycle_ms_;
    auto buf = rl.serialize();
    for (int peer = 1; peer < size_; ++peer)
      cp_->send(peer, MsgType::RESPONSE_LIST, buf);
    if (pending_shutdown_) *shutdown = true;
    if (pending_cycle_ms_ > 0) *new_cycle_ms = pending_cycle_ms_;
    pending_cycle_ms_ = 0.0;
  }
  return rs;
}

void Controller::coordinator_shutdown() { pending_shutdown_ = true; }

void Controller::ingest(const Request& q) {
  if (q.op == OpType::JOIN) {
    joined_mask_ |= (1u << q.rank);
    last_joined_ = q.rank;
    // A new join may complete tensors that were waiting on this rank (in
    // any process set the rank belongs to), and may complete the join.
    rescan_ready();
    return;
  }
  auto& pt = table_[q.name];
  if (pt.reqs.empty()) {
    pt.reqs.resize(size_);
    pt.first_seen = std::chrono::steady_clock::now();
    pt.order = order_counter_++;
    pt.pset = q.process_set;
  }
  if (pt.have & (1u << q.rank)) {
    AXON_LOG(LogLevel::WARN, "axon: duplicate request for tensor '" + q.name +
                                 "' from rank " + std::to_string(q.rank));
    return;
  }
  pt.reqs[q.rank] = q;
  pt.have |= (1u << q.rank);
  // Metadata change on a cached tensor: refresh this rank's coordinator
  // template so later bare-id submissions reconstruct the new request.
  if (cacheable(q.op)) {
    auto it = cache_.find(q.name);
    if (it != cache_.end()) templates_[it->second][q.rank] = q;
  }
  maybe_ready(q.name, pt);
}

void Controller::maybe_ready(const std::string& name, PendingTensor& pt) {
  if (pt.queued) return;
  ProcessSetInfo psi = ps_get(pt.pset);
  if (!psi.valid) {
    // id beyond the registry: may be registered later this cycle — defer.
    if (pt.pset >= ps_count()) return;
    // Explicitly removed set: emit make_response's error now instead of
    // hanging until the stall warning.
    pt.queued = true;
    ready_.push_back(name);
    return;
  }
  uint32_t need = psi.mask;
  if ((pt.have | (joined_mask_ & need)) == need) {
    pt.queued = true;
    ready_.push_back(name);
  }
}

void Controller::rescan_ready() {
  std::vector<std::string> names;
  for (auto& kv : table_)
    if (!kv.second.queued) names.push_back(kv.first);
  std::sort(names.begin(), names.end(),
            [&](const std::string& a, const std::string& b) {
              return table_[a].order < table_[b].order;
            });
  for (auto& n : names) maybe_ready(n, table_[n]);
}

void Controller::learn(const Response& r) {
  if (r.cache_ids.size() != r.names.size()) return;
  for (size_t i = 0; i < r.names.size(); ++i) {
    if (r.cache_ids[i] < 0) continue;
    cache_[r.names[i]] = static_cast<uint32_t>(r.cache_ids[i]);
    auto it = sent_fp_.find(r.names[i]);
    if (it != sent_fp_.end()) {
      cache_fp_[r.names[i]] = std::move(it->second);
      sent_fp_.erase(it);
    }
  }
}

std::vector<Response> Controller::assemble() {
  std::vector<Response> rs;
  while (!ready_.empty()) {
    std::string name = std::move(ready_.front());
    ready_.pop_front();
    auto it = table_.find(name);
    if (it == table_.end()) continue;  // already emitted
    rs.push_back(make_response(it->second));
    table_.erase(it);
  }
  // All ranks joined => emit the JOIN response and reset.
  if (joined_mask_ == (1u << size_) - 1u) {
    Response r;
    r.op = OpType::JOIN;
    r.last_joined = last_joined_;
    rs.push_back(std::move(r));
    joined_mask_ = 0;
    last_joined_ = -1;
  }
  fuse(rs);
  return rs;
}

Response Controller::make_response(PendingTensor& pt) {
  // First present request is the template.
  const Request* t0 = nullptr;
  for (auto& q : pt.reqs)
    if (!q.name.empty()) { t0 = &q; break; }
  Response r;
  r.names.push_back(t0->name);
  r.op = t0->op;
  r.red = t0->red;
  r.dtype = t0->dtype;
  r.wire_dtype = t0->wire_dtype;
  r.device = t0->device;
  r.root_rank = t0->root_rank;
  r.process_set = t0->process_set;
  ProcessSetInfo psi = ps_get(t0->process_set);
  const int set_size = psi.size();

  std::ostringstream err;
  int64_t numel0 = 1;
  for (auto d : t0->shape) numel0 *= d;
  if (!psi.valid) err << "process set " << t0->process_set << " was removed";
  else if (t0->op == OpType::BROADCAST && !psi.member(t0->root_rank))
    err << "broadcast root " << t0->root_rank << " not in process set on '"
        << t0->name << "'";
  for (int rk : psi.ranks) {
    if (err.tellp() > 0) break;
    const Request& q = pt.reqs[rk];
    if (q.name.empty()) continue;  // joined rank
    if (q.op != t0->op)
      err << "op mismatch on '" << t0->name << "' (rank " << rk << ")";
    else if (q.dtype != t0->dtype)
      err << "dtype mismatch on '" << t0->name << "' (rank " << rk << ")";
    else if (q.red != t0->red)
      err << "reduce-op mismatch on '" << t0->name << "' (rank " << rk << ")";
    else if (q.op == OpType::ALLREDUCE || q.op == OpType::BROADCAST ||
             q.op == OpType::REDUCESCATTER) {
      if (q.shape != t0->shape)
        err << "shape mismatch on '" << t0->name << "' (rank " << rk << ")";
      if (q.op == OpType::BROADCAST && q.root_rank != t0->root_rank)
        err << "root_rank mismatch on '" << t0->name << "'";
    } else if (q.op == OpType::ALLGATHER) {
      if (q.shape.size() != t0->shape.size() ||
          !std::equal(q.shape.begin() + (q.shape.empty() ? 0 : 1), q.shape.end(),
                      t0->shape.begin() + (t0->shape.empty() ? 0 : 1)))
        err << "non-first-dim shape mismatch on allgather '" << t0->name << "'";
    } else if (q.op == OpType::ALLTOALL) {
      if (static_cast<int>(q.splits.size()) != set_size)
        err << "alltoall splits size != process-set size on '" << t0->name << "'";
    }
    if (q.process_set != t0->process_set)
      err << "process-set mismatch on '" << t0->name << "'";
  }
  if (err.tellp() > 0) {
    r.error = err.str();
    return r;
  }

  r.numels.push_back(numel0);
  // Assign (or reuse) a response-cache id for cacheable ops.
  if (cacheable(t0->op)) {
    int32_t id = -1;
    auto it = cache_.find(t0->name);
    if (it != cache_.end()) {
      id = static_cast<int32_t>(it->second);
    } else if (static_cast<int64_t>(cache_.size()) < cache_capacity_) {
      id = static_cast<int32_t>(next_cache_id_++);
      cache_[t0->name] = static_cast<uint32_t>(id);
      auto& per_rank = templates_[static_cast<uint32_t>(id)];
      for (auto& q : pt.reqs)
        if (!q.name.empty()) per_rank[q.rank] = q;
    }
    r.cache_ids.push_back(id);
  } else {
    r.cache_ids.push_back(-1);
  }
  if (t0->op == OpType::ALLGATHER) {
    // first_dims[i] = dim-0 extent from the i-th SET MEMBER (set order).
    std::vector<int64_t> fd(set_size, 0);
    for (int i = 0; i < set_size; ++i) {
      const Request& q = pt.reqs[psi.ranks[i]];
      if (!q.name.empty() && !q.shape.empty()) fd[i] = q.shape[0];
    }
    r.first_dims.push_back(std::move(fd));
  } else if (t0->op == OpType::ALLTOALL) {
    // [src_member][dst_member] row counts, set-relative indexing.
    r.splits_matrix.assign(static_cast<size_t>(set_size) * set_size, 0);
    for (int i = 0; i < set_size; ++i) {
      const Request& q = pt.reqs[psi.ranks[i]];
      if (!q.name.empty())
        for (int j = 0; j < set_size; ++j)
          r.splits_matrix[static_cast<size_t>(i) * set_size + j] = q.splits[j];
    }
  } else if (t0->op == OpType::REDUCESCATTER) {
    // Dim-0 extent (rows to split across members) — lets a non-member
    // rank-0 relay compute the per-member slices without the shape.
    r.first_dims.push_back({t0->shape.empty() ? 1 : t0->shape[0]});
  }
  return r;
}

void Controller::fuse(std::vector<Response>& rs) {
  // Merge consecutive ALLREDUCE responses with identical (dtype, wire dtype,
  // device, reduce-op) up to the fusion threshold. Adasum responses are kept
  // per-tensor so the projection combine stays layer-wise.
  std::vector<Response> out;
  for (auto& r : rs) {
    bool fusable = r.op == OpType::ALLREDUCE && r.error.empty() &&
                   r.red != ReduceOp::ADASUM;
    if (fusable && !out.empty()) {
      Response& p = out.back();
      if (p.op == OpType::ALLREDUCE && p.error.empty() &&
          p.red == r.red && p.dtype == r.dtype &&
          p.wire_dtype == r.wire_dtype && p.device == r.device &&
          p.process_set == r.process_set) {
        size_t esz = dtype_size(id_dtype(r.wire_dtype ? r.wire_dtype : r.dtype));
        int64_t pbytes = 0;
        for (auto n : p.numels) pbytes += n;
        pbytes = (pbytes + r.numels[0]) * static_cast<int64_t>(esz);
        if (pbytes <= fusion_threshold_) {
          p.names.push_back(std::move(r.names[0]));
          p.numels.push_back(r.numels[0]);
          p.cache_ids.push_back(r.cache_ids.empty() ? -1 : r.cache_ids[0]);
          continue;
        }
      }
    }
    out.push_back(std::move(r));
  }
  rs = std::move(out);
}

void Controller::check_stalls() {
  if (stall_warn_sec_ <= 0) return;
  auto now = std::chrono::steady_clock::now();
  for (auto& kv : table_) {
    auto& pt = kv.second;
    if (pt.warned) continue;
    double age = std::chrono::duration<double>(now - pt.first_seen).count();
    if (age > stall_warn_sec_) {
      uint32_t need = ps_get(pt.pset).mask;
      if (!need) need = (1u << size_) - 1;
      std::ostringstream os;
      os << "axon stall inspector: tensor '" << kv.first << "' submitted by ranks [";
      for (int rk = 0; rk < size_; ++rk)
        if (pt.have & (1u << rk)) os << rk << " ";
      os << "] but missing from ranks [";
      for (int rk = 0; rk < size_; ++rk)
        if ((need & (1u << rk)) && !(pt.have & (1u << rk)) &&
            !(joined_mask_ & (1u << rk)))
          os << rk << " ";
      os << "] for " << static_cast<int>(age) << "s — one or more ranks may be stuck";
      AXON_LOG(LogLevel::WARN, os.str());
      pt.warned = true;
    }
  }
}

}  // namespace axon
