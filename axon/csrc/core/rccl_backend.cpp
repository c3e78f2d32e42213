// RCCL-over-xGMI GPU data plane (SURVEY.md §2.1 "NCCL ops -> RCCL ops",
// §2.4 N1-N6, §5.8): a dedicated non-blocking HIP comm stream per rank,
// lazily initialized RCCL communicator (unique id over the unix-socket
// control plane), fusion-buffer pack/unpack via the CDNA4 kernels in
// kernels/pack_unpack.hip, completion via a hipEvent poller thread, and a
// distributed VHDD Adasum path (kernels/adasum.hip).
//
// MI355X notes: xGMI is 7 point-to-point links per GPU — RCCL stripes one
// large buffer across links/channels, so the engine hands it FEW, LARGE
// collectives (fusion threshold default 128 MiB, controller.cpp). Average
// is folded into the collective itself (ncclAvg) instead of a separate
// scale pass (in-collective average).
#include <rccl/rccl.h>

#include <atomic>
#include <condition_variable>
#include <deque>
#include <limits>
#include <map>
#include <thread>
#include <unordered_map>
#include <unordered_set>

#include "../kernels/kernels.h"
#include "controller.h"
#include "gpu_util.h"
#include "message.h"
#include "operations.h"
#include "timeline.h"

namespace axon {

namespace {

void nccl_check(ncclResult_t r, const char* what) {
  if (r != ncclSuccess)
    throw std::runtime_error(std::string("axon RCCL error in ") + what + ": " +
                             ncclGetErrorString(r));
}

ncclDataType_t nccl_dtype(at::ScalarType t) {
  switch (t) {
    case at::ScalarType::Float: return ncclFloat32;
    case at::ScalarType::Double: return ncclFloat64;
    case at::ScalarType::Half: return ncclFloat16;
    case at::ScalarType::BFloat16: return ncclBfloat16;
    case at::ScalarType::Int: return ncclInt32;
    case at::ScalarType::Long: return ncclInt64;
    case at::ScalarType::Byte: return ncclUint8;
    case at::ScalarType::Char: return ncclInt8;
    case at::ScalarType::Bool: return ncclUint8;
    default:
      throw std::runtime_error("axon: unsupported GPU dtype " +
                               std::string(at::toString(t)));
  }
}

ncclRedOp_t nccl_red(ReduceOp r) {
  switch (r) {
    case ReduceOp::SUM: return ncclSum;
    case ReduceOp::AVERAGE: return ncclAvg;
    case ReduceOp::MIN: return ncclMin;
    case ReduceOp::MAX: return ncclMax;
    case ReduceOp::PRODUCT: return ncclProd;
    default: throw std::runtime_error("axon: bad reduce op for RCCL");
  }
}

constexpr size_t kAlign = 256;  // fusion-buffer sub-offsets: keep 16B vector
                                // paths and xGMI-friendly alignment

size_t align_up(size_t x) { return (x + kAlign - 1) & ~(kAlign - 1); }

class RcclBackend : public GpuBackend {
 public:
  RcclBackend(int rank, int size, ControlPlane* cp, Timeline* tl)
      : rank_(rank), size_(size), cp_(cp), tl_(tl) {}

  ~RcclBackend() override { finalize(); }

  void execute(const Response& r, std::vector<TensorEntry> entries) override {
    try {
      ProcessSetInfo psi = ps_get(r.process_set);
      if (!psi.member(rank_)) {
        // Rank-0 relay for a set we don't belong to: forward the set's
        // unique-id rendezvous on its first GPU op, nothing else (the data
        // plane is the set's own RCCL communicator). Deterministic: the
        // global response order is identical on every rank.
        relay_rendezvous(psi);
        return;
      }
      ensure_device(entries);
      SetComm& sc = ensure_comm(psi);
      issue(r, entries, sc);
    } catch (const std::exception& e) {
      AXON_LOG(LogLevel::ERROR, std::string("axon RCCL execute failed: ") + e.what());
      for (auto& en : entries)
        if (en.callback) en.callback(Status::Error(e.what()));
      return;
    }
    // Completion event + poller hand-off.
    hipEvent_t ev = EventPool::instance().get();
    hip_check(hipEventRecord(ev, stream_), "hipEventRecord(done)");
    {
      std::lock_guard<std::mutex> lk(mu_);
      inflight_.push_back(Inflight{ev, std::move(entries), std::move(used_slots_),
                                   std::move(keep_alive_)});
      used_slots_.clear();
      keep_alive_.clear();
    }
    cv_.notify_all();
  }

  void finalize() override {
    if (finalized_) return;
    finalized_ = true;
    {
      std::unique_lock<std::mutex> lk(mu_);
      cv_.wait(lk, [&] { return inflight_.empty(); });
      stop_ = true;
    }
    cv_.notify_all();
    if (poller_.joinable()) poller_.join();
    if (stream_) hipStreamSynchronize(stream_);
    for (auto& kv : comms_)
      if (kv.second.comm) ncclCommDestroy(kv.second.comm);
    comms_.clear();
  }

 private:
  struct SetComm {
    ncclComm_t comm = nullptr;
    int rank = 0;   // set-relative rank (== comm rank)
    int size = 1;   // number of set members
    std::vector<int> ranks;  // global ranks, set order (for root lookup)
  };
  struct Slot {
    ChunkDesc* host = nullptr;  // pinned
    ChunkDesc* dev = nullptr;
  };
  struct Inflight {
    hipEvent_t ev;
    std::vector<TensorEntry> entries;
    std::vector<int> slots;
    std::vector<at::Tensor> keep;
  };

  void ensure_device(const std::vector<TensorEntry>& entries) {
    if (device_init_) return;
    device_init_ = true;
    device_ = entries.at(0).device;
    hip_check(hipSetDevice(device_), "hipSetDevice");
    hip_check(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking),
              "hipStreamCreate(comm)");
    fuse_bytes_ = static_cast<size_t>(env_int("AXON_FUSION_THRESHOLD", 128 << 20));
    flat_allreduce_ = env_int("AXON_FLAT_ALLREDUCE", 0) != 0;
    flat_min_bytes_ = static_cast<size_t>(
        env_int("AXON_FLAT_MIN_BYTES", 4 << 20));
    hip_check(hipMalloc(&fusebuf_, fuse_bytes_ + kAlign), "hipMalloc(fusion buffer)");
    hip_check(hipMalloc(&dots_mine_, 3 * sizeof(double)), "hipMalloc(dots)");
    hip_check(hipMalloc(&dots_peer_, 3 * sizeof(double)), "hipMalloc(dots)");
    poller_ = std::thread([this] { poll_loop(); });
  }

  // Lazy per-process-set communicator. RCCL unique id rendezvous over the
  // control plane: the lowest set member generates the
  // id. All worker channels go through rank 0, so when rank 0 is not in
  // the set it relays the id (relay_rendezvous, called at the same point
  // in the global response order).
  SetComm& ensure_comm(const ProcessSetInfo& psi) {
    auto it = comms_.find(psi.id);
    if (it != comms_.end()) return it->second;
    SetComm sc;
    sc.rank = psi.my_index;
    sc.size = psi.size();
    sc.ranks = psi.ranks;
    ncclUniqueId id;
    const int root = psi.ranks.front();  // == 0 iff rank 0 is a member
    if (rank_ == root) {
      nccl_check(ncclGetUniqueId(&id), "ncclGetUniqueId");
      Writer w;
      w.raw(&id, sizeof(id));
      if (root == 0) {
        for (int p : psi.ranks)
          if (p != 0) cp_->send(p, MsgType::RENDEZVOUS, w.buf);
      } else {
        cp_->send(0, MsgType::RENDEZVOUS, w.buf);  // hub relays to members
      }
    } else {
      auto payload = cp_->recv_data(0);
      if (payload.size() != sizeof(id))
        throw std::runtime_error("axon: bad rccl unique id frame");
      std::memcpy(&id, payload.data(), sizeof(id));
    }
    nccl_check(ncclCommInitRank(&sc.comm, sc.size, id, sc.rank),
               "ncclCommInitRank");
    return comms_.emplace(psi.id, sc).first->second;
  }

  void relay_rendezvous(const ProcessSetInfo& psi) {
    if (relayed_sets_.count(psi.id)) return;
    relayed_sets_.insert(psi.id);
    auto payload = cp_->recv_data(psi.ranks.front());
    for (int p : psi.ranks)
      if (p != psi.ranks.front()) cp_->send(p, MsgType::RENDEZVOUS, payload);
  }

  int acquire_slot() {
    std::lock_guard<std::mutex> lk(mu_);  // free_slots_ shared with poller
    if (free_slots_.empty()) {
      Slot s;
      hip_check(hipHostMalloc(reinterpret_cast<void**>(&s.host),
                              kMaxChunks * sizeof(ChunkDesc)),
                "hipHostMalloc(descs)");
      hip_check(hipMalloc(reinterpret_cast<void**>(&s.dev),
                          kMaxChunks * sizeof(ChunkDesc)),
                "hipMalloc(descs)");
      slots_.push_back(s);
      free_slots_.push_back(static_cast<int>(slots_.size()) - 1);
    }
    int id = free_slots_.back();
    free_slots_.pop_back();
    used_slots_.push_back(id);
    return id;
  }

  void* ensure_scratch(size_t bytes) {
    if (scratch_bytes_ >= bytes) return scratch_;
    if (scratch_) retire_.push_back(scratch_);
    scratch_bytes_ = std::max<size_t>(bytes, 64 << 20);
    hip_check(hipMalloc(&scratch_, scratch_bytes_), "hipMalloc(scratch)");
    return scratch_;
  }

  // Stage `chunks` to the device (one or more desc slots) and launch the
  // given kernel over them.
  template <typename LaunchFn>
  void run_chunked(const std::vector<ChunkDesc>& chunks, LaunchFn&& fn) {
    size_t off = 0;
    while (off < chunks.size()) {
      int n = static_cast<int>(std::min<size_t>(chunks.size() - off, kMaxChunks));
      int sid = acquire_slot();
      std::memcpy(slots_[sid].host, chunks.data() + off, n * sizeof(ChunkDesc));
      hip_check(hipMemcpyAsync(slots_[sid].dev, slots_[sid].host,
                               n * sizeof(ChunkDesc), hipMemcpyHostToDevice,
                               stream_),
                "hipMemcpyAsync(descs)");
      fn(slots_[sid].dev, n);
      off += static_cast<size_t>(n);
    }
  }

  void wait_ready_events(std::vector<TensorEntry>& entries) {
    for (auto& e : entries) {
      if (e.ready_event) {
        hipEvent_t ev = reinterpret_cast<hipEvent_t>(e.ready_event);
        hip_check(hipStreamWaitEvent(stream_, ev, 0), "hipStreamWaitEvent(ready)");
        EventPool::instance().put(ev);
        e.ready_event = 0;
      }
    }
  }

  void issue(const Response& r, std::vector<TensorEntry>& entries, SetComm& sc) {
    hip_check(hipSetDevice(device_), "hipSetDevice");
    wait_ready_events(entries);
    for (auto& e : entries) tl_->begin(e.name, "RCCL");
    switch (r.op) {
      case OpType::ALLREDUCE:
        if (r.red == ReduceOp::ADASUM) adasum(entries, sc);
        else allreduce(r, entries, sc);
        break;
      case OpType::BROADCAST: broadcast(r, entries, sc); break;
      case OpType::ALLGATHER: allgather(r, entries, sc); break;
      case OpType::ALLTOALL: alltoall(r, entries, sc); break;
      case OpType::REDUCESCATTER: reducescatter(r, entries, sc); break;
      default:
        throw std::runtime_error("axon RCCL: unsupported op");
    }
  }

  // Split [ptr, ptr+elems) into chunk descs of <= chunk_elems elements.
  static void add_chunks(std::vector<ChunkDesc>& out, const void* src, void* dst,
                         int64_t elems, size_t esize, int64_t chunk_elems) {
    const char* s = static_cast<const char*>(src);
    char* d = static_cast<char*>(dst);
    for (int64_t off = 0; off < elems; off += chunk_elems) {
      int64_t n = std::min<int64_t>(chunk_elems, elems - off);
      out.push_back(ChunkDesc{s + off * esize, d + off * esize, n});
    }
  }

  // One all-reduced buffer. Default: a single ncclAllReduce (RCCL stripes
  // it across xGMI links/channels itself). AXON_FLAT_ALLREDUCE=1 switches
  // large buffers to the explicit flat reduce-scatter + all-gather
  // composition: on a fully-connected 8-GPU xGMI node
  // each phase moves S*(N-1)/N bytes as direct per-peer chunks over the 7
  // p2p links — the fallback if RCCL's channel striping under-saturates.
  void allreduce_buf(void* buf, int64_t elems, ncclDataType_t dt, size_t esize,
                     ncclRedOp_t op, SetComm& sc) {
    if (flat_allreduce_ && sc.size > 1 &&
        static_cast<size_t>(elems) * esize >= flat_min_bytes_) {
      char* b = static_cast<char*>(buf);
      int64_t chunk = elems / sc.size;
      if (chunk > 0) {
        // Documented NCCL in-place forms: RS recv = send + rank*chunk,
        // AG send = recv + rank*chunk.
        nccl_check(ncclReduceScatter(b, b + sc.rank * chunk * esize, chunk, dt,
                                     op, sc.comm, stream_),
                   "ncclReduceScatter(flat)");
        nccl_check(ncclAllGather(b + sc.rank * chunk * esize, b, chunk, dt,
                                 sc.comm, stream_),
                   "ncclAllGather(flat)");
      }
      int64_t tail_off = chunk * sc.size;
      if (tail_off < elems)
        nccl_check(ncclAllReduce(b + tail_off * esize, b + tail_off * esize,
                                 elems - tail_off, dt, op, sc.comm, stream_),
                   "ncclAllReduce(flat tail)");
      return;
    }
    nccl_check(ncclAllReduce(buf, buf, elems, dt, op, sc.comm, stream_),
               "ncclAllReduce");
  }

  void allreduce(const Response& r, std::vector<TensorEntry>& entries, SetComm& sc) {
    at::ScalarType dtype = id_dtype(r.dtype);
    at::ScalarType wire = r.wire_dtype ? id_dtype(r.wire_dtype) : dtype;
    bool any_scale = false;
    for (auto& e : entries)
      any_scale |= (e.prescale != 1.0 || e.postscale != 1.0);
    ncclRedOp_t op = nccl_red(r.red);

    if (entries.size() == 1 && wire == dtype && !any_scale) {
      // Zero-copy path: single tensor, no transform.
      auto& e = entries[0];
      const size_t ebytes =
          static_cast<size_t>(e.in.numel()) * e.in.element_size();
      if (flat_allreduce_ && sc.size > 1 && ebytes >= flat_min_bytes_) {
        if (e.out.data_ptr() != e.in.data_ptr())
          hip_check(hipMemcpyAsync(e.out.data_ptr(), e.in.data_ptr(), ebytes,
                                   hipMemcpyDeviceToDevice, stream_),
                    "hipMemcpyAsync(flat in->out)");
        allreduce_buf(e.out.data_ptr(), e.in.numel(), nccl_dtype(dtype),
                      e.in.element_size(), op, sc);
      } else {
        nccl_check(ncclAllReduce(e.in.data_ptr(), e.out.data_ptr(), e.in.numel(),
                                 nccl_dtype(dtype), op, sc.comm, stream_),
                   "ncclAllReduce");
      }
      return;
    }

    // Fused path: pack (cast+prescale fused) -> allreduce(wire) -> unpack
    // (cast back + postscale fused).
    size_t wsize = dtype_size(wire);
    size_t total = 0;
    std::vector<size_t> offs(entries.size());
    for (size_t i = 0; i < entries.size(); ++i) {
      offs[i] = total;
      total += align_up(static_cast<size_t>(entries[i].in.numel()) * wsize);
    }
    char* buf = static_cast<char*>(fusebuf_);
    if (total > fuse_bytes_) buf = static_cast<char*>(ensure_scratch(total));

    bool ident = wire == dtype;
    size_t esize = dtype_size(dtype);
    // Group pack chunks by prescale value (usually one group).
    std::map<double, std::vector<ChunkDesc>> packs;
    for (size_t i = 0; i < entries.size(); ++i) {
      auto& e = entries[i];
      if (ident && e.prescale == 1.0)
        add_chunks(packs[std::numeric_limits<double>::infinity()],
                   e.in.data_ptr(), buf + offs[i], e.in.numel() * esize, 1,
                   kChunkBytes);
      else
        add_chunks(packs[e.prescale], e.in.data_ptr(), buf + offs[i],
                   e.in.numel(), esize, kChunkElems);
    }
    for (auto& kv : packs) {
      if (kv.first == std::numeric_limits<double>::infinity())
        run_chunked(kv.second, [&](const ChunkDesc* d, int n) {
          launch_copy_bytes(d, n, stream_);
        });
      else {
        float sc = static_cast<float>(kv.first);
        run_chunked(kv.second, [&](const ChunkDesc* d, int n) {
          launch_xform(d, n, static_cast<int>(dtype), static_cast<int>(wire), sc,
                       stream_);
        });
      }
    }

    int64_t wire_elems = static_cast<int64_t>(total / wsize);
    allreduce_buf(buf, wire_elems, nccl_dtype(wire), wsize, op, sc);

    std::map<double, std::vector<ChunkDesc>> unpacks;
    for (size_t i = 0; i < entries.size(); ++i) {
      auto& e = entries[i];
      if (ident && e.postscale == 1.0)
        add_chunks(unpacks[std::numeric_limits<double>::infinity()],
                   buf + offs[i], e.out.data_ptr(), e.out.numel() * esize, 1,
                   kChunkBytes);
      else
        add_chunks(unpacks[e.postscale], buf + offs[i], e.out.data_ptr(),
                   e.out.numel(), wsize, kChunkElems);
    }
    for (auto& kv : unpacks) {
      if (kv.first == std::numeric_limits<double>::infinity())
        run_chunked(kv.second, [&](const ChunkDesc* d, int n) {
          launch_copy_bytes(d, n, stream_);
        });
      else {
        float sc = static_cast<float>(kv.first);
        run_chunked(kv.second, [&](const ChunkDesc* d, int n) {
          launch_xform(d, n, static_cast<int>(wire), static_cast<int>(dtype), sc,
                       stream_);
        });
      }
    }
  }

  void broadcast(const Response& r, std::vector<TensorEntry>& entries, SetComm& sc) {
    // r.root_rank is a GLOBAL rank; RCCL wants the comm-relative root.
    int root = 0;
    for (int i = 0; i < sc.size; ++i)
      if (sc.ranks[i] == r.root_rank) { root = i; break; }
    for (auto& e : entries) {
      nccl_check(ncclBroadcast(e.in.defined() ? e.in.data_ptr() : e.out.data_ptr(),
                               e.out.data_ptr(), e.out.numel(),
                               nccl_dtype(e.out.scalar_type()), root,
                               sc.comm, stream_),
                 "ncclBroadcast");
    }
  }

  void allgather(const Response& r, std::vector<TensorEntry>& entries, SetComm& sc) {
    for (size_t t = 0; t < entries.size(); ++t) {
      auto& e = entries[t];
      const auto& fd = r.first_dims.at(t);
      auto in_sizes = e.in.sizes().vec();
      int64_t row = 1;
      for (size_t d = 1; d < in_sizes.size(); ++d) row *= in_sizes[d];
      int64_t total_rows = 0;
      bool equal = true;
      for (auto x : fd) {
        total_rows += x;
        equal &= (x == fd[0]);
      }
      auto out_sizes = in_sizes;
      if (out_sizes.empty()) out_sizes.push_back(total_rows);
      else out_sizes[0] = total_rows;
      e.out.resize_(out_sizes);
      size_t esize = e.in.element_size();
      auto dt = nccl_dtype(e.in.scalar_type());
      char* ob = static_cast<char*>(e.out.data_ptr());
      if (equal && fd[0] > 0) {
        nccl_check(ncclAllGather(e.in.data_ptr(), ob, fd[0] * row, dt, sc.comm,
                                 stream_),
                   "ncclAllGather");
      } else {
        // Variable first dim: direct p2p exchange over the 7 xGMI links
        // (SURVEY.md §2.4 N3).
        int64_t my_numel = fd[sc.rank] * row;
        std::vector<int64_t> off(sc.size, 0);
        for (int p = 1; p < sc.size; ++p) off[p] = off[p - 1] + fd[p - 1] * row;
        nccl_check(ncclGroupStart(), "ncclGroupStart");
        for (int p = 0; p < sc.size; ++p) {
          if (p == sc.rank) continue;
          if (my_numel > 0)
            nccl_check(ncclSend(e.in.data_ptr(), my_numel, dt, p, sc.comm, stream_),
                       "ncclSend");
          int64_t pn = fd[p] * row;
          if (pn > 0)
            nccl_check(ncclRecv(ob + off[p] * esize, pn, dt, p, sc.comm, stream_),
                       "ncclRecv");
        }
        nccl_check(ncclGroupEnd(), "ncclGroupEnd");
        if (my_numel > 0)
          hip_check(hipMemcpyAsync(ob + off[sc.rank] * esize, e.in.data_ptr(),
                                   my_numel * esize, hipMemcpyDeviceToDevice,
                                   stream_),
                    "hipMemcpyAsync(self slab)");
      }
    }
  }

  void alltoall(const Response& r, std::vector<TensorEntry>& entries, SetComm& sc) {
    auto m = [&](int src, int dst) {
      return r.splits_matrix.at(static_cast<size_t>(src) * sc.size + dst);
    };
    for (auto& e : entries) {
      auto in_sizes = e.in.sizes().vec();
      int64_t row = 1;
      for (size_t d = 1; d < in_sizes.size(); ++d) row *= in_sizes[d];
      int64_t rows_me = 0;
      for (int src = 0; src < sc.size; ++src) rows_me += m(src, sc.rank);
      auto out_sizes = in_sizes;
      if (out_sizes.empty()) out_sizes.push_back(rows_me);
      else out_sizes[0] = rows_me;
      e.out.resize_(out_sizes);
      size_t esize = e.in.element_size();
      auto dt = nccl_dtype(e.in.scalar_type());
      const char* ib = static_cast<const char*>(e.in.data_ptr());
      char* ob = static_cast<char*>(e.out.data_ptr());
      std::vector<int64_t> soff(sc.size, 0), roff(sc.size, 0);
      for (int p = 1; p < sc.size; ++p) {
        soff[p] = soff[p - 1] + m(sc.rank, p - 1) * row;
        roff[p] = roff[p - 1] + m(p - 1, sc.rank) * row;
      }
      nccl_check(ncclGroupStart(), "ncclGroupStart");
      for (int p = 0; p < sc.size; ++p) {
        if (p == sc.rank) continue;
        int64_t sn = m(sc.rank, p) * row, rn = m(p, sc.rank) * row;
        if (sn > 0)
          nccl_check(ncclSend(ib + soff[p] * esize, sn, dt, p, sc.comm, stream_),
                     "ncclSend");
        if (rn > 0)
          nccl_check(ncclRecv(ob + roff[p] * esize, rn, dt, p, sc.comm, stream_),
                     "ncclRecv");
      }
      nccl_check(ncclGroupEnd(), "ncclGroupEnd");
      int64_t self_n = m(sc.rank, sc.rank) * row;
      if (self_n > 0)
        hip_check(hipMemcpyAsync(ob + roff[sc.rank] * esize, ib + soff[sc.rank] * esize,
                                 self_n * esize, hipMemcpyDeviceToDevice, stream_),
                  "hipMemcpyAsync(self)");
      if (e.received_splits.defined()) {
        auto rs = e.received_splits;
        rs.resize_({sc.size});
        for (int src = 0; src < sc.size; ++src) rs[src] = m(src, sc.rank);
      }
    }
  }

  void reducescatter(const Response& r, std::vector<TensorEntry>& entries, SetComm& sc) {
    for (auto& e : entries) {
      auto in_sizes = e.in.sizes().vec();
      int64_t rows = in_sizes.empty() ? 1 : in_sizes[0];
      int64_t row = 1;
      for (size_t d = 1; d < in_sizes.size(); ++d) row *= in_sizes[d];
      int64_t base = rows / sc.size, rem = rows % sc.size;
      int64_t my_rows = base + (sc.rank < rem ? 1 : 0);
      auto out_sizes = in_sizes;
      if (out_sizes.empty()) out_sizes.push_back(my_rows);
      else out_sizes[0] = my_rows;
      e.out.resize_(out_sizes);
      auto dt = nccl_dtype(e.in.scalar_type());
      ncclRedOp_t op = nccl_red(r.red);
      if (rem == 0) {
        nccl_check(ncclReduceScatter(e.in.data_ptr(), e.out.data_ptr(),
                                     base * row, dt, op, sc.comm, stream_),
                   "ncclReduceScatter");
      } else {
        // Uneven split: one rooted ncclReduce per member inside a single
        // group — every slice is reduced straight to its owner, so total
        // traffic matches a true reduce-scatter (the old allreduce+slice
        // fallback moved 2x).
        size_t esize = e.in.element_size();
        const char* ib = static_cast<const char*>(e.in.data_ptr());
        nccl_check(ncclGroupStart(), "ncclGroupStart");
        for (int p = 0; p < sc.size; ++p) {
          int64_t p_rows = base + (p < rem ? 1 : 0);
          if (p_rows == 0) continue;
          int64_t off = p * base + std::min<int64_t>(p, rem);
          nccl_check(ncclReduce(ib + off * row * esize,
                                p == sc.rank ? e.out.data_ptr() : nullptr,
                                p_rows * row, dt, op, p, sc.comm, stream_),
                     "ncclReduce(rs-uneven)");
        }
        nccl_check(ncclGroupEnd(), "ncclGroupEnd");
      }
    }
  }

  // Distributed Adasum: vector-halving distance-doubling with per-level
  // full-vector dot triples reduced over the 2^(k+1)-rank group by a
  // butterfly of 24-byte p2p exchanges (SURVEY.md §2.1 "Adasum ops"; math
  // parity with cpu_backend.cpp's fp64 oracle).
  void adasum(std::vector<TensorEntry>& entries, SetComm& sc) {
    // Non-power-of-two sets: extras (rank >= p2) fold their full vector
    // into partner rank - p2 (one pair combine), the p2-rank VHDD runs on
    // the low ranks, and the result fans back out. Association order
    // matches the CPU fp64 oracle (cpu_backend.cpp adasum).
    int p2 = 1;
    while (p2 * 2 <= sc.size) p2 *= 2;
    const int extras = sc.size - p2;
    for (auto& e : entries) {
      int dtype = static_cast<int>(e.in.scalar_type());
      size_t esize = e.in.element_size();
      int64_t n = e.in.numel();
      char* b = static_cast<char*>(e.out.data_ptr());
      if (e.out.data_ptr() != e.in.data_ptr())
        hip_check(hipMemcpyAsync(b, e.in.data_ptr(), n * esize,
                                 hipMemcpyDeviceToDevice, stream_),
                  "hipMemcpyAsync(adasum in->out)");
      if (e.prescale != 1.0) {
        std::vector<ChunkDesc> c;
        add_chunks(c, b, b, n, esize, kChunkElems);
        run_chunked(c, [&](const ChunkDesc* d, int nn) {
          launch_xform(d, nn, dtype, dtype, static_cast<float>(e.prescale), stream_);
        });
      }
      auto dt = nccl_dtype(e.in.scalar_type());
      if (sc.rank >= p2) {
        // Folded extra: hand the (prescaled) vector to the partner, get
        // the final combined result back after the VHDD finishes.
        const int partner = sc.rank - p2;
        if (n > 0) {
          nccl_check(ncclSend(b, n, dt, partner, sc.comm, stream_),
                     "ncclSend(adasum fold)");
          nccl_check(ncclRecv(b, n, dt, partner, sc.comm, stream_),
                     "ncclRecv(adasum fold result)");
        }
        if (e.postscale != 1.0) {
          std::vector<ChunkDesc> c;
          add_chunks(c, b, b, n, esize, kChunkElems);
          run_chunked(c, [&](const ChunkDesc* d, int nn) {
            launch_xform(d, nn, dtype, dtype, static_cast<float>(e.postscale),
                         stream_);
          });
        }
        continue;
      }
      const size_t half_scr = (static_cast<size_t>((n + 1) / 2) + kAlign) * esize;
      const size_t fold_scr =
          sc.rank < extras ? (static_cast<size_t>(n) + kAlign) * esize : 0;
      char* scratch =
          static_cast<char*>(ensure_scratch(std::max(half_scr, fold_scr)));
      if (sc.rank < extras && n > 0) {
        // Pair combine with my folded extra: full-vector dots, no
        // butterfly (only this rank keeps the result).
        nccl_check(ncclRecv(scratch, n, dt, p2 + sc.rank, sc.comm, stream_),
                   "ncclRecv(adasum fold)");
        hip_check(hipMemsetAsync(dots_mine_, 0, 3 * sizeof(double), stream_),
                  "hipMemset(dots)");
        launch_adasum_dots(b, scratch, n, dtype, dots_mine_, true, stream_);
        launch_adasum_apply(b, scratch, n, dtype, dots_mine_, true, stream_);
      }

      int64_t off = 0, len = n;
      for (int stride = 1; stride < p2; stride <<= 1) {
        int peer = sc.rank ^ stride;
        bool a_side = (sc.rank & stride) == 0;
        int64_t len1 = (len + 1) / 2, len2 = len - len1;
        int64_t keep_off = a_side ? off : off + len1;
        int64_t keep_len = a_side ? len1 : len2;
        int64_t send_off = a_side ? off + len1 : off;
        int64_t send_len = a_side ? len2 : len1;
        // 1) swap halves with the level peer.
        nccl_check(ncclGroupStart(), "ncclGroupStart");
        if (send_len > 0)
          nccl_check(ncclSend(b + send_off * esize, send_len, dt, peer, sc.comm,
                              stream_),
                     "ncclSend(adasum half)");
        if (keep_len > 0)
          nccl_check(ncclRecv(scratch, keep_len, dt, peer, sc.comm, stream_),
                     "ncclRecv(adasum half)");
        nccl_check(ncclGroupEnd(), "ncclGroupEnd");
        // 2) local partial dot triple (canonical a/b order).
        hip_check(hipMemsetAsync(dots_mine_, 0, 3 * sizeof(double), stream_),
                  "hipMemset(dots)");
        if (keep_len > 0)
          launch_adasum_dots(b + keep_off * esize, scratch, keep_len, dtype,
                             dots_mine_, a_side, stream_);
        // 3) butterfly-sum the triple across the 2*stride-rank group.
        for (int j = 1; j <= stride; j <<= 1) {
          int dpeer = sc.rank ^ j;
          nccl_check(ncclGroupStart(), "ncclGroupStart");
          nccl_check(ncclSend(dots_mine_, 3, ncclFloat64, dpeer, sc.comm, stream_),
                     "ncclSend(dots)");
          nccl_check(ncclRecv(dots_peer_, 3, ncclFloat64, dpeer, sc.comm, stream_),
                     "ncclRecv(dots)");
          nccl_check(ncclGroupEnd(), "ncclGroupEnd");
          launch_add3(dots_mine_, dots_peer_, stream_);
        }
        // 4) apply combine coefficients to my kept half.
        if (keep_len > 0)
          launch_adasum_apply(b + keep_off * esize, scratch, keep_len, dtype,
                              dots_mine_, a_side, stream_);
        off = keep_off;
        len = keep_len;
      }

      // 5) allgather the final segments (deterministic replay of every
      // rank's offsets) via direct p2p.
      auto segment_of = [&](int rk) {
        int64_t o = 0, l = n;
        for (int stride = 1; stride < p2; stride <<= 1) {
          int64_t l1 = (l + 1) / 2;
          if ((rk & stride) == 0) l = l1;
          else { o += l1; l -= l1; }
        }
        return std::make_pair(o, l);
      };
      nccl_check(ncclGroupStart(), "ncclGroupStart");
      for (int p = 0; p < p2; ++p) {
        if (p == sc.rank) continue;
        if (len > 0)
          nccl_check(ncclSend(b + off * esize, len, dt, p, sc.comm, stream_),
                     "ncclSend(adasum gather)");
        auto [po, pl] = segment_of(p);
        if (pl > 0)
          nccl_check(ncclRecv(b + po * esize, pl, dt, p, sc.comm, stream_),
                     "ncclRecv(adasum gather)");
      }
      nccl_check(ncclGroupEnd(), "ncclGroupEnd");
      // Fan the final vector back out to my folded extra, if any.
      if (sc.rank < extras && n > 0)
        nccl_check(ncclSend(b, n, dt, p2 + sc.rank, sc.comm, stream_),
                   "ncclSend(adasum fold result)");
      if (e.postscale != 1.0) {
        std::vector<ChunkDesc> c;
        add_chunks(c, b, b, n, esize, kChunkElems);
        run_chunked(c, [&](const ChunkDesc* d, int nn) {
          launch_xform(d, nn, dtype, dtype, static_cast<float>(e.postscale), stream_);
        });
      }
    }
  }

  void poll_loop() {
    hip_check(hipSetDevice(device_), "hipSetDevice(poller)");
    for (;;) {
      Inflight* front = nullptr;
      {
        std::unique_lock<std::mutex> lk(mu_);
        cv_.wait(lk, [&] { return stop_ || !inflight_.empty(); });
        if (stop_ && inflight_.empty()) return;
        front = &inflight_.front();
      }
      hipError_t q = hipEventQuery(front->ev);
      if (q == hipErrorNotReady) {
        std::this_thread::sleep_for(std::chrono::microseconds(100));
        continue;
      }
      Status st = q == hipSuccess
                      ? Status::OK()
                      : Status::Error(std::string("axon: comm stream error: ") +
                                      hipGetErrorString(q));
      for (auto& e : front->entries) {
        tl_->end(e.name);
        if (e.callback) e.callback(st);
      }
      EventPool::instance().put(front->ev);
      {
        std::lock_guard<std::mutex> lk(mu_);
        for (int sid : front->slots) free_slots_.push_back(sid);
        inflight_.pop_front();
        if (inflight_.empty()) {
          for (void* p : retire_) hipFree(p);
          retire_.clear();
        }
      }
      cv_.notify_all();
    }
  }

  int rank_, size_;
  ControlPlane* cp_;
  Timeline* tl_;
  int device_ = 0;
  bool device_init_ = false;
  std::unordered_map<int, SetComm> comms_;      // by process-set id
  std::unordered_set<int> relayed_sets_;        // rank-0 relay bookkeeping
  hipStream_t stream_ = nullptr;
  void* fusebuf_ = nullptr;
  size_t fuse_bytes_ = 0;
  bool flat_allreduce_ = false;
  size_t flat_min_bytes_ = 4 << 20;
  void* scratch_ = nullptr;
  size_t scratch_bytes_ = 0;
  std::vector<void*> retire_;
  double* dots_mine_ = nullptr;
  double* dots_peer_ = nullptr;

  std::vector<Slot> slots_;
  std::vector<int> free_slots_;
  std::vector<int> used_slots_;       // being assembled for current op
  std::vector<at::Tensor> keep_alive_;

  std::mutex mu_;
  std::condition_variable cv_;
  std::deque<Inflight> inflight_;
  std::thread poller_;
  std::atomic<bool> stop_{false};
  bool finalized_ = false;
};

}  // namespace

GpuBackend* create_rccl_backend(int rank, int size, ControlPlane* cp, Timeline* tl) {
  return new RcclBackend(rank, size, cp, tl);
}

}  // namespace axon
