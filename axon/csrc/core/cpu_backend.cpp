#include "cpu_backend.h"

#include <ATen/Functions.h>

#include "operations.h"  // ps_get

namespace axon {

namespace {
// Storage-flat alias: collectives are elementwise over dense storage, so a
// channels-last (or otherwise permuted-but-dense) tensor participates
// without copies — every rank has identical strides by construction.
at::Tensor flat(const at::Tensor& t) {
  if (t.is_contiguous()) return t.view({t.numel()});
  if (t.is_non_overlapping_and_dense()) {
    at::Tensor alias = at::empty({0}, t.options());
    alias.set_(t.storage(), t.storage_offset(), {t.numel()}, {1});
    return alias;
  }
  return t.contiguous().view({t.numel()});
}
}  // namespace

// All CPU collectives run hub-and-spoke over the unix-socket star: global
// rank 0 is ALWAYS the hub (it is the only rank every worker has a channel
// to). For a process set that does not contain rank 0, the hub acts as a
// pure relay: it is invoked with empty `entries` at the same globally
// agreed point in the response order as the members, receives the members'
// payloads, combines/routes them, and sends results back (SURVEY.md §2.1
// "Process sets"; the GPU data plane instead gets a per-set RCCL
// communicator and needs rank 0 only for the unique-id rendezvous).

void CpuBackend::send_tensor(int peer, const at::Tensor& t) {
  at::Tensor c = t.contiguous();
  Writer w;
  w.raw(c.data_ptr(), c.numel() * c.element_size());
  cp_->send(peer, MsgType::CPU_DATA, w.buf);
}

at::Tensor CpuBackend::recv_tensor(int peer, at::ScalarType dtype, int64_t numel) {
  auto payload = cp_->recv_data(peer);
  at::Tensor t = at::empty({numel}, at::TensorOptions().dtype(dtype));
  size_t want = static_cast<size_t>(numel) * t.element_size();
  if (payload.size() != want)
    throw std::runtime_error("axon cpu backend: size mismatch in recv (" +
                             std::to_string(payload.size()) + " vs " +
                             std::to_string(want) + ")");
  std::memcpy(t.data_ptr(), payload.data(), want);
  return t;
}

at::Tensor CpuBackend::recv_tensor_any(int peer, at::ScalarType dtype) {
  // Size inferred from the payload — used by the relay, which does not
  // always know the logical shape.
  auto payload = cp_->recv_data(peer);
  size_t esize = dtype_size(dtype);
  if (payload.size() % esize != 0)
    throw std::runtime_error("axon cpu backend: payload not a dtype multiple");
  int64_t numel = static_cast<int64_t>(payload.size() / esize);
  at::Tensor t = at::empty({numel}, at::TensorOptions().dtype(dtype));
  std::memcpy(t.data_ptr(), payload.data(), payload.size());
  return t;
}

Status CpuBackend::execute(const Response& r, std::vector<TensorEntry>& entries) {
  try {
    switch (r.op) {
      case OpType::ALLREDUCE:
        return r.red == ReduceOp::ADASUM ? adasum(r, entries) : allreduce(r, entries);
      case OpType::BROADCAST:
        return broadcast(r, entries);
      case OpType::ALLGATHER:
        return allgather(r, entries);
      case OpType::ALLTOALL:
        return alltoall(r, entries);
      case OpType::REDUCESCATTER:
        return reducescatter(r, entries);
      default:
        return Status::Error("axon cpu backend: unsupported op");
    }
  } catch (const std::exception& e) {
    return Status::Error(std::string("axon cpu backend: ") + e.what());
  }
}

Status CpuBackend::allreduce(const Response& r, std::vector<TensorEntry>& entries) {
  ProcessSetInfo psi = ps_get(r.process_set);
  const bool im_member = psi.member(rank_);
  at::ScalarType dtype = id_dtype(r.dtype);
  at::ScalarType wire = r.wire_dtype ? id_dtype(r.wire_dtype) : dtype;
  const size_t ntensors = im_member ? entries.size() : r.names.size();
  const double nmem = static_cast<double>(psi.size());

  for (size_t ti = 0; ti < ntensors; ++ti) {
    at::Tensor t;
    if (im_member) {
      auto& e = entries[ti];
      t = flat(e.in);
      if (e.prescale != 1.0) t = t.mul(e.prescale);
      if (wire != dtype) t = t.to(wire);
      t = t.contiguous();
    }

    at::Tensor result;
    if (rank_ == 0) {
      at::Tensor acc;
      bool first = true;
      if (im_member) { acc = t.clone(); first = false; }
      for (int peer : psi.ranks) {
        if (peer == 0) continue;
        at::Tensor s = first ? recv_tensor_any(peer, wire)
                             : recv_tensor(peer, wire, acc.numel());
        if (first) { acc = s; first = false; continue; }
        switch (r.red) {
          case ReduceOp::SUM:
          case ReduceOp::AVERAGE:
            acc.add_(s);
            break;
          case ReduceOp::MIN:
            acc = at::minimum(acc, s);
            break;
          case ReduceOp::MAX:
            acc = at::maximum(acc, s);
            break;
          case ReduceOp::PRODUCT:
            acc.mul_(s);
            break;
          default:
            return Status::Error("bad reduce op");
        }
      }
      if (r.red == ReduceOp::AVERAGE) acc.div_(nmem);
      for (int peer : psi.ranks)
        if (peer != 0) send_tensor(peer, acc);
      result = acc;
    } else {
      send_tensor(0, t);
      result = recv_tensor(0, wire, t.numel());
    }
    if (im_member) {
      auto& e = entries[ti];
      at::Tensor out = flat(e.out);
      out.copy_(result);  // casts wire -> dtype if compressed
      if (e.postscale != 1.0) out.mul_(e.postscale);
    }
  }
  return Status::OK();
}

at::Tensor adasum_combine_cpu(const at::Tensor& a, const at::Tensor& b) {
  at::Tensor af = a.to(at::kDouble), bf = b.to(at::kDouble);
  double dot = af.dot(bf).item<double>();
  double na = af.dot(af).item<double>();
  double nb = bf.dot(bf).item<double>();
  double ca = na > 0 ? 1.0 - dot / (2.0 * na) : 1.0;
  double cb = nb > 0 ? 1.0 - dot / (2.0 * nb) : 1.0;
  return af.mul(ca).add_(bf.mul(cb));
}

Status CpuBackend::adasum(const Response& r, std::vector<TensorEntry>& entries) {
  // Central simulation of the recursive-halving pairwise combine tree
  // (level k pairs member i with i^2^k): gather on the hub, fold the tree
  // in fp64, fan the result out. CPU path is a correctness oracle, not a
  // perf path — the GPU path (rccl_backend) does the real distributed
  // recursion.
  ProcessSetInfo psi = ps_get(r.process_set);
  const bool im_member = psi.member(rank_);
  const int nmem = psi.size();
  at::ScalarType dtype = id_dtype(r.dtype);
  const size_t ntensors = im_member ? entries.size() : r.names.size();
  for (size_t ti = 0; ti < ntensors; ++ti) {
    at::Tensor t;
    if (im_member) {
      auto& e = entries[ti];
      t = flat(e.in);
      if (e.prescale != 1.0) t = t.mul(e.prescale);
    }
    at::Tensor result;
    if (rank_ == 0) {
      std::vector<at::Tensor> v(nmem);
      for (int i = 0; i < nmem; ++i) {
        int peer = psi.ranks[i];
        v[i] = peer == 0 ? t.to(at::kDouble)
                         : recv_tensor_any(peer, dtype).to(at::kDouble);
      }
      // Non-power-of-two: fold extras (i >= p2) into partner i - p2 with a
      // full-vector pair combine FIRST, then run the power-of-two tree.
      // This is the semantic the distributed GPU VHDD implements, so the
      // oracle must use the same association order.
      int p2 = 1;
      while (p2 * 2 <= nmem) p2 *= 2;
      for (int i = 0; i + p2 < nmem; ++i)
        v[i] = adasum_combine_cpu(v[i], v[i + p2]);
      for (int stride = 1; stride < p2; stride *= 2)
        for (int i = 0; i + stride < p2; i += 2 * stride)
          v[i] = adasum_combine_cpu(v[i], v[i + stride]);
      result = v[0].to(dtype);
      for (int peer : psi.ranks)
        if (peer != 0) send_tensor(peer, result);
    } else {
      send_tensor(0, t.contiguous());
      result = recv_tensor(0, dtype, t.numel());
    }
    if (im_member) {
      auto& e = entries[ti];
      at::Tensor out = flat(e.out);
      out.copy_(result);
      if (e.postscale != 1.0) out.mul_(e.postscale);
    }
  }
  return Status::OK();
}

Status CpuBackend::broadcast(const Response& r, std::vector<TensorEntry>& entries) {
  ProcessSetInfo psi = ps_get(r.process_set);
  const bool im_member = psi.member(rank_);
  const size_t ntensors = im_member ? entries.size() : r.names.size();
  const int root = r.root_rank;
  for (size_t ti = 0; ti < ntensors; ++ti) {
    at::Tensor buf;
    if (im_member && rank_ == root) {
      buf = flat(entries[ti].in);
      if (root != 0) send_tensor(0, buf);
    }
    if (rank_ == 0) {
      if (root != 0) buf = recv_tensor_any(root, id_dtype(r.dtype));
      for (int peer : psi.ranks)
        if (peer != 0 && peer != root) send_tensor(peer, buf);
    } else if (rank_ != root) {
      buf = recv_tensor(0, id_dtype(r.dtype), entries[ti].out.numel());
    }
    if (im_member) {
      auto& e = entries[ti];
      if (rank_ != root || !e.out.is_same(e.in)) flat(e.out).copy_(buf);
    }
  }
  return Status::OK();
}

Status CpuBackend::allgather(const Response& r, std::vector<TensorEntry>& entries) {
  ProcessSetInfo psi = ps_get(r.process_set);
  const bool im_member = psi.member(rank_);
  const int nmem = psi.size();
  at::ScalarType dtype = id_dtype(r.dtype);
  const size_t ntensors = im_member ? entries.size() : r.names.size();
  for (size_t ti = 0; ti < ntensors; ++ti) {
    const auto& fd = r.first_dims.at(ti);  // indexed by set member

    at::Tensor full;
    if (rank_ == 0) {
      std::vector<at::Tensor> parts(nmem);
      int64_t total = 0;
      for (int i = 0; i < nmem; ++i) {
        int peer = psi.ranks[i];
        if (fd[i] == 0) { parts[i] = at::empty({0}, at::TensorOptions().dtype(dtype)); continue; }
        parts[i] = peer == 0 ? flat(entries[ti].in).contiguous()
                             : recv_tensor_any(peer, dtype);
        total += parts[i].numel();
      }
      full = at::empty({total}, at::TensorOptions().dtype(dtype));
      int64_t off = 0;
      for (int i = 0; i < nmem; ++i) {
        int64_t n = parts[i].numel();
        if (n) full.narrow(0, off, n).copy_(parts[i]);
        off += n;
      }
      for (int peer : psi.ranks)
        if (peer != 0) send_tensor(peer, full);
    } else {
      auto& e = entries[ti];
      // Symmetric with the hub's fd[i]==0 skip: a zero-row contribution
      // sends nothing (the hub never posts a recv for it).
      if (psi.my_index < 0 || fd.at(psi.my_index) != 0)
        send_tensor(0, flat(e.in));
      auto in_sizes = e.in.sizes().vec();
      int64_t row = 1;
      for (size_t d = 1; d < in_sizes.size(); ++d) row *= in_sizes[d];
      int64_t total_rows = 0;
      for (auto x : fd) total_rows += x;
      full = recv_tensor(0, dtype, total_rows * row);
    }
    if (im_member) {
      auto& e = entries[ti];
      auto in_sizes = e.in.sizes().vec();
      int64_t row = 1;
      for (size_t d = 1; d < in_sizes.size(); ++d) row *= in_sizes[d];
      int64_t total_rows = 0;
      for (auto x : fd) total_rows += x;
      auto out_sizes = in_sizes;
      if (out_sizes.empty()) out_sizes.push_back(total_rows);
      else out_sizes[0] = total_rows;
      e.out.resize_(out_sizes);
      flat(e.out).copy_(full);
    }
  }
  return Status::OK();
}

Status CpuBackend::alltoall(const Response& r, std::vector<TensorEntry>& entries) {
  ProcessSetInfo psi = ps_get(r.process_set);
  const bool im_member = psi.member(rank_);
  const int nmem = psi.size();
  const int my_idx = psi.my_index;
  at::ScalarType dtype = id_dtype(r.dtype);
  // splits_matrix is [src_member][dst_member] row counts.
  auto m = [&](int src, int dst) {
    return r.splits_matrix.at(static_cast<size_t>(src) * nmem + dst);
  };
  const size_t ntensors = im_member ? entries.size() : r.names.size();
  for (size_t ti = 0; ti < ntensors; ++ti) {
    int64_t row = 1;
    if (im_member) {
      auto in_sizes = entries[ti].in.sizes().vec();
      for (size_t d = 1; d < in_sizes.size(); ++d) row *= in_sizes[d];
    }

    at::Tensor mine;
    if (rank_ == 0) {
      std::vector<at::Tensor> inputs(nmem);
      for (int i = 0; i < nmem; ++i) {
        int peer = psi.ranks[i];
        inputs[i] = peer == 0 ? flat(entries[ti].in).contiguous()
                              : recv_tensor_any(peer, dtype);
        if (!im_member && row == 1) {
          // relay: derive the row element count from any member's payload
          int64_t rows_i = 0;
          for (int j = 0; j < nmem; ++j) rows_i += m(i, j);
          if (rows_i > 0) row = inputs[i].numel() / rows_i;
        }
      }
      if (row == 0) row = 1;
      for (int dst = 0; dst < nmem; ++dst) {
        int64_t rows_dst = 0;
        for (int src = 0; src < nmem; ++src) rows_dst += m(src, dst);
        at::Tensor outbuf =
            at::empty({rows_dst * row}, at::TensorOptions().dtype(dtype));
        int64_t off = 0;
        for (int src = 0; src < nmem; ++src) {
          int64_t pre = 0;  // rows src sends to dst' < dst
          for (int j = 0; j < dst; ++j) pre += m(src, j);
          int64_t n = m(src, dst) * row;
          if (n)
            outbuf.narrow(0, off, n).copy_(inputs[src].narrow(0, pre * row, n));
          off += n;
        }
        int peer = psi.ranks[dst];
        if (peer == 0) mine = outbuf;
        else send_tensor(peer, outbuf);
      }
    } else {
      send_tensor(0, flat(entries[ti].in));
      int64_t rows_me = 0;
      for (int src = 0; src < nmem; ++src) rows_me += m(src, my_idx);
      mine = recv_tensor(0, dtype, rows_me * row);
    }
    if (im_member) {
      auto& e = entries[ti];
      auto out_sizes = e.in.sizes().vec();
      int64_t rows_me = mine.numel() / std::max<int64_t>(row, 1);
      if (out_sizes.empty()) out_sizes.push_back(rows_me);
      else out_sizes[0] = rows_me;
      e.out.resize_(out_sizes);
      flat(e.out).copy_(mine);
      if (e.received_splits.defined()) {
        auto rs = e.received_splits;
        rs.resize_({nmem});
        for (int src = 0; src < nmem; ++src) rs[src] = m(src, my_idx);
      }
    }
  }
  return Status::OK();
}

Status CpuBackend::reducescatter(const Response& r, std::vector<TensorEntry>& entries) {
  ProcessSetInfo psi = ps_get(r.process_set);
  const bool im_member = psi.member(rank_);
  const int nmem = psi.size();
  const int my_idx = psi.my_index;
  at::ScalarType dtype = id_dtype(r.dtype);
  const size_t ntensors = im_member ? entries.size() : r.names.size();
  for (size_t ti = 0; ti < ntensors; ++ti) {
    // Dim-0 rows from the response (also works for the non-member relay).
    int64_t rows = r.first_dims.at(ti).at(0);
    int64_t row = r.numels.at(ti) / std::max<int64_t>(rows, 1);
    int64_t base = rows / nmem, rem = rows % nmem;
    auto member_rows = [&](int i) { return base + (i < rem ? 1 : 0); };
    auto member_off = [&](int i) {
      return static_cast<int64_t>(i) * base + std::min<int64_t>(i, rem);
    };

    at::Tensor t;
    if (im_member) {
      auto& e = entries[ti];
      t = flat(e.in);
      if (e.prescale != 1.0) t = t.mul(e.prescale);
    }
    at::Tensor mine;
    if (rank_ == 0) {
      at::Tensor acc;
      bool first = true;
      if (im_member) { acc = t.clone(); first = false; }
      for (int peer : psi.ranks) {
        if (peer == 0) continue;
        at::Tensor s = first ? recv_tensor_any(peer, dtype)
                             : recv_tensor(peer, dtype, acc.numel());
        if (first) { acc = s; first = false; }
        else acc.add_(s);
      }
      if (r.red == ReduceOp::AVERAGE) acc.div_(static_cast<double>(nmem));
      for (int i = 0; i < nmem; ++i) {
        int peer = psi.ranks[i];
        if (peer == 0) continue;
        send_tensor(peer, acc.narrow(0, member_off(i) * row, member_rows(i) * row));
      }
      if (im_member)
        mine = acc.narrow(0, member_off(my_idx) * row, member_rows(my_idx) * row);
    } else {
      send_tensor(0, t.contiguous());
      mine = recv_tensor(0, dtype, member_rows(my_idx) * row);
    }
    if (im_member) {
      auto& e = entries[ti];
      auto out_sizes = e.in.sizes().vec();
      if (out_sizes.empty()) out_sizes.push_back(member_rows(my_idx));
      else out_sizes[0] = member_rows(my_idx);
      e.out.resize_(out_sizes);
      flat(e.out).copy_(mine);
      if (e.postscale != 1.0) e.out.mul_(e.postscale);
    }
  }
  return Status::OK();
}

}  // namespace axon
