#include "cpu_backend.h"

#include <ATen/Functions.h>

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
  at::ScalarType dtype = id_dtype(r.dtype);
  at::ScalarType wire = r.wire_dtype ? id_dtype(r.wire_dtype) : dtype;
  for (auto& e : entries) {
    at::Tensor t = flat(e.in);
    if (e.prescale != 1.0) t = t.mul(e.prescale);
    if (wire != dtype) t = t.to(wire);
    t = t.contiguous();

    at::Tensor result;
    if (rank_ == 0) {
      at::Tensor acc = t.clone();
      for (int peer = 1; peer < size_; ++peer) {
        at::Tensor s = recv_tensor(peer, wire, t.numel());
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
      if (r.red == ReduceOp::AVERAGE) acc.div_(static_cast<double>(size_));
      for (int peer = 1; peer < size_; ++peer) send_tensor(peer, acc);
      result = acc;
    } else {
      send_tensor(0, t);
      result = recv_tensor(0, wire, t.numel());
    }
    at::Tensor out = flat(e.out);
    out.copy_(result);  // casts wire -> dtype if compressed
    if (e.postscale != 1.0) out.mul_(e.postscale);
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
  // (level k pairs rank r with r^2^k): gather on rank 0, fold the tree in
  // fp64, fan the result out. CPU path is a correctness oracle, not a perf
  // path — the GPU path (rccl_backend) does the real distributed recursion.
  if ((size_ & (size_ - 1)) != 0)
    return Status::Error("adasum requires a power-of-two world size");
  at::ScalarType dtype = id_dtype(r.dtype);
  for (auto& e : entries) {
    at::Tensor t = flat(e.in);
    if (e.prescale != 1.0) t = t.mul(e.prescale);
    at::Tensor result;
    if (rank_ == 0) {
      std::vector<at::Tensor> v(size_);
      v[0] = t.to(at::kDouble);
      for (int peer = 1; peer < size_; ++peer)
        v[peer] = recv_tensor(peer, dtype, t.numel()).to(at::kDouble);
      for (int stride = 1; stride < size_; stride *= 2)
        for (int i = 0; i + stride < size_; i += 2 * stride)
          v[i] = adasum_combine_cpu(v[i], v[i + stride]);
      result = v[0].to(dtype);
      for (int peer = 1; peer < size_; ++peer) send_tensor(peer, result);
    } else {
      send_tensor(0, t.contiguous());
      result = recv_tensor(0, dtype, t.numel());
    }
    at::Tensor out = flat(e.out);
    out.copy_(result);
    if (e.postscale != 1.0) out.mul_(e.postscale);
  }
  return Status::OK();
}

Status CpuBackend::broadcast(const Response& r, std::vector<TensorEntry>& entries) {
  for (auto& e : entries) {
    int root = r.root_rank;
    at::Tensor buf;
    if (rank_ == root) {
      buf = flat(e.in);
      if (root != 0) send_tensor(0, buf);
    }
    if (rank_ == 0) {
      if (root != 0) buf = recv_tensor(root, id_dtype(r.dtype), e.out.numel());
      for (int peer = 1; peer < size_; ++peer)
        if (peer != root) send_tensor(peer, buf);
    } else if (rank_ != root) {
      buf = recv_tensor(0, id_dtype(r.dtype), e.out.numel());
    }
    if (rank_ != root || !e.out.is_same(e.in))
      flat(e.out).copy_(buf);
  }
  return Status::OK();
}

Status CpuBackend::allgather(const Response& r, std::vector<TensorEntry>& entries) {
  at::ScalarType dtype = id_dtype(r.dtype);
  for (size_t ti = 0; ti < entries.size(); ++ti) {
    auto& e = entries[ti];
    const auto& fd = r.first_dims.at(ti);
    auto in_sizes = e.in.sizes().vec();
    int64_t row = 1;
    for (size_t d = 1; d < in_sizes.size(); ++d) row *= in_sizes[d];
    int64_t total_rows = 0;
    for (auto x : fd) total_rows += x;

    at::Tensor full;
    if (rank_ == 0) {
      full = at::empty({total_rows * row}, at::TensorOptions().dtype(dtype));
      int64_t off = 0;
      for (int rk = 0; rk < size_; ++rk) {
        int64_t n = fd[rk] * row;
        if (n == 0) continue;
        at::Tensor part = rk == 0 ? flat(e.in) : recv_tensor(rk, dtype, n);
        full.narrow(0, off, n).copy_(part);
        off += n;
      }
      for (int peer = 1; peer < size_; ++peer) send_tensor(peer, full);
    } else {
      send_tensor(0, flat(e.in));
      full = recv_tensor(0, dtype, total_rows * row);
    }
    auto out_sizes = in_sizes;
    if (out_sizes.empty()) out_sizes.push_back(total_rows); else out_sizes[0] = total_rows;
    e.out.resize_(out_sizes);
    flat(e.out).copy_(full);
  }
  return Status::OK();
}

Status CpuBackend::alltoall(const Response& r, std::vector<TensorEntry>& entries) {
  at::ScalarType dtype = id_dtype(r.dtype);
  auto m = [&](int src, int dst) {
    return r.splits_matrix.at(static_cast<size_t>(src) * size_ + dst);
  };
  for (auto& e : entries) {
    auto in_sizes = e.in.sizes().vec();
    int64_t row = 1;
    for (size_t d = 1; d < in_sizes.size(); ++d) row *= in_sizes[d];

    at::Tensor mine;
    if (rank_ == 0) {
      std::vector<at::Tensor> inputs(size_);
      inputs[0] = flat(e.in);
      for (int rk = 1; rk < size_; ++rk) {
        int64_t rows = 0;
        for (int j = 0; j < size_; ++j) rows += m(rk, j);
        inputs[rk] = recv_tensor(rk, dtype, rows * row);
      }
      for (int dst = 0; dst < size_; ++dst) {
        int64_t rows_dst = 0;
        for (int src = 0; src < size_; ++src) rows_dst += m(src, dst);
        at::Tensor outbuf = at::empty({rows_dst * row}, at::TensorOptions().dtype(dtype));
        int64_t off = 0;
        for (int src = 0; src < size_; ++src) {
          int64_t pre = 0;  // rows src sends to dst' < dst
          for (int j = 0; j < dst; ++j) pre += m(src, j);
          int64_t n = m(src, dst) * row;
          if (n)
            outbuf.narrow(0, off, n).copy_(inputs[src].narrow(0, pre * row, n));
          off += n;
        }
        if (dst == 0) mine = outbuf;
        else send_tensor(dst, outbuf);
      }
    } else {
      send_tensor(0, flat(e.in));
      int64_t rows_me = 0;
      for (int src = 0; src < size_; ++src) rows_me += m(src, rank_);
      mine = recv_tensor(0, dtype, rows_me * row);
    }
    auto out_sizes = in_sizes;
    int64_t rows_me = mine.numel() / std::max<int64_t>(row, 1);
    if (out_sizes.empty()) out_sizes.push_back(rows_me); else out_sizes[0] = rows_me;
    e.out.resize_(out_sizes);
    flat(e.out).copy_(mine);
    if (e.received_splits.defined()) {
      auto rs = e.received_splits;
      rs.resize_({size_});
      for (int src = 0; src < size_; ++src)
        rs[src] = m(src, rank_);
    }
  }
  return Status::OK();
}

Status CpuBackend::reducescatter(const Response& r, std::vector<TensorEntry>& entries) {
  at::ScalarType dtype = id_dtype(r.dtype);
  for (auto& e : entries) {
    auto in_sizes = e.in.sizes().vec();
    int64_t rows = in_sizes.empty() ? 1 : in_sizes[0];
    int64_t row = 1;
    for (size_t d = 1; d < in_sizes.size(); ++d) row *= in_sizes[d];
    int64_t base = rows / size_, rem = rows % size_;
    auto rank_rows = [&](int rk) { return base + (rk < rem ? 1 : 0); };
    auto rank_off = [&](int rk) {
      return static_cast<int64_t>(rk) * base + std::min<int64_t>(rk, rem);
    };

    at::Tensor t = flat(e.in);
    if (e.prescale != 1.0) t = t.mul(e.prescale);
    at::Tensor mine;
    if (rank_ == 0) {
      at::Tensor acc = t.clone();
      for (int peer = 1; peer < size_; ++peer)
        acc.add_(recv_tensor(peer, dtype, t.numel()));
      if (r.red == ReduceOp::AVERAGE) acc.div_(static_cast<double>(size_));
      for (int peer = 1; peer < size_; ++peer)
        send_tensor(peer, acc.narrow(0, rank_off(peer) * row, rank_rows(peer) * row));
      mine = acc.narrow(0, 0, rank_rows(0) * row);
    } else {
      send_tensor(0, t.contiguous());
      mine = recv_tensor(0, dtype, rank_rows(rank_) * row);
    }
    auto out_sizes = in_sizes;
    if (out_sizes.empty()) out_sizes.push_back(rank_rows(rank_));
    else out_sizes[0] = rank_rows(rank_);
    e.out.resize_(out_sizes);
    flat(e.out).copy_(mine);
    if (e.postscale != 1.0) e.out.mul_(e.postscale);
  }
  return Status::OK();
}

}  // namespace axon
