#include "message.h"

namespace axon {

namespace {
void put_i64_vec(Writer& w, const std::vector<int64_t>& v) {
  w.u32(static_cast<uint32_t>(v.size()));
  for (int64_t x : v) w.i64(x);
}
std::vector<int64_t> get_i64_vec(Reader& r) {
  uint32_t n = r.u32();
  std::vector<int64_t> v(n);
  for (uint32_t i = 0; i < n; ++i) v[i] = r.i64();
  return v;
}
}  // namespace

void Request::serialize(Writer& w) const {
  w.i32(rank);
  w.u8(static_cast<uint8_t>(op));
  w.u8(static_cast<uint8_t>(red));
  w.str(name);
  w.u8(dtype);
  w.u8(wire_dtype);
  w.i32(device);
  w.i32(root_rank);
  w.i32(process_set);
  put_i64_vec(w, shape);
  put_i64_vec(w, splits);
}

Request Request::deserialize(Reader& r) {
  Request q;
  q.rank = r.i32();
  q.op = static_cast<OpType>(r.u8());
  q.red = static_cast<ReduceOp>(r.u8());
  q.name = r.str();
  q.dtype = r.u8();
  q.wire_dtype = r.u8();
  q.device = r.i32();
  q.root_rank = r.i32();
  q.process_set = r.i32();
  q.shape = get_i64_vec(r);
  q.splits = get_i64_vec(r);
  return q;
}

void Response::serialize(Writer& w) const {
  w.u8(static_cast<uint8_t>(op));
  w.u8(static_cast<uint8_t>(red));
  w.u32(static_cast<uint32_t>(names.size()));
  for (auto& n : names) w.str(n);
  w.str(error);
  w.u8(dtype);
  w.u8(wire_dtype);
  w.i32(device);
  w.i32(root_rank);
  w.i32(process_set);
  put_i64_vec(w, numels);
  w.u32(static_cast<uint32_t>(first_dims.size()));
  for (auto& v : first_dims) put_i64_vec(w, v);
  put_i64_vec(w, splits_matrix);
  w.i32(last_joined);
  w.u32(static_cast<uint32_t>(cache_ids.size()));
  for (int32_t id : cache_ids) w.i32(id);
}

Response Response::deserialize(Reader& r) {
  Response s;
  s.op = static_cast<OpType>(r.u8());
  s.red = static_cast<ReduceOp>(r.u8());
  uint32_t n = r.u32();
  s.names.resize(n);
  for (uint32_t i = 0; i < n; ++i) s.names[i] = r.str();
  s.error = r.str();
  s.dtype = r.u8();
  s.wire_dtype = r.u8();
  s.device = r.i32();
  s.root_rank = r.i32();
  s.process_set = r.i32();
  s.numels = get_i64_vec(r);
  uint32_t fd = r.u32();
  s.first_dims.resize(fd);
  for (uint32_t i = 0; i < fd; ++i) s.first_dims[i] = get_i64_vec(r);
  s.splits_matrix = get_i64_vec(r);
  s.last_joined = r.i32();
  uint32_t nc = r.u32();
  s.cache_ids.resize(nc);
  for (uint32_t i = 0; i < nc; ++i) s.cache_ids[i] = r.i32();
  return s;
}

std::vector<uint8_t> ResponseList::serialize() const {
  Writer w;
  w.u8(shutdown ? 1 : 0);
  w.f64(new_cycle_ms);
  w.u32(static_cast<uint32_t>(responses.size()));
  for (auto& r : responses) r.serialize(w);
  return std::move(w.buf);
}

ResponseList ResponseList::deserialize(const std::vector<uint8_t>& buf) {
  Reader r(buf);
  ResponseList l;
  l.shutdown = r.u8() != 0;
  l.new_cycle_ms = r.f64();
  uint32_t n = r.u32();
  l.responses.resize(n);
  for (uint32_t i = 0; i < n; ++i) l.responses[i] = Response::deserialize(r);
  return l;
}

}  // namespace axon
