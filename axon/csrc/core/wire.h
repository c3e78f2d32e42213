// Single-node control plane: Unix-domain socket channels + binary framing.
//
// Replaces the reference engine's MPI/Gloo out-of-band coordination channel
// (SURVEY.md §2.1 "Coordination transports") with a minimal local transport:
// rank 0 listens on a unix socket, ranks 1..N-1 connect. Messages are
// length-prefixed binary frames. The same channel also carries CPU tensor
// data for the CPU collective backend (star topology through rank 0).
#pragma once

#include <cstdint>
#include <cstring>
#include <string>
#include <vector>

namespace axon {

// Simple byte-buffer serializer/deserializer (little-endian host order; the
// control plane never crosses machines, so host order is the wire order).
class Writer {
 public:
  std::vector<uint8_t> buf;
  void u8(uint8_t v) { buf.push_back(v); }
  void u32(uint32_t v) { raw(&v, 4); }
  void u64(uint64_t v) { raw(&v, 8); }
  void i32(int32_t v) { raw(&v, 4); }
  void i64(int64_t v) { raw(&v, 8); }
  void f64(double v) { raw(&v, 8); }
  void str(const std::string& s) {
    u32(static_cast<uint32_t>(s.size()));
    raw(s.data(), s.size());
  }
  void bytes(const void* p, size_t n) { u64(n); raw(p, n); }
  void raw(const void* p, size_t n) {
    const uint8_t* c = static_cast<const uint8_t*>(p);
    buf.insert(buf.end(), c, c + n);
  }
};

class Reader {
 public:
  const uint8_t* p;
  const uint8_t* end;
  explicit Reader(const std::vector<uint8_t>& b) : p(b.data()), end(b.data() + b.size()) {}
  Reader(const uint8_t* data, size_t n) : p(data), end(data + n) {}
  uint8_t u8() { uint8_t v; get(&v, 1); return v; }
  uint32_t u32() { uint32_t v; get(&v, 4); return v; }
  uint64_t u64() { uint64_t v; get(&v, 8); return v; }
  int32_t i32() { int32_t v; get(&v, 4); return v; }
  int64_t i64() { int64_t v; get(&v, 8); return v; }
  double f64() { double v; get(&v, 8); return v; }
  std::string str() {
    uint32_t n = u32();
    std::string s(reinterpret_cast<const char*>(p), n);
    p += n;
    return s;
  }
  size_t bytes_len() { return static_cast<size_t>(u64()); }
  void get(void* dst, size_t n) {
    std::memcpy(dst, p, n);
    p += n;
  }
  bool done() const { return p >= end; }
};

// Message type ids on the control channel.
enum class MsgType : uint8_t {
  REQUEST = 1,        // worker -> coord: tensor ready (Request list)
  RESPONSE_LIST = 2,  // coord -> worker: ordered responses to execute
  CPU_DATA = 3,       // CPU collective payload (either direction)
  RENDEZVOUS = 4,     // startup info exchange (rccl unique id, etc.)
  SHUTDOWN = 5,       // coord -> worker: exit background loop
  ACK = 6,
};

// A framed, blocking, point-to-point socket channel.
class Channel {
 public:
  Channel() = default;
  explicit Channel(int fd) : fd_(fd) {}
  ~Channel();
  Channel(const Channel&) = delete;
  Channel& operator=(const Channel&) = delete;
  Channel(Channel&& o) noexcept : fd_(o.fd_) { o.fd_ = -1; }
  Channel& operator=(Channel&& o) noexcept;

  bool valid() const { return fd_ >= 0; }
  int fd() const { return fd_; }

  // Send one frame: [u32 len][u8 type][payload]. Thread-compatible (callers
  // serialize). Throws std::runtime_error on socket failure.
  void send(MsgType type, const std::vector<uint8_t>& payload);
  // Blocking receive of one frame. Returns false on clean EOF.
  bool recv(MsgType* type, std::vector<uint8_t>* payload);
  // Non-blocking check: true if at least one byte is readable.
  bool readable(int timeout_ms = 0);
  void close_fd();

 private:
  int fd_ = -1;
};

// Rank 0: listen + accept `nconn` connections, each identified by the
// connecting rank (first frame is a RENDEZVOUS frame carrying the rank).
// Returns channels indexed by rank (index 0 unused).
std::vector<Channel> server_accept(const std::string& path, int world_size,
                                   double timeout_sec);
// Rank >0: connect (retry until timeout) and announce our rank.
Channel client_connect(const std::string& path, int rank, double timeout_sec);

}  // namespace axon
