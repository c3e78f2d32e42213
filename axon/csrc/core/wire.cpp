#include "wire.h"

#include <errno.h>
#include <poll.h>
#include <stdexcept>
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

#include <chrono>
#include <cstdio>
#include <thread>

namespace axon {

namespace {
void write_all(int fd, const void* buf, size_t n) {
  const char* p = static_cast<const char*>(buf);
  while (n > 0) {
    ssize_t w = ::write(fd, p, n);
    if (w < 0) {
      if (errno == EINTR) continue;
      throw std::runtime_error("axon control socket write failed: " +
                               std::string(strerror(errno)));
    }
    p += w;
    n -= static_cast<size_t>(w);
  }
}

// Returns false on clean EOF at a frame boundary.
bool read_all(int fd, void* buf, size_t n, bool eof_ok) {
  char* p = static_cast<char*>(buf);
  size_t got = 0;
  while (got < n) {
    ssize_t r = ::read(fd, p + got, n - got);
    if (r < 0) {
      if (errno == EINTR) continue;
      throw std::runtime_error("axon control socket read failed: " +
                               std::string(strerror(errno)));
    }
    if (r == 0) {
      if (eof_ok && got == 0) return false;
      throw std::runtime_error("axon control socket: unexpected EOF");
    }
    got += static_cast<size_t>(r);
  }
  return true;
}
}  // namespace

Channel::~Channel() { close_fd(); }

Channel& Channel::operator=(Channel&& o) noexcept {
  if (this != &o) {
    close_fd();
    fd_ = o.fd_;
    o.fd_ = -1;
  }
  return *this;
}

void Channel::close_fd() {
  if (fd_ >= 0) {
    ::close(fd_);
    fd_ = -1;
  }
}

void Channel::send(MsgType type, const std::vector<uint8_t>& payload) {
  uint32_t len = static_cast<uint32_t>(payload.size());
  uint8_t hdr[5];
  std::memcpy(hdr, &len, 4);
  hdr[4] = static_cast<uint8_t>(type);
  write_all(fd_, hdr, 5);
  if (len) write_all(fd_, payload.data(), len);
}

bool Channel::recv(MsgType* type, std::vector<uint8_t>* payload) {
  uint8_t hdr[5];
  if (!read_all(fd_, hdr, 5, /*eof_ok=*/true)) return false;
  uint32_t len;
  std::memcpy(&len, hdr, 4);
  *type = static_cast<MsgType>(hdr[4]);
  payload->resize(len);
  if (len) read_all(fd_, payload->data(), len, /*eof_ok=*/false);
  return true;
}

bool Channel::readable(int timeout_ms) {
  struct pollfd pfd = {fd_, POLLIN, 0};
  int r = ::poll(&pfd, 1, timeout_ms);
  return r > 0 && (pfd.revents & POLLIN);
}

std::vector<Channel> server_accept(const std::string& path, int world_size,
                                   double timeout_sec) {
  int lfd = ::socket(AF_UNIX, SOCK_STREAM, 0);
  if (lfd < 0) throw std::runtime_error("axon: socket() failed");
  struct sockaddr_un addr;
  std::memset(&addr, 0, sizeof(addr));
  addr.sun_family = AF_UNIX;
  if (path.size() >= sizeof(addr.sun_path))
    throw std::runtime_error("axon: control socket path too long: " + path);
  std::strncpy(addr.sun_path, path.c_str(), sizeof(addr.sun_path) - 1);
  ::unlink(path.c_str());
  if (::bind(lfd, reinterpret_cast<struct sockaddr*>(&addr), sizeof(addr)) < 0) {
    ::close(lfd);
    throw std::runtime_error("axon: bind(" + path + ") failed: " + strerror(errno));
  }
  if (::listen(lfd, world_size) < 0) {
    ::close(lfd);
    throw std::runtime_error("axon: listen failed");
  }
  std::vector<Channel> chans(world_size);
  auto deadline = std::chrono::steady_clock::now() +
                  std::chrono::duration<double>(timeout_sec);
  for (int i = 1; i < world_size; ++i) {
    struct pollfd pfd = {lfd, POLLIN, 0};
    auto left = std::chrono::duration_cast<std::chrono::milliseconds>(
                    deadline - std::chrono::steady_clock::now())
                    .count();
    if (left <= 0 || ::poll(&pfd, 1, static_cast<int>(left)) <= 0) {
      ::close(lfd);
      throw std::runtime_error("axon: timeout waiting for ranks to connect");
    }
    int cfd = ::accept(lfd, nullptr, nullptr);
    if (cfd < 0) {
      ::close(lfd);
      throw std::runtime_error("axon: accept failed");
    }
    Channel ch(cfd);
    MsgType t;
    std::vector<uint8_t> payload;
    if (!ch.recv(&t, &payload) || t != MsgType::RENDEZVOUS || payload.size() < 4)
      throw std::runtime_error("axon: bad rendezvous frame");
    Reader r(payload);
    int32_t rank = r.i32();
    if (rank <= 0 || rank >= world_size || chans[rank].valid())
      throw std::runtime_error("axon: invalid rank in rendezvous: " +
                               std::to_string(rank));
    chans[rank] = std::move(ch);
  }
  ::close(lfd);
  ::unlink(path.c_str());
  return chans;
}

Channel client_connect(const std::string& path, int rank, double timeout_sec) {
  auto deadline = std::chrono::steady_clock::now() +
                  std::chrono::duration<double>(timeout_sec);
  for (;;) {
    int fd = ::socket(AF_UNIX, SOCK_STREAM, 0);
    if (fd < 0) throw std::runtime_error("axon: socket() failed");
    struct sockaddr_un addr;
    std::memset(&addr, 0, sizeof(addr));
    addr.sun_family = AF_UNIX;
    std::strncpy(addr.sun_path, path.c_str(), sizeof(addr.sun_path) - 1);
    if (::connect(fd, reinterpret_cast<struct sockaddr*>(&addr), sizeof(addr)) == 0) {
      Channel ch(fd);
      Writer w;
      w.i32(rank);
      ch.send(MsgType::RENDEZVOUS, w.buf);
      return ch;
    }
    ::close(fd);
    if (std::chrono::steady_clock::now() > deadline)
      throw std::runtime_error("axon: timeout connecting to coordinator at " + path);
    std::this_thread::sleep_for(std::chrono::milliseconds(20));
  }
}

}  // namespace axon
