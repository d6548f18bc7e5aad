// Environment connections for actor threads.
//
// Two transports behind one interface:
// - SocketEnv: framed wire protocol (wire.h) over a unix domain socket to an
//   EnvServer process (capability parity with the reference's gRPC
//   StreamingEnv client, ref: src/cc/actorpool.cc:354-447).
// - NativeSyntheticEnv: in-process Atari-shaped synthetic env for
//   benchmarks — no sockets, no Python, no serialization in the hot loop;
//   selected by "synthetic:CxHxW:A[:episode_len]" addresses. This is the
//   generalization of the reference's Mock env (ref: polybeast_env.py:39-46)
//   pushed into the actor thread itself.
//
// Both return step nests shaped for the runtime: a 5-tuple
// (frame, reward, done, episode_step, episode_return), each leaf with
// leading [T=1, B=1] dims.

#pragma once

#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

#include <chrono>
#include <cstring>
#include <memory>
#include <sstream>
#include <string>
#include <thread>

#include "queues.h"
#include "wire.h"

namespace tbruntime {

// --------------------------------------------------------------------------
// Blocking exact-size socket IO.
// --------------------------------------------------------------------------

class SocketStream {
 public:
  explicit SocketStream(int fd) : fd_(fd) {}
  SocketStream(const SocketStream&) = delete;
  ~SocketStream() { close(); }

  void close() {
    if (fd_ >= 0) {
      ::shutdown(fd_, SHUT_RDWR);
      ::close(fd_);
      fd_ = -1;
    }
  }

  void write_msg(char type, const std::string& payload) {
    uint32_t len = static_cast<uint32_t>(payload.size());
    char header[5];
    std::memcpy(header, &len, 4);
    header[4] = type;
    write_all(header, 5);
    write_all(payload.data(), payload.size());
  }

  // Returns false on orderly EOF at a frame boundary.
  bool read_msg(char* type, std::string* payload) {
    char header[5];
    if (!read_all(header, 5, /*eof_ok=*/true)) return false;
    uint32_t len;
    std::memcpy(&len, header, 4);
    *type = header[4];
    payload->resize(len);
    if (len > 0) read_all(&(*payload)[0], len, /*eof_ok=*/false);
    return true;
  }

 private:
  void write_all(const void* buf, size_t n) {
    const char* p = static_cast<const char*>(buf);
    while (n > 0) {
      ssize_t w = ::send(fd_, p, n, MSG_NOSIGNAL);
      if (w <= 0) {
        if (w < 0 && errno == EINTR) continue;
        throw std::runtime_error("env socket write failed");
      }
      p += w;
      n -= w;
    }
  }

  bool read_all(void* buf, size_t n, bool eof_ok) {
    char* p = static_cast<char*>(buf);
    size_t got = 0;
    while (got < n) {
      ssize_t r = ::recv(fd_, p + got, n - got, 0);
      if (r == 0) {
        if (eof_ok && got == 0) return false;
        throw std::runtime_error("env socket closed mid-message");
      }
      if (r < 0) {
        if (errno == EINTR) continue;
        throw std::runtime_error("env socket read failed");
      }
      got += r;
    }
    return true;
  }

  int fd_;
};

inline std::string strip_unix_prefix(const std::string& address) {
  if (address.rfind("unix:", 0) == 0) return address.substr(5);
  return address;
}

// "tcp:host:port" -> (host, port); returns false for non-TCP addresses.
// The reference's gRPC env plane worked over any channel
// (ref: src/proto/rpcenv.proto:46-48); this restores the cross-machine
// capability for the framed wire protocol.
inline bool parse_tcp_address(const std::string& address, std::string* host,
                              uint16_t* port) {
  if (address.rfind("tcp:", 0) != 0) return false;
  const std::string rest = address.substr(4);
  const size_t colon = rest.rfind(':');
  if (colon == std::string::npos) {
    throw std::runtime_error("tcp address must be tcp:host:port, got " +
                             address);
  }
  *host = rest.substr(0, colon);
  *port = static_cast<uint16_t>(std::stoi(rest.substr(colon + 1)));
  return true;
}

inline int connect_tcp(const std::string& host, uint16_t port,
                       std::chrono::seconds deadline) {
  addrinfo hints;
  std::memset(&hints, 0, sizeof(hints));
  hints.ai_family = AF_UNSPEC;
  hints.ai_socktype = SOCK_STREAM;
  const std::string port_s = std::to_string(port);
  auto start = std::chrono::steady_clock::now();
  for (;;) {
    addrinfo* res = nullptr;
    if (::getaddrinfo(host.c_str(), port_s.c_str(), &hints, &res) == 0) {
      for (addrinfo* ai = res; ai != nullptr; ai = ai->ai_next) {
        int fd = ::socket(ai->ai_family, ai->ai_socktype, ai->ai_protocol);
        if (fd < 0) continue;
        if (::connect(fd, ai->ai_addr, ai->ai_addrlen) == 0) {
          int one = 1;
          ::setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
          ::freeaddrinfo(res);
          return fd;
        }
        ::close(fd);
      }
      ::freeaddrinfo(res);
    }
    if (std::chrono::steady_clock::now() - start > deadline) {
      throw std::runtime_error("timeout connecting to env server at tcp:" +
                               host + ":" + port_s);
    }
    std::this_thread::sleep_for(std::chrono::milliseconds(50));
  }
}

inline int connect_unix(const std::string& address,
                        std::chrono::seconds deadline) {
  std::string path = strip_unix_prefix(address);
  auto start = std::chrono::steady_clock::now();
  for (;;) {
    int fd = ::socket(AF_UNIX, SOCK_STREAM, 0);
    if (fd < 0) throw std::runtime_error("socket() failed");
    sockaddr_un addr;
    std::memset(&addr, 0, sizeof(addr));
    addr.sun_family = AF_UNIX;
    std::strncpy(addr.sun_path, path.c_str(), sizeof(addr.sun_path) - 1);
    if (::connect(fd, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) == 0) {
      return fd;
    }
    ::close(fd);
    if (std::chrono::steady_clock::now() - start > deadline) {
      throw std::runtime_error("timeout connecting to env server at " + path);
    }
    std::this_thread::sleep_for(std::chrono::milliseconds(50));
  }
}

// --------------------------------------------------------------------------
// Env connection interface.
// --------------------------------------------------------------------------

class EnvConnection {
 public:
  virtual ~EnvConnection() = default;
  virtual TensorNest initial() = 0;
  virtual TensorNest step(const torch::Tensor& action) = 0;
};

inline int connect_env_socket(const std::string& address,
                              std::chrono::seconds deadline) {
  std::string host;
  uint16_t port;
  if (parse_tcp_address(address, &host, &port)) {
    return connect_tcp(host, port, deadline);
  }
  return connect_unix(address, deadline);
}

class SocketEnv : public EnvConnection {
 public:
  SocketEnv(const std::string& address, std::chrono::seconds connect_deadline)
      : stream_(connect_env_socket(address, connect_deadline)) {}

  TensorNest initial() override { return read_step(); }

  TensorNest step(const torch::Tensor& action) override {
    wire::Writer w;
    w.tensor(action.squeeze());  // strip the [1,1] dims for the wire
    stream_.write_msg(wire::kMsgAction, w.buf);
    return read_step();
  }

 private:
  TensorNest read_step() {
    char type;
    std::string payload;
    if (!stream_.read_msg(&type, &payload)) {
      throw ClosedQueue("env server closed the connection");
    }
    if (type == wire::kMsgError) {
      throw std::runtime_error("env server error: " + payload);
    }
    if (type != wire::kMsgStep) {
      throw std::runtime_error("env protocol error: unexpected message");
    }
    wire::Reader r(payload.data(), payload.size());
    return r.nest(/*prepend_ones=*/2);
  }

  SocketStream stream_;
};

// --------------------------------------------------------------------------
// Native synthetic env (benchmark fast path).
// --------------------------------------------------------------------------

struct SyntheticSpec {
  std::vector<int64_t> shape{4, 84, 84};
  int64_t num_actions = 6;
  int64_t episode_length = 1000;

  // "synthetic[:CxHxW[:A[:len]]]"
  static SyntheticSpec parse(const std::string& address) {
    SyntheticSpec spec;
    std::stringstream ss(address);
    std::string part;
    int idx = 0;
    while (std::getline(ss, part, ':')) {
      if (idx == 1 && !part.empty()) {
        spec.shape.clear();
        std::stringstream dims(part);
        std::string d;
        while (std::getline(dims, d, 'x')) spec.shape.push_back(std::stoll(d));
      } else if (idx == 2 && !part.empty()) {
        spec.num_actions = std::stoll(part);
      } else if (idx == 3 && !part.empty()) {
        spec.episode_length = std::stoll(part);
      }
      ++idx;
    }
    return spec;
  }
};

class NativeSyntheticEnv : public EnvConnection {
 public:
  NativeSyntheticEnv(const SyntheticSpec& spec, uint64_t seed)
      : spec_(spec), state_(seed * 2654435761ull + 1) {}

  TensorNest initial() override {
    episode_step_ = 0;
    episode_return_ = 0.f;
    return make_step(/*reward=*/0.f, /*done=*/true);
  }

  TensorNest step(const torch::Tensor& action) override {
    ++episode_step_;
    advance();
    float reward = static_cast<float>(
        static_cast<int64_t>((state_ + action.item<int64_t>()) % 3) - 1);
    episode_return_ += reward;
    bool done = episode_step_ >= spec_.episode_length;
    TensorNest out = make_step(reward, done);
    if (done) {
      episode_step_ = 0;
      episode_return_ = 0.f;
    }
    return out;
  }

 private:
  void advance() {
    state_ ^= state_ << 13;
    state_ ^= state_ >> 7;
    state_ ^= state_ << 17;
  }

  TensorNest make_step(float reward, bool done) {
    std::vector<int64_t> shape{1, 1};
    shape.insert(shape.end(), spec_.shape.begin(), spec_.shape.end());
    torch::Tensor frame =
        torch::empty(shape, torch::TensorOptions().dtype(torch::kUInt8));
    auto* data = frame.data_ptr<uint8_t>();
    int64_t n = frame.numel();
    std::memset(data, static_cast<int>(state_ & 0xFF), n);
    uint8_t x = static_cast<uint8_t>((state_ >> 8) & 0xFF);
    for (int64_t i = 0; i < n; i += 8) data[i] ^= x;

    TensorNest::vector_t fields;
    fields.emplace_back(frame);
    fields.emplace_back(torch::full({1, 1}, reward, torch::kFloat32));
    fields.emplace_back(torch::full({1, 1}, done, torch::kBool));
    fields.emplace_back(
        torch::full({1, 1}, static_cast<int>(episode_step_), torch::kInt32));
    fields.emplace_back(torch::full({1, 1}, episode_return_, torch::kFloat32));
    return TensorNest(std::move(fields));
  }

  SyntheticSpec spec_;
  uint64_t state_;
  int64_t episode_step_ = 0;
  float episode_return_ = 0.f;
};

inline std::unique_ptr<EnvConnection> make_env_connection(
    const std::string& address, uint64_t seed,
    std::chrono::seconds connect_deadline = std::chrono::seconds(60)) {
  if (address.rfind("synthetic", 0) == 0) {
    return std::make_unique<NativeSyntheticEnv>(SyntheticSpec::parse(address),
                                                seed);
  }
  return std::make_unique<SocketEnv>(address, connect_deadline);
}

}  // namespace tbruntime
