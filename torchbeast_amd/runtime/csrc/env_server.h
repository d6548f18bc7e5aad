// EnvServer: hosts Python environments behind the unix-socket step protocol.
//
// Capability parity with the reference's gRPC EnvServer (ref:
// src/cc/rpcenv.cc:37-211): one env instance per client connection, the GIL
// is held only around env_init/reset/step (network IO runs GIL-free),
// auto-reset on done with episode_step/episode_return bookkeeping reported
// for the completed episode.

#pragma once

#include <arpa/inet.h>
#include <netinet/in.h>
#include <pybind11/pybind11.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

#include <atomic>
#include <cstring>
#include <mutex>
#include <thread>
#include <vector>

#include "env_transport.h"
#include "wire.h"

namespace tbruntime {

namespace py = pybind11;

class EnvServer {
 public:
  EnvServer(py::object env_init, std::string address)
      : env_init_(std::move(env_init)), address_(address),
        path_(strip_unix_prefix(address)) {}

  ~EnvServer() {
    try {
      stop();
    } catch (...) {
    }
  }

  void start() {
    if (listen_fd_ >= 0) throw std::runtime_error("server already running");
    std::string host;
    uint16_t port;
    if (parse_tcp_address(address_, &host, &port)) {
      // Cross-machine env plane: listen on TCP (the reference's gRPC server
      // had this for free; here it is the same framed protocol over TCP).
      listen_fd_ = ::socket(AF_INET, SOCK_STREAM, 0);
      if (listen_fd_ < 0) throw std::runtime_error("socket() failed");
      int one = 1;
      ::setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
      sockaddr_in addr;
      std::memset(&addr, 0, sizeof(addr));
      addr.sin_family = AF_INET;
      addr.sin_port = htons(port);
      addr.sin_addr.s_addr =
          host.empty() ? INADDR_ANY : ::inet_addr(host.c_str());
      if (::bind(listen_fd_, reinterpret_cast<sockaddr*>(&addr),
                 sizeof(addr)) != 0) {
        throw std::runtime_error("bind(" + address_ + ") failed");
      }
    } else {
      ::unlink(path_.c_str());
      listen_fd_ = ::socket(AF_UNIX, SOCK_STREAM, 0);
      if (listen_fd_ < 0) throw std::runtime_error("socket() failed");
      sockaddr_un addr;
      std::memset(&addr, 0, sizeof(addr));
      addr.sun_family = AF_UNIX;
      std::strncpy(addr.sun_path, path_.c_str(), sizeof(addr.sun_path) - 1);
      if (::bind(listen_fd_, reinterpret_cast<sockaddr*>(&addr),
                 sizeof(addr)) != 0) {
        throw std::runtime_error("bind(" + path_ + ") failed");
      }
    }
    if (::listen(listen_fd_, 128) != 0) {
      throw std::runtime_error("listen() failed");
    }
    running_ = true;
    accept_thread_ = std::thread([this] { accept_loop(); });
  }

  // Blocking serve (start + wait until stop()).
  void run() {
    start();
    if (accept_thread_.joinable()) accept_thread_.join();
    join_sessions();
  }

  void stop() {
    running_ = false;
    if (listen_fd_ >= 0) {
      ::shutdown(listen_fd_, SHUT_RDWR);
      ::close(listen_fd_);
      listen_fd_ = -1;
    }
    {
      std::lock_guard<std::mutex> lock(mu_);
      for (int fd : session_fds_) ::shutdown(fd, SHUT_RDWR);
    }
    if (accept_thread_.joinable() &&
        accept_thread_.get_id() != std::this_thread::get_id()) {
      accept_thread_.join();
    }
    join_sessions();
    ::unlink(path_.c_str());
  }

 private:
  void accept_loop() {
    while (running_) {
      int fd = ::accept(listen_fd_, nullptr, nullptr);
      if (fd < 0) {
        if (running_ && errno == EINTR) continue;
        break;
      }
      {
        std::lock_guard<std::mutex> lock(mu_);
        session_fds_.push_back(fd);
        session_threads_.emplace_back([this, fd] { session(fd); });
      }
    }
  }

  void join_sessions() {
    std::vector<std::thread> threads;
    {
      std::lock_guard<std::mutex> lock(mu_);
      threads.swap(session_threads_);
    }
    for (auto& t : threads) {
      if (t.joinable() && t.get_id() != std::this_thread::get_id()) t.join();
    }
  }

  // Convert an arbitrary observation (numpy array, LazyFrames, scalar...) to
  // a contiguous CPU tensor. GIL must be held.
  static torch::Tensor obs_to_tensor(const py::object& obs) {
    py::object np = py::module_::import("numpy");
    py::object arr = np.attr("ascontiguousarray")(obs);
    py::object torch_mod = py::module_::import("torch");
    return torch_mod.attr("as_tensor")(arr).cast<torch::Tensor>();
  }

  static std::string encode_step(const torch::Tensor& frame, float reward,
                                 bool done, int32_t episode_step,
                                 float episode_return) {
    wire::Writer w;
    w.u8(wire::kTagVector);
    w.u32(5);
    w.tensor(frame);
    w.tensor(torch::full({}, reward, torch::kFloat32));
    w.tensor(torch::full({}, done, torch::kBool));
    w.tensor(torch::full({}, episode_step, torch::kInt32));
    w.tensor(torch::full({}, episode_return, torch::kFloat32));
    return std::move(w.buf);
  }

  void session(int fd) {
    SocketStream stream(fd);
    py::object env;  // Destroyed under GIL below.
    try {
      torch::Tensor frame;
      {
        py::gil_scoped_acquire gil;
        env = env_init_();
        frame = obs_to_tensor(env.attr("reset")());
      }
      int32_t episode_step = 0;
      float episode_return = 0.f;
      stream.write_msg(wire::kMsgStep,
                       encode_step(frame, 0.f, true, 0, 0.f));

      char type;
      std::string payload;
      while (running_ && stream.read_msg(&type, &payload)) {
        if (type != wire::kMsgAction) {
          throw std::runtime_error("protocol error: expected action");
        }
        wire::Reader r(payload.data(), payload.size());
        torch::Tensor action = r.nest(0).front();
        int64_t a = action.item<int64_t>();

        float reward;
        bool done;
        {
          py::gil_scoped_acquire gil;
          py::tuple result = env.attr("step")(a).cast<py::tuple>();
          reward = result[1].cast<float>();
          done = result[2].cast<bool>();
          ++episode_step;
          episode_return += reward;
          if (done) {
            // Auto-reset: report completed-episode bookkeeping alongside
            // the first frame of the new episode.
            frame = obs_to_tensor(env.attr("reset")());
          } else {
            frame = obs_to_tensor(result[0]);
          }
        }
        stream.write_msg(
            wire::kMsgStep,
            encode_step(frame, reward, done, episode_step, episode_return));
        if (done) {
          episode_step = 0;
          episode_return = 0.f;
        }
      }
    } catch (const std::exception& e) {
      try {
        stream.write_msg(wire::kMsgError, e.what());
      } catch (...) {
      }
    }
    {
      py::gil_scoped_acquire gil;
      env = py::object();
    }
  }

  py::object env_init_;
  std::string address_;
  std::string path_;
  int listen_fd_ = -1;
  std::atomic<bool> running_{false};
  std::thread accept_thread_;
  std::mutex mu_;
  std::vector<int> session_fds_;
  std::vector<std::thread> session_threads_;
};

}  // namespace tbruntime
