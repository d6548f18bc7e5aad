// Framed binary wire protocol for the environment plane.
//
// The reference uses gRPC bidi streaming with protobuf ArrayNest messages
// (ref: src/proto/rpcenv.proto, src/cc/nest_serialize.h). This runtime
// replaces that with a dependency-free length-framed codec over unix domain
// sockets — the env plane is CPU-side and latency-bound, so a single
// syscall-sized frame with zero-copy payload beats a protobuf round trip.
//
// Frame:   u32le payload_length | u8 type | payload
// Types:   'S' step (array-nest), 'A' action (array-nest), 'E' error (utf8)
// ArrayNest payload (recursive):
//   tag u8: 1 = array | 2 = vector | 3 = map
//   array:  dtype u8, ndim u8, shape u32le × ndim, raw C-contiguous bytes
//   vector: count u32le, items...
//   map:    count u32le, (keylen u16le, key bytes, item)...

#pragma once

#include <torch/extension.h>

#include <cstring>
#include <string>

#include "nest.h"

namespace tbruntime {
namespace wire {

constexpr uint8_t kTagArray = 1;
constexpr uint8_t kTagVector = 2;
constexpr uint8_t kTagMap = 3;

constexpr char kMsgStep = 'S';
constexpr char kMsgAction = 'A';
constexpr char kMsgError = 'E';

inline uint8_t dtype_code(torch::ScalarType t) {
  switch (t) {
    case torch::kUInt8: return 0;
    case torch::kInt8: return 1;
    case torch::kInt16: return 2;
    case torch::kInt32: return 3;
    case torch::kInt64: return 4;
    case torch::kFloat32: return 5;
    case torch::kFloat64: return 6;
    case torch::kBool: return 7;
    default:
      throw std::runtime_error("wire: unsupported dtype");
  }
}

inline torch::ScalarType code_dtype(uint8_t c) {
  switch (c) {
    case 0: return torch::kUInt8;
    case 1: return torch::kInt8;
    case 2: return torch::kInt16;
    case 3: return torch::kInt32;
    case 4: return torch::kInt64;
    case 5: return torch::kFloat32;
    case 6: return torch::kFloat64;
    case 7: return torch::kBool;
    default:
      throw std::runtime_error("wire: bad dtype code");
  }
}

class Writer {
 public:
  std::string buf;

  void u8(uint8_t v) { buf.push_back(static_cast<char>(v)); }
  void u16(uint16_t v) { append(&v, 2); }
  void u32(uint32_t v) { append(&v, 4); }
  void bytes(const void* p, size_t n) { append(p, n); }

  void tensor(const torch::Tensor& t_in) {
    torch::Tensor t = t_in.contiguous();
    u8(kTagArray);
    u8(dtype_code(t.scalar_type()));
    u8(static_cast<uint8_t>(t.dim()));
    for (int64_t d = 0; d < t.dim(); ++d) {
      u32(static_cast<uint32_t>(t.size(d)));
    }
    bytes(t.data_ptr(), t.numel() * t.element_size());
  }

  void nest(const Nest<torch::Tensor>& n) {
    if (n.is_leaf()) {
      tensor(n.leaf());
    } else if (n.is_vector()) {
      u8(kTagVector);
      u32(static_cast<uint32_t>(n.vector().size()));
      for (const auto& child : n.vector()) nest(child);
    } else {
      u8(kTagMap);
      u32(static_cast<uint32_t>(n.map_value().size()));
      for (const auto& kv : n.map_value()) {
        u16(static_cast<uint16_t>(kv.first.size()));
        bytes(kv.first.data(), kv.first.size());
        nest(kv.second);
      }
    }
  }

 private:
  void append(const void* p, size_t n) {
    buf.append(static_cast<const char*>(p), n);
  }
};

class Reader {
 public:
  Reader(const char* data, size_t size) : p_(data), end_(data + size) {}

  uint8_t u8() { return static_cast<uint8_t>(*take(1)); }
  uint16_t u16() {
    uint16_t v;
    std::memcpy(&v, take(2), 2);
    return v;
  }
  uint32_t u32() {
    uint32_t v;
    std::memcpy(&v, take(4), 4);
    return v;
  }

  // Parse one array into a fresh CPU tensor with extra leading dims
  // prepended (the runtime wants [T=1, B=1, ...]-shaped leaves).
  torch::Tensor tensor(int64_t prepend_ones) {
    torch::ScalarType dtype = code_dtype(u8());
    int ndim = u8();
    std::vector<int64_t> shape(prepend_ones, 1);
    int64_t numel = 1;
    for (int i = 0; i < ndim; ++i) {
      int64_t d = u32();
      shape.push_back(d);
      numel *= d;
    }
    torch::Tensor t = torch::empty(
        shape, torch::TensorOptions().dtype(dtype));
    size_t nbytes = numel * t.element_size();
    std::memcpy(t.data_ptr(), take(nbytes), nbytes);
    return t;
  }

  Nest<torch::Tensor> nest(int64_t prepend_ones) {
    uint8_t tag = u8();
    if (tag == kTagArray) {
      // The array tag is consumed; tensor() parses from the dtype byte on.
      return Nest<torch::Tensor>(tensor(prepend_ones));
    }
    if (tag == kTagVector) {
      uint32_t count = u32();
      Nest<torch::Tensor>::vector_t vec;
      vec.reserve(count);
      for (uint32_t i = 0; i < count; ++i) vec.push_back(nest(prepend_ones));
      return Nest<torch::Tensor>(std::move(vec));
    }
    if (tag == kTagMap) {
      uint32_t count = u32();
      Nest<torch::Tensor>::map_t map;
      for (uint32_t i = 0; i < count; ++i) {
        uint16_t klen = u16();
        std::string key(take(klen), klen);
        map.emplace(std::move(key), nest(prepend_ones));
      }
      return Nest<torch::Tensor>(std::move(map));
    }
    throw std::runtime_error("wire: bad nest tag");
  }

 private:
  const char* take(size_t n) {
    if (p_ + n > end_) throw std::runtime_error("wire: truncated message");
    const char* r = p_;
    p_ += n;
    return r;
  }
  const char* p_;
  const char* end_;
};

}  // namespace wire
}  // namespace tbruntime
