// Deterministic fuzz of the wire codec (runtime/csrc/wire.h): random
// TensorNest structures roundtrip bit-exactly, and every truncated prefix
// of a valid message throws wire's "truncated" error instead of reading
// out of bounds. Analogue of the reference's nest_serialize_test.cc
// (src/cc/nest_serialize_test.cc:33-138), as a fuzz property instead of
// fixed cases. Build like stress_queues.cc (see README.md); add
// -fsanitize=address for the ASAN flavor.

#include <torch/torch.h>

#include <cstdio>
#include <cstring>
#include <cstdlib>
#include <random>
#include <string>
#include <vector>

#include "wire.h"

using tbruntime::wire::Reader;
using tbruntime::wire::Writer;
using TensorNest = tbruntime::Nest<torch::Tensor>;

namespace {

std::mt19937_64 rng(0xC0FFEE);

int randint(int lo, int hi) {  // inclusive
  return (int)(rng() % (uint64_t)(hi - lo + 1)) + lo;
}

torch::Tensor random_tensor() {
  static const torch::ScalarType dtypes[] = {
      torch::kUInt8, torch::kInt8,    torch::kInt16, torch::kInt32,
      torch::kInt64, torch::kFloat32, torch::kFloat64, torch::kBool};
  const auto dtype = dtypes[randint(0, 7)];
  const int dim = randint(0, 4);
  std::vector<int64_t> shape;
  for (int i = 0; i < dim; ++i) shape.push_back(randint(0, 5));  // 0-dims too
  torch::Tensor t = torch::empty(shape, torch::TensorOptions().dtype(dtype));
  if (t.numel() > 0) {
    auto* p = static_cast<uint8_t*>(t.data_ptr());
    for (int64_t i = 0; i < t.numel() * t.element_size(); ++i) {
      p[i] = (uint8_t)(rng() & 0xFF);
    }
    if (dtype == torch::kBool) t = t.ne(0);  // normalize to 0/1 bytes
  }
  // Sometimes hand the writer a non-contiguous view (it must .contiguous()).
  if (dim >= 2 && shape[0] > 1 && randint(0, 2) == 0) t = t.transpose(0, 1);
  return t;
}

TensorNest random_nest(int depth) {
  const int kind = depth == 0 ? 0 : randint(0, 3);  // leaf-biased
  if (kind <= 1) return TensorNest(random_tensor());
  if (kind == 2) {
    TensorNest::vector_t vec;
    const int n = randint(0, 3);
    for (int i = 0; i < n; ++i) vec.push_back(random_nest(depth - 1));
    return TensorNest(std::move(vec));
  }
  TensorNest::map_t map;
  const int n = randint(1, 3);
  for (int i = 0; i < n; ++i) {
    map.emplace("k" + std::to_string(randint(0, 99)), random_nest(depth - 1));
  }
  return TensorNest(std::move(map));
}

bool same(const TensorNest& a, const TensorNest& b) {
  if (a.is_leaf() != b.is_leaf() || a.is_vector() != b.is_vector()) {
    return false;
  }
  if (a.is_leaf()) {
    torch::Tensor x = a.leaf().contiguous();
    torch::Tensor y = b.leaf().contiguous();
    // Bytewise compare: random float bit patterns include NaNs, which
    // tensor.equal() would treat as unequal to themselves.
    return x.scalar_type() == y.scalar_type() && x.sizes() == y.sizes() &&
           (x.numel() == 0 ||
            std::memcmp(x.data_ptr(), y.data_ptr(),
                        x.numel() * x.element_size()) == 0);
  }
  if (a.is_vector()) {
    if (a.vector().size() != b.vector().size()) return false;
    for (size_t i = 0; i < a.vector().size(); ++i) {
      if (!same(a.vector()[i], b.vector()[i])) return false;
    }
    return true;
  }
  if (a.map_value().size() != b.map_value().size()) return false;
  auto it = b.map_value().begin();
  for (const auto& kv : a.map_value()) {
    if (kv.first != it->first || !same(kv.second, it->second)) return false;
    ++it;
  }
  return true;
}

}  // namespace

int main() {
  int truncated_checked = 0;
  for (int iter = 0; iter < 2000; ++iter) {
    TensorNest n = random_nest(3);
    Writer w;
    w.nest(n);
    Reader r(w.buf.data(), w.buf.size());
    TensorNest back = r.nest(/*prepend_ones=*/0);
    if (!same(n, back)) {
      std::fprintf(stderr, "FAIL roundtrip mismatch at iter %d\n", iter);
      return 1;
    }
    // Truncation: a strict prefix must throw, never read past the end.
    if (iter % 20 == 0 && w.buf.size() > 1) {
      for (size_t cut : {w.buf.size() / 3, w.buf.size() - 1}) {
        if (cut == 0 || cut >= w.buf.size()) continue;
        bool threw = false;
        try {
          Reader tr(w.buf.data(), cut);
          (void)tr.nest(0);
        } catch (const std::exception&) {
          threw = true;
        }
        if (!threw) {
          // A prefix CAN parse if the cut lands exactly on a nest
          // boundary of an empty-vector tail; re-serialize to verify it
          // was a legitimate shorter message, else fail.
          Reader tr(w.buf.data(), cut);
          TensorNest sub = tr.nest(0);
          Writer wv;
          wv.nest(sub);
          if (wv.buf.size() != cut) {
            std::fprintf(stderr, "FAIL truncation accepted at iter %d\n",
                         iter);
            return 1;
          }
        }
        ++truncated_checked;
      }
    }
  }
  std::printf("wire_fuzz OK: 2000 roundtrips, %d truncation checks\n",
              truncated_checked);
  return 0;
}
