// nest: a tree of leaves (T | vector | string-keyed map) with structural ops.
//
// MI355X-native reimplementation of the capability provided by the
// reference's header-only nest library (ref: nest/nest/nest.h). Fresh
// design: leaves are visited in deterministic order (vector order; map in
// sorted key order, since std::map is ordered), and the batching helpers the
// runtime needs (flatten / pack_as / map / map2 / leaf-wise columnar apply)
// are the whole surface — no iterator machinery.

#pragma once

#include <functional>
#include <map>
#include <stdexcept>
#include <string>
#include <utility>
#include <variant>
#include <vector>

namespace tbruntime {

class NestError : public std::runtime_error {
 public:
  using std::runtime_error::runtime_error;
};

template <typename T>
class Nest {
 public:
  using vector_t = std::vector<Nest>;
  using map_t = std::map<std::string, Nest>;
  using value_t = std::variant<T, vector_t, map_t>;

  Nest() : value(vector_t()) {}
  /* implicit */ Nest(T leaf) : value(std::move(leaf)) {}
  /* implicit */ Nest(vector_t v) : value(std::move(v)) {}
  /* implicit */ Nest(map_t m) : value(std::move(m)) {}

  value_t value;

  bool is_leaf() const { return std::holds_alternative<T>(value); }
  bool is_vector() const { return std::holds_alternative<vector_t>(value); }
  bool is_map() const { return std::holds_alternative<map_t>(value); }

  T& leaf() {
    if (!is_leaf()) throw NestError("not a leaf");
    return std::get<T>(value);
  }
  const T& leaf() const {
    if (!is_leaf()) throw NestError("not a leaf");
    return std::get<T>(value);
  }
  vector_t& vector() { return std::get<vector_t>(value); }
  const vector_t& vector() const { return std::get<vector_t>(value); }
  map_t& map_value() { return std::get<map_t>(value); }
  const map_t& map_value() const { return std::get<map_t>(value); }

  bool empty() const { return leaf_count() == 0; }

  int64_t leaf_count() const {
    int64_t n = 0;
    for_each([&n](const T&) { ++n; });
    return n;
  }

  const T& front() const {
    const T* result = nullptr;
    for_each([&result](const T& t) {
      if (result == nullptr) result = &t;
    });
    if (result == nullptr) throw NestError("front() of empty nest");
    return *result;
  }

  template <typename F>
  void for_each(F&& f) const {
    if (is_leaf()) {
      f(std::get<T>(value));
    } else if (is_vector()) {
      for (const Nest& n : std::get<vector_t>(value)) n.for_each(f);
    } else {
      for (const auto& kv : std::get<map_t>(value)) kv.second.for_each(f);
    }
  }

  template <typename F>
  void for_each_mut(F&& f) {
    if (is_leaf()) {
      f(std::get<T>(value));
    } else if (is_vector()) {
      for (Nest& n : std::get<vector_t>(value)) n.for_each_mut(f);
    } else {
      for (auto& kv : std::get<map_t>(value)) kv.second.for_each_mut(f);
    }
  }

  std::vector<T> flatten() const {
    std::vector<T> out;
    for_each([&out](const T& t) { out.push_back(t); });
    return out;
  }

  template <typename F>
  auto map(F&& f) const -> Nest<decltype(f(std::declval<const T&>()))> {
    using U = decltype(f(std::declval<const T&>()));
    if (is_leaf()) return Nest<U>(f(std::get<T>(value)));
    if (is_vector()) {
      typename Nest<U>::vector_t out;
      out.reserve(std::get<vector_t>(value).size());
      for (const Nest& n : std::get<vector_t>(value)) out.push_back(n.map(f));
      return Nest<U>(std::move(out));
    }
    typename Nest<U>::map_t out;
    for (const auto& kv : std::get<map_t>(value)) {
      out.emplace(kv.first, kv.second.map(f));
    }
    return Nest<U>(std::move(out));
  }

  // Structure-matching binary map.
  template <typename U, typename F>
  auto map2(const Nest<U>& other, F&& f) const
      -> Nest<decltype(f(std::declval<const T&>(), std::declval<const U&>()))> {
    using R = decltype(f(std::declval<const T&>(), std::declval<const U&>()));
    if (is_leaf() && other.is_leaf()) {
      return Nest<R>(f(std::get<T>(value), other.leaf()));
    }
    if (is_vector() && other.is_vector()) {
      const auto& a = std::get<vector_t>(value);
      const auto& b = other.vector();
      if (a.size() != b.size()) throw NestError("map2: size mismatch");
      typename Nest<R>::vector_t out;
      out.reserve(a.size());
      for (size_t i = 0; i < a.size(); ++i) out.push_back(a[i].map2(b[i], f));
      return Nest<R>(std::move(out));
    }
    if (is_map() && other.is_map()) {
      const auto& a = std::get<map_t>(value);
      const auto& b = other.map_value();
      if (a.size() != b.size()) throw NestError("map2: size mismatch");
      typename Nest<R>::map_t out;
      auto it = b.begin();
      for (const auto& kv : a) {
        if (kv.first != it->first) throw NestError("map2: key mismatch");
        out.emplace(kv.first, kv.second.map2(it->second, f));
        ++it;
      }
      return Nest<R>(std::move(out));
    }
    throw NestError("map2: structure mismatch");
  }

  // Rebuild this structure with leaves taken in order from [first, last).
  template <typename Iter>
  Nest pack_as(Iter& first, const Iter& last) const {
    if (is_leaf()) {
      if (first == last) throw NestError("pack_as: too few leaves");
      return Nest(*first++);
    }
    if (is_vector()) {
      vector_t out;
      out.reserve(std::get<vector_t>(value).size());
      for (const Nest& n : std::get<vector_t>(value)) {
        out.push_back(n.pack_as(first, last));
      }
      return Nest(std::move(out));
    }
    map_t out;
    for (const auto& kv : std::get<map_t>(value)) {
      out.emplace(kv.first, kv.second.pack_as(first, last));
    }
    return Nest(std::move(out));
  }

  Nest pack_from(std::vector<T> leaves) const {
    auto it = leaves.begin();
    Nest out = pack_as(it, leaves.end());
    if (it != leaves.end()) throw NestError("pack_as: too many leaves");
    return out;
  }

  bool same_structure(const Nest& other) const {
    if (value.index() != other.value.index()) return false;
    if (is_leaf()) return true;
    if (is_vector()) {
      const auto& a = std::get<vector_t>(value);
      const auto& b = other.vector();
      if (a.size() != b.size()) return false;
      for (size_t i = 0; i < a.size(); ++i) {
        if (!a[i].same_structure(b[i])) return false;
      }
      return true;
    }
    const auto& a = std::get<map_t>(value);
    const auto& b = other.map_value();
    if (a.size() != b.size()) return false;
    auto it = b.begin();
    for (const auto& kv : a) {
      if (kv.first != it->first || !kv.second.same_structure(it->second)) {
        return false;
      }
      ++it;
    }
    return true;
  }

  // Columnar apply over N same-structure nests: f receives, for each leaf
  // position, the vector of the N leaves at that position. This is the
  // primitive under batch assembly (cat along a dim) without recursive zip.
  template <typename F>
  static Nest apply_columns(const std::vector<const Nest*>& nests, F&& f) {
    if (nests.empty()) throw NestError("apply_columns: no nests");
    std::vector<std::vector<T>> flats;
    flats.reserve(nests.size());
    size_t n_leaves = 0;
    for (const Nest* n : nests) {
      flats.push_back(n->flatten());
      if (flats.size() == 1) {
        n_leaves = flats[0].size();
      } else if (flats.back().size() != n_leaves) {
        throw NestError("apply_columns: leaf-count mismatch");
      }
    }
    std::vector<T> out;
    out.reserve(n_leaves);
    std::vector<T> column(nests.size());
    for (size_t i = 0; i < n_leaves; ++i) {
      for (size_t j = 0; j < flats.size(); ++j) column[j] = flats[j][i];
      out.push_back(f(column));
    }
    return nests[0]->pack_from(std::move(out));
  }
};

}  // namespace tbruntime
