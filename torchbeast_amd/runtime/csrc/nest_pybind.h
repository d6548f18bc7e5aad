// pybind11 type caster: Python tuples/lists/dicts <-> tbruntime::Nest<T>.
//
// Capability parity with the reference's caster (ref:
// nest/nest/nest_pybind.h:26-96): any nesting of tuple/list/dict crossing a
// binding boundary becomes a Nest<T> (leaves cast with pybind's caster for
// T, e.g. torch::Tensor), and Nests returned to Python come back as
// tuples / lists / dicts with the original container kinds collapsed to
// tuple for vectors (matching the reference behavior).

#pragma once

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "nest.h"

namespace pybind11 {
namespace detail {

template <typename T>
struct type_caster<tbruntime::Nest<T>> {
  using NestT = tbruntime::Nest<T>;
  PYBIND11_TYPE_CASTER(NestT, _("Nest"));

  bool load(handle src, bool convert) {
    try {
      value = load_nest(src, convert);
    } catch (const cast_error&) {
      return false;
    }
    return true;
  }

  static NestT load_nest(handle src, bool convert) {
    if (isinstance<tuple>(src) || isinstance<list>(src)) {
      typename NestT::vector_t vec;
      for (handle item : reinterpret_borrow<sequence>(src)) {
        vec.push_back(load_nest(item, convert));
      }
      return NestT(std::move(vec));
    }
    if (isinstance<dict>(src)) {
      typename NestT::map_t map;
      for (auto item : reinterpret_borrow<dict>(src)) {
        map.emplace(item.first.template cast<std::string>(),
                    load_nest(item.second, convert));
      }
      return NestT(std::move(map));
    }
    return NestT(src.template cast<T>());
  }

  static handle cast_nest(const NestT& src, return_value_policy policy,
                          handle parent) {
    if (src.is_leaf()) {
      return make_caster<T>::cast(src.leaf(), policy, parent);
    }
    if (src.is_vector()) {
      const auto& vec = src.vector();
      tuple out(vec.size());
      for (size_t i = 0; i < vec.size(); ++i) {
        out[i] = reinterpret_steal<object>(cast_nest(vec[i], policy, parent));
      }
      return out.release();
    }
    dict out;
    for (const auto& kv : src.map_value()) {
      out[str(kv.first)] =
          reinterpret_steal<object>(cast_nest(kv.second, policy, parent));
    }
    return out.release();
  }

  static handle cast(const NestT& src, return_value_policy policy,
                     handle parent) {
    return cast_nest(src, policy, parent);
  }
};

}  // namespace detail
}  // namespace pybind11
