// _tbruntime: Python bindings for the MI355X-native actor-learner runtime.
//
// Python surface (capability parity with the reference's `libtorchbeast._C`
// module, ref: src/cc/libtorchbeast.cc + actorpool.cc:566-632 +
// rpcenv.cc:215-221): BatchingQueue, DynamicBatcher (+Batch), ActorPool,
// Server, exceptions ClosedBatchingQueue / AsyncError / NestError — plus the
// nest structural ops (map/map_many/map_many2/flatten/pack_as/front) that
// the reference ships as a separate `nest` extension.

#include <pybind11/functional.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>
#include <torch/extension.h>

#include "actor_pool.h"
#include "env_server.h"
#include "inference_runner.h"
#include "nest.h"
#include "nest_pybind.h"
#include "queues.h"

namespace py = pybind11;
using namespace tbruntime;

namespace {

// ---------------------------------------------------------------------------
// nest ops over arbitrary Python object trees.
// ---------------------------------------------------------------------------

bool is_container(const py::handle& h) {
  return py::isinstance<py::tuple>(h) || py::isinstance<py::list>(h) ||
         py::isinstance<py::dict>(h);
}

py::object nest_map(const py::function& f, const py::object& n) {
  if (py::isinstance<py::dict>(n)) {
    py::dict out;
    for (auto item : n.cast<py::dict>()) {
      out[item.first] = nest_map(f, py::reinterpret_borrow<py::object>(item.second));
    }
    return out;
  }
  if (py::isinstance<py::tuple>(n) || py::isinstance<py::list>(n)) {
    py::list out;
    for (auto item : n.cast<py::sequence>()) {
      out.append(nest_map(f, py::reinterpret_borrow<py::object>(item)));
    }
    if (py::isinstance<py::tuple>(n)) return py::tuple(out);
    return out;
  }
  return f(n);
}

void nest_flatten_into(const py::handle& n, py::list& out) {
  if (py::isinstance<py::dict>(n)) {
    py::list keys;
    for (auto item : n.cast<py::dict>()) keys.append(item.first);
    keys.attr("sort")();
    for (auto k : keys) nest_flatten_into(n.cast<py::dict>()[k], out);
  } else if (py::isinstance<py::tuple>(n) || py::isinstance<py::list>(n)) {
    for (auto item : n.cast<py::sequence>()) nest_flatten_into(item, out);
  } else {
    out.append(n);
  }
}

py::list nest_flatten(const py::object& n) {
  py::list out;
  nest_flatten_into(n, out);
  return out;
}

py::object nest_pack_as_impl(const py::handle& n, const py::list& flat,
                             size_t& pos) {
  if (py::isinstance<py::dict>(n)) {
    py::list keys;
    for (auto item : n.cast<py::dict>()) keys.append(item.first);
    keys.attr("sort")();
    py::dict filled;
    for (auto k : keys) {
      filled[k] = nest_pack_as_impl(n.cast<py::dict>()[k], flat, pos);
    }
    // Restore original key order.
    py::dict out;
    for (auto item : n.cast<py::dict>()) out[item.first] = filled[item.first];
    return out;
  }
  if (py::isinstance<py::tuple>(n) || py::isinstance<py::list>(n)) {
    py::list out;
    for (auto item : n.cast<py::sequence>()) {
      out.append(nest_pack_as_impl(item, flat, pos));
    }
    if (py::isinstance<py::tuple>(n)) return py::tuple(out);
    return out;
  }
  if (pos >= flat.size()) throw NestError("Too few elements to pack");
  return py::reinterpret_borrow<py::object>(flat[pos++]);
}

py::object nest_pack_as(const py::object& n, const py::list& flat) {
  size_t pos = 0;
  py::object out = nest_pack_as_impl(n, flat, pos);
  if (pos != flat.size()) throw NestError("Too many elements to pack");
  return out;
}

py::object nest_map_many(const py::function& f, const py::args& nests) {
  if (nests.size() == 0) throw NestError("map_many needs at least one nest");
  py::object first = nests[0];
  std::vector<py::list> flats;
  size_t n_leaves = 0;
  for (size_t i = 0; i < nests.size(); ++i) {
    flats.push_back(nest_flatten(nests[i]));
    if (i == 0) {
      n_leaves = flats[0].size();
    } else if (flats.back().size() != n_leaves) {
      throw NestError("nests don't match");
    }
  }
  py::list mapped;
  for (size_t leaf = 0; leaf < n_leaves; ++leaf) {
    py::list column;
    for (auto& fl : flats) column.append(fl[leaf]);
    mapped.append(f(column));
  }
  return nest_pack_as(first, mapped);
}

py::object nest_front(const py::object& n) {
  py::list flat = nest_flatten(n);
  if (flat.size() == 0) throw NestError("front() of empty nest");
  return py::reinterpret_borrow<py::object>(flat[0]);
}

}  // namespace

PYBIND11_MODULE(_tbruntime, m) {
  m.doc() = "MI355X-native torchbeast runtime (queues, actor pool, env server)";

  py::register_exception<ClosedQueue>(m, "ClosedBatchingQueue");
  py::register_exception<AsyncError>(m, "AsyncError");
  py::register_exception<NestError>(m, "NestError", PyExc_ValueError);

  // ---- nest ops ----
  m.def("map", &nest_map, py::arg("function"), py::arg("nest"));
  m.def("map_many", &nest_map_many, py::arg("function"));
  m.def(
      "map_many2",
      [](const py::function& f, const py::object& a, const py::object& b) {
        return nest_map_many(
            py::cpp_function([&f](const py::list& column) {
              return f(column[0], column[1]);
            }),
            py::make_tuple(a, b));
      },
      py::arg("function"), py::arg("nest1"), py::arg("nest2"));
  m.def("flatten", &nest_flatten, py::arg("nest"));
  m.def("pack_as", &nest_pack_as, py::arg("nest"), py::arg("sequence"));
  m.def("front", &nest_front, py::arg("nest"));

  // ---- BatchingQueue ----
  py::class_<BatchingQueue, std::shared_ptr<BatchingQueue>>(m, "BatchingQueue")
      .def(py::init<int64_t, std::optional<int64_t>, std::optional<int64_t>,
                    std::optional<int64_t>, bool, std::optional<int64_t>,
                    std::optional<std::string>>(),
           py::arg("batch_dim") = 0,
           py::arg("minimum_batch_size") = std::nullopt,
           py::arg("maximum_batch_size") = std::nullopt,
           py::arg("timeout_ms") = std::nullopt,
           py::arg("check_inputs") = true,
           py::arg("maximum_queue_size") = std::nullopt,
           py::arg("output_device") = std::nullopt)
      .def_property_readonly("batch_dim", &BatchingQueue::batch_dim)
      .def("enqueue", &BatchingQueue::enqueue, py::arg("nest"),
           py::call_guard<py::gil_scoped_release>())
      .def("size", &BatchingQueue::size)
      .def("stats", &BatchingQueue::stats)
      .def("reset_stats", &BatchingQueue::reset_stats)
      .def("close", &BatchingQueue::close,
           py::call_guard<py::gil_scoped_release>())
      .def("is_closed", &BatchingQueue::is_closed)
      .def("__iter__", [](py::object self) { return self; })
      .def("__next__", [](BatchingQueue& q) {
        std::optional<TensorNest> batch;
        {
          py::gil_scoped_release release;
          try {
            batch = q.dequeue_many().first;
          } catch (const ClosedQueue&) {
          }
        }
        if (!batch) throw py::stop_iteration();
        return *batch;
      });

  // ---- DynamicBatcher ----
  auto batcher = py::class_<DynamicBatcher, std::shared_ptr<DynamicBatcher>>(
      m, "DynamicBatcher");

  py::class_<DynamicBatcher::Batch, std::shared_ptr<DynamicBatcher::Batch>>(
      batcher, "Batch")
      .def("get_inputs", &DynamicBatcher::Batch::get_inputs)
      .def("set_outputs", &DynamicBatcher::Batch::set_outputs,
           py::arg("outputs"))
      .def("size", &DynamicBatcher::Batch::size);

  batcher
      .def(py::init<int64_t, std::optional<int64_t>, std::optional<int64_t>,
                    std::optional<int64_t>, bool>(),
           py::arg("batch_dim") = 0,
           py::arg("minimum_batch_size") = std::nullopt,
           py::arg("maximum_batch_size") = std::nullopt,
           py::arg("timeout_ms") = std::nullopt,
           py::arg("check_outputs") = true)
      .def("compute", &DynamicBatcher::compute, py::arg("inputs"),
           py::call_guard<py::gil_scoped_release>())
      .def("size", &DynamicBatcher::size)
      .def("stats", &DynamicBatcher::stats)
      .def("reset_stats", &DynamicBatcher::reset_stats)
      .def("close", &DynamicBatcher::close,
           py::call_guard<py::gil_scoped_release>())
      .def("is_closed", &DynamicBatcher::is_closed)
      .def("__iter__", [](py::object self) { return self; })
      .def("__next__", [](DynamicBatcher& b) {
        std::shared_ptr<DynamicBatcher::Batch> batch;
        {
          py::gil_scoped_release release;
          try {
            batch = b.get_batch();
          } catch (const ClosedQueue&) {
          }
        }
        if (!batch) throw py::stop_iteration();
        return batch;
      });

  // ---- ActorPool ----
  py::class_<ActorPool, std::shared_ptr<ActorPool>>(m, "ActorPool")
      .def(py::init<int64_t, std::shared_ptr<BatchingQueue>,
                    std::shared_ptr<DynamicBatcher>, std::vector<std::string>,
                    TensorNest, int64_t, bool, int64_t, int64_t>(),
           py::arg("unroll_length"), py::arg("learner_queue"),
           py::arg("inference_batcher"), py::arg("env_server_addresses"),
           py::arg("initial_agent_state"), py::arg("seed_base") = 0,
           py::arg("use_obs_slab") = false,
           py::arg("rollout_budget_mb") = 0,
           py::arg("envs_per_thread") = 1)
      .def("run", &ActorPool::run, py::call_guard<py::gil_scoped_release>())
      .def("obs_slab", &ActorPool::obs_slab,
           py::call_guard<py::gil_scoped_release>())
      .def("count", &ActorPool::count);

  // ---- InferenceRunner ----
  py::class_<InferenceRunner, std::shared_ptr<InferenceRunner>>(
      m, "InferenceRunner")
      .def(py::init<std::shared_ptr<DynamicBatcher>,
                    std::vector<torch::Tensor>, int64_t, bool, std::string>(),
           py::arg("inference_batcher"), py::arg("weights"),
           py::arg("num_lstm_layers") = 0, py::arg("greedy") = false,
           py::arg("model_type") = "shallow")
      .def("start", &InferenceRunner::start, py::arg("num_threads") = 2,
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &InferenceRunner::stop,
           py::call_guard<py::gil_scoped_release>())
      .def("batches", &InferenceRunner::batches)
      .def("steps", &InferenceRunner::steps)
      .def("set_obs_slab", &InferenceRunner::set_obs_slab)
      .def("mark_weights_dirty", &InferenceRunner::mark_weights_dirty);

  // ---- EnvServer ----
  py::class_<EnvServer, std::shared_ptr<EnvServer>>(m, "Server")
      .def(py::init<py::object, std::string>(), py::arg("env_init"),
           py::arg("address"))
      .def("run", &EnvServer::run, py::call_guard<py::gil_scoped_release>())
      .def("start", &EnvServer::start)
      .def("stop", &EnvServer::stop,
           py::call_guard<py::gil_scoped_release>());
}
