// bindings.cpp — pybind11 module `_router_core`: the native router core
// (prefix-cache hashing + index, scheduler hot loop, flow-control queues).
// The reference router core is a native Go binary (cmd/epp); here the native
// core is this C++ library driven by the Python orchestration layer, with the
// gfx950 HIP kernels (csrc/hip/) as the batched GPU path for the same ops.
#include <pybind11/pybind11.h>
#include <pybind11/numpy.h>
#include <pybind11/stl.h>

#include "prefix_index.h"
#include "scoring.h"
#include "queues.h"

namespace py = pybind11;
using namespace ldsr;

namespace {

py::array_t<uint64_t> py_hash_tokens(py::array_t<int32_t, py::array::c_style | py::array::forcecast> tokens,
                                     int block_tokens, int64_t max_blocks, uint64_t seed0) {
  if (block_tokens <= 0) throw std::invalid_argument("block_tokens must be > 0");
  int64_t n_tokens = tokens.size();
  int64_t n_blocks = std::min<int64_t>(n_tokens / block_tokens, max_blocks);
  py::array_t<uint64_t> out(n_blocks);
  if (n_blocks > 0) {
    hash_tokens(tokens.data(), n_tokens, block_tokens, max_blocks, seed0,
                out.mutable_data());
  }
  return out;
}

uint64_t py_model_seed(const std::string& model, const std::string& salt) {
  return model_seed(model.data(), model.size(), salt.data(), salt.size());
}

uint64_t py_xxh64(py::bytes data, uint64_t seed) {
  char* buf; Py_ssize_t len;
  if (PyBytes_AsStringAndSize(data.ptr(), &buf, &len) != 0) throw py::error_already_set();
  return xxh64(buf, (size_t)len, seed);
}

}  // namespace

PYBIND11_MODULE(_router_core, m) {
  m.doc() = "MI355X-native router core (C++): prefix index, scheduler hot loop, flow queues";

  m.def("xxh64", &py_xxh64, py::arg("data"), py::arg("seed") = 0);
  m.def("model_seed", &py_model_seed, py::arg("model"), py::arg("salt") = "");
  m.def("hash_tokens", &py_hash_tokens, py::arg("tokens"), py::arg("block_tokens"),
        py::arg("max_blocks"), py::arg("seed0"));

  py::class_<PrefixIndex>(m, "PrefixIndex")
      .def(py::init<int64_t>(), py::arg("lru_capacity_per_endpoint"))
      .def("add",
           [](PrefixIndex& self, int endpoint,
              py::array_t<uint64_t, py::array::c_style | py::array::forcecast> hashes) {
             return self.add(endpoint, hashes.data(), hashes.size());
           },
           py::arg("endpoint"), py::arg("hashes"),
           py::call_guard<py::gil_scoped_release>())
      .def("match_longest",
           [](const PrefixIndex& self,
              py::array_t<uint64_t, py::array::c_style | py::array::forcecast> hashes,
              int num_endpoints) {
             py::array_t<int32_t> out(num_endpoints);
             {
               py::gil_scoped_release rel;
               self.match_longest(hashes.data(), hashes.size(), num_endpoints,
                                  out.mutable_data());
             }
             return out;
           },
           py::arg("hashes"), py::arg("num_endpoints"))
      .def("remove_endpoint", &PrefixIndex::remove_endpoint)
      .def("size", &PrefixIndex::size)
      .def("endpoint_size", &PrefixIndex::endpoint_size)
      .def("set_capacity", &PrefixIndex::set_capacity)
      .def("capacity", &PrefixIndex::capacity);

  py::class_<ProfileRunner>(m, "ProfileRunner")
      .def(py::init<uint64_t>(), py::arg("seed"))
      .def("run",
           [](ProfileRunner& self,
              py::array_t<uint8_t, py::array::c_style | py::array::forcecast> roles,
              py::array_t<float, py::array::c_style | py::array::forcecast> queue_depth,
              py::array_t<float, py::array::c_style | py::array::forcecast> running,
              py::array_t<float, py::array::c_style | py::array::forcecast> kv_usage,
              py::array_t<float, py::array::c_style | py::array::forcecast> inflight_tokens,
              py::array_t<float, py::array::c_style | py::array::forcecast> active_requests,
              int role_filter,
              py::object candidate_mask,  // None or uint8 array
              const std::vector<std::tuple<int, float, float, float>>& scorers,
              py::object match_blocks,    // None or int32 array
              int total_blocks,
              py::object extra,           // None or float32 array
              int picker, int max_endpoints) {
             Snapshot s;
             s.n = (int)roles.size();
             s.roles.assign(roles.data(), roles.data() + s.n);
             s.queue_depth.assign(queue_depth.data(), queue_depth.data() + s.n);
             s.running.assign(running.data(), running.data() + s.n);
             s.kv_usage.assign(kv_usage.data(), kv_usage.data() + s.n);
             s.inflight_tokens.assign(inflight_tokens.data(), inflight_tokens.data() + s.n);
             s.active_requests.assign(active_requests.data(), active_requests.data() + s.n);

             std::vector<ScorerSpec> specs;
             for (auto& [k, w, p, p2] : scorers) specs.push_back({k, w, p, p2});

             py::array_t<uint8_t> cmask_arr;
             const uint8_t* cmask = nullptr;
             if (!candidate_mask.is_none()) {
               cmask_arr = py::cast<py::array_t<uint8_t, py::array::c_style | py::array::forcecast>>(candidate_mask);
               cmask = cmask_arr.data();
             }
             py::array_t<int32_t> mb_arr;
             const int32_t* mb = nullptr;
             if (!match_blocks.is_none()) {
               mb_arr = py::cast<py::array_t<int32_t, py::array::c_style | py::array::forcecast>>(match_blocks);
               mb = mb_arr.data();
             }
             py::array_t<float> ex_arr;
             const float* ex = nullptr;
             if (!extra.is_none()) {
               ex_arr = py::cast<py::array_t<float, py::array::c_style | py::array::forcecast>>(extra);
               ex = ex_arr.data();
             }
             ProfileResult r = self.run(s, (uint8_t)role_filter, cmask, specs, mb,
                                        total_blocks, ex, picker, max_endpoints);
             py::array_t<int32_t> picks((int64_t)r.picks.size());
             std::copy(r.picks.begin(), r.picks.end(), picks.mutable_data());
             py::array_t<float> scores((int64_t)r.scores.size());
             std::copy(r.scores.begin(), r.scores.end(), scores.mutable_data());
             return py::make_tuple(picks, scores);
           },
           py::arg("roles"), py::arg("queue_depth"), py::arg("running"),
           py::arg("kv_usage"), py::arg("inflight_tokens"), py::arg("active_requests"),
           py::arg("role_filter"), py::arg("candidate_mask"), py::arg("scorers"),
           py::arg("match_blocks"), py::arg("total_blocks"), py::arg("extra"),
           py::arg("picker"), py::arg("max_endpoints"));

  py::class_<ListQueue>(m, "ListQueue")
      .def(py::init<>())
      .def("push", [](ListQueue& q, uint64_t id, double key, int64_t b) { q.push(id, key, b); },
           py::arg("id"), py::arg("key") = 0.0, py::arg("bytes") = 0)
      .def("pop", [](ListQueue& q) -> py::object {
        uint64_t id; int64_t b;
        if (!q.pop(&id, &b)) return py::none();
        return py::make_tuple(id, b);
      })
      .def("peek", [](ListQueue& q) -> py::object {
        uint64_t id;
        if (!q.peek(&id)) return py::none();
        return py::cast(id);
      })
      .def("peek_tail", [](ListQueue& q) -> py::object {
        uint64_t id;
        if (!q.peek_tail(&id)) return py::none();
        return py::cast(id);
      })
      .def("remove", [](ListQueue& q, uint64_t id) -> py::object {
        int64_t b;
        if (!q.remove(id, &b)) return py::none();
        return py::cast(b);
      })
      .def("__len__", [](const ListQueue& q) { return q.len(); })
      .def_property_readonly("bytes", &ListQueue::bytes);

  py::class_<MaxMinHeap>(m, "MaxMinHeap")
      .def(py::init<>())
      .def("push", [](MaxMinHeap& q, uint64_t id, double key, int64_t b) { q.push(id, key, b); },
           py::arg("id"), py::arg("key"), py::arg("bytes") = 0)
      .def("pop", [](MaxMinHeap& q) -> py::object {
        uint64_t id; int64_t b;
        if (!q.pop(&id, &b)) return py::none();
        return py::make_tuple(id, b);
      })
      .def("peek", [](MaxMinHeap& q) -> py::object {
        uint64_t id;
        if (!q.peek(&id)) return py::none();
        return py::cast(id);
      })
      .def("peek_max", [](MaxMinHeap& q) -> py::object {
        uint64_t id;
        if (!q.peek_max(&id)) return py::none();
        return py::cast(id);
      })
      .def("remove", [](MaxMinHeap& q, uint64_t id) -> py::object {
        int64_t b;
        if (!q.remove(id, &b)) return py::none();
        return py::cast(b);
      })
      .def("__len__", [](const MaxMinHeap& q) { return q.len(); })
      .def_property_readonly("bytes", &MaxMinHeap::bytes);
}
