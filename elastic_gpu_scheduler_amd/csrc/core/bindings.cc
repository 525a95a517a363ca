// pybind11 bindings for the native scheduler core.
//
// The GIL is released around every verb that can fan out or block on node
// locks, so concurrent HTTP handlers schedule truly in parallel.
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "cluster.h"
#include "node.h"
#include "raters.h"
#include "search.h"
#include "topology.h"
#include "types.h"

namespace py = pybind11;
using namespace egs;

void bind_native_server(py::module_& m);

PYBIND11_MODULE(_core, m) {
  m.doc() = "MI355X-native elastic GPU scheduler core (C++)";

  m.attr("GPU_CORE_EACH_CARD") = kGPUCoreEachCard;
  m.attr("MI355X_MEMORY_BYTES") = kMI355XMemoryBytes;
  m.attr("MI355X_CARDS_PER_NODE") = kMI355XCardsPerNode;
  m.attr("SCORE_MIN") = kScoreMin;
  m.attr("SCORE_MAX") = kScoreMax;

  py::class_<Device>(m, "Device")
      .def(py::init([](int core_total, int core_avail, int64_t mem_total,
                       int64_t mem_avail) {
             Device d;
             d.core_total = core_total;
             d.core_avail = core_avail;
             d.mem_total = mem_total;
             d.mem_avail = mem_avail;
             return d;
           }),
           py::arg("core_total") = kGPUCoreEachCard,
           py::arg("core_avail") = kGPUCoreEachCard,
           py::arg("mem_total") = kMI355XMemoryBytes,
           py::arg("mem_avail") = kMI355XMemoryBytes)
      .def_readwrite("core_total", &Device::core_total)
      .def_readwrite("core_avail", &Device::core_avail)
      .def_readwrite("mem_total", &Device::mem_total)
      .def_readwrite("mem_avail", &Device::mem_avail)
      .def("whole_free", &Device::whole_free)
      .def("schedulable", &Device::schedulable)
      .def("__repr__", [](const Device& d) {
        return "Device(core " + std::to_string(d.core_avail) + "/" +
               std::to_string(d.core_total) + ", mem " + std::to_string(d.mem_avail) +
               "/" + std::to_string(d.mem_total) + ")";
      });

  py::class_<GPUUnit>(m, "GPUUnit")
      .def(py::init([](int gpu_count, int core, int64_t memory) {
             GPUUnit u;
             u.gpu_count = gpu_count;
             u.core = core;
             u.memory = memory;
             return u;
           }),
           py::arg("gpu_count") = 0, py::arg("core") = 0, py::arg("memory") = 0)
      .def_readwrite("gpu_count", &GPUUnit::gpu_count)
      .def_readwrite("core", &GPUUnit::core)
      .def_readwrite("memory", &GPUUnit::memory)
      .def("needs_gpu", &GPUUnit::needs_gpu)
      .def("__repr__", [](const GPUUnit& u) {
        return "GPUUnit(count=" + std::to_string(u.gpu_count) +
               ", core=" + std::to_string(u.core) +
               ", memory=" + std::to_string(u.memory) + ")";
      });

  py::class_<GPUOption>(m, "GPUOption")
      .def(py::init<>())
      .def_readwrite("allocated", &GPUOption::allocated)
      .def_readwrite("score", &GPUOption::score);

  py::class_<Topology>(m, "Topology")
      .def(py::init<>())
      .def(py::init<std::vector<std::vector<int>>>())
      .def("hops", &Topology::hops)
      .def("set_cost", &Topology::set_cost)
      .def("locality", &Topology::locality);

  // Standalone search entry point (used by unit tests and the bench's
  // no-HTTP microbenchmark path).
  m.def(
      "search_placement",
      [](const std::vector<Device>& devices, const GPURequest& req,
         const std::string& policy, uint64_t seed,
         std::vector<std::vector<int>> topo_hops, bool distinct) {
        auto rater = make_rater(policy, seed);
        Topology topo(std::move(topo_hops));
        RateContext ctx;
        ctx.devices = &devices;
        ctx.topo = &topo;
        ctx.salt = seed;
        SearchResult res;
        {
          py::gil_scoped_release rel;
          res = search_placement(devices, req, *rater, ctx, distinct);
        }
        return py::make_tuple(res.feasible, res.option, res.leaves_evaluated);
      },
      py::arg("devices"), py::arg("request"), py::arg("policy") = "binpack",
      py::arg("seed") = 0, py::arg("topology") = std::vector<std::vector<int>>{},
      py::arg("distinct") = false);

  bind_native_server(m);

  py::class_<ClusterState, std::shared_ptr<ClusterState>>(m, "ClusterState")
      .def(py::init<const std::string&, uint64_t, int, double>(),
           py::arg("policy") = "binpack", py::arg("seed") = 0,
           py::arg("threads") = 0, py::arg("topology_weight") = 0.3)
      .def_property_readonly("policy", &ClusterState::policy)
      .def_property_readonly("pool_size", &ClusterState::pool_size)
      .def("add_node", &ClusterState::add_node, py::arg("name"), py::arg("devices"),
           py::arg("topology") = std::vector<std::vector<int>>{},
           py::call_guard<py::gil_scoped_release>())
      .def("has_node", &ClusterState::has_node, py::call_guard<py::gil_scoped_release>())
      .def("remove_node", &ClusterState::remove_node,
           py::call_guard<py::gil_scoped_release>())
      .def("node_names", &ClusterState::node_names,
           py::call_guard<py::gil_scoped_release>())
      .def("assume", &ClusterState::assume, py::arg("nodes"), py::arg("uid"),
           py::arg("request"), py::arg("distinct") = false,
           py::call_guard<py::gil_scoped_release>())
      .def("score", &ClusterState::score, py::arg("nodes"), py::arg("uid"),
           py::arg("request"), py::arg("distinct") = false,
           py::call_guard<py::gil_scoped_release>())
      .def("allocate", &ClusterState::allocate, py::arg("node"), py::arg("uid"),
           py::arg("request"), py::arg("distinct") = false,
           py::call_guard<py::gil_scoped_release>())
      .def("add_pod", &ClusterState::add_pod, py::arg("node"), py::arg("uid"),
           py::arg("request"), py::arg("option"),
           py::call_guard<py::gil_scoped_release>())
      .def("note_pod_node", &ClusterState::note_pod_node, py::arg("uid"),
           py::arg("node"), py::call_guard<py::gil_scoped_release>())
      .def("forget_pod", &ClusterState::forget_pod,
           py::call_guard<py::gil_scoped_release>())
      .def("known_pod", &ClusterState::known_pod,
           py::call_guard<py::gil_scoped_release>())
      .def("feasible_with_victims", &ClusterState::feasible_with_victims,
           py::arg("node"), py::arg("uid"), py::arg("request"),
           py::arg("victims"), py::call_guard<py::gil_scoped_release>())
      .def("node_devices",
           [](ClusterState& cs, const std::string& name) {
             auto alloc = cs.get(name);
             if (!alloc) throw std::runtime_error("unknown node " + name);
             std::vector<Device> snap;
             {
               py::gil_scoped_release rel;
               snap = alloc->snapshot();
             }
             return snap;
           })
      .def("node_assumed_count",
           [](ClusterState& cs, const std::string& name) {
             auto alloc = cs.get(name);
             if (!alloc) throw std::runtime_error("unknown node " + name);
             py::gil_scoped_release rel;
             return alloc->assumed_count();
           })
      .def("node_pod_placements",
           [](ClusterState& cs, const std::string& name) {
             auto alloc = cs.get(name);
             if (!alloc) throw std::runtime_error("unknown node " + name);
             std::vector<std::pair<std::string, std::vector<std::vector<int>>>> out;
             {
               py::gil_scoped_release rel;
               out = alloc->pod_placements();
             }
             py::dict d;
             for (auto& [uid, cards] : out) d[py::str(uid)] = cards;
             return d;
           })
      .def("node_pods", [](ClusterState& cs, const std::string& name) {
        auto alloc = cs.get(name);
        if (!alloc) throw std::runtime_error("unknown node " + name);
        std::vector<std::string> uids;
        {
          py::gil_scoped_release rel;
          uids = alloc->pod_uids();
        }
        return uids;
      });
}
