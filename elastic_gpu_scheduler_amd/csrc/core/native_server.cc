// Python bindings for the native extender HTTP server.
//
// The server shares the scheduler's ClusterState: filter/priorities are
// answered entirely in C++ (no GIL), everything else — bind (apiserver
// writes), status, version, metrics, debug — is delegated to a Python
// fallback callable `(method, path, body-bytes) -> (status, content_type,
// body-bytes)`.
#include <pybind11/functional.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <memory>

#include "../httpd/extender.h"
#include "../httpd/http_server.h"
#include "cluster.h"

namespace py = pybind11;
using namespace egs;

namespace {

BareUnit parse_bare_unit(const std::string& s) {
  if (s == "bytes") return BareUnit::Bytes;
  if (s == "GiB") return BareUnit::GiB;
  if (s == "MiB") return BareUnit::MiB;
  return BareUnit::Auto;
}

class NativeExtenderServer {
 public:
  NativeExtenderServer(std::shared_ptr<ClusterState> state,
                       const std::string& bare_unit, const std::string& host,
                       int port, py::object fallback,
                       const std::string& tls_cert = "",
                       const std::string& tls_key = "",
                       const std::string& tls_client_ca = "")
      : core_(std::make_shared<ExtenderCore>(state, parse_bare_unit(bare_unit))),
        fallback_(std::move(fallback)) {
    auto core = core_;
    py::object fb = fallback_;
    egshttp::Handler handler = [core, fb](const egshttp::Request& req,
                                          egshttp::Response* resp) {
      if (req.method == "POST" && req.path == "/scheduler/filter") {
        if (core->filter(req.body, &resp->body) == HandleStatus::Handled) return;
      } else if (req.method == "POST" && req.path == "/scheduler/priorities") {
        if (core->priorities(req.body, &resp->body) == HandleStatus::Handled)
          return;
      }
      // Python fallback (bind, status, metrics, errors, cold nodes...).
      core->counters.fallback.fetch_add(1, std::memory_order_relaxed);
      py::gil_scoped_acquire gil;
      try {
        py::tuple out = fb(req.method, req.path, py::bytes(req.body));
        resp->status = out[0].cast<int>();
        resp->content_type = out[1].cast<std::string>();
        resp->body = out[2].cast<std::string>();
      } catch (const std::exception& e) {
        core->counters.errors.fetch_add(1, std::memory_order_relaxed);
        resp->status = 500;
        resp->body = std::string("{\"error\": \"fallback failed: ") + e.what() +
                     "\"}";
      }
    };
    egshttp::TlsConfig tls;
    tls.cert_file = tls_cert;
    tls.key_file = tls_key;
    tls.client_ca_file = tls_client_ca;
    server_ = std::make_unique<egshttp::HttpServer>(host, port, handler,
                                                    /*max_connections=*/512,
                                                    std::move(tls));
  }

  bool tls_enabled() const { return server_->tls_enabled(); }

  ~NativeExtenderServer() { stop(); }

  void start() { server_->start(); }
  void stop() {
    if (server_) server_->stop();
  }
  int port() const { return server_->port(); }

  void note_filter(const std::string& uid) { core_->tracker.note(uid); }
  double pop_filter_seconds(const std::string& uid) {
    return core_->tracker.pop(uid);
  }

  py::dict stats() const {
    py::dict d;
    d["filter_native"] = core_->counters.filter_native.load();
    d["priorities_native"] = core_->counters.priorities_native.load();
    d["fallback"] = core_->counters.fallback.load();
    d["errors"] = core_->counters.errors.load();
    return d;
  }

  // Per-verb latency histograms computed IN the GIL-free fast path
  // (log2 us buckets): verb -> {count, sum_us, buckets: [(le_us, n), ...]}.
  py::dict latency_histograms() const {
    py::dict out;
    auto dump = [](const LatencyHist& h) {
      py::dict d;
      d["count"] = h.count.load(std::memory_order_relaxed);
      d["sum_us"] = h.sum_us.load(std::memory_order_relaxed);
      py::list buckets;
      for (int b = 0; b < LatencyHist::kBuckets; ++b) {
        uint64_t le_us = 1ULL << b;
        buckets.append(py::make_tuple(
            le_us, h.buckets[b].load(std::memory_order_relaxed)));
      }
      d["buckets"] = std::move(buckets);
      return d;
    };
    out["filter"] = dump(core_->filter_hist);
    out["priorities"] = dump(core_->priorities_hist);
    return out;
  }

 private:
  std::shared_ptr<ExtenderCore> core_;
  py::object fallback_;
  std::unique_ptr<egshttp::HttpServer> server_;
};

}  // namespace

void bind_native_server(py::module_& m) {
  py::class_<NativeExtenderServer>(m, "NativeExtenderServer")
      .def(py::init<std::shared_ptr<ClusterState>, const std::string&,
                    const std::string&, int, py::object, const std::string&,
                    const std::string&, const std::string&>(),
           py::arg("state"), py::arg("bare_unit") = "auto",
           py::arg("host") = "0.0.0.0", py::arg("port") = 0,
           py::arg("fallback"), py::arg("tls_cert") = "",
           py::arg("tls_key") = "", py::arg("tls_client_ca") = "")
      .def_property_readonly("tls_enabled",
                             &NativeExtenderServer::tls_enabled)
      .def("start", &NativeExtenderServer::start,
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &NativeExtenderServer::stop,
           py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("port", &NativeExtenderServer::port)
      .def("note_filter", &NativeExtenderServer::note_filter)
      .def("pop_filter_seconds", &NativeExtenderServer::pop_filter_seconds)
      .def("stats", &NativeExtenderServer::stats)
      .def("latency_histograms", &NativeExtenderServer::latency_histograms);

  // JSON codec round-trip (exposed for tests of the native parser).
  m.def("json_roundtrip", [](const std::string& s) {
    return egsjson::dump(egsjson::parse(s));
  });

  // Bare-number "auto" threshold: Python utils/quantity.py is the single
  // source of truth; package import pushes it here so both request paths
  // (C++ fast path, Python fallback) always agree.
  m.def("set_bare_auto_gib_threshold", [](int64_t v) {
    bare_auto_gib_threshold().store(v, std::memory_order_relaxed);
  });
  m.def("get_bare_auto_gib_threshold", []() {
    return bare_auto_gib_threshold().load(std::memory_order_relaxed);
  });

  // De-herded protocol score (exposed for the Python<->C++ parity test).
  m.def("jittered_int_score", &jittered_int_score, py::arg("score"),
        py::arg("uid"), py::arg("node"));
}
