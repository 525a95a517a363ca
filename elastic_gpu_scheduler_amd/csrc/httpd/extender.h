// Native extender handlers: the filter / priorities hot path in pure C++.
//
// Mirrors the Python handlers (server/app.py) and codecs (k8s/objects.py,
// utils/quantity.py) exactly — same wire JSON, same request semantics — but
// runs without the GIL: JSON parse -> GPURequest extraction -> ClusterState
// verbs -> JSON response. Requests the C++ path cannot serve (unknown nodes
// needing an apiserver fetch, non-POST routes, bind) return NeedFallback and
// are handled by the Python app.
#pragma once

#include <algorithm>
#include <atomic>
#include <chrono>
#include <cmath>
#include <cstdint>
#include <mutex>
#include <string>
#include <unordered_map>
#include <vector>

#include "../core/cluster.h"
#include "../core/types.h"
#include "json.h"

namespace egs {

// ---- quantity parsing (ports utils/quantity.py semantics) ----

inline bool parse_quantity_ll(const std::string& s, int64_t* out,
                              bool* had_suffix) {
  // number [suffix]; suffixes: Ki Mi Gi Ti Pi Ei k M G T P E m
  size_t i = 0;
  while (i < s.size() && (isspace(static_cast<unsigned char>(s[i])))) ++i;
  size_t start = i;
  while (i < s.size() &&
         (isdigit(static_cast<unsigned char>(s[i])) || s[i] == '.' ||
          s[i] == '+' || s[i] == '-' || s[i] == 'e' || s[i] == 'E')) {
    // 'E' is also a decimal suffix; only treat as exponent if followed by
    // digit/sign AND preceded by a digit start — keep it simple: stop at 'E'
    // unless next char is a digit or sign and we've already consumed digits.
    if (s[i] == 'e' || s[i] == 'E') {
      if (i + 1 < s.size() && (isdigit(static_cast<unsigned char>(s[i + 1])) ||
                               s[i + 1] == '+' || s[i + 1] == '-')) {
        ++i;
        continue;
      }
      break;
    }
    ++i;
  }
  if (i == start) return false;
  std::string num_s = s.substr(start, i - start);
  bool is_int = num_s.find_first_of(".eE") == std::string::npos;
  double num = 0.0;
  int64_t num_i = 0;
  try {
    if (is_int) num_i = std::stoll(num_s);
    else num = std::stod(num_s);
  } catch (const std::exception&) {
    return false;
  }
  std::string suffix = s.substr(i);
  while (!suffix.empty() && isspace(static_cast<unsigned char>(suffix.back())))
    suffix.pop_back();
  double mult = 1.0;
  *had_suffix = !suffix.empty();
  if (suffix.empty()) mult = 1.0;
  else if (suffix == "Ki") mult = 1024.0;
  else if (suffix == "Mi") mult = 1048576.0;
  else if (suffix == "Gi") mult = 1073741824.0;
  else if (suffix == "Ti") mult = 1099511627776.0;
  else if (suffix == "Pi") mult = 1125899906842624.0;
  else if (suffix == "Ei") mult = 1152921504606846976.0;
  else if (suffix == "k") mult = 1e3;
  else if (suffix == "M") mult = 1e6;
  else if (suffix == "G") mult = 1e9;
  else if (suffix == "T") mult = 1e12;
  else if (suffix == "P") mult = 1e15;
  else if (suffix == "E") mult = 1e18;
  else if (suffix == "m") mult = 1e-3;
  else return false;
  if (is_int) {
    // exact integer scaling for integral multipliers (large byte counts)
    if (mult >= 1.0) *out = num_i * static_cast<int64_t>(mult);
    else *out = num_i / 1000;  // "m" milli
  } else {
    *out = static_cast<int64_t>(num * mult);
  }
  return true;
}

enum class BareUnit { Auto, Bytes, GiB, MiB };

// Bare-number "auto" heuristic threshold: bare values below it are GiB,
// larger are bytes. The SINGLE source of truth is Python
// utils/quantity.py:BARE_AUTO_GIB_THRESHOLD — package import pushes it here
// (bindings set_bare_auto_gib_threshold), so the C++ fast path and the
// Python fallback can never drift per request path.
inline std::atomic<int64_t>& bare_auto_gib_threshold() {
  static std::atomic<int64_t> v{8192};
  return v;
}

inline int64_t memory_bytes(const egsjson::Value& v, BareUnit bare) {
  int64_t raw = 0;
  bool had_suffix = false;
  if (v.is_string()) {
    if (!parse_quantity_ll(v.as_string(), &raw, &had_suffix)) return 0;
  } else if (v.is_number()) {
    raw = v.as_int();
  } else {
    return 0;
  }
  if (had_suffix) return raw;
  switch (bare) {
    case BareUnit::Bytes: return raw;
    case BareUnit::GiB: return raw * (1LL << 30);
    case BareUnit::MiB: return raw * (1LL << 20);
    case BareUnit::Auto: {
      int64_t thr = bare_auto_gib_threshold().load(std::memory_order_relaxed);
      return (raw > 0 && raw < thr) ? raw * (1LL << 30) : raw;
    }
  }
  return raw;
}

inline int64_t int_quantity(const egsjson::Value& v) {
  if (v.is_number()) return v.as_int();
  if (v.is_string()) {
    int64_t out = 0;
    bool suf = false;
    if (parse_quantity_ll(v.as_string(), &out, &suf)) return out;
  }
  return 0;
}

// ---- pod -> GPURequest (ports k8s/objects.py container_gpu_unit) ----

struct PodInfo {
  std::string uid;
  GPURequest request;
  bool is_gpu_pod = false;
  bool spread_containers = false;  // elasticgpu.io/spread-containers=true
};

inline const egsjson::Value& merged_resource(const egsjson::Value& resources,
                                             const std::string& name) {
  const egsjson::Value& req = resources.get("requests").get(name);
  if (!req.is_null()) return req;
  return resources.get("limits").get(name);
}

inline PodInfo parse_pod(const egsjson::Value& pod, BareUnit bare) {
  static const std::string kGpuCore = "elasticgpu.io/gpu-core";
  static const std::string kGpuMem = "elasticgpu.io/gpu-memory";
  static const std::string kQCore = "elasticgpu.io/qgpu-core";
  static const std::string kQMem = "elasticgpu.io/qgpu-memory";
  static const std::string kPgpu = "elasticgpu.io/pgpu";

  PodInfo info;
  info.uid = pod.get("metadata").get("uid").as_string();
  static const std::string kSpread = "elasticgpu.io/spread-containers";
  if (pod.get("metadata").get("annotations").get(kSpread).as_string() ==
          "true" ||
      pod.get("metadata").get("labels").get(kSpread).as_string() == "true")
    info.spread_containers = true;
  const auto& containers = pod.get("spec").get("containers").as_array();
  for (const auto& c : containers) {
    const egsjson::Value& res = c.get("resources");
    int64_t core = int_quantity(merged_resource(res, kGpuCore)) +
                   int_quantity(merged_resource(res, kQCore));
    int64_t mem = 0;
    const egsjson::Value& m1 = merged_resource(res, kGpuMem);
    const egsjson::Value& m2 = merged_resource(res, kQMem);
    if (!m1.is_null()) mem += memory_bytes(m1, bare);
    if (!m2.is_null()) mem += memory_bytes(m2, bare);
    int64_t pgpu = int_quantity(merged_resource(res, kPgpu));

    GPUUnit u;
    if (pgpu > 0) {
      u.gpu_count = static_cast<int>(pgpu);
      info.is_gpu_pod = true;
    } else if (core == 0 && mem == 0) {
      // no GPU for this container; but pgpu/mem/core key PRESENCE still
      // marks the pod as a GPU pod for routing purposes
    } else if (core >= kGPUCoreEachCard) {
      u.gpu_count = static_cast<int>(core / kGPUCoreEachCard);
      info.is_gpu_pod = true;
    } else {
      u.core = static_cast<int>(core);
      u.memory = mem;
      info.is_gpu_pod = true;
    }
    if (!info.is_gpu_pod) {
      // presence of any of the 5 resource names marks a GPU pod
      for (const std::string* name : {&kGpuCore, &kGpuMem, &kQCore, &kQMem, &kPgpu}) {
        if (!merged_resource(res, *name).is_null()) {
          info.is_gpu_pod = true;
          break;
        }
      }
    }
    info.request.push_back(u);
  }
  return info;
}

// ---- score -> extender-protocol integer (de-herded) ----
//
// The protocol wants ints 0..10, so equally-packed nodes used to tie and
// every concurrent pod raced for the same node (r1 soak: ~3-4% binpack
// bind retries, profiles/r01_results.md). Stochastic rounding with a
// stable per-(pod, node) hash offset splits those ties per pod without
// changing any ordering that differs by >= 1 score level: floor(s + u)
// where u in [0,1) is pseudo-random in (uid, node) — deterministic, so
// Assume/Score/Bind agree, and different pods order tied nodes
// differently, spreading the herd. MUST stay bit-identical with the
// Python fallback (server/app.py _jittered_int_score; parity-tested).
inline int64_t jittered_int_score(double s, const std::string& uid,
                                  const std::string& node) {
  uint64_t h = 14695981039346656037ULL;
  for (unsigned char c : uid) {
    h ^= c;
    h *= 1099511628211ULL;
  }
  h ^= 0x9e3779b97f4a7c15ULL;
  for (unsigned char c : node) {
    h ^= c;
    h *= 1099511628211ULL;
  }
  double u = static_cast<double>(h % 4096) / 4096.0;
  double v = std::min(std::max(s, 0.0), 10.0);
  auto out = static_cast<int64_t>(std::floor(v + u));
  return std::min<int64_t>(out, 10);
}

// ---- filter -> first-filter timestamp tracker (p50 filter->bind) ----

class FilterTracker {
 public:
  void note(const std::string& uid) {
    double now = monotonic();
    std::lock_guard<std::mutex> g(mu_);
    if (first_.size() > 65536) first_.clear();
    first_.emplace(uid, now);
  }
  // seconds since the first filter, or -1 when unknown; erases the entry
  double pop(const std::string& uid) {
    std::lock_guard<std::mutex> g(mu_);
    auto it = first_.find(uid);
    if (it == first_.end()) return -1.0;
    double t0 = it->second;
    first_.erase(it);
    return monotonic() - t0;
  }
  static double monotonic() {
    return std::chrono::duration<double>(
               std::chrono::steady_clock::now().time_since_epoch())
        .count();
  }

 private:
  std::mutex mu_;
  std::unordered_map<std::string, double> first_;
};

// ---- native handler outcome ----

struct NativeCounters {
  std::atomic<uint64_t> filter_native{0};
  std::atomic<uint64_t> priorities_native{0};
  std::atomic<uint64_t> fallback{0};
  std::atomic<uint64_t> errors{0};
};

// Lock-free log2 latency histogram for the GIL-free fast path. r1's
// /debug/profile sampler could only see Python threads, leaving the C++
// 95% of traffic invisible; these histograms
// are computed where the work happens and exported through stats() and
// /metrics. Bucket b counts requests with latency in (2^(b-1), 2^b] us.
struct LatencyHist {
  static constexpr int kBuckets = 24;  // 1 us .. ~8.4 s
  std::atomic<uint64_t> buckets[kBuckets];
  std::atomic<uint64_t> count{0};
  std::atomic<uint64_t> sum_us{0};

  LatencyHist() {
    for (auto& b : buckets) b.store(0, std::memory_order_relaxed);
  }
  void record_us(uint64_t us) {
    int b = us <= 1 ? 0
                    : std::min<int>(kBuckets - 1,
                                    64 - __builtin_clzll(us - 1));
    buckets[b].fetch_add(1, std::memory_order_relaxed);
    count.fetch_add(1, std::memory_order_relaxed);
    sum_us.fetch_add(us, std::memory_order_relaxed);
  }
};

class ScopedLatency {
 public:
  explicit ScopedLatency(LatencyHist* h) : h_(h), t0_(Clock::now()) {}
  ~ScopedLatency() {
    auto us = std::chrono::duration_cast<std::chrono::microseconds>(
                  Clock::now() - t0_)
                  .count();
    h_->record_us(static_cast<uint64_t>(us < 0 ? 0 : us));
  }

 private:
  using Clock = std::chrono::steady_clock;
  LatencyHist* h_;
  Clock::time_point t0_;
};

enum class HandleStatus { Handled, NeedFallback };

class ExtenderCore {
 public:
  ExtenderCore(std::shared_ptr<ClusterState> state, BareUnit bare)
      : state_(std::move(state)), bare_(bare) {}

  // POST /scheduler/filter. Returns Handled + response JSON, or NeedFallback.
  HandleStatus filter(const std::string& body, std::string* response) {
    ScopedLatency lat(&filter_hist);
    egsjson::Value args;
    try {
      args = egsjson::parse(body);
    } catch (const egsjson::ParseError&) {
      return HandleStatus::NeedFallback;  // Python returns the 400
    }
    const egsjson::Value& pod = args.get("pod");
    if (pod.is_null()) return HandleStatus::NeedFallback;
    const egsjson::Value& nodenames = args.get("nodenames");
    if (!nodenames.is_array()) return HandleStatus::NeedFallback;  // error path

    std::vector<std::string> names;
    names.reserve(nodenames.as_array().size());
    for (const auto& n : nodenames.as_array()) names.push_back(n.as_string());
    if (!state_->has_all_nodes(names)) return HandleStatus::NeedFallback;

    PodInfo info = parse_pod(pod, bare_);
    egsjson::Value out = egsjson::Value::make_object();
    if (!info.is_gpu_pod) {
      // pass-through
      egsjson::Array ok;
      for (auto& n : names) ok.push_back(egsjson::Value(n));
      out.set("nodenames", egsjson::Value(std::move(ok)));
      out.set("failedNodes", egsjson::Value::make_object());
      *response = egsjson::dump(out);
      counters.filter_native.fetch_add(1, std::memory_order_relaxed);
      return HandleStatus::Handled;
    }
    tracker.note(info.uid);
    std::vector<int> verdicts =
        state_->assume(names, info.uid, info.request, info.spread_containers);
    egsjson::Array ok;
    egsjson::Value failed = egsjson::Value::make_object();
    for (size_t i = 0; i < names.size(); ++i) {
      if (verdicts[i] == 0) {
        ok.push_back(egsjson::Value(names[i]));
      } else if (verdicts[i] == 1) {
        failed.set(names[i], egsjson::Value("insufficient GPU resources"));
      } else {
        failed.set(names[i], egsjson::Value("node has no GPU inventory"));
      }
    }
    out.set("nodenames", egsjson::Value(std::move(ok)));
    out.set("failedNodes", std::move(failed));
    *response = egsjson::dump(out);
    counters.filter_native.fetch_add(1, std::memory_order_relaxed);
    return HandleStatus::Handled;
  }

  // POST /scheduler/priorities.
  HandleStatus priorities(const std::string& body, std::string* response) {
    ScopedLatency lat(&priorities_hist);
    egsjson::Value args;
    try {
      args = egsjson::parse(body);
    } catch (const egsjson::ParseError&) {
      return HandleStatus::NeedFallback;
    }
    const egsjson::Value& pod = args.get("pod");
    if (pod.is_null()) return HandleStatus::NeedFallback;
    std::vector<std::string> names;
    for (const auto& n : args.get("nodenames").as_array())
      names.push_back(n.as_string());
    if (!state_->has_all_nodes(names)) return HandleStatus::NeedFallback;

    PodInfo info = parse_pod(pod, bare_);
    egsjson::Array result;
    if (!info.is_gpu_pod) {
      for (auto& n : names) {
        egsjson::Value e = egsjson::Value::make_object();
        e.set("host", egsjson::Value(n));
        e.set("score", egsjson::Value(static_cast<int64_t>(0)));
        result.push_back(std::move(e));
      }
    } else {
      std::vector<double> scores =
          state_->score(names, info.uid, info.request, info.spread_containers);
      for (size_t i = 0; i < names.size(); ++i) {
        egsjson::Value e = egsjson::Value::make_object();
        e.set("host", egsjson::Value(names[i]));
        e.set("score", egsjson::Value(
                           jittered_int_score(scores[i], info.uid, names[i])));
        result.push_back(std::move(e));
      }
    }
    *response = egsjson::dump(egsjson::Value(std::move(result)));
    counters.priorities_native.fetch_add(1, std::memory_order_relaxed);
    return HandleStatus::Handled;
  }

  FilterTracker tracker;
  NativeCounters counters;
  LatencyHist filter_hist;
  LatencyHist priorities_hist;

 private:
  std::shared_ptr<ClusterState> state_;
  BareUnit bare_;
};

}  // namespace egs
