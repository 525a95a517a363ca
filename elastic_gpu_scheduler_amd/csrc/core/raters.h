// Placement scoring policies.
//
// The reference ships a Binpack whose score can exceed the extender protocol's
// 0..10 range (pkg/scheduler/rater.go:18-51 multiplies by 100) and a Spread
// that is an unimplemented stub returning 0 (rater.go:56-59). Here all raters
// return calibrated scores in [0, 10], Spread is real, Random exists (the
// upstream README promises "binpack, spread, random and other policies"), and
// multi-card placements are blended with an xGMI-locality term.
#pragma once

#include <algorithm>
#include <cmath>
#include <cstdint>
#include <memory>
#include <string>

#include "topology.h"
#include "types.h"

namespace egs {

constexpr double kScoreMin = 0.0;
constexpr double kScoreMax = 10.0;

// Default weight of the xGMI-locality term for multi-card containers
// (operators tune via --topology-weight).
constexpr double kDefaultTopologyWeight = 0.3;

struct RateContext {
  const std::vector<Device>* devices = nullptr;  // state BEFORE the placement
  const Topology* topo = nullptr;
  uint64_t salt = 0;  // stable per (node, pod): differentiates Random scores
  double topology_weight = kDefaultTopologyWeight;
};

namespace detail {

// Apply `option` for `req` to a copy of the devices; returns post-state.
inline std::vector<Device> apply(const std::vector<Device>& devices,
                                 const GPURequest& req, const GPUOption& option) {
  std::vector<Device> after = devices;
  for (size_t c = 0; c < option.allocated.size() && c < req.size(); ++c) {
    const GPUUnit& u = req[c];
    for (int idx : option.allocated[c]) {
      if (idx < 0 || idx >= static_cast<int>(after.size())) continue;
      Device& d = after[idx];
      if (u.whole_cards()) {
        d.core_avail = 0;
        d.mem_avail = 0;
      } else {
        d.core_avail -= u.core;
        d.mem_avail -= u.memory;
      }
    }
  }
  return after;
}

inline double utilization(const Device& d) {
  double core_used =
      d.core_total > 0 ? 1.0 - static_cast<double>(d.core_avail) / d.core_total : 0.0;
  double mem_used =
      d.mem_total > 0 ? 1.0 - static_cast<double>(d.mem_avail) / d.mem_total : 0.0;
  return 0.5 * (core_used + mem_used);
}

// Mean post-placement utilization over the touched cards, in [0,1].
inline double touched_utilization(const std::vector<Device>& after,
                                  const GPUOption& option) {
  double sum = 0.0;
  int n = 0;
  std::vector<bool> seen(after.size(), false);
  for (const auto& per_container : option.allocated) {
    for (int idx : per_container) {
      if (idx < 0 || idx >= static_cast<int>(after.size()) || seen[idx]) continue;
      seen[idx] = true;
      sum += utilization(after[idx]);
      ++n;
    }
  }
  return n > 0 ? sum / n : 0.0;
}

inline double topology_locality(const RateContext& ctx, const GPUOption& option) {
  double sum = 0.0;
  int n = 0;
  for (const auto& per_container : option.allocated) {
    if (per_container.size() >= 2) {
      sum += ctx.topo ? ctx.topo->locality(per_container) : 1.0;
      ++n;
    }
  }
  return n > 0 ? sum / n : -1.0;  // -1: no multi-card container
}

// Blend a base policy score with xGMI locality for multi-card placements.
inline double blend(const RateContext& ctx, const GPUOption& option, double base) {
  double loc = topology_locality(ctx, option);
  double w = std::clamp(ctx.topology_weight, 0.0, 1.0);
  if (loc < 0.0 || w == 0.0) return std::clamp(base, kScoreMin, kScoreMax);
  double s = (1.0 - w) * base + w * kScoreMax * loc;
  return std::clamp(s, kScoreMin, kScoreMax);
}

inline uint64_t fnv1a(uint64_t h, uint64_t v) {
  for (int i = 0; i < 8; ++i) {
    h ^= (v >> (i * 8)) & 0xff;
    h *= 1099511628211ULL;
  }
  return h;
}

}  // namespace detail

class Rater {
 public:
  virtual ~Rater() = default;
  virtual double rate(const RateContext& ctx, const GPURequest& req,
                      const GPUOption& option) const = 0;
  virtual std::string name() const = 0;
};

// Tight packing: maximise post-placement utilization of the touched cards and
// penalise spilling over more distinct cards than the request demands.
class Binpack : public Rater {
 public:
  double rate(const RateContext& ctx, const GPURequest& req,
              const GPUOption& option) const override {
    auto after = detail::apply(*ctx.devices, req, option);
    double base = kScoreMax * detail::touched_utilization(after, option);
    return detail::blend(ctx, option, base);
  }
  std::string name() const override { return "binpack"; }
};

// Load balancing: prefer placements that land on the emptiest cards.
class Spread : public Rater {
 public:
  double rate(const RateContext& ctx, const GPURequest& req,
              const GPUOption& option) const override {
    auto after = detail::apply(*ctx.devices, req, option);
    double base = kScoreMax * (1.0 - detail::touched_utilization(after, option));
    return detail::blend(ctx, option, base);
  }
  std::string name() const override { return "spread"; }
};

// Deterministic pseudo-random: stable for a given (seed, node, pod, placement)
// so Assume/Score/Bind agree, but uncorrelated across pods and nodes.
class Random : public Rater {
 public:
  explicit Random(uint64_t seed) : seed_(seed) {}
  double rate(const RateContext& ctx, const GPURequest& /*req*/,
              const GPUOption& option) const override {
    uint64_t h = detail::fnv1a(14695981039346656037ULL, seed_);
    h = detail::fnv1a(h, ctx.salt);
    for (const auto& per_container : option.allocated) {
      for (int idx : per_container) h = detail::fnv1a(h, static_cast<uint64_t>(idx) + 1);
      h = detail::fnv1a(h, 0x5e9a7a70ULL);  // container separator
    }
    double base = static_cast<double>(h % 10000) / 1000.0;  // [0, 10)
    return detail::blend(ctx, option, base);
  }
  std::string name() const override { return "random"; }

 private:
  uint64_t seed_;
};

inline std::unique_ptr<Rater> make_rater(const std::string& policy, uint64_t seed) {
  if (policy == "spread") return std::make_unique<Spread>();
  if (policy == "random") return std::make_unique<Random>(seed);
  return std::make_unique<Binpack>();
}

}  // namespace egs
