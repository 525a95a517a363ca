// Core domain model for the MI355X-native elastic GPU scheduler.
//
// Re-designed from scratch against the behavior of the reference
// (elastic-ai/elastic-gpu-scheduler pkg/scheduler/{gpu,allocate}.go) with
// deliberate fixes:
//   * memory is held in BYTES (the reference keeps a unitless int and divides
//     node allocatable evenly across cards, pkg/scheduler/node.go:25-40);
//   * the per-node model carries an xGMI adjacency matrix (the reference GPU
//     struct, pkg/scheduler/gpu.go:19-25, has no topology at all);
//   * MI355X defaults (8 cards / 288 GiB HBM3E per card) are explicit.
#pragma once

#include <cstdint>
#include <string>
#include <vector>

namespace egs {

// One whole card == 100 "gpu-core" units (reference pkg/utils/types.go:6).
constexpr int kGPUCoreEachCard = 100;

// MI355X: 288 GB HBM3E per card, 8 OAM cards per node, fully connected
// xGMI hive (7 point-to-point links per card at ~153 GB/s each).
constexpr int64_t kMI355XMemoryBytes = 288LL * 1024 * 1024 * 1024;
constexpr int kMI355XCardsPerNode = 8;
constexpr double kXGMILinkGBps = 153.0;

// One physical GPU card on a node.
struct Device {
  int core_total = kGPUCoreEachCard;
  int core_avail = kGPUCoreEachCard;
  int64_t mem_total = kMI355XMemoryBytes;
  int64_t mem_avail = kMI355XMemoryBytes;

  // Zero-capacity devices are placeholders the agent publishes for sick
  // cards (HBM health gate) so list position keeps equalling the physical
  // card index; they must never look schedulable — without the core_total
  // guard a 0/0 card is "whole free" (0 == 0) and a whole-card request
  // would land on the sick card.
  bool schedulable() const { return core_total > 0; }
  bool whole_free() const {
    return schedulable() && core_avail == core_total && mem_avail == mem_total;
  }
  bool can_fit(int core, int64_t mem) const {
    return schedulable() && core_avail >= core && mem_avail >= mem;
  }
};

// One container's GPU demand.
// Mirrors the semantics of reference NewGPURequest (pkg/scheduler/allocate.go:35-58):
//   core == 0 && mem == 0          -> no GPU needed
//   core >= 100                    -> gpu_count = core / 100 whole cards
//   otherwise                      -> fractional share of a single card
struct GPUUnit {
  int gpu_count = 0;   // whole cards wanted (exclusive use)
  int core = 0;        // fractional core percent, 0..99
  int64_t memory = 0;  // bytes

  bool needs_gpu() const { return gpu_count > 0 || core > 0 || memory > 0; }
  bool whole_cards() const { return gpu_count > 0; }
};

using GPURequest = std::vector<GPUUnit>;

// A chosen placement: per container, the list of device indexes.
struct GPUOption {
  std::vector<std::vector<int>> allocated;  // [container][device index]
  double score = 0.0;                       // calibrated 0..10
};

inline bool request_needs_gpu(const GPURequest& req) {
  for (const auto& u : req) {
    if (u.needs_gpu()) return true;
  }
  return false;
}

}  // namespace egs
