// xGMI topology model for MI355X nodes.
//
// The reference scheduler has no notion of inter-GPU topology (its multi-card
// path just takes the first N free cards, pkg/scheduler/gpu.go:96-108). On an
// 8x MI355X OAM board every card has 7 point-to-point xGMI links (~153 GB/s
// each), so ring collectives inside a placed pod are per-link bound and the
// scheduler must co-locate a pod's cards on directly-linked sets whenever the
// node is partitioned or heterogeneous (e.g. partitioned hives, PCIe-attached
// expansion cards). This module scores candidate card subsets by link
// locality.
#pragma once

#include <algorithm>
#include <cstdint>
#include <vector>

#include "types.h"

namespace egs {

// Pairwise "hop" matrix between cards. hops[i][j]:
//   0  -> same card
//   1  -> direct xGMI link
//   2+ -> routed (other card / CPU / PCIe) — increasingly bad
// Default (empty matrix) means fully-connected single hive: all pairs 1 hop.
class Topology {
 public:
  Topology() = default;
  explicit Topology(std::vector<std::vector<int>> hops) : hops_(std::move(hops)) {}

  bool empty() const { return hops_.empty(); }
  int size() const { return static_cast<int>(hops_.size()); }

  int hops(int i, int j) const {
    if (i == j) return 0;
    if (hops_.empty() || i >= size() || j >= size()) return 1;  // single hive
    return hops_[i][j];
  }

  // Sum of pairwise hops over a card set; lower is better.
  int set_cost(const std::vector<int>& cards) const {
    int cost = 0;
    for (size_t a = 0; a < cards.size(); ++a)
      for (size_t b = a + 1; b < cards.size(); ++b) cost += hops(cards[a], cards[b]);
    return cost;
  }

  // Worst possible pairwise cost for a set of k cards drawn from n, used to
  // normalise set_cost into [0,1]. Assumes hop values are small (<= 4).
  int worst_pair_cost(int n, int k) const {
    if (k < 2) return 0;
    int pairs = k * (k - 1) / 2;
    int max_hop = 1;
    for (int i = 0; i < size(); ++i)
      for (int j = i + 1; j < size(); ++j) max_hop = std::max(max_hop, hops(i, j));
    (void)n;
    return pairs * max_hop;
  }

  // 1.0 = perfectly local (all direct links), 0.0 = worst case.
  double locality(const std::vector<int>& cards) const {
    if (cards.size() < 2) return 1.0;
    int pairs = static_cast<int>(cards.size() * (cards.size() - 1) / 2);
    int worst = worst_pair_cost(size(), static_cast<int>(cards.size()));
    if (worst <= pairs) return 1.0;  // uniform topology: everything is local
    int cost = set_cost(cards);
    return 1.0 - static_cast<double>(cost - pairs) / static_cast<double>(worst - pairs);
  }

 private:
  std::vector<std::vector<int>> hops_;
};

}  // namespace egs
