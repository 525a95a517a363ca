// Placement search: choose device indexes for every container of a pod.
//
// Semantics follow the reference's GPUs.Trade (pkg/scheduler/gpu.go:65-129):
//   * a whole-card container (gpu_count > 0) takes N fully-free cards
//     exclusively;
//   * a fractional container takes a share (core%, memory bytes) of exactly
//     one card; multiple containers may share a card;
//   * every complete assignment is scored by the Rater and the best is kept.
// Improvements over the reference:
//   * whole-card sets are CHOSEN (xGMI-topology- and policy-scored k-subsets),
//     not just the first N free cards (gpu.go:96-108);
//   * symmetric fractional branches are deduplicated and the DFS is bounded
//     by a deterministic leaf budget with greedy candidate ordering, so worst
//     cases stay microseconds instead of cards^containers;
//   * ties break deterministically (first-best in candidate order), so
//     Assume/Score/Bind always agree on the same placement.
#pragma once

#include <algorithm>
#include <cstdint>
#include <functional>
#include <utility>
#include <vector>

#include "raters.h"
#include "topology.h"
#include "types.h"

namespace egs {

// Upper bound on scored complete assignments per search. Candidates are
// ordered greedily, so truncation degrades score quality, never feasibility
// (the first feasible leaf is always reached if one exists in the explored
// prefix; the greedy order explores plausible placements first).
constexpr int kMaxLeafEvals = 4096;
// Upper bound on TOTAL dfs node visits (complete assignments AND dead-end
// partial paths). kMaxLeafEvals alone only counts complete assignments: a
// multi-container pod whose last container never fits would explore up to
// cards^containers dead ends with zero leaves, stalling the node mutex.
constexpr int kMaxDfsVisits = 65536;
// Upper bound on whole-card k-subsets enumerated per container.
constexpr int kMaxWholeCardCandidates = 128;

struct SearchResult {
  bool feasible = false;
  GPUOption option;
  int leaves_evaluated = 0;
};

namespace search_detail {

// Below this many raw k-subsets the enumeration is EXHAUSTIVE (exact
// best-by-topology ordering); above it, greedy-seeded generation takes
// over (see below).
constexpr long long kExhaustiveSubsetCap = 4096;

// Enumerate k-subsets of `free_cards`, best-topology-first, capped.
//
// r1 enumerated lexicographically and stopped GENERATING at 1024
// candidates before sorting, so on a fragmented large node (CPX 64-way)
// every candidate shared a low-index prefix and the best-locality set
// could simply never be generated. Now:
//   * C(n,k) <= kExhaustiveSubsetCap: full enumeration — exact;
//   * larger: one candidate GREEDY-GROWN FROM EVERY free card (seed ->
//     repeatedly add the card with the lowest incremental hop cost), then
//     a single swap-improvement pass per candidate. A seed inside the
//     best hive always exists, and min-incremental-hop growth stays
//     inside that hive while it has free cards — so fragmentation cannot
//     hide the minimum-hop set behind a lexicographic prefix.
// Everything ties-breaks on card index, so the result is deterministic.
inline void enumerate_subsets(const std::vector<int>& free_cards, int k,
                              const Topology& topo,
                              std::vector<std::vector<int>>* out) {
  const int n = static_cast<int>(free_cards.size());
  if (k > n) return;
  if (k <= 0) return;
  struct Cand {
    std::vector<int> cards;
    int cost;
  };
  std::vector<Cand> cands;

  long long total = 1;  // C(n,k), saturated
  for (int i = 0; i < k; ++i) {
    total = total * (n - i) / (i + 1);
    if (total > kExhaustiveSubsetCap) break;
  }

  if (total <= kExhaustiveSubsetCap) {
    std::vector<int> pick;
    pick.reserve(k);
    std::function<void(int)> rec = [&](int start) {
      if (static_cast<int>(pick.size()) == k) {
        std::vector<int> cards;
        cards.reserve(k);
        for (int i : pick) cards.push_back(free_cards[i]);
        int cost = topo.set_cost(cards);
        cands.push_back({std::move(cards), cost});
        return;
      }
      for (int i = start; i < n; ++i) {
        pick.push_back(i);
        rec(i + 1);
        pick.pop_back();
      }
    };
    rec(0);
  } else {
    for (int s = 0; s < n; ++s) {
      std::vector<int> set = {free_cards[s]};
      std::vector<char> used(n, 0);
      used[s] = 1;
      while (static_cast<int>(set.size()) < k) {
        int best = -1;
        long long best_inc = 0;
        for (int j = 0; j < n; ++j) {
          if (used[j]) continue;
          long long inc = 0;
          for (int c : set) inc += topo.hops(c, free_cards[j]);
          if (best < 0 || inc < best_inc) {
            best = j;
            best_inc = inc;
          }
        }
        used[best] = 1;
        set.push_back(free_cards[best]);
      }
      std::sort(set.begin(), set.end());
      // One swap-improvement pass: replace a member with a non-member
      // when it strictly lowers the pairwise cost (first-improvement,
      // ascending order -> deterministic).
      int cost = topo.set_cost(set);
      for (size_t a = 0; a < set.size(); ++a) {
        for (int j = 0; j < n; ++j) {
          int cand = free_cards[j];
          if (std::find(set.begin(), set.end(), cand) != set.end()) continue;
          int delta = 0;
          for (size_t b = 0; b < set.size(); ++b) {
            if (b == a) continue;
            delta += topo.hops(set[b], cand) - topo.hops(set[b], set[a]);
          }
          if (delta < 0) {
            set[a] = cand;
            cost += delta;
          }
        }
      }
      std::sort(set.begin(), set.end());
      cands.push_back({std::move(set), cost});
    }
    // dedupe identical sets (many seeds converge to the same optimum)
    std::sort(cands.begin(), cands.end(),
              [](const Cand& a, const Cand& b) { return a.cards < b.cards; });
    cands.erase(std::unique(cands.begin(), cands.end(),
                            [](const Cand& a, const Cand& b) {
                              return a.cards == b.cards;
                            }),
                cands.end());
  }
  std::stable_sort(cands.begin(), cands.end(), [](const Cand& a, const Cand& b) {
    return a.cost != b.cost ? a.cost < b.cost : a.cards < b.cards;
  });
  int keep = std::min<int>(static_cast<int>(cands.size()), kMaxWholeCardCandidates);
  for (int i = 0; i < keep; ++i) out->push_back(std::move(cands[i].cards));
}

struct DfsState {
  std::vector<Device> devices;
  const GPURequest* req;
  const Rater* rater;
  const RateContext* ctx;
  GPUOption current;
  GPUOption best;
  bool found = false;
  int leaves = 0;
  int visits = 0;  // every dfs() entry, including dead ends
  // distinct_containers: no card may serve two containers of this pod
  // (elasticgpu.io/spread-containers annotation)
  bool distinct_containers = false;
  std::vector<bool> used_by_pod;  // card -> taken by an earlier container
};

inline void dfs(DfsState& st, size_t c);

inline void try_candidate(DfsState& st, size_t c, const std::vector<int>& cards,
                          bool whole) {
  const GPUUnit& u = (*st.req)[c];
  // Apply.
  std::vector<std::pair<int, Device>> saved;
  saved.reserve(cards.size());
  for (int idx : cards) {
    saved.emplace_back(idx, st.devices[idx]);
    Device& d = st.devices[idx];
    if (whole) {
      d.core_avail = 0;
      d.mem_avail = 0;
    } else {
      d.core_avail -= u.core;
      d.mem_avail -= u.memory;
    }
    if (st.distinct_containers) st.used_by_pod[idx] = true;
  }
  st.current.allocated[c] = cards;
  dfs(st, c + 1);
  // Undo.
  st.current.allocated[c].clear();
  for (auto& [idx, d] : saved) st.devices[idx] = d;
  if (st.distinct_containers)
    for (int idx : cards) st.used_by_pod[idx] = false;
}

inline void dfs(DfsState& st, size_t c) {
  if (st.leaves >= kMaxLeafEvals || ++st.visits > kMaxDfsVisits) return;
  const GPURequest& req = *st.req;
  if (c == req.size()) {
    ++st.leaves;
    double score = st.rater->rate(*st.ctx, req, st.current);
    // Strictly-greater keeps the FIRST best in deterministic candidate order.
    if (!st.found || score > st.best.score) {
      st.best = st.current;
      st.best.score = score;
      st.found = true;
    }
    return;
  }
  const GPUUnit& u = req[c];
  if (!u.needs_gpu()) {
    st.current.allocated[c].clear();
    dfs(st, c + 1);
    return;
  }
  if (u.whole_cards()) {
    std::vector<int> free_cards;
    for (int i = 0; i < static_cast<int>(st.devices.size()); ++i)
      if (st.devices[i].whole_free() &&
          !(st.distinct_containers && st.used_by_pod[i]))
        free_cards.push_back(i);
    if (static_cast<int>(free_cards.size()) < u.gpu_count) return;  // infeasible here
    std::vector<std::vector<int>> subsets;
    const Topology* topo = st.ctx->topo;
    static const Topology kEmpty;
    enumerate_subsets(free_cards, u.gpu_count, topo ? *topo : kEmpty, &subsets);
    for (const auto& cards : subsets) {
      try_candidate(st, c, cards, /*whole=*/true);
      if (st.leaves >= kMaxLeafEvals) return;
    }
  } else {
    // Fractional: one card among those that fit; dedupe cards with an
    // identical availability signature when topology is uniform (symmetric
    // branches produce identical scores).
    bool uniform = st.ctx->topo == nullptr || st.ctx->topo->empty();
    std::vector<std::pair<int64_t, int64_t>> seen;
    for (int i = 0; i < static_cast<int>(st.devices.size()); ++i) {
      const Device& d = st.devices[i];
      if (!d.can_fit(u.core, u.memory)) continue;
      if (st.distinct_containers && st.used_by_pod[i]) continue;
      if (uniform) {
        auto sig = std::make_pair(static_cast<int64_t>(d.core_avail), d.mem_avail);
        if (std::find(seen.begin(), seen.end(), sig) != seen.end()) continue;
        seen.push_back(sig);
      }  // (used_by_pod cards were excluded above, so dedupe stays sound)
      try_candidate(st, c, {i}, /*whole=*/false);
      if (st.leaves >= kMaxLeafEvals) return;
    }
  }
}

}  // namespace search_detail

inline SearchResult search_placement(const std::vector<Device>& devices,
                                     const GPURequest& req, const Rater& rater,
                                     const RateContext& ctx,
                                     bool distinct_containers = false) {
  // Fail fast when some container can never fit even on a completely FREE
  // node: without this the dfs explores every placement of the earlier
  // containers before discovering the doomed one (the kMaxDfsVisits cap
  // bounds that walk; this check removes it entirely for the common case of
  // an oversized container).
  {
    int schedulable = 0;
    for (const auto& d : devices)
      if (d.schedulable()) ++schedulable;
    for (const auto& u : req) {
      if (!u.needs_gpu()) continue;
      if (u.whole_cards()) {
        if (u.gpu_count > schedulable) return SearchResult{};
      } else {
        bool fits_somewhere = false;
        for (const auto& d : devices) {
          if (d.core_total >= u.core && d.mem_total >= u.memory &&
              d.schedulable()) {
            fits_somewhere = true;
            break;
          }
        }
        if (!fits_somewhere) return SearchResult{};
      }
    }
  }
  search_detail::DfsState st;
  st.devices = devices;
  st.req = &req;
  st.rater = &rater;
  st.ctx = &ctx;
  st.distinct_containers = distinct_containers;
  st.used_by_pod.assign(devices.size(), false);
  st.current.allocated.resize(req.size());
  search_detail::dfs(st, 0);
  SearchResult res;
  res.feasible = st.found;
  res.option = std::move(st.best);
  res.leaves_evaluated = st.leaves;
  return res;
}

}  // namespace egs
