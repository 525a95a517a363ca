// Cluster-wide scheduler state.
//
// The reference serialises every verb (Assume/Score/Bind/AddPod/ForgetPod)
// behind one global mutex per scheduler (pkg/scheduler/scheduler.go:44,113)
// and fans the per-node feasibility check over a fixed 4-goroutine pool
// (scheduler.go:129-156). Here the node map is read-mostly under a
// shared_mutex, each node carries its own lock, and the filter fan-out runs
// on a persistent thread pool sized to the host — concurrent pods scheduling
// onto different nodes never contend.
#pragma once

#include <atomic>
#include <condition_variable>
#include <functional>
#include <memory>
#include <shared_mutex>
#include <string>
#include <thread>
#include <unordered_map>
#include <vector>

#include "node.h"
#include "raters.h"
#include "types.h"

namespace egs {

// Small fixed thread pool for the filter fan-out.
class ThreadPool {
 public:
  explicit ThreadPool(int n) {
    if (n <= 0) n = static_cast<int>(std::thread::hardware_concurrency());
    if (n <= 0) n = 4;
    for (int i = 0; i < n; ++i)
      workers_.emplace_back([this] { loop(); });
  }
  ~ThreadPool() {
    {
      std::lock_guard<std::mutex> g(mu_);
      stop_ = true;
    }
    cv_.notify_all();
    for (auto& t : workers_) t.join();
  }
  int size() const { return static_cast<int>(workers_.size()); }

  // Run fn(i) for i in [0, n) across the pool; blocks until done. Every
  // worker gets a self-draining copy (an atomic index feeds items) and the
  // caller sleeps on a condvar — measured faster under many CONCURRENT
  // large fan-outs than caller-participates/bounded-copies variants (the
  // sleeping caller frees its core for pool workers).
  void parallel_for(int n, const std::function<void(int)>& fn) {
    if (n <= 0) return;
    if (n == 1 || workers_.empty()) {
      for (int i = 0; i < n; ++i) fn(i);
      return;
    }
    std::atomic<int> next{0};
    int done = 0;  // guarded by done_mu
    std::mutex done_mu;
    std::condition_variable done_cv;
    auto task = [&, n] {
      int i;
      while ((i = next.fetch_add(1)) < n) fn(i);
      // The final increment happens UNDER done_mu: if it were a bare atomic
      // the waiting caller could observe done==size() on a spurious wakeup,
      // return, and destroy done_mu/done_cv while this worker is about to
      // lock them (use-after-destroy race).
      std::lock_guard<std::mutex> g(done_mu);
      if (++done == size()) done_cv.notify_one();
    };
    {
      std::lock_guard<std::mutex> g(mu_);
      for (int i = 0; i < size(); ++i) queue_.push_back(task);
    }
    cv_.notify_all();
    std::unique_lock<std::mutex> g(done_mu);
    done_cv.wait(g, [&] { return done == size(); });
  }

 private:
  void loop() {
    for (;;) {
      std::function<void()> task;
      {
        std::unique_lock<std::mutex> g(mu_);
        cv_.wait(g, [this] { return stop_ || !queue_.empty(); });
        if (stop_ && queue_.empty()) return;
        task = std::move(queue_.front());
        // Deliberately vector + erase-front: an A/B on a 256-core MI355X
        // host (16 concurrent 256-node fan-outs) measured 378 pods/s with
        // this form vs 89 pods/s with deque::pop_front — the longer
        // critical section throttles 256 workers' mutex thrash on short holds
        // (profiles/r01_results.md). Do not "optimise" to deque without
        // re-measuring that scenario.
        queue_.erase(queue_.begin());
      }
      task();
    }
  }
  std::vector<std::thread> workers_;
  std::vector<std::function<void()>> queue_;
  std::mutex mu_;
  std::condition_variable cv_;
  bool stop_ = false;
};

// Below this many nodes, the filter fan-out runs inline on the calling
// thread (pool dispatch overhead > the per-node search cost).
constexpr size_t kInlineFanout = 64;

enum class AssumeVerdict : int {
  kOk = 0,
  kInfeasible = 1,
  kUnknownNode = 2,
};

class ClusterState {
 public:
  ClusterState(const std::string& policy, uint64_t seed, int threads,
               double topology_weight = kDefaultTopologyWeight)
      : rater_(make_rater(policy, seed)), pool_(threads),
        topology_weight_(topology_weight) {}

  std::string policy() const { return rater_->name(); }

  void add_node(const std::string& name, std::vector<Device> devices,
                std::vector<std::vector<int>> topo_hops) {
    auto alloc = std::make_shared<NodeAllocator>(name, std::move(devices),
                                                Topology(std::move(topo_hops)),
                                                topology_weight_);
    std::unique_lock<std::shared_mutex> g(mu_);
    nodes_[name] = std::move(alloc);  // replaces any stale entry
  }

  bool has_node(const std::string& name) {
    std::shared_lock<std::shared_mutex> g(mu_);
    return nodes_.count(name) > 0;
  }

  // One lock acquisition for a whole candidate list (a 256-node filter
  // would otherwise take the shared lock 256 times per request).
  bool has_all_nodes(const std::vector<std::string>& names) {
    std::shared_lock<std::shared_mutex> g(mu_);
    for (const auto& n : names)
      if (nodes_.count(n) == 0) return false;
    return true;
  }

  void remove_node(const std::string& name) {
    std::unique_lock<std::shared_mutex> g(mu_);
    nodes_.erase(name);
  }

  std::vector<std::string> node_names() {
    std::shared_lock<std::shared_mutex> g(mu_);
    std::vector<std::string> out;
    out.reserve(nodes_.size());
    for (const auto& [n, _] : nodes_) out.push_back(n);
    return out;
  }

  // Filter fan-out: feasibility of `req` on every node in `names`.
  std::vector<int> assume(const std::vector<std::string>& names,
                          const std::string& uid, const GPURequest& req,
                          bool distinct = false) {
    std::vector<std::shared_ptr<NodeAllocator>> allocs(names.size());
    {
      std::shared_lock<std::shared_mutex> g(mu_);
      for (size_t i = 0; i < names.size(); ++i) {
        auto it = nodes_.find(names[i]);
        if (it != nodes_.end()) allocs[i] = it->second;
      }
    }
    std::vector<int> verdicts(names.size(), static_cast<int>(AssumeVerdict::kUnknownNode));
    // Inline cache pass first: warm nodes answer with two map lookups, so
    // only the nodes that genuinely need a placement search pay the
    // thread-pool dispatch.
    std::vector<int> pending;
    for (size_t i = 0; i < names.size(); ++i) {
      if (!allocs[i]) continue;
      int v = allocs[i]->assume_cached(uid, req, distinct);
      if (v >= 0)
        verdicts[i] = v ? static_cast<int>(AssumeVerdict::kOk)
                        : static_cast<int>(AssumeVerdict::kInfeasible);
      else
        pending.push_back(static_cast<int>(i));
    }
    auto task = [&](int pi) {
      int i = pending[pi];
      verdicts[i] = allocs[i]->assume(uid, req, *rater_, distinct)
                        ? static_cast<int>(AssumeVerdict::kOk)
                        : static_cast<int>(AssumeVerdict::kInfeasible);
    };
    run_fanout(pending.size(), task);
    return verdicts;
  }

  std::vector<double> score(const std::vector<std::string>& names,
                            const std::string& uid, const GPURequest& req,
                            bool distinct = false) {
    std::vector<std::shared_ptr<NodeAllocator>> allocs(names.size());
    {
      std::shared_lock<std::shared_mutex> g(mu_);
      for (size_t i = 0; i < names.size(); ++i) {
        auto it = nodes_.find(names[i]);
        if (it != nodes_.end()) allocs[i] = it->second;
      }
    }
    std::vector<double> scores(names.size(), kScoreMin);
    std::vector<int> pending;
    for (size_t i = 0; i < names.size(); ++i) {
      if (!allocs[i]) continue;
      if (!allocs[i]->score_cached(uid, req, distinct, &scores[i]))
        pending.push_back(static_cast<int>(i));
    }
    auto task = [&](int pi) {
      int i = pending[pi];
      scores[i] = allocs[i]->score(uid, req, *rater_, distinct);
    };
    run_fanout(pending.size(), task);
    return scores;
  }

  GPUOption allocate(const std::string& node, const std::string& uid,
                     const GPURequest& req, bool distinct = false) {
    auto alloc = get(node);
    if (!alloc) throw std::runtime_error("unknown node " + node);
    GPUOption option = alloc->allocate(uid, req, *rater_, distinct);
    {
      std::lock_guard<std::mutex> g(pod_node_mu_);
      pod_node_[uid] = node;
    }
    return option;
  }

  void add_pod(const std::string& node, const std::string& uid,
               const GPURequest& req, const GPUOption& option) {
    auto alloc = get(node);
    if (!alloc) throw std::runtime_error("unknown node " + node);
    alloc->add_pod(uid, req, option);
    std::lock_guard<std::mutex> g(pod_node_mu_);
    pod_node_[uid] = node;
  }

  void note_pod_node(const std::string& uid, const std::string& node) {
    std::lock_guard<std::mutex> g(pod_node_mu_);
    pod_node_[uid] = node;
  }

  void forget_pod(const std::string& uid) {
    std::string node;
    {
      std::lock_guard<std::mutex> g(pod_node_mu_);
      auto it = pod_node_.find(uid);
      if (it == pod_node_.end()) return;
      node = it->second;
      pod_node_.erase(it);
    }
    auto alloc = get(node);
    if (alloc) alloc->forget_pod(uid);
  }

  bool known_pod(const std::string& uid) {
    std::lock_guard<std::mutex> g(pod_node_mu_);
    return pod_node_.count(uid) > 0;
  }

  bool feasible_with_victims(const std::string& node, const std::string& /*uid*/,
                             const GPURequest& req,
                             const std::vector<std::string>& victims) {
    auto alloc = get(node);
    if (!alloc) return false;
    return alloc->feasible_with_victims(req, victims, *rater_);
  }

  // Adaptive fan-out: small node counts run inline (pool dispatch costs
  // more than the searches). Large fan-outs use the pool while it still has
  // spare parallelism relative to the requests already fanning out; once
  // concurrent requests saturate the pool (many clients on a small host),
  // each request runs inline on its own server thread instead of queueing
  // (measured: always-inline under concurrency halves throughput on a
  // 256-core host, always-pool thrashes on an 8-core one).
  void run_fanout(size_t n, const std::function<void(int)>& task) {
    struct Guard {
      std::atomic<int>& c;
      explicit Guard(std::atomic<int>& c_) : c(c_) { c.fetch_add(1); }
      ~Guard() { c.fetch_sub(1); }
    } guard(inflight_);
    bool use_pool = n > kInlineFanout &&
                    inflight_.load() * 4 <= pool_.size();
    if (use_pool) {
      pool_.parallel_for(static_cast<int>(n), task);
    } else {
      for (size_t i = 0; i < n; ++i) task(static_cast<int>(i));
    }
  }

  std::shared_ptr<NodeAllocator> get(const std::string& name) {
    std::shared_lock<std::shared_mutex> g(mu_);
    auto it = nodes_.find(name);
    return it == nodes_.end() ? nullptr : it->second;
  }

  int pool_size() const { return pool_.size(); }

 private:
  std::unique_ptr<Rater> rater_;
  ThreadPool pool_;
  double topology_weight_ = kDefaultTopologyWeight;
  std::atomic<int> inflight_{0};
  std::shared_mutex mu_;
  std::unordered_map<std::string, std::shared_ptr<NodeAllocator>> nodes_;
  std::mutex pod_node_mu_;
  std::unordered_map<std::string, std::string> pod_node_;  // uid -> node name
};

}  // namespace egs
