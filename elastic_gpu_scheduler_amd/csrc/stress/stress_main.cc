// Standalone allocator stress binary for ThreadSanitizer runs.
//
// The reference configures no race detection at all (no -race anywhere,
// SURVEY.md §5); this binary is the rebuild's race-detection harness: it
// hammers ClusterState's verbs from many threads and is built with
// -fsanitize=thread by `make tsan-stress` (and exercised by
// tests/test_tsan_stress.py). Exit 0 = clean under TSAN.
#include <atomic>
#include <cstdio>
#include <string>
#include <thread>
#include <vector>

#include "../core/cluster.h"

using namespace egs;

int main() {
  constexpr int64_t GiB = 1024LL * 1024 * 1024;
  ClusterState cs("binpack", 0, 4);
  const int kNodes = 4;
  for (int i = 0; i < kNodes; ++i) {
    std::vector<Device> devs(8);
    for (auto& d : devs) {
      d.mem_total = d.mem_avail = 288 * GiB;
    }
    cs.add_node("n" + std::to_string(i), devs, {});
  }
  std::vector<std::string> names;
  for (int i = 0; i < kNodes; ++i) names.push_back("n" + std::to_string(i));

  std::atomic<int> errors{0};
  auto worker = [&](int wid) {
    GPURequest req{GPUUnit{0, 10 + (wid % 5) * 10, 4 * GiB}};
    for (int it = 0; it < 200; ++it) {
      std::string uid = "w" + std::to_string(wid) + "-" + std::to_string(it);
      cs.assume(names, uid, req);
      cs.score(names, uid, req);
      try {
        cs.allocate(names[(wid + it) % kNodes], uid, req);
      } catch (const std::exception&) {
        continue;  // node full: fine
      }
      if (it % 3 != 0) cs.forget_pod(uid);
    }
  };
  std::vector<std::thread> threads;
  for (int w = 0; w < 16; ++w) threads.emplace_back(worker, w);
  // concurrent node churn: add/remove an extra node while scheduling runs
  threads.emplace_back([&] {
    for (int i = 0; i < 50; ++i) {
      std::vector<Device> devs(8);
      for (auto& d : devs) d.mem_total = d.mem_avail = 288 * GiB;
      cs.add_node("hot", devs, {});
      cs.remove_node("hot");
    }
  });
  for (auto& t : threads) t.join();

  // availability must never be negative
  for (const auto& n : names) {
    auto alloc = cs.get(n);
    for (const auto& d : alloc->snapshot()) {
      if (d.core_avail < 0 || d.mem_avail < 0) {
        std::fprintf(stderr, "negative availability on %s\n", n.c_str());
        return 1;
      }
    }
  }

  // ThreadPool::parallel_for teardown race: each iteration
  // creates fresh done_mu/done_cv on the caller's stack; the caller must
  // never outrun the last worker's notify (use-after-destroy). Hammer it
  // with concurrent short fan-outs — exactly the shape that raced.
  {
    ThreadPool pool(4);
    std::vector<std::thread> drivers;
    for (int d = 0; d < 8; ++d) {
      drivers.emplace_back([&pool] {
        std::atomic<int> sum{0};
        for (int it = 0; it < 500; ++it)
          pool.parallel_for(3, [&](int i) { sum.fetch_add(i); });
        (void)sum;
      });
    }
    for (auto& t : drivers) t.join();
  }

  // Large-cluster fan-out: enough nodes that ClusterState::assume takes
  // the POOL path (> kInlineFanout pending searches), with concurrent
  // binds bumping generations so shape-cache hits and misses interleave.
  {
    ClusterState big("binpack", 0, 4);
    std::vector<std::string> bignames;
    for (int i = 0; i < 96; ++i) {
      std::vector<Device> devs(8);
      for (auto& d : devs) d.mem_total = d.mem_avail = 288 * GiB;
      std::string n = "b" + std::to_string(i);
      big.add_node(n, devs, {});
      bignames.push_back(n);
    }
    std::vector<std::thread> ts;
    for (int w = 0; w < 8; ++w) {
      ts.emplace_back([&, w] {
        GPURequest req{GPUUnit{0, 10 + (w % 3) * 10, 2 * GiB}};
        for (int it = 0; it < 60; ++it) {
          std::string uid =
              "big" + std::to_string(w) + "-" + std::to_string(it);
          big.assume(bignames, uid, req);
          big.score(bignames, uid, req);
          try {
            big.allocate(bignames[(w * 31 + it) % bignames.size()], uid, req);
          } catch (const std::exception&) {
          }
          if (it % 2) big.forget_pod(uid);
        }
      });
    }
    for (auto& t : ts) t.join();
  }

  std::printf("stress ok\n");
  return errors.load() ? 1 : 0;
}
