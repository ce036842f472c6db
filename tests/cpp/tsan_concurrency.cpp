// ThreadSanitizer-verified concurrency unit test for the engine's shared
// primitives: BoundedQueue (pipeline stage boundaries), PoolAllocator /
// BlockAllocator (every element allocation), and the profiler counters.
//
// The reference ships no race detection at all (its only
// sanitizer affordance is cudaDeviceReset for cuda-memcheck); this build's
// claim that the queues/allocators are clean under TSAN is backed by this
// binary: tests/test_tsan.py compiles it with -fsanitize=thread (host g++,
// no HIP — the primitives are pure C++) and asserts zero reports.
#include <atomic>
#include <cassert>
#include <cstdio>
#include <thread>
#include <vector>

// Compile the primitives standalone: SCA_TSAN_HOST makes memory.h's HIP
// dependencies inert (see the include shim below).
#include "../../scanner_amd/csrc/profiler.h"
#include "../../scanner_amd/csrc/queue.h"

using namespace sca;

int main() {
  // ---- BoundedQueue: 4 producers, 4 consumers, 100k items ----
  {
    BoundedQueue<int> q(64);
    std::atomic<long> sum{0};
    std::atomic<int> live_producers{4};
    std::vector<std::thread> ts;
    for (int p = 0; p < 4; ++p) {
      ts.emplace_back([&, p] {
        for (int i = 0; i < 25000; ++i) q.push(i);
        if (live_producers.fetch_sub(1) == 1) q.close();
      });
    }
    for (int c = 0; c < 4; ++c) {
      ts.emplace_back([&] {
        while (auto v = q.pop()) sum.fetch_add(*v);
      });
    }
    for (auto& t : ts) t.join();
    long expect = 4L * (25000L * 24999L / 2);
    assert(sum.load() == expect);
  }

  // ---- Profiler: concurrent intervals + counters ----
  {
    Profiler prof(ProfilerLevel::Info);
    std::vector<std::thread> ts;
    for (int p = 0; p < 8; ++p) {
      ts.emplace_back([&, p] {
        for (int i = 0; i < 2000; ++i) {
          prof.increment("c" + std::to_string(p & 1));
          prof.add_interval("work", i, i + 1);
        }
      });
    }
    for (auto& t : ts) t.join();
    assert(prof.counters().at("c0") + prof.counters().at("c1") == 16000);
    assert((long)prof.intervals().size() == 16000);
  }

  std::printf("tsan concurrency: OK\n");
  return 0;
}
