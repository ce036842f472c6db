#include "span_cache.h"

#include <hip/hip_runtime.h>

#include <atomic>
#include <condition_variable>
#include <list>
#include <map>
#include <mutex>

#include "../memory.h"

namespace sca {

CachedSpan::~CachedSpan() {
  // The allocator generation can have rotated (executor torn down and
  // re-initialized) while a stale handle lived on; freeing into the new
  // allocators would corrupt them, so only release same-generation buffers.
  if (ptr && mem_gen == memory_generation()) delete_buffer(dev, ptr);
}

namespace {

struct Cache {
  std::mutex mu;
  std::condition_variable cv;
  size_t budget = 0;
  size_t bytes = 0;
  std::atomic<u64> hits{0}, misses{0};
  // key -> (handle, LRU iterator). LRU list front = most recent.
  struct Entry {
    SpanHandle h;
    std::list<std::pair<i32, SpanKey>>::iterator lru_it;
  };
  std::map<std::pair<i32, SpanKey>, Entry> map;  // keyed (device id, key)
  std::list<std::pair<i32, SpanKey>> lru;
  bool teardown_registered = false;

  // caller holds mu
  void evict_until(size_t need) {
    auto it = lru.end();
    while (bytes + need > budget && it != lru.begin()) {
      --it;
      auto mit = map.find(*it);
      if (mit == map.end()) {
        it = lru.erase(it);
        continue;
      }
      // evict only idle, settled entries (in-flight decodes hold a ref)
      if (mit->second.h.use_count() > 1 || mit->second.h->state == 0)
        continue;
      bytes -= mit->second.h->size;
      map.erase(mit);
      it = lru.erase(it);
    }
  }

  // caller holds mu; frees >= want idle bytes if possible, LRU-first.
  // Returns bytes actually released (allocator-pressure path).
  size_t evict_bytes(size_t want) {
    size_t freed = 0;
    auto it = lru.end();
    while (freed < want && it != lru.begin()) {
      --it;
      auto mit = map.find(*it);
      if (mit == map.end()) {
        it = lru.erase(it);
        continue;
      }
      if (mit->second.h.use_count() > 1 || mit->second.h->state == 0)
        continue;
      freed += mit->second.h->size;
      bytes -= mit->second.h->size;
      map.erase(mit);
      it = lru.erase(it);
    }
    return freed;
  }
};

Cache& cache() {
  static Cache c;
  return c;
}

void drop_all() {
  Cache& c = cache();
  std::lock_guard<std::mutex> l(c.mu);
  c.map.clear();
  c.lru.clear();
  c.bytes = 0;
}

}  // namespace

std::pair<SpanHandle, bool> span_cache_acquire(DeviceHandle dev,
                                               const SpanKey& key,
                                               size_t bytes) {
  Cache& c = cache();
  std::unique_lock<std::mutex> l(c.mu);
  if (c.budget == 0 || bytes > c.budget) return {nullptr, false};
  if (!c.teardown_registered) {
    c.teardown_registered = true;
    register_memory_teardown_callback(drop_all);
    // Under pool pressure the cache is the first thing to shrink: frame
    // working sets must win over cached input spans.
    register_memory_pressure_callback([](size_t want) -> bool {
      Cache& cc = cache();
      std::lock_guard<std::mutex> pl(cc.mu);
      return cc.evict_bytes(want) > 0;
    });
  }
  auto k = std::make_pair(dev.id, key);
  auto it = c.map.find(k);
  if (it != c.map.end()) {
    c.hits.fetch_add(1, std::memory_order_relaxed);
    // bump to MRU
    c.lru.erase(it->second.lru_it);
    c.lru.push_front(k);
    it->second.lru_it = c.lru.begin();
    return {it->second.h, false};
  }
  c.misses.fetch_add(1, std::memory_order_relaxed);
  if (c.bytes + bytes > c.budget) c.evict_until(bytes);
  if (c.bytes + bytes > c.budget) return {nullptr, false};
  auto h = std::make_shared<CachedSpan>();
  h->dev = dev;
  h->size = bytes;
  h->mem_gen = memory_generation();
  // Pool allocation can fail under pressure (PoolAllocator throws) — the
  // cache then simply declines and the caller stays on the host path.
  l.unlock();
  try {
    h->ptr = new_buffer(dev, bytes);
  } catch (const std::exception&) {
    return {nullptr, false};
  }
  l.lock();
  // lost a race? another thread may have inserted while we allocated
  auto it2 = c.map.find(k);
  if (it2 != c.map.end()) {
    c.hits.fetch_add(1, std::memory_order_relaxed);
    return {it2->second.h, false};
  }
  c.lru.push_front(k);
  c.map[k] = {h, c.lru.begin()};
  c.bytes += bytes;
  return {h, true};
}

void span_mark_ready(const SpanHandle& h) {
  Cache& c = cache();
  {
    std::lock_guard<std::mutex> l(c.mu);
    h->state = 1;
  }
  c.cv.notify_all();
}

void span_mark_failed(const SpanHandle& h) {
  Cache& c = cache();
  {
    std::lock_guard<std::mutex> l(c.mu);
    h->state = 2;
    // remove so later acquires retry
    for (auto it = c.map.begin(); it != c.map.end(); ++it) {
      if (it->second.h == h) {
        c.bytes -= h->size;
        c.lru.erase(it->second.lru_it);
        c.map.erase(it);
        break;
      }
    }
  }
  c.cv.notify_all();
}

bool span_wait_ready(const SpanHandle& h) {
  Cache& c = cache();
  std::unique_lock<std::mutex> l(c.mu);
  c.cv.wait(l, [&] { return h->state != 0; });
  return h->state == 1;
}

void span_cache_set_budget(size_t bytes) {
  Cache& c = cache();
  std::lock_guard<std::mutex> l(c.mu);
  c.budget = bytes;
  if (c.bytes > c.budget) c.evict_until(0);
}

size_t span_cache_budget() {
  Cache& c = cache();
  std::lock_guard<std::mutex> l(c.mu);
  return c.budget;
}

void span_cache_clear() { drop_all(); }

void span_cache_upload(const SpanHandle& h, const u8* host, size_t bytes) {
  // Per-thread per-device upload streams: load workers serve tasks bound to
  // any GPU, and a HIP stream belongs to the device that was current at
  // creation, so one cached stream per (thread, device).
  thread_local std::map<i32, hipStream_t> streams;
  auto it = streams.find(h->dev.id);
  if (it == streams.end()) {
    int prev = 0;
    (void)hipGetDevice(&prev);
    SCA_CHECK(hipSetDevice(h->dev.id) == hipSuccess, "hipSetDevice failed");
    hipStream_t s;
    SCA_CHECK(hipStreamCreateWithFlags(&s, hipStreamNonBlocking) == hipSuccess,
              "upload stream create failed");
    (void)hipSetDevice(prev);
    it = streams.emplace(h->dev.id, s).first;
  }
  hipError_t e = hipMemcpyAsync(h->ptr, host, bytes, hipMemcpyHostToDevice,
                                it->second);
  if (e == hipSuccess) e = hipStreamSynchronize(it->second);
  if (e != hipSuccess) {
    span_mark_failed(h);
    throw ScannerError(std::string("span upload failed: ") +
                       hipGetErrorString(e));
  }
  span_mark_ready(h);
}

u64 span_cache_hits() { return cache().hits.load(std::memory_order_relaxed); }
u64 span_cache_misses() {
  return cache().misses.load(std::memory_order_relaxed);
}
size_t span_cache_bytes_live() {
  Cache& c = cache();
  std::lock_guard<std::mutex> l(c.mu);
  return c.bytes;
}

}  // namespace sca
