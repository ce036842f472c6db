// HBM-resident LRU cache of compressed video byte spans.
//
// The MI355X has 288 GB of HBM3E per GPU — far more than any working set of
// compressed video — while the host link (~58 GB/s measured) is the
// bottleneck of decode-heavy pipelines. This cache keeps the keyframe-aligned
// encoded GOP spans the engine reads (reference analogue: the encoded ranges
// column_source.cpp:209 reads per task) resident in device memory, so a span
// is read from storage and crosses PCIe exactly once; every later task that
// needs it decodes straight out of HBM. Capability note: the reference has no
// analogue (its data plane re-reads storage every task); this is the
// MI355X-native redesign of the load path.
//
// Concurrency: load workers and pipeline instances race on the same spans.
// acquire() returns (handle, is_new); the is_new caller uploads the bytes and
// calls mark_ready(); everyone else blocks in wait_ready() (uploads are
// a few ms). Handles are shared_ptrs — an entry is evictable only when the
// cache holds the sole reference, and its device buffer is returned to the
// pool when the last handle drops (decoders sync their stream before
// releasing, so the buffer is never freed under an in-flight kernel).
#pragma once

#include <memory>

#include "../common.h"

namespace sca {

struct SpanKey {
  u64 db = 0;  // hash of the database root (table ids are unique per db)
  i32 table = 0;
  i32 col = 0;
  i32 item = 0;
  u64 lo = 0, hi = 0;

  bool operator<(const SpanKey& o) const {
    if (db != o.db) return db < o.db;
    if (table != o.table) return table < o.table;
    if (col != o.col) return col < o.col;
    if (item != o.item) return item < o.item;
    if (lo != o.lo) return lo < o.lo;
    return hi < o.hi;
  }
};

struct CachedSpan {
  u8* ptr = nullptr;
  DeviceHandle dev{DeviceType::CPU, 0};
  size_t size = 0;
  u64 mem_gen = 0;  // memory_generation() at allocation
  ~CachedSpan();

  // state guarded by the cache mutex; see span_cache.cpp
  int state = 0;  // 0=uploading, 1=ready, 2=failed
};

using SpanHandle = std::shared_ptr<CachedSpan>;

// Lookup-or-reserve. Returns {handle, is_new}:
//   * hit: {handle, false} — call span_wait_ready() before reading ptr.
//   * new: {handle, true}  — upload `bytes` into handle->ptr, then
//     span_mark_ready(handle) (or span_mark_failed on error).
//   * {nullptr, false} — caching disabled / span exceeds budget / device
//     pool exhausted; caller uses its host-path fallback.
std::pair<SpanHandle, bool> span_cache_acquire(DeviceHandle dev,
                                               const SpanKey& key,
                                               size_t bytes);
void span_mark_ready(const SpanHandle& h);
void span_mark_failed(const SpanHandle& h);
// Blocks until the uploader finished; returns false if the upload failed
// (entry is already removed; caller falls back to the host path).
bool span_wait_ready(const SpanHandle& h);

// Total byte budget across devices (0 disables caching). Safe to call any
// time; shrinking evicts LRU entries immediately.
void span_cache_set_budget(size_t bytes);
size_t span_cache_budget();
void span_cache_clear();

// Upload helper: async H2D copy of pinned host bytes into h->ptr on a
// per-thread per-device non-blocking stream, synced before returning, then
// mark_ready. Safe from any thread regardless of its current HIP device.
void span_cache_upload(const SpanHandle& h, const u8* host, size_t bytes);

// stats (monotonic, for tests and profiling)
u64 span_cache_hits();
u64 span_cache_misses();
size_t span_cache_bytes_live();

}  // namespace sca
