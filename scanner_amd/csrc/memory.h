// Three-level memory system for the MI355X engine.
//
// Capability parity with the reference's scanner/util/memory.{h,cpp}
// (SystemAllocator / PoolAllocator / BlockAllocator, ref-counted block
// buffers, copy-or-ref semantics), re-designed for MI355X:
//   * GPU allocations come from one up-front hipMalloc slab per device —
//     hipMalloc synchronizes the device, so steady-state allocation must
//     never touch the driver. Slabs are sized for 288 GB HBM3E.
//   * CPU pool memory is allocated with hipHostMalloc (pinned) whenever a
//     GPU is visible, so every host<->device transfer is a true async DMA
//     with no bounce staging (the reference staged through 32 MB pinned
//     buffers instead).
//   * Copies run on a pool of per-thread HIP streams (hipMemcpyAsync) and
//     same-block element copies are coalesced into one memcpy, like the
//     reference's memcpy_vec.
#pragma once

#include <functional>
#include <map>
#include <mutex>
#include <unordered_map>
#include <vector>

#include "common.h"

namespace sca {

class Allocator {
 public:
  virtual ~Allocator() = default;
  virtual u8* allocate(size_t size) = 0;
  virtual void free(u8* ptr) = 0;
};

// Direct allocator: new/delete on CPU, hipMalloc/hipFree on GPU,
// hipHostMalloc for pinned host memory.
class SystemAllocator : public Allocator {
 public:
  SystemAllocator(DeviceHandle device, bool pinned);
  ~SystemAllocator() override;
  u8* allocate(size_t size) override;
  void free(u8* ptr) override;

 private:
  DeviceHandle device_;
  bool pinned_;
};

// Single-slab pool with best-fit free list + coalescing. All engine
// allocations in steady state come from here.
class PoolAllocator : public Allocator {
 public:
  PoolAllocator(DeviceHandle device, SystemAllocator* system, size_t pool_size);
  ~PoolAllocator() override;
  u8* allocate(size_t size) override;
  void free(u8* ptr) override;

  size_t bytes_in_use() const;
  size_t pool_size() const { return pool_size_; }

 private:
  static constexpr size_t kAlign = 256;
  DeviceHandle device_;
  SystemAllocator* system_;
  size_t pool_size_;
  u8* slab_ = nullptr;
  mutable std::mutex mu_;
  // offset -> size for free and used chunks
  std::map<size_t, size_t> free_chunks_;
  std::unordered_map<size_t, size_t> used_chunks_;
};

// Ref-counted allocations. A "block buffer" holds a whole batch of elements
// in one allocation; each element's pointer is an interior pointer into the
// block, and the block is freed when the refcount (initialized to the number
// of elements) reaches zero. Matches the zero-copy semantics the reference's
// engine relies on (copy_or_ref_buffers).
class BlockAllocator {
 public:
  explicit BlockAllocator(std::unique_ptr<Allocator> base);

  u8* allocate(size_t size, i32 refs);
  // True if ptr points anywhere inside one of our live allocations.
  bool owns(const u8* ptr) const;
  void add_ref(const u8* ptr, i32 n = 1);
  void release(const u8* ptr);

  size_t num_live() const;
  // Live/peak byte accounting (streaming-executor memory tests).
  size_t bytes_live() const;
  size_t bytes_peak() const;
  void reset_peak();

 private:
  struct Block {
    u8* base;
    size_t size;
    i32 refs;
  };
  size_t live_bytes_ = 0;
  size_t peak_bytes_ = 0;
  // Find block containing ptr; caller holds mu_. Returns iterator into
  // blocks_ (keyed by base address).
  std::map<const u8*, Block>::iterator find_block(const u8* ptr);

  std::unique_ptr<Allocator> base_;
  mutable std::mutex mu_;
  std::map<const u8*, Block> blocks_;
};

// ---- global memory API (mirrors memory.h:39-76 in the reference) ----

struct MemoryConfig {
  // GPU pool bytes per device; 0 = system allocator only (no pool).
  size_t gpu_pool_size = 0;
  // CPU pool bytes; 0 = system allocator only.
  size_t cpu_pool_size = 0;
  // Pin CPU pool with hipHostMalloc when a GPU is present.
  bool pin_cpu_pool = true;
  std::vector<i32> gpu_ids;
};

void init_memory_allocators(const MemoryConfig& cfg);
void destroy_memory_allocators();
bool memory_initialized();
// Increments on every init_memory_allocators; buffers allocated under an
// older generation must not be freed into the current allocators.
u64 memory_generation();

// Callbacks run at the START of destroy_memory_allocators() so caches
// holding allocator-backed buffers (e.g. the DNN model cache) can release
// them while the allocators still exist. Survives re-init.
void register_memory_teardown_callback(std::function<void()> cb);

// Pressure callbacks run when a pool allocation fails; a cache (HBM span
// cache) frees idle entries and returns true if it released anything, and
// the allocation is retried once. Survives re-init.
void register_memory_pressure_callback(std::function<bool(size_t)> cb);
bool run_memory_pressure_callbacks(size_t want);

u8* new_buffer(DeviceHandle device, size_t size);
u8* new_block_buffer(DeviceHandle device, size_t size, i32 refs);
void add_buffer_ref(DeviceHandle device, u8* buffer, i32 n = 1);
void delete_buffer(DeviceHandle device, u8* buffer);

// Synchronous-on-return copy between any two devices (uses a per-thread HIP
// stream internally for D<->H and D<->D).
void memcpy_buffer(u8* dest, DeviceHandle dest_device, const u8* src,
                   DeviceHandle src_device, size_t size);

// Batched copy: coalesces runs of adjacent src/dest and issues all chunks
// async on the per-thread stream before one sync.
void memcpy_vec(const std::vector<u8*>& dests, DeviceHandle dest_device,
                const std::vector<const u8*>& srcs, DeviceHandle src_device,
                const std::vector<size_t>& sizes);

size_t gpu_pool_bytes_in_use(i32 gpu_id);

// Live/peak block-allocated bytes for a device (peak resets via
// mem_reset_peak; used by the streaming-executor memory-bound tests).
size_t mem_bytes_live(DeviceHandle dev);
size_t mem_bytes_peak(DeviceHandle dev);
void mem_reset_peak(DeviceHandle dev);

// The calling thread's HIP stream (lazily created). All of a pipeline
// instance's kernels and copies are issued on its thread's stream, so
// ordering within an instance is by construction.
void* per_thread_hip_stream();
void sync_per_thread_stream();
// Auxiliary per-thread streams (i mod 4) for intra-op parallel chains
// (GOP-parallel decode); fork/join with events against the main stream.
void* per_thread_aux_stream(int i);

}  // namespace sca
