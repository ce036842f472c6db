#include "memory.h"

#include <atomic>

#include <algorithm>
#include <cstdlib>

#include "hip_util.h"

namespace sca {

namespace {
std::atomic<u64> g_mem_generation{0};
}  // namespace

// ---------------- SystemAllocator ----------------

SystemAllocator::SystemAllocator(DeviceHandle device, bool pinned)
    : device_(device), pinned_(pinned) {}

SystemAllocator::~SystemAllocator() = default;

u8* SystemAllocator::allocate(size_t size) {
  if (size == 0) size = 1;
  if (device_.is_gpu()) {
    DeviceGuard g(device_.id);
    void* p = nullptr;
    HIP_CHECK(hipMalloc(&p, size));
    return static_cast<u8*>(p);
  }
  if (pinned_ && have_gpu()) {
    void* p = nullptr;
    HIP_CHECK(hipHostMalloc(&p, size, hipHostMallocDefault));
    return static_cast<u8*>(p);
  }
  void* p = nullptr;
  if (posix_memalign(&p, 256, size) != 0) {
    throw ScannerError("posix_memalign failed for " + std::to_string(size));
  }
  return static_cast<u8*>(p);
}

void SystemAllocator::free(u8* ptr) {
  if (ptr == nullptr) return;
  if (device_.is_gpu()) {
    DeviceGuard g(device_.id);
    (void)hipFree(ptr);
    return;
  }
  if (pinned_ && have_gpu()) {
    (void)hipHostFree(ptr);
    return;
  }
  std::free(ptr);
}

// ---------------- PoolAllocator ----------------

PoolAllocator::PoolAllocator(DeviceHandle device, SystemAllocator* system,
                             size_t pool_size)
    : device_(device), system_(system), pool_size_(pool_size) {
  slab_ = system_->allocate(pool_size_);
  free_chunks_[0] = pool_size_;
}

PoolAllocator::~PoolAllocator() { system_->free(slab_); }

u8* PoolAllocator::allocate(size_t size) {
  size_t want = (size + kAlign - 1) / kAlign * kAlign;
  if (want == 0) want = kAlign;
  std::lock_guard<std::mutex> l(mu_);
  // best fit
  auto best = free_chunks_.end();
  for (auto it = free_chunks_.begin(); it != free_chunks_.end(); ++it) {
    if (it->second >= want &&
        (best == free_chunks_.end() || it->second < best->second)) {
      best = it;
    }
  }
  if (best == free_chunks_.end()) {
    throw ScannerError("PoolAllocator (" + device_.to_string() +
                       ") out of memory: want " + std::to_string(want) +
                       " bytes, pool " + std::to_string(pool_size_));
  }
  size_t off = best->first;
  size_t chunk = best->second;
  free_chunks_.erase(best);
  if (chunk > want) {
    free_chunks_[off + want] = chunk - want;
  }
  used_chunks_[off] = want;
  return slab_ + off;
}

void PoolAllocator::free(u8* ptr) {
  if (ptr == nullptr) return;
  std::lock_guard<std::mutex> l(mu_);
  size_t off = static_cast<size_t>(ptr - slab_);
  auto it = used_chunks_.find(off);
  SCA_CHECK(it != used_chunks_.end(), "free of unknown pool pointer");
  size_t sz = it->second;
  used_chunks_.erase(it);
  // insert and coalesce with neighbors
  auto ins = free_chunks_.emplace(off, sz).first;
  if (ins != free_chunks_.begin()) {
    auto prev = std::prev(ins);
    if (prev->first + prev->second == ins->first) {
      prev->second += ins->second;
      free_chunks_.erase(ins);
      ins = prev;
    }
  }
  auto next = std::next(ins);
  if (next != free_chunks_.end() && ins->first + ins->second == next->first) {
    ins->second += next->second;
    free_chunks_.erase(next);
  }
}

size_t PoolAllocator::bytes_in_use() const {
  std::lock_guard<std::mutex> l(mu_);
  size_t total = 0;
  for (auto& kv : used_chunks_) total += kv.second;
  return total;
}

// ---------------- BlockAllocator ----------------

BlockAllocator::BlockAllocator(std::unique_ptr<Allocator> base)
    : base_(std::move(base)) {}

u8* BlockAllocator::allocate(size_t size, i32 refs) {
  SCA_CHECK(refs > 0, "block buffer needs >=1 ref");
  u8* p = nullptr;
  try {
    p = base_->allocate(size);
  } catch (const std::exception&) {
    // Pool pressure: give registered caches (HBM span cache) a chance to
    // release idle entries, then retry once before propagating.
    if (!run_memory_pressure_callbacks(size)) throw;
    p = base_->allocate(size);
  }
  std::lock_guard<std::mutex> l(mu_);
  blocks_.emplace(p, Block{p, size == 0 ? 1 : size, refs});
  live_bytes_ += size == 0 ? 1 : size;
  if (live_bytes_ > peak_bytes_) peak_bytes_ = live_bytes_;
  return p;
}

std::map<const u8*, BlockAllocator::Block>::iterator BlockAllocator::find_block(
    const u8* ptr) {
  // blocks_ keyed by base; upper_bound-1 gives candidate containing block.
  auto it = blocks_.upper_bound(ptr);
  if (it == blocks_.begin()) return blocks_.end();
  --it;
  const Block& b = it->second;
  if (ptr >= b.base && ptr < b.base + b.size) return it;
  return blocks_.end();
}

bool BlockAllocator::owns(const u8* ptr) const {
  auto* self = const_cast<BlockAllocator*>(this);
  std::lock_guard<std::mutex> l(mu_);
  return self->find_block(ptr) != self->blocks_.end();
}

void BlockAllocator::add_ref(const u8* ptr, i32 n) {
  std::lock_guard<std::mutex> l(mu_);
  auto it = find_block(ptr);
  SCA_CHECK(it != blocks_.end(), "add_ref on unknown buffer");
  it->second.refs += n;
}

void BlockAllocator::release(const u8* ptr) {
  u8* to_free = nullptr;
  {
    std::lock_guard<std::mutex> l(mu_);
    auto it = find_block(ptr);
    SCA_CHECK(it != blocks_.end(), "release of unknown buffer");
    if (--it->second.refs == 0) {
      to_free = it->second.base;
      live_bytes_ -= it->second.size;
      blocks_.erase(it);
    }
  }
  if (to_free) base_->free(to_free);
}

size_t BlockAllocator::num_live() const {
  std::lock_guard<std::mutex> l(mu_);
  return blocks_.size();
}

size_t BlockAllocator::bytes_live() const {
  std::lock_guard<std::mutex> l(mu_);
  return live_bytes_;
}

size_t BlockAllocator::bytes_peak() const {
  std::lock_guard<std::mutex> l(mu_);
  return peak_bytes_;
}

void BlockAllocator::reset_peak() {
  std::lock_guard<std::mutex> l(mu_);
  peak_bytes_ = live_bytes_;
}

// ---------------- global state ----------------

namespace {

struct DeviceAllocators {
  std::unique_ptr<SystemAllocator> system;
  std::unique_ptr<PoolAllocator> pool;  // may be null
  std::unique_ptr<BlockAllocator> block;
  Allocator* data() {  // allocator used for plain buffers
    return pool ? static_cast<Allocator*>(pool.get())
                : static_cast<Allocator*>(system.get());
  }
};

struct GlobalMemory {
  bool initialized = false;
  DeviceAllocators cpu;
  std::unordered_map<i32, DeviceAllocators> gpus;
  std::mutex mu;
};

GlobalMemory g_mem;

// An adapter so the BlockAllocator can share the pool (or system) allocator
// without owning it.
class BorrowedAllocator : public Allocator {
 public:
  explicit BorrowedAllocator(Allocator* a) : a_(a) {}
  u8* allocate(size_t size) override { return a_->allocate(size); }
  void free(u8* ptr) override { a_->free(ptr); }

 private:
  Allocator* a_;
};

void setup_device(DeviceAllocators& da, DeviceHandle dev, size_t pool_size,
                  bool pinned);

DeviceAllocators& allocators_for(DeviceHandle device) {
  if (!g_mem.initialized) {
    // Lazy default init (CPU only, no pools) so unit tests and simple tools
    // work without an explicit init call.
    MemoryConfig cfg;
    init_memory_allocators(cfg);
  }
  if (!device.is_gpu()) return g_mem.cpu;
  auto it = g_mem.gpus.find(device.id);
  if (it == g_mem.gpus.end()) {
    // GPU not covered by the explicit init (or lazy default init): give it
    // system+block allocators (no pool) on demand.
    std::lock_guard<std::mutex> l(g_mem.mu);
    it = g_mem.gpus.find(device.id);
    if (it == g_mem.gpus.end()) {
      SCA_CHECK(device.id < gpu_device_count(),
                "no such GPU " + std::to_string(device.id));
      DeviceAllocators da;
      setup_device(da, device, 0, false);
      it = g_mem.gpus.emplace(device.id, std::move(da)).first;
    }
  }
  return it->second;
}

void setup_device(DeviceAllocators& da, DeviceHandle dev, size_t pool_size,
                  bool pinned) {
  da.system = std::make_unique<SystemAllocator>(dev, pinned);
  if (pool_size > 0) {
    da.pool = std::make_unique<PoolAllocator>(dev, da.system.get(), pool_size);
  }
  da.block =
      std::make_unique<BlockAllocator>(std::make_unique<BorrowedAllocator>(da.data()));
}

// Per-thread HIP stream for copies (created lazily; reference rotated 32
// global streams — per-thread is simpler and at least as parallel since each
// pipeline stage is its own thread).
hipStream_t thread_copy_stream() {
  thread_local hipStream_t stream = nullptr;
  if (stream == nullptr) {
    HIP_CHECK(hipStreamCreateWithFlags(&stream, hipStreamNonBlocking));
  }
  return stream;
}

}  // namespace

void init_memory_allocators(const MemoryConfig& cfg) {
  std::lock_guard<std::mutex> l(g_mem.mu);
  if (g_mem.initialized) return;
  setup_device(g_mem.cpu, CPU_DEVICE, cfg.cpu_pool_size, cfg.pin_cpu_pool);
  for (i32 id : cfg.gpu_ids) {
    DeviceAllocators da;
    setup_device(da, DeviceHandle{DeviceType::GPU, id}, cfg.gpu_pool_size,
                 false);
    g_mem.gpus.emplace(id, std::move(da));
  }
  g_mem.initialized = true;
  g_mem_generation.fetch_add(1);
}

namespace {
std::mutex g_teardown_mu;
std::vector<std::function<void()>>& teardown_callbacks() {
  static std::vector<std::function<void()>> cbs;
  return cbs;
}
std::mutex g_pressure_mu;
std::vector<std::function<bool(size_t)>>& pressure_callbacks() {
  static std::vector<std::function<bool(size_t)>> cbs;
  return cbs;
}
}  // namespace

void register_memory_teardown_callback(std::function<void()> cb) {
  std::lock_guard<std::mutex> l(g_teardown_mu);
  teardown_callbacks().push_back(std::move(cb));
}

void register_memory_pressure_callback(std::function<bool(size_t)> cb) {
  std::lock_guard<std::mutex> l(g_pressure_mu);
  pressure_callbacks().push_back(std::move(cb));
}

bool run_memory_pressure_callbacks(size_t want) {
  std::lock_guard<std::mutex> l(g_pressure_mu);
  bool any = false;
  for (auto& cb : pressure_callbacks()) any |= cb(want);
  return any;
}

void destroy_memory_allocators() {
  // Let caches drop allocator-backed buffers while allocators still exist
  // (outside g_mem.mu: the callbacks call delete_buffer).
  {
    std::lock_guard<std::mutex> l(g_teardown_mu);
    for (auto& cb : teardown_callbacks()) cb();
  }
  std::vector<i32> gpu_ids;
  {
    std::lock_guard<std::mutex> l(g_mem.mu);
    for (auto& kv : g_mem.gpus) gpu_ids.push_back(kv.first);
    g_mem.gpus.clear();
    g_mem.cpu = DeviceAllocators{};
    g_mem.initialized = false;
  }
  // Sanitizer affordance (parity: the reference's SystemAllocator dtor
  // calls cudaDeviceReset so cuda-memcheck can flush its state,
  // memory.cpp:110-117): SCANNER_DEVICE_RESET=1 resets every device at
  // teardown so rocprof-compute / sanitizer runs see a clean shutdown.
  static const bool kReset = []() {
    const char* e = std::getenv("SCANNER_DEVICE_RESET");
    return e && e[0] == '1';
  }();
  if (kReset) {
    int prev = 0;
    (void)hipGetDevice(&prev);
    for (i32 id : gpu_ids) {
      (void)hipSetDevice(id);
      (void)hipDeviceReset();
    }
    (void)hipSetDevice(prev);
  }
}

bool memory_initialized() { return g_mem.initialized; }

u64 memory_generation() { return g_mem_generation.load(); }

u8* new_buffer(DeviceHandle device, size_t size) {
  // Plain buffers are block buffers with one ref — uniform delete path.
  return new_block_buffer(device, size, 1);
}

u8* new_block_buffer(DeviceHandle device, size_t size, i32 refs) {
  return allocators_for(device).block->allocate(size, refs);
}

void add_buffer_ref(DeviceHandle device, u8* buffer, i32 n) {
  allocators_for(device).block->add_ref(buffer, n);
}

void delete_buffer(DeviceHandle device, u8* buffer) {
  allocators_for(device).block->release(buffer);
}

void memcpy_buffer(u8* dest, DeviceHandle dest_device, const u8* src,
                   DeviceHandle src_device, size_t size) {
  if (size == 0) return;
  if (!dest_device.is_gpu() && !src_device.is_gpu()) {
    std::memcpy(dest, src, size);
    return;
  }
  hipStream_t s = thread_copy_stream();
  DeviceGuard g(dest_device.is_gpu() ? dest_device.id : src_device.id);
  HIP_CHECK(hipMemcpyAsync(dest, src, size, hipMemcpyDefault, s));
  HIP_CHECK(hipStreamSynchronize(s));
}

void memcpy_vec(const std::vector<u8*>& dests, DeviceHandle dest_device,
                const std::vector<const u8*>& srcs, DeviceHandle src_device,
                const std::vector<size_t>& sizes) {
  SCA_CHECK(dests.size() == srcs.size() && srcs.size() == sizes.size(),
            "memcpy_vec size mismatch");
  size_t n = dests.size();
  if (n == 0) return;
  bool any_gpu = dest_device.is_gpu() || src_device.is_gpu();
  size_t i = 0;
  hipStream_t s = any_gpu ? thread_copy_stream() : nullptr;
  while (i < n) {
    // Coalesce adjacent runs (block buffers lay elements contiguously).
    size_t run = sizes[i];
    size_t j = i + 1;
    while (j < n && srcs[j] == srcs[j - 1] + sizes[j - 1] &&
           dests[j] == dests[j - 1] + sizes[j - 1]) {
      run += sizes[j];
      ++j;
    }
    if (!any_gpu) {
      std::memcpy(dests[i], srcs[i], run);
    } else {
      HIP_CHECK(hipMemcpyAsync(dests[i], srcs[i], run, hipMemcpyDefault, s));
    }
    i = j;
  }
  if (any_gpu) HIP_CHECK(hipStreamSynchronize(s));
}

void* per_thread_hip_stream() { return (void*)thread_copy_stream(); }

// Small per-thread pool of auxiliary streams for intra-op parallelism
// (e.g. GOP-parallel video decode chains). Lazily created, never freed.
void* per_thread_aux_stream(int i) {
  constexpr int kAux = 4;
  thread_local hipStream_t aux[kAux] = {nullptr, nullptr, nullptr, nullptr};
  int k = ((i % kAux) + kAux) % kAux;
  if (aux[k] == nullptr) {
    HIP_CHECK(hipStreamCreateWithFlags(&aux[k], hipStreamNonBlocking));
  }
  return (void*)aux[k];
}

void sync_per_thread_stream() {
  HIP_CHECK(hipStreamSynchronize(thread_copy_stream()));
}

size_t gpu_pool_bytes_in_use(i32 gpu_id) {
  auto it = g_mem.gpus.find(gpu_id);
  if (it == g_mem.gpus.end() || !it->second.pool) return 0;
  return it->second.pool->bytes_in_use();
}

size_t mem_bytes_live(DeviceHandle dev) {
  return allocators_for(dev).block->bytes_live();
}
size_t mem_bytes_peak(DeviceHandle dev) {
  return allocators_for(dev).block->bytes_peak();
}
void mem_reset_peak(DeviceHandle dev) {
  allocators_for(dev).block->reset_peak();
}

}  // namespace sca
