// First-party HIP/CDNA4 image ops for gfx950 (MI355X): histogram, resize.
// (Capability parity: the reference keeps GPU image ops in the external
// scannertools repo; scanner/util/image.cu holds only NV12->RGB + HWC->CHW,
// whose HIP equivalents live in svc_codec.hip / color.hip.)
//
// Design notes (per /opt/skills/guides/cdna_hip_programming.md):
//  * wave64; block sizes are multiples of 64 (256 default).
//  * memory-bound ops use dword4-vectorized grid-stride loops, grid capped
//    at ~2048 workgroups so the scheduler keeps all 8 XCDs fed.
//  * histogram: per-workgroup LDS histogram + one global atomic merge per
//    bin (Guideline 12 — per-block reduction before atomics).
#include <hip/hip_runtime.h>

#include "../csrc/memory.h"
#include "../csrc/msgpack.h"
#include "../csrc/ops/kernel.h"

namespace sca {

namespace {

inline hipStream_t cur_stream() {
  return (hipStream_t)per_thread_hip_stream();
}

#define HIPK_CHECK(expr)                                                 \
  do {                                                                   \
    hipError_t _e = (expr);                                              \
    if (_e != hipSuccess) {                                              \
      throw ScannerError(std::string("HIP error in op kernel: ") +       \
                         hipGetErrorString(_e));                         \
    }                                                                    \
  } while (0)

// ---------------- histogram ----------------
// RGB24 (HWC u8, c==3) -> 3x256 u32, batched over a work packet.
// Two-stage: (1) per-block LDS histogram of a frame chunk -> partials in
// scratch (plain stores, no global atomics — 2048 blocks atomically merging
// into one 3 KB output was the first profile's hotspot); (2) tiny reduce
// kernel sums partials per frame. Frames arrive as a device pointer array
// (engine elements are separate allocations); outputs go to one block
// buffer, one 3 KB slice per frame.
__global__ void __launch_bounds__(256)
    histogram_rgb_partial_kernel(const u8* const* __restrict__ frames,
                                 u64 nbytes, u32* __restrict__ partials,
                                 u32 chunks) {
  // 4-way sub-histogram replication: real video is spatially smooth, so
  // neighboring lanes hit the SAME bin and serialize on one LDS atomic
  // (measured 2.1 TB/s); spreading lanes over 4 copies by (tid & 3) cuts
  // that contention 4x for 12 KB of LDS.
  __shared__ u32 lhist[4][3 * 256];
  u32* lh_flat = &lhist[0][0];
  for (u32 i = threadIdx.x; i < 4 * 3 * 256; i += blockDim.x) lh_flat[i] = 0;
  __syncthreads();
  u32 rep = threadIdx.x & 3;

  u32 frame = blockIdx.y;
  u32 chunk = blockIdx.x;
  const u8* in = frames[frame];
  const uint4* in16 = reinterpret_cast<const uint4*>(in);
  u64 nvec = nbytes / 16;
  // grid-stride over this frame's vectors, chunk-interleaved so chunks
  // read coalesced interleaved spans
  for (u64 v = chunk * 256 + threadIdx.x; v < nvec;
       v += (u64)chunks * 256) {
    uint4 x = in16[v];
    u64 byte0 = v * 16;
    const u32 words[4] = {x.x, x.y, x.z, x.w};
#pragma unroll
    for (int wi = 0; wi < 4; ++wi) {
      u32 wv = words[wi];
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        u32 ch = (u32)((byte0 + wi * 4 + k) % 3);
        atomicAdd(&lhist[rep][ch * 256 + ((wv >> (8 * k)) & 0xff)], 1u);
      }
    }
  }
  // tail bytes (nbytes % 16) handled by chunk 0
  if (chunk == 0) {
    u64 t0 = nvec * 16;
    if (threadIdx.x < nbytes - t0) {
      u64 i = t0 + threadIdx.x;
      atomicAdd(&lhist[rep][(u32)(i % 3) * 256 + in[i]], 1u);
    }
  }
  __syncthreads();
  u32* dst = partials + ((u64)frame * chunks + chunk) * 768;
  for (u32 i = threadIdx.x; i < 768; i += blockDim.x) {
    dst[i] = lhist[0][i] + lhist[1][i] + lhist[2][i] + lhist[3][i];
  }
}

// Merge partials: blockIdx.x spans frames, blockIdx.y splits the chunk
// dimension so the launch fills the chip (a frame-only grid left 16 of
// 256 CUs busy and ran at 0.1 TB/s); each block sums its chunk slice and
// atomically accumulates into the (pre-zeroed) output.
__global__ void __launch_bounds__(256)
    histogram_reduce_kernel(const u32* __restrict__ partials, u32 chunks,
                            u32 chunks_per_block, u32* __restrict__ out) {
  u32 frame = blockIdx.x;
  u32 c0 = blockIdx.y * chunks_per_block;
  u32 c1 = min(c0 + chunks_per_block, chunks);
  bool sole = gridDim.y == 1;
  for (u32 i = threadIdx.x; i < 768; i += blockDim.x) {
    u32 sum = 0;
    for (u32 c = c0; c < c1; ++c) {
      sum += partials[((u64)frame * chunks + c) * 768 + i];
    }
    if (sole) {
      out[(u64)frame * 768 + i] = sum;
    } else if (sum) {
      atomicAdd(&out[(u64)frame * 768 + i], sum);
    }
  }
}

// Generic channel count fallback (c != 3 or c == 1).
__global__ void __launch_bounds__(256)
    histogram_any_kernel(const u8* __restrict__ in, u64 npixels, u32 c,
                         u32* __restrict__ out) {
  extern __shared__ u32 lh[];
  for (u32 i = threadIdx.x; i < c * 256; i += blockDim.x) lh[i] = 0;
  __syncthreads();
  u64 stride = (u64)gridDim.x * blockDim.x;
  for (u64 p = (u64)blockIdx.x * blockDim.x + threadIdx.x; p < npixels;
       p += stride) {
    for (u32 ch = 0; ch < c; ++ch)
      atomicAdd(&lh[ch * 256 + in[p * c + ch]], 1u);
  }
  __syncthreads();
  for (u32 i = threadIdx.x; i < c * 256; i += blockDim.x) {
    if (lh[i]) atomicAdd(&out[i], lh[i]);
  }
}

class HistogramKernelGPU : public BatchedKernel {
 public:
  using BatchedKernel::BatchedKernel;
  void execute_batch(const BatchedElements& in, BatchedElements& out) override {
    hipStream_t s = cur_stream();
    size_t n = in[0].size();
    if (n == 0) return;
    const Element& f0 = in[0][0];
    SCA_CHECK(f0.is_frame && f0.device.is_gpu(),
              "GPU Histogram needs GPU frame input");
    i32 c = f0.frame_info.shape[2];
    u64 nbytes = f0.frame_info.size();
    if (c != 3) {  // per-frame fallback for non-RGB
      for (const Element& f : in[0]) {
        size_t out_size = (size_t)c * 256 * sizeof(u32);
        Element e;
        e.buffer = new_buffer(config_.device, out_size);
        e.size = out_size;
        e.device = config_.device;
        HIPK_CHECK(hipMemsetAsync(e.buffer, 0, out_size, s));
        i64 npix = (i64)f.frame_info.shape[0] * f.frame_info.shape[1];
        int blocks = (int)std::min<i64>(2048, (npix + 255) / 256);
        histogram_any_kernel<<<blocks, 256, c * 256 * 4, s>>>(
            f.buffer, (u64)npix, (u32)c, (u32*)e.buffer);
        HIPK_CHECK(hipGetLastError());
        out[0].push_back(e);
      }
      return;
    }
    // chunks sized so the whole batch fills the chip (>=2048 workgroups
    // when the batch allows)
    u32 chunks = (u32)std::max<size_t>(1, 2048 / n);
    std::vector<const u8*> ptrs(n);
    for (size_t i = 0; i < n; ++i) {
      SCA_CHECK(in[0][i].frame_info.size() == nbytes,
                "Histogram batch with mixed frame sizes");
      ptrs[i] = in[0][i].buffer;
    }
    u8* d_ptrs = new_buffer(config_.device, n * sizeof(u8*));
    HIPK_CHECK(hipMemcpyAsync(d_ptrs, ptrs.data(), n * sizeof(u8*),
                              hipMemcpyHostToDevice, s));
    u8* scratch = new_buffer(config_.device, (u64)n * chunks * 768 * 4);
    u8* out_block =
        new_block_buffer(config_.device, (u64)n * 768 * 4, (i32)n);
    dim3 grid(chunks, (u32)n);
    histogram_rgb_partial_kernel<<<grid, 256, 0, s>>>(
        (const u8* const*)d_ptrs, nbytes, (u32*)scratch, chunks);
    HIPK_CHECK(hipGetLastError());
    // reduce fills the chip by splitting the chunk dimension; >1 slice
    // needs a zeroed output for the atomic merge
    u32 gy = std::max<u32>(1, std::min<u32>(chunks,
                                            (u32)(512 / std::max<size_t>(
                                                      1, n))));
    u32 cpb = (chunks + gy - 1) / gy;
    gy = (chunks + cpb - 1) / cpb;
    if (gy > 1) {
      HIPK_CHECK(hipMemsetAsync(out_block, 0, (u64)n * 768 * 4, s));
    }
    histogram_reduce_kernel<<<dim3((u32)n, gy), 256, 0, s>>>(
        (const u32*)scratch, chunks, cpb, (u32*)out_block);
    HIPK_CHECK(hipGetLastError());
    // scratch + pointer array feed kernels on this stream; sync before
    // returning them to the (shared) pool.
    HIPK_CHECK(hipStreamSynchronize(s));
    delete_buffer(config_.device, scratch);
    delete_buffer(config_.device, d_ptrs);
    for (size_t i = 0; i < n; ++i) {
      Element e;
      e.buffer = out_block + (u64)i * 768 * 4;
      e.size = 768 * 4;
      e.device = config_.device;
      out[0].push_back(e);
    }
  }
};

// ---------------- resize (bilinear u8, any channel count) ----------------
__global__ void __launch_bounds__(256)
    resize_bilinear_kernel(const u8* __restrict__ src, int sh, int sw,
                           u8* __restrict__ dst, int dh, int dw, int c) {
  u64 total = (u64)dh * dw * c;
  u64 stride = (u64)gridDim.x * blockDim.x;
  for (u64 i = (u64)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int ch = (int)(i % c);
    u64 pix = i / c;
    int x = (int)(pix % dw);
    int y = (int)(pix / dw);
    float sy = (y + 0.5f) * sh / dh - 0.5f;
    float sx = (x + 0.5f) * sw / dw - 0.5f;
    int y0 = max(0, (int)floorf(sy));
    int x0 = max(0, (int)floorf(sx));
    int y1 = min(sh - 1, y0 + 1);
    int x1 = min(sw - 1, x0 + 1);
    float fy = fminf(fmaxf(sy - y0, 0.f), 1.f);
    float fx = fminf(fmaxf(sx - x0, 0.f), 1.f);
    float v00 = src[((u64)y0 * sw + x0) * c + ch];
    float v01 = src[((u64)y0 * sw + x1) * c + ch];
    float v10 = src[((u64)y1 * sw + x0) * c + ch];
    float v11 = src[((u64)y1 * sw + x1) * c + ch];
    float v = v00 * (1 - fy) * (1 - fx) + v01 * (1 - fy) * fx +
              v10 * fy * (1 - fx) + v11 * fy * fx;
    dst[i] = (u8)(v + 0.5f);
  }
}

class ResizeKernelGPU : public BatchedKernel {
 public:
  explicit ResizeKernelGPU(const KernelConfig& cfg) : BatchedKernel(cfg) {
    auto a = mp::decode(cfg.args);
    out_w_ = (i32)a.get_int("width", 0);
    out_h_ = (i32)a.get_int("height", 0);
    SCA_CHECK(out_w_ > 0 && out_h_ > 0, "Resize needs width/height args");
  }
  void execute_batch(const BatchedElements& in, BatchedElements& out) override {
    hipStream_t s = cur_stream();
    for (const Element& f : in[0]) {
      SCA_CHECK(f.is_frame && f.device.is_gpu(), "GPU Resize needs GPU input");
      i32 h = f.frame_info.shape[0], w = f.frame_info.shape[1],
          c = f.frame_info.shape[2];
      Element e;
      e.is_frame = true;
      e.frame_info.shape[0] = out_h_;
      e.frame_info.shape[1] = out_w_;
      e.frame_info.shape[2] = c;
      e.frame_info.type = FrameType::U8;
      e.size = e.frame_info.size();
      e.buffer = new_buffer(config_.device, e.size);
      e.device = config_.device;
      u64 total = (u64)out_h_ * out_w_ * c;
      int blocks = (int)std::min<u64>(2048, (total + 255) / 256);
      resize_bilinear_kernel<<<blocks, 256, 0, s>>>(f.buffer, h, w, e.buffer,
                                                    out_h_, out_w_, c);
      HIPK_CHECK(hipGetLastError());
      out[0].push_back(e);
    }
  }

 private:
  i32 out_w_, out_h_;
};

}  // namespace

void register_gpu_ops() {
  static bool done = false;
  if (done) return;
  done = true;
  {
    KernelFactory f;
    f.op_name = "Histogram";
    f.device_type = DeviceType::GPU;
    f.preferred_batch = 16;
    f.make = [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
      return std::make_unique<HistogramKernelGPU>(c);
    };
    kernel_registry().add(f);
  }
  {
    KernelFactory f;
    f.op_name = "Resize";
    f.device_type = DeviceType::GPU;
    f.preferred_batch = 16;
    f.make = [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
      return std::make_unique<ResizeKernelGPU>(c);
    };
    kernel_registry().add(f);
  }
}

}  // namespace sca
