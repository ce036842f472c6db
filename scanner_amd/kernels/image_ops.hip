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
// RGB24 (HWC u8, c==3) -> 3x256 u32. One LDS histogram per workgroup.
// Input is read as dwords (4 bytes/lane/iter); channel of byte i is i%3.
__global__ void __launch_bounds__(256)
    histogram_rgb_kernel(const u8* __restrict__ in, u64 nbytes,
                         u32* __restrict__ out) {
  __shared__ u32 lhist[3 * 256];
  for (u32 i = threadIdx.x; i < 3 * 256; i += blockDim.x) lhist[i] = 0;
  __syncthreads();

  u64 ndwords = nbytes / 4;
  const u32* in32 = reinterpret_cast<const u32*>(in);
  u64 stride = (u64)gridDim.x * blockDim.x;
  for (u64 d = (u64)blockIdx.x * blockDim.x + threadIdx.x; d < ndwords;
       d += stride) {
    u32 v = in32[d];
    u64 byte0 = d * 4;
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      u32 ch = (u32)((byte0 + k) % 3);
      u32 val = (v >> (8 * k)) & 0xff;
      atomicAdd(&lhist[ch * 256 + val], 1u);
    }
  }
  // tail bytes
  if (blockIdx.x == 0 && threadIdx.x < nbytes - ndwords * 4) {
    u64 i = ndwords * 4 + threadIdx.x;
    atomicAdd(&lhist[(u32)(i % 3) * 256 + in[i]], 1u);
  }
  __syncthreads();
  for (u32 i = threadIdx.x; i < 3 * 256; i += blockDim.x) {
    if (lhist[i]) atomicAdd(&out[i], lhist[i]);
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
    for (const Element& f : in[0]) {
      SCA_CHECK(f.is_frame && f.device.is_gpu(),
                "GPU Histogram needs GPU frame input");
      i32 h = f.frame_info.shape[0], w = f.frame_info.shape[1],
          c = f.frame_info.shape[2];
      size_t out_size = (size_t)c * 256 * sizeof(u32);
      Element e;
      e.buffer = new_buffer(config_.device, out_size);
      e.size = out_size;
      e.device = config_.device;
      HIPK_CHECK(hipMemsetAsync(e.buffer, 0, out_size, s));
      u64 nbytes = (u64)h * w * c;
      int blocks = (int)std::min<u64>(2048, (nbytes / 4 + 255) / 256 + 1);
      if (c == 3) {
        histogram_rgb_kernel<<<blocks, 256, 0, s>>>(f.buffer, nbytes,
                                                    (u32*)e.buffer);
      } else {
        histogram_any_kernel<<<blocks, 256, c * 256 * 4, s>>>(
            f.buffer, (u64)h * w, (u32)c, (u32*)e.buffer);
      }
      HIPK_CHECK(hipGetLastError());
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
