// Color / geometry image ops for gfx950: ColorConvert (gray / yuv /
// planar), Crop, Blur. GPU counterparts of the CPU kernels in
// stdlib_cpu.cpp (same math, so device-parity tests compare directly).
// Capability parity: the reference's GPU image surface is
// scanner/util/image.cu (NV12_to_RGB at 109-200, RGB_interleaved_to_planar
// at 202-220) plus the scannertools image ops; here they are first-party.
// All memory-bound: dword-vectorizable grid-stride loops, grids capped so
// the scheduler keeps all 8 XCDs fed; launches go on the pipeline
// instance's per-thread HIP stream.
#include <hip/hip_runtime.h>

#include <vector>

#include "../csrc/memory.h"
#include "../csrc/msgpack.h"
#include "../csrc/ops/kernel.h"

namespace sca {

namespace {

#define COL_CHECK(expr)                                                  \
  do {                                                                   \
    hipError_t _e = (expr);                                              \
    if (_e != hipSuccess) {                                              \
      throw ScannerError(std::string("HIP error in color op: ") +        \
                         hipGetErrorString(_e));                         \
    }                                                                    \
  } while (0)

inline hipStream_t cur_stream() {
  return (hipStream_t)per_thread_hip_stream();
}

inline int grid_1d(i64 total, int block = 256, int cap = 4096) {
  return (int)std::min<i64>(cap, (total + block - 1) / block);
}

__device__ inline u8 clamp_u8(float v) {
  return (u8)fminf(fmaxf(v, 0.f), 255.f);
}

// RGB HWC u8 -> gray HW u8 (BT.601 luma, same formula as the CPU op).
__global__ void __launch_bounds__(256)
    rgb_to_gray_kernel(const u8* __restrict__ src, i64 npix,
                       u8* __restrict__ dst) {
  i64 gs = (i64)gridDim.x * blockDim.x;
  for (i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x; i < npix;
       i += gs) {
    const u8* p = src + i * 3;
    dst[i] = clamp_u8(0.299f * p[0] + 0.587f * p[1] + 0.114f * p[2] + 0.5f);
  }
}

// RGB HWC u8 -> YUV (BT.601 full range) HWC u8.
__global__ void __launch_bounds__(256)
    rgb_to_yuv_kernel(const u8* __restrict__ src, i64 npix,
                      u8* __restrict__ dst) {
  i64 gs = (i64)gridDim.x * blockDim.x;
  for (i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x; i < npix;
       i += gs) {
    float r = src[i * 3], g = src[i * 3 + 1], b = src[i * 3 + 2];
    dst[i * 3 + 0] = clamp_u8(0.299f * r + 0.587f * g + 0.114f * b + 0.5f);
    dst[i * 3 + 1] =
        clamp_u8(-0.169f * r - 0.331f * g + 0.5f * b + 128.5f);
    dst[i * 3 + 2] =
        clamp_u8(0.5f * r - 0.419f * g - 0.081f * b + 128.5f);
  }
}

// HWC u8 -> CHW u8 planar (reference parity: image.cu:202-220). LDS-tiled
// transpose: a 32x32 pixel tile per channel is staged so both the HWC
// reads and the CHW writes are coalesced along x.
__global__ void __launch_bounds__(256)
    hwc_to_chw_kernel(const u8* __restrict__ src, int h, int w, int c,
                      u8* __restrict__ dst) {
  __shared__ u8 tile[32][33];
  int tx0 = blockIdx.x * 32, ty0 = blockIdx.y * 32, ch = blockIdx.z;
  for (int r = threadIdx.y; r < 32; r += blockDim.y) {
    int y = ty0 + r;
    for (int cc = threadIdx.x; cc < 32; cc += blockDim.x) {
      int x = tx0 + cc;
      if (y < h && x < w) tile[r][cc] = src[((i64)y * w + x) * c + ch];
    }
  }
  __syncthreads();
  for (int r = threadIdx.y; r < 32; r += blockDim.y) {
    int y = ty0 + r;
    if (y >= h) continue;
    for (int cc = threadIdx.x; cc < 32; cc += blockDim.x) {
      int x = tx0 + cc;
      if (x < w) dst[(i64)ch * h * w + (i64)y * w + x] = tile[r][cc];
    }
  }
}

__global__ void __launch_bounds__(256)
    crop_kernel(const u8* __restrict__ src, int w, int c, int cx, int cy,
                int cw, int chh, u8* __restrict__ dst) {
  i64 total = (i64)chh * cw * c;
  i64 gs = (i64)gridDim.x * blockDim.x;
  for (i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gs) {
    int ch = (int)(i % c);
    i64 pix = i / c;
    int x = (int)(pix % cw), y = (int)(pix / cw);
    dst[i] = src[(((i64)(cy + y)) * w + cx + x) * c + ch];
  }
}

// Box blur with edge clipping, identical averaging to BlurKernelCPU.
__global__ void __launch_bounds__(256)
    box_blur_kernel(const u8* __restrict__ src, int h, int w, int c, int R,
                    u8* __restrict__ dst) {
  i64 total = (i64)h * w * c;
  i64 gs = (i64)gridDim.x * blockDim.x;
  for (i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gs) {
    int ch = (int)(i % c);
    i64 pix = i / c;
    int x = (int)(pix % w), y = (int)(pix / w);
    int sum = 0, cnt = 0;
    for (int dy = -R; dy <= R; ++dy) {
      int yy = y + dy;
      if (yy < 0 || yy >= h) continue;
      for (int dx = -R; dx <= R; ++dx) {
        int xx = x + dx;
        if (xx < 0 || xx >= w) continue;
        sum += src[((i64)yy * w + xx) * c + ch];
        ++cnt;
      }
    }
    dst[i] = (u8)(sum / cnt);
  }
}

// ---------------- host kernels ----------------

class ColorConvertKernelGPU : public BatchedKernel {
 public:
  explicit ColorConvertKernelGPU(const KernelConfig& cfg)
      : BatchedKernel(cfg) {
    auto a = mp::decode(cfg.args);
    mode_ = a.get_str("format", "gray");
    SCA_CHECK(mode_ == "gray" || mode_ == "yuv" || mode_ == "planar",
              "ColorConvert format must be gray|yuv|planar");
  }
  void execute_batch(const BatchedElements& in, BatchedElements& out) override {
    hipStream_t s = cur_stream();
    for (const Element& f : in[0]) {
      SCA_CHECK(f.is_frame && f.device.is_gpu(),
                "GPU ColorConvert needs GPU frames");
      int h = f.frame_info.shape[0], w = f.frame_info.shape[1],
          c = f.frame_info.shape[2];
      i64 npix = (i64)h * w;
      Element e;
      e.is_frame = true;
      e.frame_info.type = FrameType::U8;
      e.device = config_.device;
      if (mode_ == "gray") {
        SCA_CHECK(c == 3, "gray conversion needs RGB input");
        e.frame_info.shape[0] = h;
        e.frame_info.shape[1] = w;
        e.frame_info.shape[2] = 1;
        e.size = e.frame_info.size();
        e.buffer = new_buffer(config_.device, e.size);
        rgb_to_gray_kernel<<<grid_1d(npix), 256, 0, s>>>(f.buffer, npix,
                                                         e.buffer);
      } else if (mode_ == "yuv") {
        SCA_CHECK(c == 3, "yuv conversion needs RGB input");
        e.frame_info = f.frame_info;
        e.size = e.frame_info.size();
        e.buffer = new_buffer(config_.device, e.size);
        rgb_to_yuv_kernel<<<grid_1d(npix), 256, 0, s>>>(f.buffer, npix,
                                                        e.buffer);
      } else {  // planar: HWC -> CHW
        e.frame_info.shape[0] = c;
        e.frame_info.shape[1] = h;
        e.frame_info.shape[2] = w;
        e.size = e.frame_info.size();
        e.buffer = new_buffer(config_.device, e.size);
        dim3 grid((w + 31) / 32, (h + 31) / 32, c);
        dim3 block(32, 8);
        hwc_to_chw_kernel<<<grid, block, 0, s>>>(f.buffer, h, w, c,
                                                 e.buffer);
      }
      COL_CHECK(hipGetLastError());
      out[0].push_back(e);
    }
  }

 private:
  std::string mode_;
};

class CropKernelGPU : public BatchedKernel {
 public:
  explicit CropKernelGPU(const KernelConfig& cfg) : BatchedKernel(cfg) {
    auto a = mp::decode(cfg.args);
    x_ = (int)a.get_int("x", 0);
    y_ = (int)a.get_int("y", 0);
    w_ = (int)a.get_int("width", 0);
    h_ = (int)a.get_int("height", 0);
    SCA_CHECK(w_ > 0 && h_ > 0, "Crop needs width/height args");
    SCA_CHECK(x_ >= 0 && y_ >= 0, "Crop x/y must be >= 0");
  }
  void execute_batch(const BatchedElements& in, BatchedElements& out) override {
    hipStream_t s = cur_stream();
    for (const Element& f : in[0]) {
      int h = f.frame_info.shape[0], w = f.frame_info.shape[1],
          c = f.frame_info.shape[2];
      SCA_CHECK(x_ + w_ <= w && y_ + h_ <= h, "Crop outside frame bounds");
      Element e;
      e.is_frame = true;
      e.frame_info.shape[0] = h_;
      e.frame_info.shape[1] = w_;
      e.frame_info.shape[2] = c;
      e.frame_info.type = FrameType::U8;
      e.size = e.frame_info.size();
      e.device = config_.device;
      e.buffer = new_buffer(config_.device, e.size);
      crop_kernel<<<grid_1d((i64)h_ * w_ * c), 256, 0, s>>>(
          f.buffer, w, c, x_, y_, w_, h_, e.buffer);
      COL_CHECK(hipGetLastError());
      out[0].push_back(e);
    }
  }

 private:
  int x_, y_, w_, h_;
};

class BlurKernelGPU : public BatchedKernel {
 public:
  explicit BlurKernelGPU(const KernelConfig& cfg) : BatchedKernel(cfg) {
    auto a = mp::decode(cfg.args);
    int ks = (int)a.get_int("kernel_size", 3);
    SCA_CHECK(ks >= 1, "Blur kernel_size must be >= 1");
    radius_ = ks / 2;
  }
  void execute_batch(const BatchedElements& in, BatchedElements& out) override {
    hipStream_t s = cur_stream();
    for (const Element& f : in[0]) {
      int h = f.frame_info.shape[0], w = f.frame_info.shape[1],
          c = f.frame_info.shape[2];
      Element e;
      e.is_frame = true;
      e.frame_info = f.frame_info;
      e.size = e.frame_info.size();
      e.device = config_.device;
      e.buffer = new_buffer(config_.device, e.size);
      box_blur_kernel<<<grid_1d((i64)h * w * c), 256, 0, s>>>(
          f.buffer, h, w, c, radius_, e.buffer);
      COL_CHECK(hipGetLastError());
      out[0].push_back(e);
    }
  }

 private:
  int radius_;
};

}  // namespace

void register_color_gpu() {
  static bool done = false;
  if (done) return;
  done = true;
  auto add = [](const char* name, auto maker, int batch) {
    KernelFactory f;
    f.op_name = name;
    f.device_type = DeviceType::GPU;
    f.preferred_batch = batch;
    f.make = maker;
    kernel_registry().add(f);
  };
  add("ColorConvert",
      [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
        return std::make_unique<ColorConvertKernelGPU>(c);
      },
      16);
  add("Crop",
      [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
        return std::make_unique<CropKernelGPU>(c);
      },
      16);
  add("Blur",
      [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
        return std::make_unique<BlurKernelGPU>(c);
      },
      8);
}

}  // namespace sca
