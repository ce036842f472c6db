// CPU implementations of the first-party op stdlib + test ops.
// Capability parity: the reference keeps image ops in the external
// `scannertools` repo and test ops in tests/test_ops.cpp (Histogram,
// OpticalFlow, Resize, TestIncrement bounded/unbounded, Blur, Sleep);
// here they are first-party, with GPU (HIP) counterparts in
// kernels/image_ops.hip registered by ops/stdlib_gpu.cpp.
#include <cmath>
#include <thread>

#include "../memory.h"
#include "../msgpack.h"
#include "kernel.h"

namespace sca {

namespace {

Element alloc_bytes(DeviceHandle dev, size_t size) {
  Element e;
  e.buffer = new_buffer(dev, size);
  e.size = size;
  e.device = dev;
  return e;
}

Element alloc_frame(DeviceHandle dev, i32 h, i32 w, i32 c, FrameType t) {
  Element e;
  e.is_frame = true;
  e.frame_info.shape[0] = h;
  e.frame_info.shape[1] = w;
  e.frame_info.shape[2] = c;
  e.frame_info.type = t;
  e.size = e.frame_info.size();
  e.buffer = new_buffer(dev, e.size);
  e.device = dev;
  return e;
}

// ---- Histogram: HWC u8 frame -> C x 256 u32 bins ----
class HistogramKernelCPU : public BatchedKernel {
 public:
  using BatchedKernel::BatchedKernel;
  void execute_batch(const BatchedElements& in, BatchedElements& out) override {
    for (const Element& f : in[0]) {
      SCA_CHECK(f.is_frame, "Histogram needs a frame input");
      i32 h = f.frame_info.shape[0], w = f.frame_info.shape[1],
          c = f.frame_info.shape[2];
      Element e = alloc_bytes(config_.device, (size_t)c * 256 * sizeof(u32));
      u32* hist = reinterpret_cast<u32*>(e.buffer);
      std::memset(hist, 0, e.size);
      const u8* p = f.buffer;
      for (i64 i = 0; i < (i64)h * w; ++i) {
        for (i32 ch = 0; ch < c; ++ch) {
          hist[ch * 256 + p[i * c + ch]]++;
        }
      }
      out[0].push_back(e);
    }
  }
};

// ---- Resize: bilinear HWC u8 ----
class ResizeKernelCPU : public BatchedKernel {
 public:
  explicit ResizeKernelCPU(const KernelConfig& cfg) : BatchedKernel(cfg) {
    auto a = mp::decode(cfg.args);
    out_w_ = (i32)a.get_int("width", 0);
    out_h_ = (i32)a.get_int("height", 0);
    SCA_CHECK(out_w_ > 0 && out_h_ > 0, "Resize needs width/height args");
  }
  void execute_batch(const BatchedElements& in, BatchedElements& out) override {
    for (const Element& f : in[0]) {
      SCA_CHECK(f.is_frame, "Resize needs a frame input");
      i32 h = f.frame_info.shape[0], w = f.frame_info.shape[1],
          c = f.frame_info.shape[2];
      Element e = alloc_frame(config_.device, out_h_, out_w_, c, FrameType::U8);
      u8* dst = e.buffer;
      const u8* src = f.buffer;
      for (i32 y = 0; y < out_h_; ++y) {
        f32 sy = (y + 0.5f) * h / out_h_ - 0.5f;
        i32 y0 = std::max(0, (i32)std::floor(sy));
        i32 y1 = std::min(h - 1, y0 + 1);
        f32 fy = sy - y0;
        if (fy < 0) fy = 0;
        for (i32 x = 0; x < out_w_; ++x) {
          f32 sx = (x + 0.5f) * w / out_w_ - 0.5f;
          i32 x0 = std::max(0, (i32)std::floor(sx));
          i32 x1 = std::min(w - 1, x0 + 1);
          f32 fx = sx - x0;
          if (fx < 0) fx = 0;
          for (i32 ch = 0; ch < c; ++ch) {
            f32 v00 = src[((i64)y0 * w + x0) * c + ch];
            f32 v01 = src[((i64)y0 * w + x1) * c + ch];
            f32 v10 = src[((i64)y1 * w + x0) * c + ch];
            f32 v11 = src[((i64)y1 * w + x1) * c + ch];
            f32 v = v00 * (1 - fy) * (1 - fx) + v01 * (1 - fy) * fx +
                    v10 * fy * (1 - fx) + v11 * fy * fx;
            dst[((i64)y * out_w_ + x) * c + ch] = (u8)(v + 0.5f);
          }
        }
      }
      out[0].push_back(e);
    }
  }

 private:
  i32 out_w_, out_h_;
};

// ---- Blur: separable box blur (test-op parity with reference Blur) ----
class BlurKernelCPU : public BatchedKernel {
 public:
  explicit BlurKernelCPU(const KernelConfig& cfg) : BatchedKernel(cfg) {
    auto a = mp::decode(cfg.args);
    radius_ = (i32)a.get_int("kernel_size", 3) / 2;
  }
  void execute_batch(const BatchedElements& in, BatchedElements& out) override {
    for (const Element& f : in[0]) {
      i32 h = f.frame_info.shape[0], w = f.frame_info.shape[1],
          c = f.frame_info.shape[2];
      Element e = alloc_frame(config_.device, h, w, c, FrameType::U8);
      const u8* src = f.buffer;
      u8* dst = e.buffer;
      for (i32 y = 0; y < h; ++y) {
        for (i32 x = 0; x < w; ++x) {
          for (i32 ch = 0; ch < c; ++ch) {
            i32 sum = 0, cnt = 0;
            for (i32 dy = -radius_; dy <= radius_; ++dy) {
              for (i32 dx = -radius_; dx <= radius_; ++dx) {
                i32 yy = y + dy, xx = x + dx;
                if (yy < 0 || yy >= h || xx < 0 || xx >= w) continue;
                sum += src[((i64)yy * w + xx) * c + ch];
                cnt++;
              }
            }
            dst[((i64)y * w + x) * c + ch] = (u8)(sum / cnt);
          }
        }
      }
      out[0].push_back(e);
    }
  }

 private:
  i32 radius_;
};

// ---- TestIncrement: stateful op over i64 blobs (test parity with
// reference TestIncrement[Unbounded]State, tests/test_ops.cpp:173-237) ----
class TestIncrementKernel : public Kernel {
 public:
  using Kernel::Kernel;
  void reset() override { state_ = 0; }
  void execute_row(const ElementVector& in, ElementVector& out) override {
    i64 v = 0;
    if (!in[0].is_null) {
      SCA_CHECK(in[0].size == 8, "TestIncrement wants i64 input");
      std::memcpy(&v, in[0].buffer, 8);
    }
    state_ += 1;
    i64 r = v + state_;
    Element e = alloc_bytes(config_.device, 8);
    std::memcpy(e.buffer, &r, 8);
    out[0] = e;
  }

 private:
  i64 state_ = 0;
};

// ---- Sleep: scheduling-test op ----
class SleepKernel : public Kernel {
 public:
  explicit SleepKernel(const KernelConfig& cfg) : Kernel(cfg) {
    auto a = mp::decode(cfg.args);
    ms_ = a.get_int("ms", 10);
  }
  void execute_row(const ElementVector& in, ElementVector& out) override {
    std::this_thread::sleep_for(std::chrono::milliseconds(ms_));
    Element e = alloc_bytes(config_.device, in[0].size);
    std::memcpy(e.buffer, in[0].buffer, in[0].size);
    out[0] = e;
  }

 private:
  i64 ms_;
};

// ---- OpticalFlowCPU: coarse diamond-search block matching, stencil [0,1].
// (Parity role of the reference's Farneback test op; the production dense
// flow lives in kernels/optflow.hip.) Output: H/16 x W/16 x 2 f32 motion
// vectors per 16x16 block.
class OpticalFlowKernelCPU : public StenciledKernel {
 public:
  using StenciledKernel::StenciledKernel;
  void execute_stencil(const BatchedElements& in, ElementVector& out) override {
    const Element& f0 = in[0][0];
    const Element& f1 = in[0][1];
    i32 h = f0.frame_info.shape[0], w = f0.frame_info.shape[1],
        c = f0.frame_info.shape[2];
    i32 bh = h / 16, bw = w / 16;
    Element e = alloc_frame(config_.device, bh, bw, 2, FrameType::F32);
    f32* flow = reinterpret_cast<f32*>(e.buffer);
    auto lum = [&](const u8* p, i32 y, i32 x) -> i32 {
      i64 off = ((i64)y * w + x) * c;
      i32 s = 0;
      for (i32 ch = 0; ch < c; ++ch) s += p[off + ch];
      return s;
    };
    for (i32 by = 0; by < bh; ++by) {
      for (i32 bx = 0; bx < bw; ++bx) {
        i32 y0 = by * 16, x0 = bx * 16;
        i32 best_dy = 0, best_dx = 0;
        i64 best = INT64_MAX;
        for (i32 dy = -8; dy <= 8; dy += 4) {
          for (i32 dx = -8; dx <= 8; dx += 4) {
            i64 sad = 0;
            for (i32 y = 0; y < 16; y += 4) {
              for (i32 x = 0; x < 16; x += 4) {
                i32 sy = y0 + y + dy, sx = x0 + x + dx;
                if (sy < 0 || sy >= h || sx < 0 || sx >= w) {
                  sad += 255;
                  continue;
                }
                sad += std::abs(lum(f1.buffer, sy, sx) -
                                lum(f0.buffer, y0 + y, x0 + x));
              }
            }
            if (sad < best) {
              best = sad;
              best_dy = dy;
              best_dx = dx;
            }
          }
        }
        flow[((i64)by * bw + bx) * 2 + 0] = (f32)best_dx;
        flow[((i64)by * bw + bx) * 2 + 1] = (f32)best_dy;
      }
    }
    out[0] = e;
  }
};

}  // namespace

void register_stdlib_ops() {
  static bool done = false;
  if (done) return;
  done = true;

  auto frame_in = OpColumnDef{"frame", ColumnType::Video};
  auto frame_out = OpColumnDef{"frame", ColumnType::Video};

  {
    OpInfo o;
    o.name = "Histogram";
    o.input_columns = {frame_in};
    o.output_columns = {{"histogram", ColumnType::Bytes}};
    op_registry().add(o);
    KernelFactory f;
    f.op_name = "Histogram";
    f.device_type = DeviceType::CPU;
    f.preferred_batch = 8;
    f.make = [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
      return std::make_unique<HistogramKernelCPU>(c);
    };
    kernel_registry().add(f);
  }
  {
    OpInfo o;
    o.name = "Resize";
    o.input_columns = {frame_in};
    o.output_columns = {frame_out};
    op_registry().add(o);
    KernelFactory f;
    f.op_name = "Resize";
    f.device_type = DeviceType::CPU;
    f.preferred_batch = 8;
    f.make = [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
      return std::make_unique<ResizeKernelCPU>(c);
    };
    kernel_registry().add(f);
  }
  {
    OpInfo o;
    o.name = "Blur";
    o.input_columns = {frame_in};
    o.output_columns = {frame_out};
    op_registry().add(o);
    KernelFactory f;
    f.op_name = "Blur";
    f.device_type = DeviceType::CPU;
    f.preferred_batch = 4;
    f.make = [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
      return std::make_unique<BlurKernelCPU>(c);
    };
    kernel_registry().add(f);
  }
  {
    OpInfo o;
    o.name = "TestIncrement";
    o.input_columns = {{"ignore", ColumnType::Bytes}};
    o.output_columns = {{"integer", ColumnType::Bytes}};
    o.has_bounded_state = true;
    o.warmup = 0;
    op_registry().add(o);
    KernelFactory f;
    f.op_name = "TestIncrement";
    f.device_type = DeviceType::CPU;
    f.make = [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
      return std::make_unique<TestIncrementKernel>(c);
    };
    kernel_registry().add(f);
  }
  {
    OpInfo o;
    o.name = "TestIncrementUnbounded";
    o.input_columns = {{"ignore", ColumnType::Bytes}};
    o.output_columns = {{"integer", ColumnType::Bytes}};
    o.has_unbounded_state = true;
    op_registry().add(o);
    KernelFactory f;
    f.op_name = "TestIncrementUnbounded";
    f.device_type = DeviceType::CPU;
    f.make = [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
      return std::make_unique<TestIncrementKernel>(c);
    };
    kernel_registry().add(f);
  }
  {
    OpInfo o;
    o.name = "Sleep";
    o.input_columns = {{"ignore", ColumnType::Bytes}};
    o.output_columns = {{"out", ColumnType::Bytes}};
    op_registry().add(o);
    KernelFactory f;
    f.op_name = "Sleep";
    f.device_type = DeviceType::CPU;
    f.make = [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
      return std::make_unique<SleepKernel>(c);
    };
    kernel_registry().add(f);
  }
  {
    OpInfo o;
    o.name = "OpticalFlow";
    o.input_columns = {frame_in};
    o.output_columns = {{"flow", ColumnType::Video}};
    o.stencil = {0, 1};
    op_registry().add(o);
    KernelFactory f;
    f.op_name = "OpticalFlow";
    f.device_type = DeviceType::CPU;
    f.make = [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
      return std::make_unique<OpticalFlowKernelCPU>(c);
    };
    kernel_registry().add(f);
  }
}

}  // namespace sca
