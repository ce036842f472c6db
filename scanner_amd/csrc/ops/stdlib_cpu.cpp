// CPU implementations of the first-party op stdlib + test ops.
// Capability parity: the reference keeps image ops in the external
// `scannertools` repo and test ops in tests/test_ops.cpp (Histogram,
// OpticalFlow, Resize, TestIncrement bounded/unbounded, Blur, Sleep);
// here they are first-party, with GPU (HIP) counterparts in
// kernels/image_ops.hip registered by ops/stdlib_gpu.cpp.
#include <algorithm>
#include <cmath>
#include <cstring>
#include <thread>
#include <vector>

#include "../memory.h"
#include "../msgpack.h"
#include "kernel.h"
#include "optflow_common.h"

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
    i32 ks = (i32)a.get_int("kernel_size", 3);
    // ks < 1 made radius negative -> empty tap loop -> division by zero
    // (UB; observed as a wedged pipeline instance)
    SCA_CHECK(ks >= 1, "Blur kernel_size must be >= 1");
    radius_ = ks / 2;
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

// ---- OpticalFlowCPU: dense pyramidal Lucas-Kanade, stencil [0,1].
// Same algorithm and constants (optflow_common.h) as the CDNA4 kernel in
// kernels/optflow.hip, so GPU numerics tests compare against this directly.
// Parity: the reference's OpticalFlow op wraps OpenCV Farneback
// (tests/test_ops.cpp:63-113) with the same dense H x W x 2 f32 contract.
class OpticalFlowKernelCPU : public StenciledKernel {
 public:
  explicit OpticalFlowKernelCPU(const KernelConfig& cfg)
      : StenciledKernel(cfg) {
    auto a = mp::decode(cfg.args);
    radius_ = (int)a.get_int("radius", optflow::kDefaultRadius);
    iters_ = (int)a.get_int("iters", optflow::kDefaultIters);
    max_levels_ = (int)a.get_int("levels", optflow::kDefaultMaxLevels);
  }
  void execute_stencil(const BatchedElements& in, ElementVector& out) override {
    const Element& f0 = in[0][0];
    const Element& f1 = in[0][1];
    i32 h = f0.frame_info.shape[0], w = f0.frame_info.shape[1],
        c = f0.frame_info.shape[2];
    int levels = optflow::num_levels(h, w, max_levels_);
    // Gray pyramids.
    std::vector<std::vector<f32>> p0(levels), p1(levels);
    std::vector<int> lh(levels), lw(levels);
    lh[0] = h;
    lw[0] = w;
    to_gray(f0.buffer, h, w, c, p0[0]);
    to_gray(f1.buffer, h, w, c, p1[0]);
    for (int l = 1; l < levels; ++l) {
      lh[l] = lh[l - 1] / 2;
      lw[l] = lw[l - 1] / 2;
      down2x(p0[l - 1], lh[l - 1], lw[l - 1], p0[l], lh[l], lw[l]);
      down2x(p1[l - 1], lh[l - 1], lw[l - 1], p1[l], lh[l], lw[l]);
    }
    std::vector<f32> flow(2 * (size_t)lh[levels - 1] * lw[levels - 1], 0.f);
    std::vector<f32> next;
    for (int l = levels - 1; l >= 0; --l) {
      for (int it = 0; it < iters_; ++it) {
        lk_iter(p0[l], p1[l], lh[l], lw[l], flow, next);
        std::swap(flow, next);
      }
      if (l > 0) {
        upsample(flow, lh[l], lw[l], next, lh[l - 1], lw[l - 1]);
        std::swap(flow, next);
      }
    }
    Element e = alloc_frame(config_.device, h, w, 2, FrameType::F32);
    std::memcpy(e.buffer, flow.data(), flow.size() * sizeof(f32));
    out[0] = e;
  }

 private:
  static void to_gray(const u8* p, int h, int w, int c, std::vector<f32>& g) {
    g.resize((size_t)h * w);
    for (i64 i = 0; i < (i64)h * w; ++i) {
      const u8* px = p + i * c;
      g[i] = c == 3 ? 0.299f * px[0] + 0.587f * px[1] + 0.114f * px[2]
                    : (f32)px[0];
    }
  }
  static void down2x(const std::vector<f32>& s, int sh, int sw,
                     std::vector<f32>& d, int dh, int dw) {
    d.resize((size_t)dh * dw);
    for (int y = 0; y < dh; ++y)
      for (int x = 0; x < dw; ++x) {
        int y0 = 2 * y, x0 = 2 * x;
        int y1 = std::min(y0 + 1, sh - 1), x1 = std::min(x0 + 1, sw - 1);
        d[(size_t)y * dw + x] =
            0.25f * (s[(size_t)y0 * sw + x0] + s[(size_t)y0 * sw + x1] +
                     s[(size_t)y1 * sw + x0] + s[(size_t)y1 * sw + x1]);
      }
  }
  static f32 bilerp(const std::vector<f32>& im, int h, int w, f32 fx, f32 fy) {
    int x0 = (int)std::floor(fx), y0 = (int)std::floor(fy);
    f32 ax = fx - x0, ay = fy - y0;
    int x0c = std::min(std::max(x0, 0), w - 1);
    int x1c = std::min(std::max(x0 + 1, 0), w - 1);
    int y0c = std::min(std::max(y0, 0), h - 1);
    int y1c = std::min(std::max(y0 + 1, 0), h - 1);
    f32 v00 = im[(size_t)y0c * w + x0c], v01 = im[(size_t)y0c * w + x1c];
    f32 v10 = im[(size_t)y1c * w + x0c], v11 = im[(size_t)y1c * w + x1c];
    return v00 * (1 - ay) * (1 - ax) + v01 * (1 - ay) * ax +
           v10 * ay * (1 - ax) + v11 * ay * ax;
  }
  void lk_iter(const std::vector<f32>& I0, const std::vector<f32>& I1, int h,
               int w, const std::vector<f32>& fin, std::vector<f32>& fout) {
    fout.resize((size_t)h * w * 2);
    int R = radius_;
    auto at = [&](const std::vector<f32>& im, int y, int x) {
      return im[(size_t)std::min(std::max(y, 0), h - 1) * w +
                std::min(std::max(x, 0), w - 1)];
    };
    for (int y = 0; y < h; ++y) {
      for (int x = 0; x < w; ++x) {
        f32 u = fin[((size_t)y * w + x) * 2 + 0];
        f32 v = fin[((size_t)y * w + x) * 2 + 1];
        f32 a11 = 0, a12 = 0, a22 = 0, b1 = 0, b2 = 0;
        for (int dy = -R; dy <= R; ++dy)
          for (int dx = -R; dx <= R; ++dx) {
            int yy = y + dy, xx = x + dx;
            f32 ix = 0.5f * (at(I0, yy, xx + 1) - at(I0, yy, xx - 1));
            f32 iy = 0.5f * (at(I0, yy + 1, xx) - at(I0, yy - 1, xx));
            f32 it = bilerp(I1, h, w, xx + u, yy + v) - at(I0, yy, xx);
            a11 += ix * ix;
            a12 += ix * iy;
            a22 += iy * iy;
            b1 += ix * it;
            b2 += iy * it;
          }
        f32 det = a11 * a22 - a12 * a12;
        if (det > optflow::kDetEps) {
          u -= (a22 * b1 - a12 * b2) / det;
          v -= (a11 * b2 - a12 * b1) / det;
        }
        fout[((size_t)y * w + x) * 2 + 0] = u;
        fout[((size_t)y * w + x) * 2 + 1] = v;
      }
    }
  }
  void upsample(const std::vector<f32>& s, int sh, int sw,
                std::vector<f32>& d, int dh, int dw) {
    d.resize((size_t)dh * dw * 2);
    f32 fx_scale = (f32)dw / sw, fy_scale = (f32)dh / sh;
    for (int y = 0; y < dh; ++y)
      for (int x = 0; x < dw; ++x) {
        f32 sxf = (x + 0.5f) / fx_scale - 0.5f;
        f32 syf = (y + 0.5f) / fy_scale - 0.5f;
        int x0 = (int)std::floor(sxf), y0 = (int)std::floor(syf);
        f32 ax = sxf - x0, ay = syf - y0;
        int x0c = std::min(std::max(x0, 0), sw - 1);
        int x1c = std::min(std::max(x0 + 1, 0), sw - 1);
        int y0c = std::min(std::max(y0, 0), sh - 1);
        int y1c = std::min(std::max(y0 + 1, 0), sh - 1);
        for (int ch = 0; ch < 2; ++ch) {
          f32 v00 = s[((size_t)y0c * sw + x0c) * 2 + ch];
          f32 v01 = s[((size_t)y0c * sw + x1c) * 2 + ch];
          f32 v10 = s[((size_t)y1c * sw + x0c) * 2 + ch];
          f32 v11 = s[((size_t)y1c * sw + x1c) * 2 + ch];
          f32 vv = v00 * (1 - ay) * (1 - ax) + v01 * (1 - ay) * ax +
                   v10 * ay * (1 - ax) + v11 * ay * ax;
          d[((size_t)y * dw + x) * 2 + ch] =
              vv * (ch == 0 ? fx_scale : fy_scale);
        }
      }
  }

  int radius_, iters_, max_levels_;
};

// ---- Crop: HWC u8 frame -> window (x, y, width, height) ----
class CropKernelCPU : public BatchedKernel {
 public:
  explicit CropKernelCPU(const KernelConfig& cfg) : BatchedKernel(cfg) {
    auto a = mp::decode(cfg.args);
    x_ = (i32)a.get_int("x", 0);
    y_ = (i32)a.get_int("y", 0);
    w_ = (i32)a.get_int("width", 0);
    h_ = (i32)a.get_int("height", 0);
    SCA_CHECK(w_ > 0 && h_ > 0, "Crop needs width/height args");
    // negative offsets passed the (x + w <= frame_w) check and read
    // before the source buffer
    SCA_CHECK(x_ >= 0 && y_ >= 0, "Crop x/y must be >= 0");
  }
  void execute_batch(const BatchedElements& in, BatchedElements& out) override {
    for (const Element& f : in[0]) {
      i32 h = f.frame_info.shape[0], w = f.frame_info.shape[1],
          c = f.frame_info.shape[2];
      SCA_CHECK(x_ + w_ <= w && y_ + h_ <= h, "Crop outside frame bounds");
      Element e = alloc_frame(config_.device, h_, w_, c, FrameType::U8);
      for (i32 y = 0; y < h_; ++y) {
        std::memcpy(e.buffer + (i64)y * w_ * c,
                    f.buffer + (((i64)(y_ + y)) * w + x_) * c,
                    (size_t)w_ * c);
      }
      out[0].push_back(e);
    }
  }

 private:
  i32 x_, y_, w_, h_;
};

// ---- ColorConvert: gray (BT.601 luma) | yuv (BT.601 full) | planar ----
class ColorConvertKernelCPU : public BatchedKernel {
 public:
  explicit ColorConvertKernelCPU(const KernelConfig& cfg)
      : BatchedKernel(cfg) {
    auto a = mp::decode(cfg.args);
    mode_ = a.get_str("format", "gray");
    SCA_CHECK(mode_ == "gray" || mode_ == "yuv" || mode_ == "planar",
              "ColorConvert format must be gray|yuv|planar");
  }
  void execute_batch(const BatchedElements& in, BatchedElements& out) override {
    auto cl = [](f32 v) -> u8 {
      return (u8)std::min(std::max(v, 0.f), 255.f);
    };
    for (const Element& f : in[0]) {
      i32 h = f.frame_info.shape[0], w = f.frame_info.shape[1],
          c = f.frame_info.shape[2];
      i64 npix = (i64)h * w;
      const u8* p = f.buffer;
      Element e;
      if (mode_ == "gray") {
        SCA_CHECK(c == 3, "gray conversion needs RGB input");
        e = alloc_frame(config_.device, h, w, 1, FrameType::U8);
        for (i64 i = 0; i < npix; ++i) {
          e.buffer[i] = cl(0.299f * p[i * 3] + 0.587f * p[i * 3 + 1] +
                           0.114f * p[i * 3 + 2] + 0.5f);
        }
      } else if (mode_ == "yuv") {
        SCA_CHECK(c == 3, "yuv conversion needs RGB input");
        e = alloc_frame(config_.device, h, w, 3, FrameType::U8);
        for (i64 i = 0; i < npix; ++i) {
          f32 r = p[i * 3], g = p[i * 3 + 1], b = p[i * 3 + 2];
          e.buffer[i * 3 + 0] =
              cl(0.299f * r + 0.587f * g + 0.114f * b + 0.5f);
          e.buffer[i * 3 + 1] =
              cl(-0.169f * r - 0.331f * g + 0.5f * b + 128.5f);
          e.buffer[i * 3 + 2] =
              cl(0.5f * r - 0.419f * g - 0.081f * b + 128.5f);
        }
      } else {  // planar HWC -> CHW (reference parity: image.cu:202-220)
        e = alloc_frame(config_.device, c, h, w, FrameType::U8);
        for (i32 ch = 0; ch < c; ++ch) {
          for (i64 i = 0; i < npix; ++i) {
            e.buffer[(i64)ch * npix + i] = p[i * c + ch];
          }
        }
      }
      out[0].push_back(e);
    }
  }

 private:
  std::string mode_;
};

// ---- ConcatBytes: variadic test op — concatenates all input columns'
//      blobs per row (parity: reference variadic ops, op.h:77
//      variadic_inputs + py_test.py:695) ----
class ConcatBytesKernel : public BaseKernel {
 public:
  using BaseKernel::BaseKernel;
  void execute(const StenciledElements& in, BatchedElements& out) override {
    size_t rows = in.empty() ? 0 : in[0].size();
    for (size_t r = 0; r < rows; ++r) {
      size_t total = 0;
      for (auto& col : in) total += col[r][0].size;
      Element e = alloc_bytes(config_.device, total);
      size_t off = 0;
      for (auto& col : in) {
        const Element& src = col[r][0];
        std::memcpy(e.buffer + off, src.buffer, src.size);
        off += src.size;
      }
      out[0].push_back(e);
    }
  }
};

// ---- FlowStats: dense flow frame -> {mean|u|, mean|v|, max|u|, max|v|} ----
class FlowStatsKernelCPU : public Kernel {
 public:
  using Kernel::Kernel;
  void execute_row(const ElementVector& in, ElementVector& out) override {
    const Element& f = in[0];
    SCA_CHECK(f.is_frame && f.frame_info.type == FrameType::F32 &&
                  f.frame_info.shape[2] == 2,
              "FlowStats expects H x W x 2 f32 flow frames");
    i64 npix = (i64)f.frame_info.shape[0] * f.frame_info.shape[1];
    const f32* p = reinterpret_cast<const f32*>(f.buffer);
    // match the GPU reduction: sums accumulate in f32
    f32 su = 0, sv = 0, mu = 0, mv = 0;
    for (i64 i = 0; i < npix; ++i) {
      f32 u = std::fabs(p[i * 2]), v = std::fabs(p[i * 2 + 1]);
      su += u;
      sv += v;
      mu = std::max(mu, u);
      mv = std::max(mv, v);
    }
    Element e = alloc_bytes(config_.device, 4 * sizeof(f32));
    f32* o = reinterpret_cast<f32*>(e.buffer);
    o[0] = su / npix;
    o[1] = sv / npix;
    o[2] = mu;
    o[3] = mv;
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
    o.name = "Crop";
    o.input_columns = {frame_in};
    o.output_columns = {frame_out};
    op_registry().add(o);
    KernelFactory f;
    f.op_name = "Crop";
    f.device_type = DeviceType::CPU;
    f.preferred_batch = 8;
    f.make = [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
      return std::make_unique<CropKernelCPU>(c);
    };
    kernel_registry().add(f);
  }
  {
    OpInfo o;
    o.name = "ColorConvert";
    o.input_columns = {frame_in};
    o.output_columns = {frame_out};
    op_registry().add(o);
    KernelFactory f;
    f.op_name = "ColorConvert";
    f.device_type = DeviceType::CPU;
    f.preferred_batch = 8;
    f.make = [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
      return std::make_unique<ColorConvertKernelCPU>(c);
    };
    kernel_registry().add(f);
  }
  {
    OpInfo o;
    o.name = "ConcatBytes";
    o.variadic_inputs = true;
    o.output_columns = {{"out", ColumnType::Bytes}};
    op_registry().add(o);
    KernelFactory f;
    f.op_name = "ConcatBytes";
    f.device_type = DeviceType::CPU;
    f.preferred_batch = 8;
    f.make = [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
      return std::make_unique<ConcatBytesKernel>(c);
    };
    kernel_registry().add(f);
  }
  {
    OpInfo o;
    o.name = "FlowStats";
    o.input_columns = {{"flow", ColumnType::Video}};
    o.output_columns = {{"stats", ColumnType::Bytes}};
    op_registry().add(o);
    KernelFactory f;
    f.op_name = "FlowStats";
    f.device_type = DeviceType::CPU;
    f.make = [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
      return std::make_unique<FlowStatsKernelCPU>(c);
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
