// Dense optical flow for gfx950 (MI355X): pyramidal Lucas-Kanade.
//
// Capability parity: the reference's OpticalFlow op (tests/test_ops.cpp:
// 63-113) wraps OpenCV's Farneback — stencil [0,1], dense H x W x 2 f32
// flow. This is a from-scratch CDNA4 implementation of the same contract:
//   * all frames of a work packet are processed in ONE set of batched
//     kernel launches per pyramid level (blocks carry a frame/pair index),
//     so coarse levels still put >> 256 workgroups on the 8 XCDs;
//   * consecutive stencil windows share frames — gray pyramids are built
//     once per unique frame, not once per pair;
//   * the LK iteration stages the reference image tile (incl. gradient +
//     window halo) in LDS; warped-image taps go through L2 (they are
//     flow-dependent gather reads, uncacheable in LDS by construction).
// All launches go on the pipeline instance's per-thread HIP stream; the
// kernel never synchronizes the device.
#include <hip/hip_runtime.h>

#include <map>
#include <type_traits>
#include <vector>

#include "../csrc/memory.h"
#include "../csrc/msgpack.h"
#include "../csrc/ops/kernel.h"
#include "../csrc/ops/optflow_common.h"

namespace sca {

namespace {

namespace of = ::sca::optflow;

#define OF_CHECK(expr)                                                    \
  do {                                                                    \
    hipError_t _e = (expr);                                               \
    if (_e != hipSuccess) {                                               \
      throw ScannerError(std::string("HIP error in OpticalFlow: ") +      \
                         hipGetErrorString(_e));                          \
    }                                                                     \
  } while (0)

inline hipStream_t cur_stream() {
  return (hipStream_t)per_thread_hip_stream();
}

// u8 HWC (c=1 or 3) -> f32 luma planes, one plane per unique frame.
// frames: device array of device pointers. out: [frame][h*w].
__global__ void __launch_bounds__(256)
    of_gray_kernel(const u8* const* __restrict__ frames, int nframes, int h,
                   int w, int c, float* __restrict__ out) {
  i64 total = (i64)nframes * h * w;
  i64 stride = (i64)gridDim.x * blockDim.x;
  for (i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int f = (int)(i / ((i64)h * w));
    i64 pix = i % ((i64)h * w);
    const u8* p = frames[f] + pix * c;
    float v;
    if (c == 3)
      v = 0.299f * p[0] + 0.587f * p[1] + 0.114f * p[2];
    else
      v = p[0];
    out[i] = v;
  }
}

// 2x2 box downsample of nframes planes. src stride sh*sw, dst stride dh*dw.
__global__ void __launch_bounds__(256)
    of_down2x_kernel(const float* __restrict__ src, int sh, int sw,
                     float* __restrict__ dst, int dh, int dw, int nframes) {
  i64 total = (i64)nframes * dh * dw;
  i64 stride = (i64)gridDim.x * blockDim.x;
  for (i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int f = (int)(i / ((i64)dh * dw));
    i64 pix = i % ((i64)dh * dw);
    int y = (int)(pix / dw), x = (int)(pix % dw);
    const float* s = src + (i64)f * sh * sw;
    int y0 = 2 * y, x0 = 2 * x;
    int y1 = min(y0 + 1, sh - 1), x1 = min(x0 + 1, sw - 1);
    dst[i] = 0.25f * (s[(i64)y0 * sw + x0] + s[(i64)y0 * sw + x1] +
                      s[(i64)y1 * sw + x0] + s[(i64)y1 * sw + x1]);
  }
}


// One LK iteration at one pyramid level for all pairs. Block = 16x16 pixel
// tile of one pair's plane (blockIdx.z = pair); I0 tile + halo staged in
// LDS. RADIUS is a compile-time template so the window loops fully unroll.
// When `coarse` is non-null, the input flow is read from the NEXT
// pyramid level (ch x cw) with the upsample bilerp fused in — the
// standalone of_upsample_kernel pass and its flow round-trip through HBM
// are skipped for the first iteration of every level.
template <int RADIUS, int PY>
__global__ void __launch_bounds__(256, 4)
    of_lk_kernel(const float* __restrict__ gray, i64 level_off, int h, int w,
                 const int* __restrict__ pair_f0,
                 const int* __restrict__ pair_f1,
                 const float* __restrict__ flow_in,
                 float* __restrict__ flow_out,
                 const float* __restrict__ coarse, int ch, int cw) {
  constexpr int TILE = 16;
  constexpr int TH = TILE * PY;              // output rows per block
  constexpr int HALO = RADIUS + 1;           // gradient needs +-1 past window
  constexpr int LW = TILE + 2 * HALO;        // staged tile width
  constexpr int LH = TH + 2 * HALO;
  constexpr int GW = TILE + 2 * RADIUS;      // gradient tile width
  constexpr int GH = TH + 2 * RADIUS;
  __shared__ float lds_i0[LH * LW];
  __shared__ float lds_gx[GH * GW];
  __shared__ float lds_gy[GH * GW];

  int pair = blockIdx.z;
  const float* I0 = gray + level_off + (i64)pair_f0[pair] * h * w;
  const float* I1 = gray + level_off + (i64)pair_f1[pair] * h * w;

  int tx0 = blockIdx.x * TILE, ty0 = blockIdx.y * TH;
  int x = tx0 + threadIdx.x;
  // Issue the flow reads (all PY pixels) before the LDS fill + gradient
  // barriers so their ~200-900 cycle latency hides under the cooperative
  // staging work. PY > 1 amortizes the fill/barriers over PY output rows
  // and doubles the per-wave load chains in the window loop.
  float u[PY], v[PY];
#pragma unroll
  for (int py = 0; py < PY; ++py) {
    u[py] = 0.f;
    v[py] = 0.f;
    int y = ty0 + py * TILE + (int)threadIdx.y;
    if (x >= w || y >= h) continue;
    if (coarse != nullptr) {
      // fused upsample from the coarser level (same math as
      // of_upsample_kernel: bilerp + per-axis magnitude scaling)
      const float* src = coarse + (i64)pair * ch * cw * 2;
      float fx_scale = (float)w / cw, fy_scale = (float)h / ch;
      float sxf = (x + 0.5f) / fx_scale - 0.5f;
      float syf = (y + 0.5f) / fy_scale - 0.5f;
      int x0 = (int)floorf(sxf), y0 = (int)floorf(syf);
      float axc = sxf - x0, ayc = syf - y0;
      int x0c = min(max(x0, 0), cw - 1), x1c = min(max(x0 + 1, 0), cw - 1);
      int y0c = min(max(y0, 0), ch - 1), y1c = min(max(y0 + 1, 0), ch - 1);
#pragma unroll
      for (int chn = 0; chn < 2; ++chn) {
        float v00 = src[((i64)y0c * cw + x0c) * 2 + chn];
        float v01 = src[((i64)y0c * cw + x1c) * 2 + chn];
        float v10 = src[((i64)y1c * cw + x0c) * 2 + chn];
        float v11 = src[((i64)y1c * cw + x1c) * 2 + chn];
        float vv = v00 * (1 - ayc) * (1 - axc) + v01 * (1 - ayc) * axc +
                   v10 * ayc * (1 - axc) + v11 * ayc * axc;
        float sc = chn == 0 ? fx_scale : fy_scale;
        if (chn == 0)
          u[py] = vv * sc;
        else
          v[py] = vv * sc;
      }
    } else {
      const float* fin = flow_in + (i64)pair * h * w * 2;
      u[py] = fin[((i64)y * w + x) * 2 + 0];
      v[py] = fin[((i64)y * w + x) * 2 + 1];
    }
  }
  // Cooperative LDS fill with clamped loads.
  for (int i = threadIdx.y * TILE + threadIdx.x; i < LH * LW;
       i += TILE * TILE) {
    int ly = i / LW, lx = i % LW;
    int gy = min(max(ty0 + ly - HALO, 0), h - 1);
    int gx = min(max(tx0 + lx - HALO, 0), w - 1);
    lds_i0[i] = I0[(i64)gy * w + gx];
  }
  __syncthreads();
  // Gradient tile computed once per block (each cell is read by up to 49
  // window taps otherwise).
  for (int i = threadIdx.y * TILE + threadIdx.x; i < GH * GW;
       i += TILE * TILE) {
    int gy = i / GW + 1, gx = i % GW + 1;  // offsets into lds_i0
    lds_gx[i] = 0.5f * (lds_i0[gy * LW + gx + 1] - lds_i0[gy * LW + gx - 1]);
    lds_gy[i] = 0.5f * (lds_i0[(gy + 1) * LW + gx] -
                        lds_i0[(gy - 1) * LW + gx]);
  }
  __syncthreads();

  float* fout = flow_out + (i64)pair * h * w * 2;
  constexpr int RW = 2 * RADIUS + 2;

#pragma unroll
  for (int py = 0; py < PY; ++py) {
    int lyb = py * TILE + (int)threadIdx.y;  // row within the tile space
    int y = ty0 + lyb;
    if (x >= w || y >= h) continue;
    float uu = u[py], vv = v[py];

    // The warp offset (u,v) is constant across the window, so the bilinear
    // fractions are too: the 49 taps read a contiguous (2R+2)^2 region of
    // I1 at one fractional offset. Interior fast path: pipeline three row
    // register arrays down that region (row dy+2 fetched while rows
    // (dy,dy+1) are consumed). Border threads take the clamped slow path.
    float a11 = 0, a12 = 0, a22 = 0, b1 = 0, b2 = 0;
    float wxf = x + uu, wyf = y + vv;
    int xi = (int)floorf(wxf), yi = (int)floorf(wyf);
    float ax = wxf - xi, ay = wyf - yi;
    bool interior = xi - RADIUS >= 0 && xi + RADIUS + 1 < w &&
                    yi - RADIUS >= 0 && yi + RADIUS + 1 < h;
    if (interior) {
      float r0[RW], r1[RW], r2[RW];
      const float* row = I1 + (i64)(yi - RADIUS) * w + (xi - RADIUS);
#pragma unroll
      for (int j = 0; j < RW; ++j) r0[j] = row[j];
      const float* row1 = row + w;
#pragma unroll
      for (int j = 0; j < RW; ++j) r1[j] = row1[j];
#pragma unroll
      for (int dy = -RADIUS; dy <= RADIUS; ++dy) {
        if (dy < RADIUS) {
          const float* p2 = I1 + (i64)(yi + dy + 2) * w + (xi - RADIUS);
#pragma unroll
          for (int j = 0; j < RW; ++j) r2[j] = p2[j];
        }
        int ly = lyb + HALO + dy;
        int gy_ = lyb + RADIUS + dy;
#pragma unroll
        for (int dx = -RADIUS; dx <= RADIUS; ++dx) {
          int lx = (int)threadIdx.x + HALO + dx;
          int gx_ = (int)threadIdx.x + RADIUS + dx;
          float ix = lds_gx[gy_ * GW + gx_];
          float iy = lds_gy[gy_ * GW + gx_];
          int j = dx + RADIUS;
          float top = r0[j] + ax * (r0[j + 1] - r0[j]);
          float bot = r1[j] + ax * (r1[j + 1] - r1[j]);
          float it = top + ay * (bot - top) - lds_i0[ly * LW + lx];
          a11 += ix * ix;
          a12 += ix * iy;
          a22 += iy * iy;
          b1 += ix * it;
          b2 += iy * it;
        }
#pragma unroll
        for (int j = 0; j < RW; ++j) {
          r0[j] = r1[j];
          r1[j] = r2[j];
        }
      }
    } else {
#pragma unroll
      for (int dy = -RADIUS; dy <= RADIUS; ++dy) {
#pragma unroll
        for (int dx = -RADIUS; dx <= RADIUS; ++dx) {
          int lx = (int)threadIdx.x + HALO + dx;
          int ly = lyb + HALO + dy;
          float ix = lds_gx[(lyb + RADIUS + dy) * GW +
                            threadIdx.x + RADIUS + dx];
          float iy = lds_gy[(lyb + RADIUS + dy) * GW +
                            threadIdx.x + RADIUS + dx];
          float i0v = lds_i0[ly * LW + lx];
          int x0c = min(max(xi + dx, 0), w - 1);
          int x1c = min(max(xi + dx + 1, 0), w - 1);
          int y0c = min(max(yi + dy, 0), h - 1);
          int y1c = min(max(yi + dy + 1, 0), h - 1);
          float w00 = I1[(i64)y0c * w + x0c], w01 = I1[(i64)y0c * w + x1c];
          float w10 = I1[(i64)y1c * w + x0c], w11 = I1[(i64)y1c * w + x1c];
          float i1v = w00 * (1 - ay) * (1 - ax) + w01 * (1 - ay) * ax +
                      w10 * ay * (1 - ax) + w11 * ay * ax;
          float it = i1v - i0v;
          a11 += ix * ix;
          a12 += ix * iy;
          a22 += iy * iy;
          b1 += ix * it;
          b2 += iy * it;
        }
      }
    }
    float det = a11 * a22 - a12 * a12;
    if (det > of::kDetEps) {
      uu -= (a22 * b1 - a12 * b2) / det;
      vv -= (a11 * b2 - a12 * b1) / det;
    }
    fout[((i64)y * w + x) * 2 + 0] = uu;
    fout[((i64)y * w + x) * 2 + 1] = vv;
  }
}

inline int grid_1d(i64 total, int block = 256, int cap = 4096) {
  return (int)std::min<i64>(cap, (total + block - 1) / block);
}

// ---------------- FlowStats: dense flow -> {mean|u|,mean|v|,max|u|,max|v|}
// Two-stage reduction (partials per chunk, then per-frame merge) — one
// workgroup can't fill an XCD, and f32 global atomics would serialize.
__global__ void __launch_bounds__(256)
    flowstats_partial_kernel(const float* const* __restrict__ flows, i64 npix,
                             float* __restrict__ partials, int chunks) {
  __shared__ float lsum[2][64], lmax[2][64];
  int frame = blockIdx.y, chunk = blockIdx.x;
  const float* f = flows[frame];
  float su = 0, sv = 0, mu = 0, mv = 0;
  for (i64 i = (i64)chunk * 256 + threadIdx.x; i < npix;
       i += (i64)chunks * 256) {
    float u = fabsf(f[i * 2]), v = fabsf(f[i * 2 + 1]);
    su += u;
    sv += v;
    mu = fmaxf(mu, u);
    mv = fmaxf(mv, v);
  }
  // wave64 shuffle reduce, then LDS across the 4 waves
  for (int off = 32; off; off >>= 1) {
    su += __shfl_down(su, off, 64);
    sv += __shfl_down(sv, off, 64);
    mu = fmaxf(mu, __shfl_down(mu, off, 64));
    mv = fmaxf(mv, __shfl_down(mv, off, 64));
  }
  int lane = threadIdx.x % 64, wave = threadIdx.x / 64;
  if (lane == 0) {
    lsum[0][wave] = su;
    lsum[1][wave] = sv;
    lmax[0][wave] = mu;
    lmax[1][wave] = mv;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float tsu = 0, tsv = 0, tmu = 0, tmv = 0;
    for (int wv = 0; wv < (int)blockDim.x / 64; ++wv) {
      tsu += lsum[0][wv];
      tsv += lsum[1][wv];
      tmu = fmaxf(tmu, lmax[0][wv]);
      tmv = fmaxf(tmv, lmax[1][wv]);
    }
    float* p = partials + ((i64)frame * chunks + chunk) * 4;
    p[0] = tsu;
    p[1] = tsv;
    p[2] = tmu;
    p[3] = tmv;
  }
}

__global__ void flowstats_reduce_kernel(const float* __restrict__ partials,
                                        int chunks, i64 npix,
                                        float* __restrict__ out) {
  int frame = blockIdx.x;
  if (threadIdx.x != 0) return;
  const float* p = partials + (i64)frame * chunks * 4;
  float su = 0, sv = 0, mu = 0, mv = 0;
  for (int cc = 0; cc < chunks; ++cc) {
    su += p[cc * 4 + 0];
    sv += p[cc * 4 + 1];
    mu = fmaxf(mu, p[cc * 4 + 2]);
    mv = fmaxf(mv, p[cc * 4 + 3]);
  }
  float* o = out + (i64)frame * 4;
  o[0] = su / npix;
  o[1] = sv / npix;
  o[2] = mu;
  o[3] = mv;
}

class FlowStatsKernelGPU : public BatchedKernel {
 public:
  using BatchedKernel::BatchedKernel;
  void execute_batch(const BatchedElements& in, BatchedElements& out) override {
    hipStream_t s = cur_stream();
    size_t n = in[0].size();
    if (n == 0) return;
    const Element& f0 = in[0][0];
    SCA_CHECK(f0.is_frame && f0.frame_info.type == FrameType::F32 &&
                  f0.frame_info.shape[2] == 2,
              "FlowStats expects H x W x 2 f32 flow frames");
    i64 npix = (i64)f0.frame_info.shape[0] * f0.frame_info.shape[1];
    int chunks = 32;
    auto dev = config_.device;
    std::vector<const u8*> ptrs(n);
    for (size_t i = 0; i < n; ++i) ptrs[i] = in[0][i].buffer;
    u8* d_ptrs = new_buffer(dev, n * sizeof(u8*));
    OF_CHECK(hipMemcpyAsync(d_ptrs, ptrs.data(), n * sizeof(u8*),
                            hipMemcpyHostToDevice, s));
    u8* scratch = new_buffer(dev, n * chunks * 4 * sizeof(float));
    u8* out_block = new_block_buffer(dev, n * 4 * sizeof(float), (i32)n);
    dim3 grid(chunks, (u32)n);
    flowstats_partial_kernel<<<grid, 256, 0, s>>>(
        (const float* const*)d_ptrs, npix, (float*)scratch, chunks);
    OF_CHECK(hipGetLastError());
    flowstats_reduce_kernel<<<(u32)n, 64, 0, s>>>((const float*)scratch,
                                                  chunks, npix,
                                                  (float*)out_block);
    OF_CHECK(hipGetLastError());
    OF_CHECK(hipStreamSynchronize(s));
    delete_buffer(dev, scratch);
    delete_buffer(dev, d_ptrs);
    for (size_t i = 0; i < n; ++i) {
      Element e;
      e.buffer = out_block + i * 4 * sizeof(float);
      e.size = 4 * sizeof(float);
      e.device = dev;
      out[0].push_back(e);
    }
  }
};

// ---------------- host kernel ----------------

class OpticalFlowKernelGPU : public BaseKernel {
 public:
  explicit OpticalFlowKernelGPU(const KernelConfig& cfg) : BaseKernel(cfg) {
    auto a = mp::decode(cfg.args);
    radius_ = (int)a.get_int("radius", of::kDefaultRadius);
    iters_ = (int)a.get_int("iters", of::kDefaultIters);
    max_levels_ = (int)a.get_int("levels", of::kDefaultMaxLevels);
    SCA_CHECK(radius_ == 2 || radius_ == 3 || radius_ == 4,
              "OpticalFlow radius must be 2, 3 or 4");
  }

  void execute(const StenciledElements& in, BatchedElements& out) override {
    hipStream_t s = cur_stream();
    size_t n = in[0].size();  // pairs
    if (n == 0) return;
    const Element& f00 = in[0][0][0];
    SCA_CHECK(f00.is_frame && f00.device.is_gpu(),
              "GPU OpticalFlow needs GPU frame input");
    int h = f00.frame_info.shape[0], w = f00.frame_info.shape[1],
        c = f00.frame_info.shape[2];
    SCA_CHECK(f00.frame_info.type == FrameType::U8,
              "OpticalFlow expects u8 frames");

    // Dedup unique frames across stencil windows (frame i closes pair i-1
    // and opens pair i).
    std::map<const u8*, int> uniq;
    std::vector<const u8*> frame_ptrs;
    std::vector<int> pf0(n), pf1(n);
    for (size_t i = 0; i < n; ++i) {
      SCA_CHECK(in[0][i].size() == 2, "OpticalFlow stencil must be [0,1]");
      for (int j = 0; j < 2; ++j) {
        const Element& e = in[0][i][j];
        SCA_CHECK(e.frame_info.shape[0] == h && e.frame_info.shape[1] == w,
                  "OpticalFlow batch with mixed frame sizes");
        auto it = uniq.find(e.buffer);
        int idx;
        if (it == uniq.end()) {
          idx = (int)frame_ptrs.size();
          uniq.emplace(e.buffer, idx);
          frame_ptrs.push_back(e.buffer);
        } else {
          idx = it->second;
        }
        (j == 0 ? pf0 : pf1)[i] = idx;
      }
    }
    int nf = (int)frame_ptrs.size();

    int levels = num_levels_clamped(h, w);
    // Pyramid geometry + one gray buffer holding all levels of all frames:
    // [level][frame][h_l*w_l], levels concatenated.
    std::vector<int> lh(levels), lw(levels);
    std::vector<i64> loff(levels);
    i64 gray_elems = 0;
    for (int l = 0; l < levels; ++l) {
      lh[l] = l ? lh[l - 1] / 2 : h;
      lw[l] = l ? lw[l - 1] / 2 : w;
      loff[l] = gray_elems;
      gray_elems += (i64)nf * lh[l] * lw[l];
    }

    auto dev = config_.device;
    u8* gray = new_buffer(dev, gray_elems * sizeof(float));
    u8* d_ptrs = new_buffer(dev, nf * sizeof(u8*));
    u8* d_pf = new_buffer(dev, 2 * n * sizeof(int));
    OF_CHECK(hipMemcpyAsync(d_ptrs, frame_ptrs.data(), nf * sizeof(u8*),
                            hipMemcpyHostToDevice, s));
    OF_CHECK(hipMemcpyAsync(d_pf, pf0.data(), n * sizeof(int),
                            hipMemcpyHostToDevice, s));
    OF_CHECK(hipMemcpyAsync(d_pf + n * sizeof(int), pf1.data(),
                            n * sizeof(int), hipMemcpyHostToDevice, s));

    float* gbase = (float*)gray;
    of_gray_kernel<<<grid_1d((i64)nf * h * w), 256, 0, s>>>(
        (const u8* const*)d_ptrs, nf, h, w, c, gbase);
    OF_CHECK(hipGetLastError());
    for (int l = 1; l < levels; ++l) {
      of_down2x_kernel<<<grid_1d((i64)nf * lh[l] * lw[l]), 256, 0, s>>>(
          gbase + loff[l - 1], lh[l - 1], lw[l - 1], gbase + loff[l], lh[l],
          lw[l], nf);
      OF_CHECK(hipGetLastError());
    }

    // Flow ping-pong buffers sized for level 0; output is a block buffer so
    // the whole packet's flow is one allocation.
    i64 flow_bytes = (i64)n * h * w * 2 * sizeof(float);
    u8* ping = new_buffer(dev, flow_bytes);
    u8* pong = new_buffer(dev, flow_bytes);
    u8* out_block = new_block_buffer(dev, flow_bytes, (i32)n);
    float* cur = (float*)ping;
    float* alt = (float*)pong;

    int top = levels - 1;
    OF_CHECK(hipMemsetAsync(cur, 0,
                            (i64)n * lh[top] * lw[top] * 2 * sizeof(float),
                            s));
    const int* d_pf0 = (const int*)d_pf;
    const int* d_pf1 = (const int*)(d_pf + n * sizeof(int));
    for (int l = top; l >= 0; --l) {
      // multiple output rows per thread when the level is tall enough
      // to keep the grid chip-filling; tiny coarse levels stay at 1
      // (SCANNER_LK_PY overrides for A/B)
      static const int py_max = []() {
        const char* e = std::getenv("SCANNER_LK_PY");
        int v = e ? atoi(e) : 2;
        return v == 1 || v == 2 || v == 4 ? v : 2;
      }();
      int py = (lh[l] >= 64 && (i64)n * ((lw[l] + 15) / 16) *
                                   ((lh[l] + 31) / 32) >= 512) ? py_max : 1;
      if (py == 4 && (lh[l] < 128 || (i64)n * ((lw[l] + 15) / 16) *
                                         ((lh[l] + 63) / 64) < 512)) {
        py = 2;
      }
      dim3 grid((lw[l] + 15) / 16, (lh[l] + 16 * py - 1) / (16 * py),
                (u32)n);
      dim3 block(16, 16);
      for (int it = 0; it < iters_; ++it) {
        float* dst = (l == 0 && it == iters_ - 1) ? (float*)out_block : alt;
        if (l < top && it == 0) {
          // first iteration of each finer level: fused upsample read from
          // the coarser level's result (cur holds level l+1 flow)
          launch_lk(py, grid, block, s, gbase, loff[l], lh[l], lw[l],
                    d_pf0, d_pf1, nullptr, dst, cur, lh[l + 1], lw[l + 1]);
        } else {
          launch_lk(py, grid, block, s, gbase, loff[l], lh[l], lw[l],
                    d_pf0, d_pf1, cur, dst);
        }
        std::swap(cur, alt);
        if (dst == (float*)out_block) cur = (float*)out_block;
      }
    }

    // Scratch feeds kernels on this stream; sync before returning it to the
    // shared pool.
    OF_CHECK(hipStreamSynchronize(s));
    delete_buffer(dev, gray);
    delete_buffer(dev, d_ptrs);
    delete_buffer(dev, d_pf);
    delete_buffer(dev, ping);
    delete_buffer(dev, pong);

    i64 per_pair = (i64)h * w * 2 * sizeof(float);
    for (size_t i = 0; i < n; ++i) {
      Element e;
      e.is_frame = true;
      e.frame_info.shape[0] = h;
      e.frame_info.shape[1] = w;
      e.frame_info.shape[2] = 2;
      e.frame_info.type = FrameType::F32;
      e.size = per_pair;
      e.buffer = out_block + i * per_pair;
      e.device = dev;
      out[0].push_back(e);
    }
  }

 private:
  int num_levels_clamped(int h, int w) const {
    return of::num_levels(h, w, max_levels_);
  }

  void launch_lk(int py, dim3 grid, dim3 block, hipStream_t s,
                 const float* gray, i64 off, int h, int w, const int* pf0,
                 const int* pf1, const float* fin, float* fout,
                 const float* coarse = nullptr, int ch = 0, int cw = 0) {
    auto go = [&](auto rad, auto pyc) {
      of_lk_kernel<decltype(rad)::value, decltype(pyc)::value>
          <<<grid, block, 0, s>>>(gray, off, h, w, pf0, pf1, fin, fout,
                                  coarse, ch, cw);
    };
    auto by_r = [&](auto pyc) {
      switch (radius_) {
        case 2: go(std::integral_constant<int, 2>{}, pyc); break;
        case 3: go(std::integral_constant<int, 3>{}, pyc); break;
        default: go(std::integral_constant<int, 4>{}, pyc); break;
      }
    };
    if (py == 4)
      by_r(std::integral_constant<int, 4>{});
    else if (py == 2)
      by_r(std::integral_constant<int, 2>{});
    else
      by_r(std::integral_constant<int, 1>{});
    OF_CHECK(hipGetLastError());
  }

  int radius_, iters_, max_levels_;
};

}  // namespace

void register_optflow_gpu() {
  static bool done = false;
  if (done) return;
  done = true;
  KernelFactory f;
  f.op_name = "OpticalFlow";
  f.device_type = DeviceType::GPU;
  f.preferred_batch = 16;
  f.make = [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
    return std::make_unique<OpticalFlowKernelGPU>(c);
  };
  kernel_registry().add(f);
  {
    // OpInfo for FlowStats is registered by register_stdlib_ops().
    KernelFactory fs;
    fs.op_name = "FlowStats";
    fs.device_type = DeviceType::GPU;
    fs.preferred_batch = 32;
    fs.make = [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
      return std::make_unique<FlowStatsKernelGPU>(c);
    };
    kernel_registry().add(fs);
  }
}

}  // namespace sca
