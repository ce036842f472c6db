// Auxiliary DNN kernels for gfx950: fused preprocess (resize+normalize to
// bf16), im2col, pooling. All memory-bound: grid-stride loops, vectorized
// where the channel count allows (cdna_hip_programming.md Guideline 13).
#include <hip/hip_runtime.h>

#include "../csrc/memory.h"
#include "dnn.h"

namespace sca {

namespace {

using bf16 = __bf16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;

#define DNN_CHECK()                                                      \
  do {                                                                   \
    hipError_t _e = hipGetLastError();                                   \
    if (_e != hipSuccess) {                                              \
      throw ScannerError(std::string("dnn kernel launch: ") +            \
                         hipGetErrorString(_e));                         \
    }                                                                    \
  } while (0)

__global__ void __launch_bounds__(256)
    preprocess_kernel(const u8* const* __restrict__ frames, int n, int ih,
                      int iw, int ic, int ohw, int oc,
                      bf16* __restrict__ out,
                      const float* __restrict__ mean,
                      const float* __restrict__ stdv) {
  // oc > 3 writes zero-padded channels: the consumer is the implicit-GEMM
  // conv path, whose 16-byte LDS-DMA gather needs c % 8 == 0 — padding at
  // preprocess removes conv1's explicit im2col HBM round trip entirely.
  i64 total = (i64)n * ohw * ohw * oc;
  i64 stride = (i64)gridDim.x * blockDim.x;
  for (i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int c = (int)(i % oc);
    if (c >= 3) {
      out[i] = (bf16)0.f;
      continue;
    }
    i64 pix = i / oc;
    int x = (int)(pix % ohw);
    i64 t = pix / ohw;
    int y = (int)(t % ohw);
    int f = (int)(t / ohw);
    const u8* src = frames[f];
    float sy = (y + 0.5f) * ih / ohw - 0.5f;
    float sx = (x + 0.5f) * iw / ohw - 0.5f;
    int y0 = max(0, (int)floorf(sy)), x0 = max(0, (int)floorf(sx));
    int y1 = min(ih - 1, y0 + 1), x1 = min(iw - 1, x0 + 1);
    float fy = fminf(fmaxf(sy - y0, 0.f), 1.f);
    float fx = fminf(fmaxf(sx - x0, 0.f), 1.f);
    int cc = c < ic ? c : ic - 1;
    float v00 = src[((i64)y0 * iw + x0) * ic + cc];
    float v01 = src[((i64)y0 * iw + x1) * ic + cc];
    float v10 = src[((i64)y1 * iw + x0) * ic + cc];
    float v11 = src[((i64)y1 * iw + x1) * ic + cc];
    float v = v00 * (1 - fy) * (1 - fx) + v01 * (1 - fy) * fx +
              v10 * fy * (1 - fx) + v11 * fy * fx;
    v = (v / 255.f - mean[c]) / stdv[c];
    out[i] = (bf16)v;
  }
}

// One thread per 8 output elements along the K axis when c % 8 == 0,
// else scalar. Output rows are [n*oh*ow][k_padded], zero past r*s*c.
__global__ void __launch_bounds__(256)
    im2col_kernel(const bf16* __restrict__ in, int n, int h, int w, int c,
                  int r, int s, int stride_, int pad,
                  bf16* __restrict__ out, int oh, int ow, int kp) {
  i64 rows = (i64)n * oh * ow;
  i64 total = rows * kp;
  i64 gstride = (i64)gridDim.x * blockDim.x * 8;
  for (i64 base = ((i64)blockIdx.x * blockDim.x + threadIdx.x) * 8;
       base < total; base += gstride) {
    i64 row = base / kp;
    int k = (int)(base % kp);
    int q = (int)(row % ow);
    i64 t = row / ow;
    int p = (int)(t % oh);
    int f = (int)(t / oh);
    int krs = r * s * c;
    // 8 consecutive k: when c%8==0 and k+8 <= within one (dr,ds) cell,
    // they map to 8 consecutive input channels.
    if (k + 8 <= krs && (k % c) + 8 <= c && (k / c) == ((k + 7) / c)) {
      int cell = k / c;
      int dr = cell / s, ds = cell % s;
      int hh = p * stride_ - pad + dr;
      int ww = q * stride_ - pad + ds;
      if (hh < 0 || hh >= h || ww < 0 || ww >= w) {
        for (int j = 0; j < 8; ++j) out[base + j] = (bf16)0.f;
      } else {
        const bf16* src =
            in + (((i64)f * h + hh) * w + ww) * c + (k % c);
        *reinterpret_cast<ulonglong2*>(out + base) =
            *reinterpret_cast<const ulonglong2*>(src);
      }
    } else {
      for (int j = 0; j < 8 && base + j < total; ++j) {
        int kj = k + j;
        if (kj >= krs) {
          out[base + j] = (bf16)0.f;
          continue;
        }
        int cj = kj % c;
        int cell = kj / c;
        int dr = cell / s, ds = cell % s;
        int hh = p * stride_ - pad + dr;
        int ww = q * stride_ - pad + ds;
        if (hh < 0 || hh >= h || ww < 0 || ww >= w) {
          out[base + j] = (bf16)0.f;
        } else {
          out[base + j] = in[(((i64)f * h + hh) * w + ww) * c + cj];
        }
      }
    }
  }
}

__global__ void __launch_bounds__(256)
    maxpool_kernel(const bf16* __restrict__ in, int n, int h, int w, int c,
                   bf16* __restrict__ out, int oh, int ow) {
  i64 total = (i64)n * oh * ow * c;
  i64 stride = (i64)gridDim.x * blockDim.x;
  for (i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int cc = (int)(i % c);
    i64 t = i / c;
    int x = (int)(t % ow);
    t /= ow;
    int y = (int)(t % oh);
    int f = (int)(t / oh);
    float best = -1e30f;
    for (int dy = 0; dy < 3; ++dy) {
      for (int dx = 0; dx < 3; ++dx) {
        int yy = y * 2 - 1 + dy, xx = x * 2 - 1 + dx;
        if (yy < 0 || yy >= h || xx < 0 || xx >= w) continue;
        float v = (float)in[(((i64)f * h + yy) * w + xx) * c + cc];
        best = fmaxf(best, v);
      }
    }
    out[i] = (bf16)best;
  }
}

// Vectorized variant for c % 8 == 0 (the resnet path, c=64): one thread
// owns 8 channels, 16-byte loads per window tap — the scalar kernel's
// 2-byte gathers ran at 2.3 TB/s.
__global__ void __launch_bounds__(256)
    maxpool8_kernel(const bf16* __restrict__ in, int n, int h, int w, int c,
                    bf16* __restrict__ out, int oh, int ow) {
  int c8 = c / 8;
  i64 total = (i64)n * oh * ow * c8;
  i64 stride = (i64)gridDim.x * blockDim.x;
  for (i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int cc = (int)(i % c8) * 8;
    i64 t = i / c8;
    int x = (int)(t % ow);
    t /= ow;
    int y = (int)(t % oh);
    int f = (int)(t / oh);
    float best[8];
#pragma unroll
    for (int k = 0; k < 8; ++k) best[k] = -1e30f;
    for (int dy = 0; dy < 3; ++dy) {
      int yy = y * 2 - 1 + dy;
      if (yy < 0 || yy >= h) continue;
      for (int dx = 0; dx < 3; ++dx) {
        int xx = x * 2 - 1 + dx;
        if (xx < 0 || xx >= w) continue;
        bf16x8 v = *reinterpret_cast<const bf16x8*>(
            in + (((i64)f * h + yy) * w + xx) * c + cc);
#pragma unroll
        for (int k = 0; k < 8; ++k) best[k] = fmaxf(best[k], (float)v[k]);
      }
    }
    bf16x8 o;
#pragma unroll
    for (int k = 0; k < 8; ++k) o[k] = (bf16)best[k];
    *reinterpret_cast<bf16x8*>(out + i * 8) = o;
  }
}

__global__ void __launch_bounds__(256)
    avgpool_kernel(const bf16* __restrict__ in, int n, int h, int w, int c,
                   bf16* __restrict__ out) {
  i64 total = (i64)n * c;
  i64 stride = (i64)gridDim.x * blockDim.x;
  for (i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int cc = (int)(i % c);
    int f = (int)(i / c);
    float sum = 0.f;
    for (int p = 0; p < h * w; ++p) {
      sum += (float)in[((i64)f * h * w + p) * c + cc];
    }
    out[i] = (bf16)(sum / (h * w));
  }
}

inline int grid_for(i64 total, int per_thread = 1) {
  i64 blocks = (total / per_thread + 255) / 256;
  return (int)std::min<i64>(4096, std::max<i64>(1, blocks));
}

}  // namespace

void preprocess_frames_bf16(const void* frames_ptr_array, int n, int in_h,
                            int in_w, int in_c, int out_hw, void* out,
                            const float* mean, const float* std_,
                            void* stream, int out_c) {
  i64 total = (i64)n * out_hw * out_hw * out_c;
  preprocess_kernel<<<grid_for(total), 256, 0, (hipStream_t)stream>>>(
      (const u8* const*)frames_ptr_array, n, in_h, in_w, in_c, out_hw,
      out_c, (bf16*)out, mean, std_);
  DNN_CHECK();
}

void im2col_bf16(const void* in, int n, int h, int w, int c, int r, int s,
                 int stride, int pad, void* out, int out_h, int out_w,
                 int k_padded, void* stream) {
  i64 total = (i64)n * out_h * out_w * k_padded;
  im2col_kernel<<<grid_for(total, 8), 256, 0, (hipStream_t)stream>>>(
      (const bf16*)in, n, h, w, c, r, s, stride, pad, (bf16*)out, out_h,
      out_w, k_padded);
  DNN_CHECK();
}

void maxpool3x3s2_bf16(const void* in, int n, int h, int w, int c, void* out,
                       int out_h, int out_w, void* stream) {
  if (c % 8 == 0) {
    i64 total = (i64)n * out_h * out_w * (c / 8);
    maxpool8_kernel<<<grid_for(total), 256, 0, (hipStream_t)stream>>>(
        (const bf16*)in, n, h, w, c, (bf16*)out, out_h, out_w);
  } else {
    i64 total = (i64)n * out_h * out_w * c;
    maxpool_kernel<<<grid_for(total), 256, 0, (hipStream_t)stream>>>(
        (const bf16*)in, n, h, w, c, (bf16*)out, out_h, out_w);
  }
  DNN_CHECK();
}

void global_avgpool_bf16(const void* in, int n, int h, int w, int c,
                         void* out, void* stream) {
  i64 total = (i64)n * c;
  avgpool_kernel<<<grid_for(total), 256, 0, (hipStream_t)stream>>>(
      (const bf16*)in, n, h, w, c, (bf16*)out);
  DNN_CHECK();
}

namespace {
__global__ void __launch_bounds__(256)
    cast_rows_kernel(const bf16* __restrict__ in, i64 n, int stride_cols,
                     int ncols, float* __restrict__ out) {
  i64 total = n * ncols;
  i64 gs = (i64)gridDim.x * blockDim.x;
  for (i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gs) {
    i64 row = i / ncols;
    int col = (int)(i % ncols);
    out[i] = (float)in[row * stride_cols + col];
  }
}
}  // namespace

void bf16_rows_to_f32(const void* in, int n, int stride_cols, int ncols,
                      void* out, void* stream) {
  i64 total = (i64)n * ncols;
  cast_rows_kernel<<<grid_for(total), 256, 0, (hipStream_t)stream>>>(
      (const bf16*)in, n, stride_cols, ncols, (float*)out);
  DNN_CHECK();
}


// ---- pose-net helpers (csrc/ops/pose.cpp) ----

namespace {
// NHWC channel concat of three tensors (b and c channel counts may read
// from GEMM-padded buffers via stride args).
__global__ void __launch_bounds__(256)
    concat3_kernel(const bf16* __restrict__ a, int ca, int sa,
                   const bf16* __restrict__ b, int cb, int sb,
                   const bf16* __restrict__ c, int cc, int scc, i64 npix,
                   bf16* __restrict__ out, int ct) {
  // ct >= ca+cb+cc: channels past the concat are zero-filled so the
  // consumer can run with a c%8==0 stride (implicit-GEMM requirement)
  i64 total = npix * ct;
  i64 gs = (i64)gridDim.x * blockDim.x;
  for (i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gs) {
    i64 pix = i / ct;
    int ch = (int)(i % ct);
    bf16 v;
    if (ch < ca)
      v = a[pix * sa + ch];
    else if (ch < ca + cb)
      v = b[pix * sb + (ch - ca)];
    else if (ch < ca + cb + cc)
      v = c[pix * scc + (ch - ca - cb)];
    else
      v = (bf16)0.f;
    out[i] = v;
  }
}

// Per-(frame, channel) spatial argmax over an NHWC bf16 map with padded
// channel stride: out[frame][ch] = {x, y, score} f32. One workgroup per
// (frame, channel); wave64 shuffle reduction.
__global__ void __launch_bounds__(256)
    heatmap_argmax_kernel(const bf16* __restrict__ maps, int h, int w,
                          int c_stride, int nch, float* __restrict__ out) {
  int frame = blockIdx.y, ch = blockIdx.x;
  const bf16* m = maps + (i64)frame * h * w * c_stride + ch;
  float best = -1e30f;
  int besti = 0;
  for (int i = threadIdx.x; i < h * w; i += blockDim.x) {
    float v = (float)m[(i64)i * c_stride];
    if (v > best) {
      best = v;
      besti = i;
    }
  }
  __shared__ float lbest[4];
  __shared__ int lidx[4];
  for (int off = 32; off; off >>= 1) {
    float ov = __shfl_down(best, off, 64);
    int oi = __shfl_down(besti, off, 64);
    if (ov > best) {
      best = ov;
      besti = oi;
    }
  }
  int lane = threadIdx.x % 64, wave = threadIdx.x / 64;
  if (lane == 0) {
    lbest[wave] = best;
    lidx[wave] = besti;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int wv = 1; wv < (int)blockDim.x / 64; ++wv) {
      if (lbest[wv] > best) {
        best = lbest[wv];
        besti = lidx[wv];
      }
    }
    float* o = out + ((i64)frame * nch + ch) * 3;
    o[0] = (float)(besti % w);
    o[1] = (float)(besti / w);
    o[2] = best;
  }
}
}  // namespace

void concat3_bf16(const void* a, int ca, int stride_a, const void* b, int cb,
                  int stride_b, const void* c, int cc, int stride_c, i64 npix,
                  void* out, int out_stride, void* stream) {
  i64 total = npix * out_stride;
  concat3_kernel<<<grid_for(total), 256, 0, (hipStream_t)stream>>>(
      (const bf16*)a, ca, stride_a, (const bf16*)b, cb, stride_b,
      (const bf16*)c, cc, stride_c, npix, (bf16*)out, out_stride);
  DNN_CHECK();
}

void heatmap_argmax(const void* maps, int n, int h, int w, int c_stride,
                    int nch, void* out, void* stream) {
  dim3 grid(nch, n);
  heatmap_argmax_kernel<<<grid, 256, 0, (hipStream_t)stream>>>(
      (const bf16*)maps, h, w, c_stride, nch, (float*)out);
  DNN_CHECK();
}

}  // namespace sca
