// Launch APIs for the DNN kernels (implemented in gemm_mfma.hip and
// dnn_ops.hip; host orchestration in csrc/ops/resnet50.cpp). All pointers
// are device memory; all launches go onto the caller's HIP stream.
#pragma once

#include "../csrc/common.h"

namespace sca {

struct GemmArgs {
  const void* A = nullptr;  // bf16 [M][K] row-major
  const void* B = nullptr;  // bf16 [N][K] row-major (transposed weights)
  void* C = nullptr;        // bf16 [M][N]
  int M = 0, N = 0, K = 0;  // K, N multiples of 64
  const float* scale = nullptr;     // per-col (folded BN gamma/sqrt(var))
  const float* bias = nullptr;      // per-col
  const void* residual = nullptr;   // bf16 [M][N] added before activation
  bool relu = false;
  // Optional split-K workspace (f32 partials). When set and the shape is
  // launch-bound (few output tiles, deep K), gemm_bf16 splits K across
  // workgroups and reduces. Caller guarantees the buffer lives until the
  // stream syncs (the DNN ops pass a slice of their per-execute
  // workspace).
  void* splitk_scratch = nullptr;
  size_t splitk_scratch_bytes = 0;
};

void gemm_bf16(const GemmArgs& g, void* stream);

// Zero the fused-reduce tile-counter region at the TAIL of a split-K
// scratch buffer. Owners must call this ONCE when they allocate the
// scratch (the kernel self-restores the counters to zero after each
// launch, so no per-call work is needed). stream may be null (synchronous
// memset).
void splitk_scratch_init(void* scratch, size_t bytes, void* stream);

// Implicit-GEMM convolution: the GEMM's A operand is gathered straight
// from the NHWC activation tensor during LDS staging (no im2col buffer in
// HBM). g.A = activations (NHWC bf16); g.M = n*oh*ow; g.K = kp (padded
// r*s*c). Requires c % 8 == 0 (each 16-byte staged chunk must stay within
// one (r,s) cell's channel run).
struct ConvDesc {
  int n, h, w, c;       // input NHWC
  int r, s, stride, pad;
  int oh, ow;
};

void conv_gemm_bf16(const GemmArgs& g, const ConvDesc& d, void* stream);

// u8 HWC frames (device pointer array) -> normalized bf16 NHWC out_hw x
// out_hw x 3, bilinear resize. mean/std per channel (device, 3 floats).
void preprocess_frames_bf16(const void* frames_ptr_array, int n, int in_h,
                            int in_w, int in_c, int out_hw, void* out,
                            const float* mean, const float* std_,
                            void* stream, int out_c = 3);

// NHWC bf16 -> im2col rows [n*out_h*out_w][k_padded] (zero-padded past
// r*s*c). stride/pad symmetric.
void im2col_bf16(const void* in, int n, int h, int w, int c, int r, int s,
                 int stride, int pad, void* out, int out_h, int out_w,
                 int k_padded, void* stream);

// 3x3 stride-2 pad-1 max pool, NHWC bf16.
void maxpool3x3s2_bf16(const void* in, int n, int h, int w, int c, void* out,
                       int out_h, int out_w, void* stream);

// global average pool NHWC -> [n][c] bf16
void global_avgpool_bf16(const void* in, int n, int h, int w, int c,
                         void* out, void* stream);

// [n][stride_cols] bf16 -> [n][ncols] f32 (row-sliced cast)
void bf16_rows_to_f32(const void* in, int n, int stride_cols, int ncols,
                      void* out, void* stream);

// NHWC channel concat of three bf16 tensors sharing npix pixels; per-input
// channel strides let branches read from GEMM-padded buffers.
void concat3_bf16(const void* a, int ca, int stride_a, const void* b, int cb,
                  int stride_b, const void* c, int cc, int stride_c, i64 npix,
                  void* out, int out_stride, void* stream);

// Per-(frame,channel) spatial argmax of NHWC bf16 maps (padded channel
// stride c_stride): out[n][nch][3] f32 = {x, y, peak}.
void heatmap_argmax(const void* maps, int n, int h, int w, int c_stride,
                    int nch, void* out, void* stream);

}  // namespace sca
