// Hand-written bf16 GEMM on MFMA for gfx950 — the conv/GEMM engine of the
// DNN-inference op (ResNet-50). Structure follows the CDNA4 canonical GEMM
// anatomy (cdna_hip_programming.md §5): 128-wide tiles, global_load_lds
// 16-byte staging into double-buffered LDS, v_mfma_f32_16x16x32_bf16 inner
// loop accumulating fp32, fused epilogue (per-channel scale+bias = folded
// BN, ReLU, residual add) writing bf16.
//
// C[M,N] = A[M,K] @ B[N,K]^T   (A row-major [M][K]; B stored [N][K] so
// both operands are contiguous in K — weights are laid out at init time).
// K and N must be multiples of 64 (callers pad); M is arbitrary.
#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstdlib>
#include <map>
#include <mutex>

#include "../csrc/memory.h"
#include "dnn.h"

namespace sca {

namespace {

using bf16 = __bf16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__device__ inline float bf16_to_f32(bf16 v) { return (float)v; }

__device__ inline bf16 f32_to_bf16(float v) { return (bf16)v; }

// Coalesced epilogue: the MFMA fragment layout leaves each lane holding
// 4 rows x 1 col, so direct stores are 2-byte scattered (measured 1.4 TB/s
// on store-bound shapes — 1/6 of HBM). Transpose each 16-row block through
// per-wave LDS scratch (write col-major with scale/bias applied, read back
// 16 consecutive cols per lane), then store 32 contiguous bytes per lane —
// full 64-byte-per-4-lane coalescing. Scratch is per-wave, so only
// intra-wave lgkmcnt ordering is needed: no block barrier, which keeps the
// small-K kernel's cross-tile A prefetch in flight.
template <int FM, int FN, bool RELU, bool RESIDUAL>
__device__ inline void epilogue_store(f32x4 (&acc)[FM][FN], float* scratch,
                                      bf16* __restrict__ C,
                                      const bf16* __restrict__ residual,
                                      int M, int N, int m0, int n0, int wrow,
                                      int wcol, int lane,
                                      const float* __restrict__ scale,
                                      const float* __restrict__ bias) {
  constexpr int COLS = FN * 16;
  constexpr int STRIDE = COLS + 4;  // keeps 16 B row alignment, breaks banks
#pragma unroll
  for (int i = 0; i < FM; ++i) {
    asm volatile("s_waitcnt lgkmcnt(0)");  // WAR: prior reads done
#pragma unroll
    for (int j = 0; j < FN; ++j) {
      int cw = j * 16 + (lane & 15);
      int col = n0 + wcol * COLS + cw;
      float sc = scale ? scale[col] : 1.f;
      float bi = bias ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        scratch[((lane >> 4) * 4 + r) * STRIDE + cw] =
            acc[i][j][r] * sc + bi;
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)");  // writes visible to own reads
    constexpr int CHUNK = COLS / 4;        // contiguous cols per lane (>=8)
    static_assert(CHUNK >= 8 && CHUNK % 8 == 0, "wave tile too narrow");
    int r2 = lane >> 2;                    // 16 rows
    int q = lane & 3;                      // 4 chunks per row
    int row = m0 + wrow * (FM * 16) + i * 16 + r2;
    if (row < M) {
      int colbase = n0 + wcol * COLS + q * CHUNK;
      const float* src = scratch + r2 * STRIDE + q * CHUNK;
      bf16 out[CHUNK];
#pragma unroll
      for (int k = 0; k < CHUNK; ++k) {
        float v = src[k];
        if constexpr (RESIDUAL) {
          v += bf16_to_f32(residual[(size_t)row * N + colbase + k]);
        }
        if constexpr (RELU) v = v > 0.f ? v : 0.f;
        out[k] = f32_to_bf16(v);
      }
      // 16-byte stores, contiguous across the 4 q-lanes of each row
#pragma unroll
      for (int c8 = 0; c8 < CHUNK; c8 += 8) {
        *reinterpret_cast<bf16x8*>(C + (size_t)row * N + colbase + c8) =
            *reinterpret_cast<const bf16x8*>(out + c8);
      }
    }
  }
}

// One workgroup = 256 threads = 4 waves in a WM x WN grid; each wave owns
// a (BM/WM) x (BN/WN) output sub-tile as FM x FN fragments of 16x16.
// When IMPLICIT, A is the NHWC activation tensor and the A staging
// gathers im2col rows on the fly (ConvDesc d gives the geometry; zero
// points at 16+ zero bytes for padding / k-pad lanes). Lane-addressed
// global_load_lds makes the gather free of any HBM im2col round trip.
template <int BM, int BN, int WM, int WN, bool RELU, bool RESIDUAL,
          bool IMPLICIT>
__global__ void __launch_bounds__(256, 2)
    gemm_bf16_kernel(const bf16* __restrict__ A, const bf16* __restrict__ B,
                     bf16* __restrict__ C, int M, int N, int K,
                     const float* __restrict__ scale,
                     const float* __restrict__ bias,
                     const bf16* __restrict__ residual, ConvDesc d,
                     const bf16* __restrict__ zero) {
  constexpr int BK = 64;
  constexpr int FM = BM / WM / 16;
  constexpr int FN = BN / WN / 16;
  // LDS: [2 buffers][A BM rows + B BN rows][BK]
  __shared__ bf16 lds[2 * (BM + BN) * BK];

  int tid = threadIdx.x;
  int lane = tid & 63;
  int wave = tid >> 6;
  int wrow = wave / WN;
  int wcol = wave % WN;

  // XCD-aware swizzle (T1): consecutive blockIdx.x stay on one XCD's share
  // of the N dimension when the grid allows.
  int nwg = gridDim.x;
  int wgid = blockIdx.x;
  if (nwg % 8 == 0) {
    int q = nwg / 8;
    wgid = (wgid % q) * 8 + wgid / q;
  }
  int ntiles_n = (N + BN - 1) / BN;
  int m0 = (wgid / ntiles_n) * BM;
  int n0 = (wgid % ntiles_n) * BN;

  f32x4 acc[FM][FN];
#pragma unroll
  for (int i = 0; i < FM; ++i)
#pragma unroll
    for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // Staging: each wave covers 8 rows x 64 k per glds instruction
  // (64 lanes x 16 B = 8 bf16 each). A tile: BM rows -> BM/8 instrs
  // over 4 waves; same for B.
  constexpr int A_INSTRS = BM / 8 / 4;  // per wave
  constexpr int B_INSTRS = BN / 8 / 4;
  // Implicit-conv staging geometry: the output-pixel part of each lane's
  // gather address is k-independent, and (cell, cj, dr, ds) advance by a
  // constant per BK step — precompute the former, step the latter
  // incrementally so the hot stage() does ZERO integer divisions (3 divs
  // per LDS-DMA instruction otherwise; K/64 steps x 4 instrs adds up).
  const int lrow = lane >> 3;          // 0..7 row within instr block
  const int lk = (lane & 7) * 8;       // k offset (8 bf16 = 16B)
  i64 imp_rowbase[IMPLICIT ? A_INSTRS : 1];  // nn*h*w*c + cj later
  int imp_hh0[IMPLICIT ? A_INSTRS : 1];      // p*stride - pad
  int imp_ww0[IMPLICIT ? A_INSTRS : 1];      // q*stride - pad
  int imp_cell = 0, imp_cj = 0, imp_dr = 0, imp_ds = 0;
  if constexpr (IMPLICIT) {
#pragma unroll
    for (int i = 0; i < A_INSTRS; ++i) {
      int row = (wave * A_INSTRS + i) * 8 + lrow;
      int grow = m0 + row;
      if (grow >= M) grow = M - 1;
      int q = grow % d.ow;
      int t = grow / d.ow;
      int p = t % d.oh;
      int nn = t / d.oh;
      imp_rowbase[i] = (i64)nn * d.h * d.w * d.c;
      imp_hh0[i] = p * d.stride - d.pad;
      imp_ww0[i] = q * d.stride - d.pad;
    }
    imp_cell = lk / d.c;
    imp_cj = lk - imp_cell * d.c;
    imp_dr = imp_cell / d.s;
    imp_ds = imp_cell - imp_dr * d.s;
  }
  auto imp_advance = [&](int dk) {
    // advance (cell, cj) by dk k-positions; c >= 8 so the wrap loop is
    // short and branch-uniform across the wave (same cj for all lanes of
    // one lk — lanes differ only via lk, folded into the initial state)
    imp_cj += dk;
    while (imp_cj >= d.c) {
      imp_cj -= d.c;
      ++imp_cell;
      ++imp_ds;
      if (imp_ds == d.s) {
        imp_ds = 0;
        ++imp_dr;
      }
    }
  };
  auto stage = [&](int buf, int k0) {
    bf16* lds_a = lds + buf * (BM + BN) * BK;
    bf16* lds_b = lds_a + BM * BK;
#pragma unroll
    for (int i = 0; i < A_INSTRS; ++i) {
      int row = (wave * A_INSTRS + i) * 8 + lrow;
      int grow = m0 + row;
      if (grow >= M) grow = M - 1;  // clamp: garbage rows masked at store
      const bf16* src;
      if constexpr (IMPLICIT) {
        if (imp_cell >= d.r * d.s) {
          src = zero;
        } else {
          int hh = imp_hh0[i] + imp_dr;
          int ww = imp_ww0[i] + imp_ds;
          src = (hh >= 0 && hh < d.h && ww >= 0 && ww < d.w)
                    ? A + (imp_rowbase[i] + ((i64)hh * d.w + ww) * d.c +
                           imp_cj)
                    : zero;
        }
      } else {
        src = A + (size_t)grow * K + k0 + lk;
      }
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)src,
          (__attribute__((address_space(3)))
               uint32_t*)(lds_a + (size_t)(wave * A_INSTRS + i) * 8 * BK),
          16, 0, 0);
    }
#pragma unroll
    for (int i = 0; i < B_INSTRS; ++i) {
      int row = (wave * B_INSTRS + i) * 8 + lrow;
      const bf16* src = B + (size_t)(n0 + row) * K + k0 + lk;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)src,
          (__attribute__((address_space(3)))
               uint32_t*)(lds_b + (size_t)(wave * B_INSTRS + i) * 8 * BK),
          16, 0, 0);
    }
  };

  int ksteps = K / BK;
  stage(0, 0);
  for (int ks = 0; ks < ksteps; ++ks) {
    int buf = ks & 1;
    // prefetch next while computing current
    __builtin_amdgcn_s_waitcnt(/*vmcnt(0) lgkmcnt(0)*/ 0);
    __syncthreads();
    if (ks + 1 < ksteps) {
      if constexpr (IMPLICIT) imp_advance(BK);
      stage(buf ^ 1, (ks + 1) * BK);
    }

    const bf16* lds_a = lds + buf * (BM + BN) * BK;
    const bf16* lds_b = lds_a + BM * BK;
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {  // two K=32 chunks per BK
      int kbase = kk * 32 + (lane >> 4) * 8;
      bf16x8 afrag[FM], bfrag[FN];
#pragma unroll
      for (int i = 0; i < FM; ++i) {
        int row = wrow * (BM / WM) + i * 16 + (lane & 15);
        afrag[i] = *reinterpret_cast<const bf16x8*>(
            lds_a + (size_t)row * BK + kbase);
      }
#pragma unroll
      for (int j = 0; j < FN; ++j) {
        int col = wcol * (BN / WN) + j * 16 + (lane & 15);
        bfrag[j] = *reinterpret_cast<const bf16x8*>(
            lds_b + (size_t)col * BK + kbase);
      }
#pragma unroll
      for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    }
    // No trailing barrier: the next iteration's leading waitcnt+barrier
    // orders everything, and a __syncthreads() here would drain the
    // in-flight LDS-DMA prefetch (vmcnt(0) in its fence).
  }

  // epilogue: D = act(acc * scale[col] + bias[col] [+ residual]), stored
  // coalesced via per-wave LDS transpose. The scratch overlays the staging
  // LDS (no extra LDS), so one barrier first: every wave must be done
  // reading its final fragments before any wave overwrites the buffer.
  // Nothing is in flight here (the last iteration stages no prefetch), so
  // this barrier drains nothing.
  __syncthreads();
  constexpr int EP_STRIDE = BN / WN + 4;
  float* scratch =
      reinterpret_cast<float*>(lds) + (size_t)wave * 16 * EP_STRIDE;
  epilogue_store<FM, FN, RELU, RESIDUAL>(acc, scratch, C, residual, M, N,
                                         m0, n0, wrow, wcol, lane, scale,
                                         bias);
}

// Small-K variant (K = 64: the DNN ops' 1x1 convolutions out of 64-channel
// activations). A single K-step leaves the main kernel with no prefetch
// overlap — every A tile is a cold HBM read the MFMAs must wait on
// (measured 69 TF/s at M=50176,N=256,K=64). Here a workgroup walks
// MULTIPLE M-tiles of one N-strip, double-buffering the A staging across
// tiles: tile t+1's global_load_lds runs while tile t's MFMAs and stores
// execute. B ([BN][64], 16 KB) is staged once and reused for the whole
// walk. K=64 keeps the 128-byte LDS row layout the lane-contiguous
// LDS-DMA scatter requires.
template <int BM, int BN, int WM, int WN, bool RELU, bool RESIDUAL>
__global__ void __launch_bounds__(256, 2)
    gemm_bf16_smallk_kernel(const bf16* __restrict__ A,
                            const bf16* __restrict__ B, bf16* __restrict__ C,
                            int M, int N, const float* __restrict__ scale,
                            const float* __restrict__ bias,
                            const bf16* __restrict__ residual,
                            int tiles_per_wg) {
  constexpr int K = 64;
  constexpr int FM = BM / WM / 16;
  constexpr int FN = BN / WN / 16;
  __shared__ bf16 lds_b[BN * K];
  __shared__ bf16 lds_a[2 * BM * K];
  __shared__ float lds_ep[4 * 16 * (BN / WN + 4)];

  int tid = threadIdx.x;
  int lane = tid & 63;
  int wave = tid >> 6;
  int wrow = wave / WN;
  int wcol = wave % WN;

  int ntiles_n = (N + BN - 1) / BN;
  int ntiles_m = (M + BM - 1) / BM;
  int strip = blockIdx.x / ntiles_n;
  int n0 = (blockIdx.x % ntiles_n) * BN;
  int mt0 = strip * tiles_per_wg;
  int mt_end = min(mt0 + tiles_per_wg, ntiles_m);
  if (mt0 >= ntiles_m) return;

  const int lrow = lane >> 3;
  const int lk = (lane & 7) * 8;
  constexpr int B_INSTRS = BN / 8 / 4;
  constexpr int A_INSTRS = BM / 8 / 4;

#pragma unroll
  for (int i = 0; i < B_INSTRS; ++i) {
    int row = (wave * B_INSTRS + i) * 8 + lrow;
    const bf16* src = B + (size_t)(n0 + row) * K + lk;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t*)src,
        (__attribute__((address_space(3)))
             uint32_t*)(lds_b + (size_t)(wave * B_INSTRS + i) * 8 * K),
        16, 0, 0);
  }

  auto stage_a = [&](int buf, int mt) {
    int m0 = mt * BM;
    bf16* dst = lds_a + buf * BM * K;
#pragma unroll
    for (int i = 0; i < A_INSTRS; ++i) {
      int row = (wave * A_INSTRS + i) * 8 + lrow;
      int grow = m0 + row;
      if (grow >= M) grow = M - 1;
      const bf16* src = A + (size_t)grow * K + lk;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)src,
          (__attribute__((address_space(3)))
               uint32_t*)(dst + (size_t)(wave * A_INSTRS + i) * 8 * K),
          16, 0, 0);
    }
  };

  stage_a(0, mt0);
  for (int mt = mt0; mt < mt_end; ++mt) {
    int buf = (mt - mt0) & 1;
    __builtin_amdgcn_s_waitcnt(0);
    __syncthreads();
    if (mt + 1 < mt_end) stage_a(buf ^ 1, mt + 1);

    f32x4 acc[FM][FN];
#pragma unroll
    for (int i = 0; i < FM; ++i)
#pragma unroll
      for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

    const bf16* la = lds_a + buf * BM * K;
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      int kbase = kk * 32 + (lane >> 4) * 8;
      bf16x8 afrag[FM], bfrag[FN];
#pragma unroll
      for (int i = 0; i < FM; ++i) {
        int row = wrow * (BM / WM) + i * 16 + (lane & 15);
        afrag[i] =
            *reinterpret_cast<const bf16x8*>(la + (size_t)row * K + kbase);
      }
#pragma unroll
      for (int j = 0; j < FN; ++j) {
        int col = wcol * (BN / WN) + j * 16 + (lane & 15);
        bfrag[j] = *reinterpret_cast<const bf16x8*>(lds_b +
                                                    (size_t)col * K + kbase);
      }
#pragma unroll
      for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    }

    // Coalesced store through dedicated per-wave scratch (lds_a/lds_b stay
    // live — the next tile's A prefetch is in flight right now, and the
    // wave-local lgkmcnt waits inside don't drain it).
    epilogue_store<FM, FN, RELU, RESIDUAL>(
        acc, lds_ep + (size_t)wave * 16 * (BN / WN + 4), C, residual, M, N,
        mt * BM, n0, wrow, wcol, lane, scale, bias);
    // No trailing barrier (see main kernel note): next iteration's leading
    // waitcnt+barrier is the only ordering needed, and keeping the A
    // prefetch un-drained across the epilogue is the whole point.
  }
}

// ---- split-K path ----
// For launch-bound shapes (few output tiles, deep K — e.g. ResNet stage-4
// conv2: M-tiles x N-tiles = 52 workgroups on a 256-CU chip, measured
// ~59 TF/s), S workgroup groups each compute a K-slice partial into f32
// scratch and a reduce kernel applies the epilogue. An OPT-IN fused
// variant (SCANNER_SPLITK_FUSED=1) instead has the LAST workgroup to
// finish a tile (tile-completion counter) reduce the partials in-kernel —
// G16-safe since no workgroup ever WAITS on another (each increments its
// tile's counter exactly once and exits; only the one observing
// count == splits-1 does the extra work) — but it measured 2.2x SLOWER on
// the flagship (see the kFused comment in try_splitk), so the separate
// reduce launch stays the default.
struct SplitkEp {
  unsigned int* counters = nullptr;  // null => unfused (write partials only)
  int splits = 0;
  const float* scale = nullptr;
  const float* bias = nullptr;
  const bf16* residual = nullptr;
  bf16* C = nullptr;
  int relu = 0;
};

template <int BM, int BN, int WM, int WN, bool IMPLICIT, bool ATOMIC>
__global__ void __launch_bounds__(256, 2)
    gemm_bf16_splitk_kernel(const bf16* __restrict__ A,
                            const bf16* __restrict__ B, int M, int N, int K,
                            int ksteps_per_split,
                            float* __restrict__ partials, ConvDesc d,
                            const bf16* __restrict__ zero, SplitkEp ep) {
  constexpr int BK = 64;
  constexpr int FM = BM / WM / 16;
  constexpr int FN = BN / WN / 16;
  __shared__ bf16 lds[2 * (BM + BN) * BK];

  int tid = threadIdx.x;
  int lane = tid & 63;
  int wave = tid >> 6;
  int wrow = wave / WN;
  int wcol = wave % WN;

  int ntiles_n = (N + BN - 1) / BN;
  int ntiles_m = (M + BM - 1) / BM;
  int tiles = ntiles_m * ntiles_n;
  int split = blockIdx.x / tiles;
  int tile = blockIdx.x % tiles;
  int m0 = (tile / ntiles_n) * BM;
  int n0 = (tile % ntiles_n) * BN;
  int ks_begin = split * ksteps_per_split;
  int ks_end = min(ks_begin + ksteps_per_split, K / BK);
  if (ks_begin >= ks_end) return;

  f32x4 acc[FM][FN];
#pragma unroll
  for (int i = 0; i < FM; ++i)
#pragma unroll
    for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  constexpr int A_INSTRS = BM / 8 / 4;
  constexpr int B_INSTRS = BN / 8 / 4;
  auto stage = [&](int buf, int k0) {
    const int lrow = lane >> 3;
    const int lk = (lane & 7) * 8;
    bf16* lds_a = lds + buf * (BM + BN) * BK;
    bf16* lds_b = lds_a + BM * BK;
#pragma unroll
    for (int i = 0; i < A_INSTRS; ++i) {
      int row = (wave * A_INSTRS + i) * 8 + lrow;
      int grow = m0 + row;
      if (grow >= M) grow = M - 1;
      const bf16* src;
      if constexpr (IMPLICIT) {
        int k = k0 + lk;
        int cell = k / d.c;
        int cj = k - cell * d.c;
        if (cell >= d.r * d.s) {
          src = zero;
        } else {
          int dr = cell / d.s, ds = cell - dr * d.s;
          int q = grow % d.ow;
          int t = grow / d.ow;
          int p = t % d.oh;
          int nn = t / d.oh;
          int hh = p * d.stride - d.pad + dr;
          int ww = q * d.stride - d.pad + ds;
          src = (hh >= 0 && hh < d.h && ww >= 0 && ww < d.w)
                    ? A + ((((i64)nn * d.h + hh) * d.w + ww) * d.c + cj)
                    : zero;
        }
      } else {
        src = A + (size_t)grow * K + k0 + lk;
      }
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)src,
          (__attribute__((address_space(3)))
               uint32_t*)(lds_a + (size_t)(wave * A_INSTRS + i) * 8 * BK),
          16, 0, 0);
    }
#pragma unroll
    for (int i = 0; i < B_INSTRS; ++i) {
      int row = (wave * B_INSTRS + i) * 8 + lrow;
      const bf16* src = B + (size_t)(n0 + row) * K + k0 + lk;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)src,
          (__attribute__((address_space(3)))
               uint32_t*)(lds_b + (size_t)(wave * B_INSTRS + i) * 8 * BK),
          16, 0, 0);
    }
  };

  stage(0, ks_begin * BK);
  for (int ks = ks_begin; ks < ks_end; ++ks) {
    int buf = (ks - ks_begin) & 1;
    __builtin_amdgcn_s_waitcnt(0);
    __syncthreads();
    if (ks + 1 < ks_end) stage(buf ^ 1, (ks + 1) * BK);

    const bf16* lds_a = lds + buf * (BM + BN) * BK;
    const bf16* lds_b = lds_a + BM * BK;
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      int kbase = kk * 32 + (lane >> 4) * 8;
      bf16x8 afrag[FM], bfrag[FN];
#pragma unroll
      for (int i = 0; i < FM; ++i) {
        int row = wrow * (BM / WM) + i * 16 + (lane & 15);
        afrag[i] = *reinterpret_cast<const bf16x8*>(
            lds_a + (size_t)row * BK + kbase);
      }
#pragma unroll
      for (int j = 0; j < FN; ++j) {
        int col = wcol * (BN / WN) + j * 16 + (lane & 15);
        bfrag[j] = *reinterpret_cast<const bf16x8*>(
            lds_b + (size_t)col * BK + kbase);
      }
#pragma unroll
      for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  if constexpr (ATOMIC) {
    // Atomic split-K: all splits add into ONE pre-zeroed f32 [M][N]
    // buffer (f32 atomic adds are commutative — no inter-WG ordering,
    // G16-safe); a finalize kernel applies the epilogue. Vs the
    // partials+reduce pair this removes the S x M x N reduce read
    // (measured ~12% of all flagship GPU time).
#pragma unroll
    for (int i = 0; i < FM; ++i) {
#pragma unroll
      for (int j = 0; j < FN; ++j) {
        int col = n0 + wcol * (BN / WN) + j * 16 + (lane & 15);
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int row = m0 + wrow * (BM / WM) + i * 16 + (lane >> 4) * 4 + r;
          if (row < M)
            atomicAdd(&partials[(size_t)row * N + col], acc[i][j][r]);
        }
      }
    }
  } else {
    // f32 partials: [split][M][N]; lanes 0-15 write 64-byte runs
    float* out = partials + (size_t)split * M * N;
#pragma unroll
    for (int i = 0; i < FM; ++i) {
#pragma unroll
      for (int j = 0; j < FN; ++j) {
        int col = n0 + wcol * (BN / WN) + j * 16 + (lane & 15);
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int row = m0 + wrow * (BM / WM) + i * 16 + (lane >> 4) * 4 + r;
          if (row < M) out[(size_t)row * N + col] = acc[i][j][r];
        }
      }
    }
    if (ep.counters) {
      // Fused reduce: release this split's partial (per-thread fence,
      // then a block barrier so tid 0's atomic is ordered after EVERY
      // thread's stores), count the tile's completions, and if this
      // workgroup is the last one, sum the partials in fixed split order
      // (bitwise identical to splitk_reduce_kernel) + epilogue + bf16
      // store. The counter is restored to 0 for the next launch that
      // reuses this scratch (launches on the stream are ordered, and
      // hipGraph replays see the same all-zero state).
      __threadfence();
      __syncthreads();
      __shared__ unsigned int s_old;
      if (tid == 0) s_old = atomicAdd(ep.counters + tile, 1u);
      __syncthreads();
      if (s_old != (unsigned int)ep.splits - 1) return;
      __threadfence();  // acquire: other splits' partials now visible
      int rows = min(BM, M - m0);
      constexpr int Q = BN / 4;  // float4 quads per tile row
      for (int idx = tid; idx < rows * Q; idx += 256) {
        int r = idx / Q, q = idx - r * Q;
        size_t base = (size_t)(m0 + r) * N + n0 + q * 4;
        float4 av = make_float4(0.f, 0.f, 0.f, 0.f);
        for (int sp = 0; sp < ep.splits; ++sp) {
          float4 p = *reinterpret_cast<const float4*>(
              partials + (size_t)sp * M * N + base);
          av.x += p.x;
          av.y += p.y;
          av.z += p.z;
          av.w += p.w;
        }
        float v4[4] = {av.x, av.y, av.z, av.w};
        bf16 o4[4];
#pragma unroll
        for (int k = 0; k < 4; ++k) {
          float v = v4[k];
          int col = n0 + q * 4 + k;
          if (ep.scale) v *= ep.scale[col];
          if (ep.bias) v += ep.bias[col];
          if (ep.residual) v += bf16_to_f32(ep.residual[base + k]);
          if (ep.relu) v = v > 0.f ? v : 0.f;
          o4[k] = f32_to_bf16(v);
        }
        *reinterpret_cast<uint64_t*>(ep.C + base) =
            *reinterpret_cast<const uint64_t*>(o4);
      }
      if (tid == 0) ep.counters[tile] = 0;
    }
  }
}

template <bool RELU, bool RESIDUAL>
__global__ void __launch_bounds__(256)
    splitk_finalize_kernel(const float* __restrict__ acc, i64 mn, int N,
                           const float* __restrict__ scale,
                           const float* __restrict__ bias,
                           const bf16* __restrict__ residual,
                           bf16* __restrict__ C) {
  i64 quads = mn / 4;
  i64 gs = (i64)gridDim.x * blockDim.x;
  for (i64 q = (i64)blockIdx.x * blockDim.x + threadIdx.x; q < quads;
       q += gs) {
    i64 i = q * 4;
    float4 a4 = reinterpret_cast<const float4*>(acc)[q];
    float v4[4] = {a4.x, a4.y, a4.z, a4.w};
    bf16 o4[4];
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float v = v4[k];
      int col = (int)((i + k) % N);
      if (scale) v *= scale[col];
      if (bias) v += bias[col];
      if constexpr (RESIDUAL) v += bf16_to_f32(residual[i + k]);
      if constexpr (RELU) v = v > 0.f ? v : 0.f;
      o4[k] = f32_to_bf16(v);
    }
    *reinterpret_cast<uint64_t*>(C + i) =
        *reinterpret_cast<const uint64_t*>(o4);
  }
}

template <bool RELU, bool RESIDUAL>
__global__ void __launch_bounds__(256)
    splitk_reduce_kernel(const float* __restrict__ partials, int splits,
                         i64 mn, int N, const float* __restrict__ scale,
                         const float* __restrict__ bias,
                         const bf16* __restrict__ residual,
                         bf16* __restrict__ C) {
  // 4 elements per thread via float4 (N is a multiple of 64, so mn % 4 ==
  // 0 and every row stays 16-byte aligned): the scalar version measured
  // only ~2.2 TB/s on S x M x N partial streams.
  i64 quads = mn / 4;
  i64 gs = (i64)gridDim.x * blockDim.x;
  for (i64 q = (i64)blockIdx.x * blockDim.x + threadIdx.x; q < quads;
       q += gs) {
    i64 i = q * 4;
    float4 acc = make_float4(0.f, 0.f, 0.f, 0.f);
    for (int sp = 0; sp < splits; ++sp) {
      float4 p = reinterpret_cast<const float4*>(
          partials + (size_t)sp * mn)[q];
      acc.x += p.x;
      acc.y += p.y;
      acc.z += p.z;
      acc.w += p.w;
    }
    float v4[4] = {acc.x, acc.y, acc.z, acc.w};
    bf16 o4[4];
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float v = v4[k];
      int col = (int)((i + k) % N);
      if (scale) v *= scale[col];
      if (bias) v += bias[col];
      if constexpr (RESIDUAL) v += bf16_to_f32(residual[i + k]);
      if constexpr (RELU) v = v > 0.f ? v : 0.f;
      o4[k] = f32_to_bf16(v);
    }
    *reinterpret_cast<uint64_t*>(C + i) =
        *reinterpret_cast<const uint64_t*>(o4);
  }
}

const bf16* device_zero_chunk();

bool try_splitk(const GemmArgs& g, hipStream_t s,
                const ConvDesc* dd = nullptr) {
  constexpr int BM = 128, BN = 128;
  // SCANNER_SPLITK_OFF=1: skip split-K entirely (A/B: with multiple
  // pipeline instances co-running, their kernels may fill the chip
  // without paying the partial+reduce HBM traffic).
  static const bool kOff = []() {
    const char* e = std::getenv("SCANNER_SPLITK_OFF");
    return e && e[0] == '1';
  }();
  if (kOff) return false;
  if (!g.splitk_scratch || g.N % BN != 0 || g.K < 1024) return false;
  int ntiles_m = (g.M + BM - 1) / BM;
  int tiles = ntiles_m * (g.N / BN);
  if (tiles >= 256) return false;  // already fills the chip
  static const int kMaxSplits = []() {
    const char* e = std::getenv("SCANNER_SPLITK_MAX");
    // A/B on the flagship: 6 > 8 > 16 (18.0k / 17.95k / 15.4k f/s) —
    // fewer splits cut the partial+reduce HBM traffic and the chip stays
    // fed by the co-running pipeline instances.
    return e ? std::max(2, atoi(e)) : 6;
  }();
  int ksteps = g.K / 64;
  int want = std::min({kMaxSplits, ksteps / 4, (2048 + tiles - 1) / tiles});
  size_t per_split = (size_t)g.M * g.N * 4;
  // The scratch tail holds the fused-reduce tile counters (tiles < 256 so
  // 4 KB is ample); the owner zeroes it once at allocation
  // (splitk_scratch_init) and the kernel self-restores it after each use.
  constexpr size_t kCtrBytes = 4096;
  size_t avail = g.splitk_scratch_bytes > kCtrBytes
                     ? g.splitk_scratch_bytes - kCtrBytes
                     : 0;
  int fit = (int)(avail / per_split);
  int splits = std::min(want, fit);
  if (splits < 2) return false;
  int ksteps_per_split = (ksteps + splits - 1) / splits;
  splits = (ksteps + ksteps_per_split - 1) / ksteps_per_split;
  i64 mn = (i64)g.M * g.N;
  // Atomic accumulation (opt-in, SCANNER_SPLITK_ATOMIC=1): one pre-zeroed
  // f32 [M][N] buffer, no S x M x N reduce read — but f32 atomic-add
  // ORDER is nondeterministic, so results vary at the last bf16 ulp
  // between runs. The default stays the deterministic partials+reduce
  // pair (run-to-run reproducibility is a framework property our own
  // regression tests rely on).
  static const bool kAtomic = []() {
    const char* e = std::getenv("SCANNER_SPLITK_ATOMIC");
    return e && e[0] == '1';
  }();
  // Fused reduce (opt-in, SCANNER_SPLITK_FUSED=1; default OFF). Measured
  // NEGATIVE on MI355X: flagship 7.6k vs 16.6k f/s, resnet 7.9k vs 18.0k
  // (same box, 3-step A/B, profiles/r02_results.md update 12). Numerics
  // are correct (27 GPU tests green with it on), but the device-scope
  // release fence every workgroup needs before bumping its tile counter
  // forces an L2 writeback per WG — on 8 per-XCD L2s that evicts the
  // B-operand and co-running instances' working sets — and the tail
  // reduce runs with one 256-thread WG per tile vs the standalone
  // kernel's chip-wide parallelism. Kept as a documented experiment; the
  // separate splitk_reduce_kernel launch is the default.
  static const bool kFused = []() {
    const char* e = std::getenv("SCANNER_SPLITK_FUSED");
    return e && e[0] == '1';
  }();
  bool fused = kFused && !kAtomic;
  SplitkEp ep{};
  if (fused) {
    ep.counters = reinterpret_cast<unsigned int*>(
        (u8*)g.splitk_scratch + g.splitk_scratch_bytes - kCtrBytes);
    ep.splits = splits;
    ep.scale = g.scale;
    ep.bias = g.bias;
    ep.residual = (const bf16*)g.residual;
    ep.C = (bf16*)g.C;
    ep.relu = g.relu ? 1 : 0;
  }
  hipError_t e;
  if (kAtomic) {
    e = hipMemsetAsync(g.splitk_scratch, 0, (size_t)mn * 4, s);
    if (e != hipSuccess) {
      throw ScannerError(std::string("splitk memset failed: ") +
                         hipGetErrorString(e));
    }
    if (dd) {
      gemm_bf16_splitk_kernel<BM, BN, 2, 2, true, true>
          <<<tiles * splits, 256, 0, s>>>(
              (const bf16*)g.A, (const bf16*)g.B, g.M, g.N, g.K,
              ksteps_per_split, (float*)g.splitk_scratch, *dd,
              device_zero_chunk(), SplitkEp{});
    } else {
      gemm_bf16_splitk_kernel<BM, BN, 2, 2, false, true>
          <<<tiles * splits, 256, 0, s>>>(
              (const bf16*)g.A, (const bf16*)g.B, g.M, g.N, g.K,
              ksteps_per_split, (float*)g.splitk_scratch, ConvDesc{},
              nullptr, SplitkEp{});
    }
  } else if (dd) {
    gemm_bf16_splitk_kernel<BM, BN, 2, 2, true, false>
        <<<tiles * splits, 256, 0, s>>>(
            (const bf16*)g.A, (const bf16*)g.B, g.M, g.N, g.K,
            ksteps_per_split, (float*)g.splitk_scratch, *dd,
            device_zero_chunk(), ep);
  } else {
    gemm_bf16_splitk_kernel<BM, BN, 2, 2, false, false>
        <<<tiles * splits, 256, 0, s>>>(
            (const bf16*)g.A, (const bf16*)g.B, g.M, g.N, g.K,
            ksteps_per_split, (float*)g.splitk_scratch, ConvDesc{}, nullptr,
            ep);
  }
  e = hipGetLastError();
  if (e != hipSuccess) {
    throw ScannerError(std::string("splitk launch failed: ") +
                       hipGetErrorString(e));
  }
  if (fused) return true;  // epilogue ran in-kernel; no reduce launch
  // 4x more threads than quads: the extra waves are pure memory-level
  // parallelism for this latency-bound pass (measured 18.2 us avg vs a
  // ~1.3 us bandwidth bound)
  int grid = (int)std::min<i64>(4096, (mn + 255) / 256);
  auto disp = [&](auto relu, auto res) {
    if (kAtomic) {
      splitk_finalize_kernel<decltype(relu)::value, decltype(res)::value>
          <<<grid, 256, 0, s>>>((const float*)g.splitk_scratch, mn, g.N,
                                g.scale, g.bias, (const bf16*)g.residual,
                                (bf16*)g.C);
    } else {
      splitk_reduce_kernel<decltype(relu)::value, decltype(res)::value>
          <<<grid, 256, 0, s>>>((const float*)g.splitk_scratch, splits, mn,
                                g.N, g.scale, g.bias,
                                (const bf16*)g.residual, (bf16*)g.C);
    }
  };
  if (g.relu && g.residual)
    disp(std::true_type{}, std::true_type{});
  else if (g.relu)
    disp(std::true_type{}, std::false_type{});
  else if (g.residual)
    disp(std::false_type{}, std::true_type{});
  else
    disp(std::false_type{}, std::false_type{});
  e = hipGetLastError();
  if (e != hipSuccess) {
    throw ScannerError(std::string("splitk reduce launch failed: ") +
                       hipGetErrorString(e));
  }
  return true;
}

void launch_smallk(const GemmArgs& g, hipStream_t s) {
  constexpr int BM = 128, BN = 128;
  int ntiles_m = (g.M + BM - 1) / BM;
  int ntiles_n = (g.N + BN - 1) / BN;
  // Enough strips to fill the chip (>=2048 workgroups when the shape
  // allows), each walking a run of consecutive M-tiles.
  static const int target_wgs = []() {
    const char* e = std::getenv("SCANNER_SMALLK_WGS");
    return e ? std::max(256, atoi(e)) : 2048;
  }();
  int strips = std::max(1, std::min(ntiles_m, target_wgs / ntiles_n));
  int tiles_per_wg = (ntiles_m + strips - 1) / strips;
  strips = (ntiles_m + tiles_per_wg - 1) / tiles_per_wg;
  int grid = strips * ntiles_n;
  auto disp = [&](auto relu, auto res) {
    gemm_bf16_smallk_kernel<BM, BN, 2, 2, decltype(relu)::value,
                            decltype(res)::value><<<grid, 256, 0, s>>>(
        (const bf16*)g.A, (const bf16*)g.B, (bf16*)g.C, g.M, g.N, g.scale,
        g.bias, (const bf16*)g.residual, tiles_per_wg);
  };
  if (g.relu && g.residual)
    disp(std::true_type{}, std::true_type{});
  else if (g.relu)
    disp(std::true_type{}, std::false_type{});
  else if (g.residual)
    disp(std::false_type{}, std::true_type{});
  else
    disp(std::false_type{}, std::false_type{});
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) {
    throw ScannerError(std::string("gemm smallk launch failed: ") +
                       hipGetErrorString(e));
  }
}

template <int BM, int BN, int WM, int WN>
void launch_variant(const GemmArgs& g, hipStream_t s) {
  int grid = ((g.M + BM - 1) / BM) * (g.N / BN);
  auto disp = [&](auto relu, auto res) {
    gemm_bf16_kernel<BM, BN, WM, WN, decltype(relu)::value,
                     decltype(res)::value, false><<<grid, 256, 0, s>>>(
        (const bf16*)g.A, (const bf16*)g.B, (bf16*)g.C, g.M, g.N, g.K,
        g.scale, g.bias, (const bf16*)g.residual, ConvDesc{}, nullptr);
  };
  if (g.relu && g.residual)
    disp(std::true_type{}, std::true_type{});
  else if (g.relu)
    disp(std::true_type{}, std::false_type{});
  else if (g.residual)
    disp(std::false_type{}, std::true_type{});
  else
    disp(std::false_type{}, std::false_type{});
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) {
    throw ScannerError(std::string("gemm launch failed: ") +
                       hipGetErrorString(e));
  }
}

// 16+ zero bytes in device memory for implicit-conv padding lanes; one
// per device, allocated outside the pool so memory teardown cycles never
// invalidate it.
const bf16* device_zero_chunk() {
  static std::mutex mu;
  static std::map<int, bf16*> per_dev;
  int dev = 0;
  (void)hipGetDevice(&dev);
  std::lock_guard<std::mutex> l(mu);
  auto it = per_dev.find(dev);
  if (it != per_dev.end()) return it->second;
  void* p = nullptr;
  hipError_t e = hipMalloc(&p, 256);
  if (e != hipSuccess || hipMemset(p, 0, 256) != hipSuccess) {
    throw ScannerError("failed to allocate implicit-conv zero chunk");
  }
  per_dev[dev] = (bf16*)p;
  return (bf16*)p;
}

template <int BM, int BN, int WM, int WN>
void launch_conv_variant(const GemmArgs& g, const ConvDesc& d,
                         hipStream_t s) {
  int grid = ((g.M + BM - 1) / BM) * (g.N / BN);
  const bf16* zero = device_zero_chunk();
  auto disp = [&](auto relu, auto res) {
    gemm_bf16_kernel<BM, BN, WM, WN, decltype(relu)::value,
                     decltype(res)::value, true><<<grid, 256, 0, s>>>(
        (const bf16*)g.A, (const bf16*)g.B, (bf16*)g.C, g.M, g.N, g.K,
        g.scale, g.bias, (const bf16*)g.residual, d, zero);
  };
  if (g.relu && g.residual)
    disp(std::true_type{}, std::true_type{});
  else if (g.relu)
    disp(std::true_type{}, std::false_type{});
  else if (g.residual)
    disp(std::false_type{}, std::true_type{});
  else
    disp(std::false_type{}, std::false_type{});
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) {
    throw ScannerError(std::string("conv gemm launch failed: ") +
                       hipGetErrorString(e));
  }
}

}  // namespace

void conv_gemm_bf16(const GemmArgs& g, const ConvDesc& d, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  SCA_CHECK(g.K % 64 == 0 && g.N % 64 == 0, "conv gemm K/N must be x64");
  SCA_CHECK(d.c % 8 == 0, "implicit conv needs c % 8 == 0");
  SCA_CHECK(g.M == d.n * d.oh * d.ow, "conv gemm M mismatch");
  if (try_splitk(g, s, &d)) {
    return;
  }
  if (g.N % 128 == 0) {
    launch_conv_variant<128, 128, 2, 2>(g, d, s);
  } else {
    launch_conv_variant<64, 64, 2, 2>(g, d, s);
  }
}

void splitk_scratch_init(void* scratch, size_t bytes, void* stream) {
  if (!scratch || bytes < 4096) return;
  u8* tail = (u8*)scratch + bytes - 4096;
  hipError_t e = stream
                     ? hipMemsetAsync(tail, 0, 4096, (hipStream_t)stream)
                     : hipMemset(tail, 0, 4096);
  SCA_CHECK(e == hipSuccess, "splitk_scratch_init memset failed");
}

void gemm_bf16(const GemmArgs& g, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  SCA_CHECK(g.K % 64 == 0, "gemm K must be a multiple of 64");
  SCA_CHECK(g.N % 64 == 0, "gemm N must be a multiple of 64");
  if (g.K == 64 && g.N % 128 == 0 && g.M >= 1024) {
    launch_smallk(g, s);
  } else if (try_splitk(g, s)) {
    // handled
  } else if (g.N % 128 == 0 && g.M > 64) {
    launch_variant<128, 128, 2, 2>(g, s);
  } else {
    launch_variant<64, 64, 2, 2>(g, s);
  }
}

}  // namespace sca
