// GPU decoder for the SVC codec (format: csrc/video/svc.h) — the MI355X
// analogue of the reference's NVDEC path (nvidia_video_decoder.cpp), which
// decoded into GPU surfaces; here decoded frames land directly in HBM as
// engine elements.
//
// Kernel shape: one workgroup per supergroup (128 groups x 32 bytes = 4 KiB
// of payload). The block loads the 128 group bit-widths, prefix-sums them in
// LDS for packed offsets, then each lane unpacks its group's 32 residuals
// and applies the predictor (delta: previous frame byte — fully parallel;
// key: serial 32-byte chain within the lane).
#include <hip/hip_runtime.h>

#include "../csrc/memory.h"
#include "../csrc/video/svc.h"

namespace sca {

namespace {

#define SVC_CHECK(expr)                                                  \
  do {                                                                   \
    hipError_t _e = (expr);                                              \
    if (_e != hipSuccess) {                                              \
      throw ScannerError(std::string("HIP error in svc decode: ") +      \
                         hipGetErrorString(_e));                         \
    }                                                                    \
  } while (0)

__device__ inline u8 dev_unzigzag(u8 z) {
  i8 v = (i8)((z >> 1) ^ (-(i32)(z & 1)));
  return (u8)v;
}

// One block per supergroup, 128 threads (one lane per group). Group
// payloads are 4*width bytes and 4-byte aligned, so the unpack runs on u32
// loads; the delta path reads prev and writes cur as u32s (the frame base
// of a group is 32-byte aligned).
__global__ void __launch_bounds__(128)
    svc_decode_frame_kernel(const u8* __restrict__ pkt_widths,
                            const u32* __restrict__ pkt_super_off,
                            const u8* __restrict__ pkt_packed,
                            const u8* __restrict__ prev, bool is_key,
                            u32 nbytes, u32 ngroups,
                            u8* __restrict__ cur) {
  u32 s = blockIdx.x;
  u32 g0 = s * 128;
  u32 local_n = min(128u, ngroups - g0);
  u32 tid = threadIdx.x;
  u32 lane = tid & 63;
  u32 wave = tid >> 6;

  u32 w = tid < local_n ? (u32)pkt_widths[g0 + tid] : 0;
  // Packed offsets by prefix sum of 4*w: wave64 shuffle scan (no LDS
  // traffic), one barrier to carry wave 0's total into wave 1 — vs the 14
  // barriers of a 128-wide Hillis-Steele LDS scan (measured 35 us/frame
  // before, decode-bound histogram pipeline).
  u32 val = 4u * w;
  u32 x = val;
#pragma unroll
  for (u32 d = 1; d < 64; d <<= 1) {
    u32 y = __shfl_up(x, d, 64);
    if (lane >= d) x += y;
  }
  __shared__ u32 wave_total;
  if (wave == 0 && lane == 63) wave_total = x;
  __syncthreads();
  u32 my_off = x - val + (wave ? wave_total : 0);
  if (tid >= local_n) return;

  u32 g = g0 + tid;
  const u32* q =
      reinterpret_cast<const u32*>(pkt_packed + pkt_super_off[s] + my_off);
  // Fully unrolled unpack straight into packed u32 output words: res bytes
  // never touch memory (a partially-unrolled byte array spills to
  // scratch).
  u32 res32[8];
  if (w == 0) {
#pragma unroll
    for (int k = 0; k < 8; ++k) res32[k] = 0;
  } else {
    u64 acc = 0;
    u32 nacc = 0;
    u32 qi = 0;
    u32 mask = (1u << w) - 1;
#pragma unroll
    for (int k = 0; k < 32; ++k) {
      if (nacc < w) {
        acc |= ((u64)q[qi++]) << nacc;
        nacc += 32;
      }
      u32 r = (u32)(acc & mask);
      if ((k & 3) == 0)
        res32[k / 4] = r;
      else
        res32[k / 4] |= r << (8 * (k & 3));
      acc >>= w;
      nacc -= w;
    }
  }
  auto res_at = [&](int k) -> u8 {
    return (u8)((res32[k / 4] >> (8 * (k & 3))) & 0xff);
  };
  u32 base = g * 32;
  if (base + 32 > nbytes) {  // scalar tail group
    u32 n = nbytes - base;
    if (is_key) {
      u8 p = 128;
      for (u32 k = 0; k < n; ++k) {
        p = (u8)(p + dev_unzigzag(res_at(k)));
        cur[base + k] = p;
      }
    } else {
      for (u32 k = 0; k < n; ++k)
        cur[base + k] = (u8)(prev[base + k] + dev_unzigzag(res_at(k)));
    }
    return;
  }
  u32 out[8];
  if (is_key) {
    u8 p = 128;
#pragma unroll
    for (int k = 0; k < 32; ++k) {
      p = (u8)(p + dev_unzigzag(res_at(k)));
      out[k / 4] = (k % 4 == 0) ? p : (out[k / 4] | ((u32)p << (8 * (k % 4))));
    }
  } else {
    const u32* prev32 = reinterpret_cast<const u32*>(prev + base);
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      u32 pv = prev32[k];
      u32 rz = res32[k];
      u32 o = 0;
#pragma unroll
      for (int b = 0; b < 4; ++b) {
        u8 byte = (u8)((pv >> (8 * b)) & 0xff);
        byte = (u8)(byte + dev_unzigzag((u8)((rz >> (8 * b)) & 0xff)));
        o |= (u32)byte << (8 * b);
      }
      out[k] = o;
    }
  }
  uint4* dst = reinterpret_cast<uint4*>(cur + base);
  dst[0] = make_uint4(out[0], out[1], out[2], out[3]);
  dst[1] = make_uint4(out[4], out[5], out[6], out[7]);
}

}  // namespace

// Shared decode core: `d_stream` holds device-resident encoded bytes
// covering [lo, hi) of the item's stream. Packet geometry (nbytes, ngroups,
// nsuper, keyframe-ness) is fully determined by the VideoMetadata, so no
// host-side packet parsing is needed — this is what lets the HBM span cache
// (video/span_cache.h) feed decode without the bytes ever revisiting the
// host. `up_ev`, when non-null, is an event the upload was recorded on;
// decode chains wait on it.
static std::vector<Element> svc_decode_gpu_impl(
    const u8* d_stream, u64 lo, const VideoMetadata& vm,
    const std::vector<i64>& span, const std::vector<i64>& want,
    DeviceHandle dev, hipEvent_t up_ev) {
  hipStream_t s = (hipStream_t)per_thread_hip_stream();
  u32 nbytes = (u32)((i64)vm.height * vm.width * vm.channels);
  u32 ngroups = (nbytes + 31) / 32;
  u32 nsuper = (ngroups + 127) / 128;
  std::vector<Element> out;

  // Frames inside a GOP are a serial prediction chain, but GOPs are
  // independent: each keyframe starts a new chain on one of 4 auxiliary
  // streams (fork/join with events), so a work packet's GOPs decode
  // concurrently. (The reference's decoder automaton is serial per item —
  // decoder_automata.cpp; GOP concurrency is the MI355X-native upgrade.)
  constexpr int kChains = 4;
  hipStream_t chain_stream[kChains];
  bool chain_used[kChains] = {false, false, false, false};
  std::vector<u8*> scratches;
  size_t wi = 0;
  int chain = -1;
  hipStream_t cs = s;
  u8* prev = nullptr;
  u8* chain_scratch[2] = {nullptr, nullptr};
  int flip = 0;
  size_t ki = 0;  // cursor into sorted keyframe_indices
  for (i64 f : span) {
    while (ki < vm.keyframe_indices.size() && vm.keyframe_indices[ki] < f)
      ++ki;
    bool is_key =
        ki < vm.keyframe_indices.size() && vm.keyframe_indices[ki] == f;
    SvcPacketView v;
    v.is_key = is_key;
    v.nbytes = nbytes;
    v.ngroups = ngroups;
    v.nsuper = nsuper;
    if (v.is_key) {
      // new chain
      chain = (chain + 1);
      int ci = chain % kChains;
      cs = (hipStream_t)per_thread_aux_stream(ci);
      if (!chain_used[ci]) {
        chain_used[ci] = true;
        chain_stream[ci] = cs;
        if (up_ev) SVC_CHECK(hipStreamWaitEvent(cs, up_ev, 0));
      }
      prev = nullptr;
      chain_scratch[0] = chain_scratch[1] = nullptr;
      flip = 0;
    }
    u64 pkt_off = vm.sample_offsets[f] - lo;
    const u8* pkt_d = d_stream + pkt_off;
    const u32* super_off_d = reinterpret_cast<const u32*>(pkt_d + 20);
    const u8* widths_d = pkt_d + 20 + v.nsuper * 4;
    const u8* packed_d = widths_d + (v.ngroups + 3) / 4 * 4;

    bool wanted = wi < want.size() && want[wi] == f;
    u8* cur;
    Element e;
    if (wanted) {
      e.is_frame = true;
      e.frame_info.shape[0] = vm.height;
      e.frame_info.shape[1] = vm.width;
      e.frame_info.shape[2] = vm.channels;
      e.frame_info.type = vm.frame_type;
      e.size = nbytes;
      e.buffer = new_buffer(dev, nbytes);
      e.device = dev;
      e.index = f;
      cur = e.buffer;
    } else {
      if (!chain_scratch[flip]) {
        chain_scratch[flip] = new_buffer(dev, nbytes);
        scratches.push_back(chain_scratch[flip]);
      }
      cur = chain_scratch[flip];
    }
    u32 blocks = v.nsuper;
    svc_decode_frame_kernel<<<blocks, 128, 0, cs>>>(
        widths_d, super_off_d, packed_d, prev, v.is_key, nbytes, v.ngroups,
        cur);
    SVC_CHECK(hipGetLastError());
    if (wanted) {
      out.push_back(e);
      ++wi;
    }
    prev = cur;
    flip ^= 1;
  }
  // Join: main stream waits every used chain, then sync before returning
  // scratch + stream buffers to the shared pool.
  for (int ci = 0; ci < kChains; ++ci) {
    if (!chain_used[ci]) continue;
    hipEvent_t ev;
    SVC_CHECK(hipEventCreateWithFlags(&ev, hipEventDisableTiming));
    SVC_CHECK(hipEventRecord(ev, chain_stream[ci]));
    SVC_CHECK(hipStreamWaitEvent(s, ev, 0));
    SVC_CHECK(hipEventDestroy(ev));
  }
  SVC_CHECK(hipStreamSynchronize(s));
  for (u8* sc : scratches) delete_buffer(dev, sc);
  SCA_CHECK(wi == want.size(), "svc gpu decode: not all frames produced");
  return out;
}

std::vector<Element> svc_decode_gpu(const u8* stream_host, size_t size,
                                    const VideoMetadata& vm,
                                    const std::vector<i64>& want,
                                    DeviceHandle dev, u64 stream_offset) {
  std::vector<i64> span = svc_decode_span(vm, want);
  if (span.empty()) return {};
  hipStream_t s = (hipStream_t)per_thread_hip_stream();

  // Upload the byte range covering the span asynchronously on the main
  // stream; decode chains fork off it via an event. stream_host is pinned
  // (CPU pool = hipHostMalloc) and outlives the end-of-function sync.
  u64 lo = vm.sample_offsets[span.front()];
  u64 hi = vm.sample_offsets[span.back()] + vm.sample_sizes[span.back()];
  SCA_CHECK(lo >= stream_offset && hi <= stream_offset + size,
            "svc stream range does not cover decode span");
  // Sanity-check the first packet against metadata while host bytes exist
  // (the device path below trusts VideoMetadata alone).
  SvcPacketView v0 = svc_parse_packet(stream_host + (lo - stream_offset),
                                      vm.sample_sizes[span.front()]);
  SCA_CHECK(v0.nbytes == (u32)((i64)vm.height * vm.width * vm.channels),
            "svc frame size mismatch");
  u8* d_stream = new_buffer(dev, hi - lo);
  SVC_CHECK(hipMemcpyAsync(d_stream, stream_host + (lo - stream_offset),
                           hi - lo, hipMemcpyHostToDevice, s));
  hipEvent_t up_ev;
  SVC_CHECK(hipEventCreateWithFlags(&up_ev, hipEventDisableTiming));
  SVC_CHECK(hipEventRecord(up_ev, s));
  std::vector<Element> out;
  try {
    out = svc_decode_gpu_impl(d_stream, lo, vm, span, want, dev, up_ev);
  } catch (...) {
    (void)hipEventDestroy(up_ev);
    delete_buffer(dev, d_stream);
    throw;
  }
  SVC_CHECK(hipEventDestroy(up_ev));
  delete_buffer(dev, d_stream);
  return out;
}

std::vector<Element> svc_decode_gpu_dev(const u8* stream_dev, u64 dev_lo,
                                        const VideoMetadata& vm,
                                        const std::vector<i64>& want,
                                        DeviceHandle dev) {
  std::vector<i64> span = svc_decode_span(vm, want);
  if (span.empty()) return {};
  u64 lo = vm.sample_offsets[span.front()];
  u64 hi = vm.sample_offsets[span.back()] + vm.sample_sizes[span.back()];
  SCA_CHECK(lo >= dev_lo, "svc device stream range does not cover span");
  (void)hi;
  return svc_decode_gpu_impl(stream_dev + (lo - dev_lo), lo, vm, span, want,
                             dev, nullptr);
}

}  // namespace sca
