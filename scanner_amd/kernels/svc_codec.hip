// GPU decoder for the SVC codec (format: csrc/video/svc.h) — the MI355X
// analogue of the reference's NVDEC path (nvidia_video_decoder.cpp), which
// decoded into GPU surfaces; here decoded frames land directly in HBM as
// engine elements.
//
// Kernel shape: one workgroup per supergroup (128 groups x 32 bytes = 4 KiB
// of payload) decoding a WHOLE GOP chain segment in one launch: the block
// loads each frame's 128 group bit-widths, prefix-sums them for packed
// offsets, unpacks its lane's 32 residuals and applies the predictor —
// with the inter-frame prediction chain carried in REGISTERS across the
// frame loop. Versus the earlier one-launch-per-frame design this removes
// 15/16 of the kernel launches and every HBM round trip of intermediate
// chain state (an unwanted frame between two wanted ones never touches
// memory at all).
#include <hip/hip_runtime.h>

#include <cstdlib>

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

constexpr int kGopBatch = 16;

struct SvcGopArgs {
  u64 pkt_off[kGopBatch];  // packet offsets within the device stream
  u8* out[kGopBatch];      // frame output; null = keep in registers only
  const u8* prev;          // chain state entering the launch (delta start)
  u32 key_mask;            // bit f = frame f is a keyframe
  i32 nframes;
  u32 nbytes, ngroups, nsuper;
};

// One block per supergroup, 128 threads (one lane per group). Group
// payloads are 4*width bytes and 4-byte aligned, so the unpack runs on u32
// loads. The frame loop keeps the 32 decoded bytes per lane in registers
// as the next frame's prediction source.
__global__ void __launch_bounds__(128)
    svc_decode_gop_kernel(const u8* __restrict__ stream, SvcGopArgs a,
                          u32* __restrict__ dbg = nullptr) {
  u32 s = blockIdx.x;
  u32 g0 = s * 128;
  u32 local_n = min(128u, a.ngroups - g0);
  u32 tid = threadIdx.x;
  u32 lane = tid & 63;
  u32 wave = tid >> 6;
  bool valid = tid < local_n;
  u32 g = g0 + tid;
  u32 base = g * 32;
  bool tail = valid && base + 32 > a.nbytes;
  u32 ntail = tail ? a.nbytes - base : 32;

  u32 cur32[8];
#pragma unroll
  for (int k = 0; k < 8; ++k) cur32[k] = 0;
  if (valid && a.prev && !(a.key_mask & 1)) {
    if (!tail) {
      const uint4* p4 = reinterpret_cast<const uint4*>(a.prev + base);
      uint4 v0 = p4[0], v1 = p4[1];
      cur32[0] = v0.x; cur32[1] = v0.y; cur32[2] = v0.z; cur32[3] = v0.w;
      cur32[4] = v1.x; cur32[5] = v1.y; cur32[6] = v1.z; cur32[7] = v1.w;
    } else {
      for (u32 k = 0; k < ntail; ++k) {
        cur32[k / 4] |= (u32)a.prev[base + k] << (8 * (k & 3));
      }
    }
  }

  // widths/packed offsets relative to the packet start, fully u64
  const u64 widths_off = 20ull + (u64)a.nsuper * 4;
  const u64 packed_off = widths_off + (u64)((a.ngroups + 3) / 4 * 4);
  for (i32 f = 0; f < a.nframes; ++f) {
    const u8* pkt = stream + a.pkt_off[f];
    const u32* super_off = reinterpret_cast<const u32*>(pkt + 20);

    u32 w = valid ? (u32)pkt[widths_off + g0 + tid] : 0;
    // Packed offsets by prefix sum of 4*w: wave64 shuffle scan. The
    // cross-wave carry is NOT passed through LDS: wave 1 re-loads wave
    // 0's 64 widths (one byte per lane) and butterfly-reduces them — a
    // barrier-free kernel, so a block's 16-frame loop never synchronizes.
    u32 val = 4u * w;
    u32 x = val;
#pragma unroll
    for (u32 d = 1; d < 64; d <<= 1) {
      u32 y = __shfl_up(x, d, 64);
      if (lane >= d) x += y;
    }
    u32 carry = 0;
    if (wave == 1) {
      u32 w0 = (g0 + lane < a.ngroups)
                   ? (u32)pkt[widths_off + g0 + lane]
                   : 0;
      u32 sum = 4u * w0;
#pragma unroll
      for (u32 d = 32; d >= 1; d >>= 1) sum += __shfl_xor(sum, d, 64);
      carry = sum;
    }
    u32 my_off = x - val + carry;

    // Fully unrolled unpack straight into packed u32 words: residual bytes
    // never touch memory.
    u32 res32[8];
    if (!valid || w == 0) {
#pragma unroll
      for (int k = 0; k < 8; ++k) res32[k] = 0;
    } else {
      const u32* q = reinterpret_cast<const u32*>(
          pkt + (packed_off + (u64)super_off[s] + my_off));
      if (dbg && f == 0 && tid == 0 && s >= 255 && s <= 257) {
        u32* d = dbg + (s - 255) * 8;
        d[0] = super_off[s];
        d[1] = my_off;
        d[2] = w;
        d[3] = q[0];
        d[4] = q[1];
        d[5] = (u32)packed_off;
        d[6] = (u32)a.pkt_off[0];
        d[7] = 0xd00dfeed;
      }
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

    if (valid) {
      bool is_key = (a.key_mask >> f) & 1;
      u32 nb = tail ? ntail : 32;
      if (is_key) {
        u8 p = 128;
        u32 out[8] = {0, 0, 0, 0, 0, 0, 0, 0};
        for (u32 k = 0; k < nb; ++k) {
          u8 rz = (u8)((res32[k / 4] >> (8 * (k & 3))) & 0xff);
          p = (u8)(p + dev_unzigzag(rz));
          out[k / 4] |= (u32)p << (8 * (k & 3));
        }
#pragma unroll
        for (int k = 0; k < 8; ++k) cur32[k] = out[k];
      } else if (!tail) {
#pragma unroll
        for (int k = 0; k < 8; ++k) {
          u32 pv = cur32[k];
          u32 rz = res32[k];
          u32 o = 0;
#pragma unroll
          for (int b = 0; b < 4; ++b) {
            u8 byte = (u8)((pv >> (8 * b)) & 0xff);
            byte = (u8)(byte + dev_unzigzag((u8)((rz >> (8 * b)) & 0xff)));
            o |= (u32)byte << (8 * b);
          }
          cur32[k] = o;
        }
      } else {
        u32 out[8] = {0, 0, 0, 0, 0, 0, 0, 0};
        for (u32 k = 0; k < nb; ++k) {
          u8 pv = (u8)((cur32[k / 4] >> (8 * (k & 3))) & 0xff);
          u8 rz = (u8)((res32[k / 4] >> (8 * (k & 3))) & 0xff);
          out[k / 4] |= (u32)((u8)(pv + dev_unzigzag(rz))) << (8 * (k & 3));
        }
#pragma unroll
        for (int k = 0; k < 8; ++k) cur32[k] = out[k];
      }

      if (a.out[f]) {
        if (!tail) {
          uint4* dst = reinterpret_cast<uint4*>(a.out[f] + base);
          dst[0] = make_uint4(cur32[0], cur32[1], cur32[2], cur32[3]);
          dst[1] = make_uint4(cur32[4], cur32[5], cur32[6], cur32[7]);
        } else {
          for (u32 k = 0; k < ntail; ++k) {
            a.out[f][base + k] =
                (u8)((cur32[k / 4] >> (8 * (k & 3))) & 0xff);
          }
        }
      }
    }
  }
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
  if (span.empty()) return out;

  // GOP chains are independent: each keyframe starts a new chain on one of
  // 4 auxiliary streams (fork/join with events) so a work packet's GOPs
  // decode concurrently; within a chain, kGopBatch frames go down in ONE
  // launch with the prediction chain in registers. (The reference's
  // decoder automaton is serial per item — decoder_automata.cpp.)
  constexpr int kChains = 4;
  hipStream_t chain_stream[kChains];
  bool chain_used[kChains] = {false, false, false, false};
  std::vector<u8*> scratches;

  auto make_elem = [&](i64 f) {
    Element e;
    e.is_frame = true;
    e.frame_info.shape[0] = vm.height;
    e.frame_info.shape[1] = vm.width;
    e.frame_info.shape[2] = vm.channels;
    e.frame_info.type = vm.frame_type;
    e.size = nbytes;
    e.buffer = new_buffer(dev, nbytes);
    e.device = dev;
    e.index = f;
    return e;
  };

  SvcGopArgs a{};
  a.nbytes = nbytes;
  a.ngroups = ngroups;
  a.nsuper = nsuper;
  a.nframes = 0;
  a.prev = nullptr;
  a.key_mask = 0;

  int chain = -1;
  hipStream_t cs = s;
  u8* chain_scratch = nullptr;
  size_t wi = 0;
  size_t ki = 0;  // cursor into sorted keyframe_indices

  auto flush = [&](bool chain_continues) {
    if (a.nframes == 0) return;
    int last = a.nframes - 1;
    if (chain_continues && !a.out[last]) {
      // the next launch needs this frame as its prediction source
      if (!chain_scratch) {
        chain_scratch = new_buffer(dev, nbytes);
        scratches.push_back(chain_scratch);
      }
      a.out[last] = chain_scratch;
    }
    svc_decode_gop_kernel<<<nsuper, 128, 0, cs>>>(d_stream, a);
    SVC_CHECK(hipGetLastError());
    const u8* next_prev = a.out[last];
    a.nframes = 0;
    a.key_mask = 0;
    a.prev = next_prev;
  };

  for (i64 f : span) {
    while (ki < vm.keyframe_indices.size() && vm.keyframe_indices[ki] < f)
      ++ki;
    bool is_key =
        ki < vm.keyframe_indices.size() && vm.keyframe_indices[ki] == f;
    static const int kBatchLimit = []() {
      const char* e = std::getenv("SCANNER_SVC_BATCH");
      int v = e ? atoi(e) : kGopBatch;
      return v < 1 ? 1 : (v > kGopBatch ? kGopBatch : v);
    }();
    if (is_key) {
      flush(false);
      a.prev = nullptr;
      chain_scratch = nullptr;
      ++chain;
      int ci = chain % kChains;
      cs = (hipStream_t)per_thread_aux_stream(ci);
      if (!chain_used[ci]) {
        chain_used[ci] = true;
        chain_stream[ci] = cs;
        if (up_ev) SVC_CHECK(hipStreamWaitEvent(cs, up_ev, 0));
      }
    } else if (a.nframes == kBatchLimit) {
      flush(true);
    }
    SCA_CHECK(is_key || a.nframes > 0 || a.prev,
              "svc chain does not start at a keyframe");
    int fi = a.nframes++;
    a.pkt_off[fi] = vm.sample_offsets[f] - lo;
    if (is_key) a.key_mask |= 1u << fi;
    bool wanted = wi < want.size() && want[wi] == f;
    if (wanted) {
      Element e = make_elem(f);
      out.push_back(e);
      a.out[fi] = e.buffer;
      ++wi;
    } else {
      a.out[fi] = nullptr;
    }
  }
  flush(false);

  // Join: main stream waits every used chain, then sync before returning
  // scratch buffers to the shared pool.
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

std::vector<u32> svc_gpu_debug_dump(const std::vector<u8>& stream,
                                    const VideoMetadata& vm) {
  DeviceHandle dev{DeviceType::GPU, 0};
  hipStream_t s = (hipStream_t)per_thread_hip_stream();
  u32 nbytes = (u32)((i64)vm.height * vm.width * vm.channels);
  u32 ngroups = (nbytes + 31) / 32;
  u32 nsuper = (ngroups + 127) / 128;
  u8* d_stream = new_buffer(dev, stream.size());
  SVC_CHECK(hipMemcpy(d_stream, stream.data(), stream.size(),
                      hipMemcpyHostToDevice));
  u8* d_out = new_buffer(dev, nbytes);
  u8* d_dbg = new_buffer(dev, 3 * 8 * 4);
  SVC_CHECK(hipMemset(d_dbg, 0, 3 * 8 * 4));
  SvcGopArgs a{};
  a.nbytes = nbytes;
  a.ngroups = ngroups;
  a.nsuper = nsuper;
  a.nframes = 1;
  a.key_mask = 1;
  a.pkt_off[0] = vm.sample_offsets[0];
  a.out[0] = d_out;
  svc_decode_gop_kernel<<<nsuper, 128, 0, s>>>(d_stream, a, (u32*)d_dbg);
  SVC_CHECK(hipGetLastError());
  SVC_CHECK(hipStreamSynchronize(s));
  std::vector<u32> dbg(3 * 8);
  SVC_CHECK(hipMemcpy(dbg.data(), d_dbg, 3 * 8 * 4,
                      hipMemcpyDeviceToHost));
  delete_buffer(dev, d_stream);
  delete_buffer(dev, d_out);
  delete_buffer(dev, d_dbg);
  return dbg;
}

}  // namespace sca
