#include "svc.h"

#include <algorithm>
#include <cstring>

namespace sca {

namespace {

constexpr u32 kMagic = 0x31435653;  // 'SVC1' LE

inline u8 zigzag(i32 r) {
  // r in [-255, 255] reduced mod 256 as i8; computed in unsigned
  // arithmetic ((v<<1) on a negative int is UB before C++20 — UBSan
  // caught the signed form via tests/cpp/asan_parsers.cpp)
  i8 v = (i8)(u8)(r & 0xff);
  u8 doubled = (u8)((u32)(u8)v << 1);  // (v << 1) mod 256
  u8 sign = v < 0 ? 0xff : 0x00;       // arithmetic v >> 7
  return (u8)(doubled ^ sign);
}
inline u8 unzigzag(u8 z) {
  i8 v = (i8)((z >> 1) ^ (-(i32)(z & 1)));
  return (u8)v;
}

inline u32 bits_needed(u8 maxz) {
  u32 b = 0;
  while ((1u << b) <= maxz && b < 8) ++b;
  // (1<<b) > maxz required; maxz up to 255 needs 8
  if (b == 8 && maxz >= (1u << 7) && maxz > ((1u << 8) - 1)) b = 8;
  return b;
}

void encode_frame(const u8* cur, const u8* prev, u32 nbytes, bool key,
                  std::vector<u8>& out) {
  u32 ngroups = (nbytes + 31) / 32;
  u32 nsuper = (ngroups + 127) / 128;

  std::vector<u8> residual(ngroups * 32, 0);
  for (u32 g = 0; g < ngroups; ++g) {
    for (u32 k = 0; k < 32; ++k) {
      u32 i = g * 32 + k;
      if (i >= nbytes) break;
      u8 pred;
      if (key) {
        pred = (k == 0) ? 128 : cur[i - 1];
      } else {
        pred = prev[i];
      }
      residual[g * 32 + k] = zigzag((i32)cur[i] - (i32)pred);
    }
  }

  std::vector<u8> widths(ngroups);
  for (u32 g = 0; g < ngroups; ++g) {
    u8 mz = 0;
    for (u32 k = 0; k < 32; ++k) mz = std::max(mz, residual[g * 32 + k]);
    widths[g] = (u8)bits_needed(mz);
  }

  std::vector<u32> super_off(nsuper, 0);
  {
    u32 off = 0;
    for (u32 s = 0; s < nsuper; ++s) {
      super_off[s] = off;
      u32 ge = std::min(ngroups, (s + 1) * 128);
      for (u32 g = s * 128; g < ge; ++g) off += 4u * widths[g];
    }
  }

  size_t header = 4 + 4 + 4 + 4 + 4;
  size_t packed_bytes = 0;
  for (u32 g = 0; g < ngroups; ++g) packed_bytes += 4u * widths[g];
  size_t widths_padded = (ngroups + 3) / 4 * 4;  // 4B-align packed region
  size_t total = header + nsuper * 4 + widths_padded + packed_bytes;
  size_t base = out.size();
  out.resize(base + total, 0);
  u8* p = out.data() + base;
  auto put32 = [&](u32 v) {
    std::memcpy(p, &v, 4);
    p += 4;
  };
  put32(kMagic);
  *p++ = key ? 0 : 1;
  *p++ = 0; *p++ = 0; *p++ = 0;
  put32(nbytes);
  put32(ngroups);
  put32(nsuper);
  std::memcpy(p, super_off.data(), nsuper * 4);
  p += nsuper * 4;
  std::memcpy(p, widths.data(), ngroups);
  p += widths_padded;
  // pack
  for (u32 g = 0; g < ngroups; ++g) {
    u32 w = widths[g];
    if (w == 0) continue;
    u64 acc = 0;
    u32 nacc = 0;
    u8* q = p;
    for (u32 k = 0; k < 32; ++k) {
      acc |= ((u64)(residual[g * 32 + k] & ((1u << w) - 1))) << nacc;
      nacc += w;
      while (nacc >= 8) {
        *q++ = (u8)(acc & 0xff);
        acc >>= 8;
        nacc -= 8;
      }
    }
    if (nacc > 0) *q++ = (u8)(acc & 0xff);
    p += 4 * w;  // exactly 4*w bytes per group
  }
}

void decode_frame(const SvcPacketView& v, const u8* prev, u8* cur) {
  for (u32 g = 0; g < v.ngroups; ++g) {
    u32 w = v.widths[g];
    // corrupt-stream guards: widths > 8 would shift out of range and
    // read past the group's 4*w payload; arbitrary super_off values
    // would walk the read cursor outside the packet (the fuzz test
    // drives both — tests/cpp/asan_parsers.cpp)
    SCA_CHECK(w <= 8, "svc packet: group width > 8");
    // packed offset: supergroup base + local prefix
    u32 s = g / 128;
    u64 off = v.super_off[s];
    for (u32 gg = s * 128; gg < g; ++gg) off += 4u * v.widths[gg];
    SCA_CHECK(off + 4u * w <= v.packed_size,
              "svc packet: packed region overrun");
    const u8* q = v.packed + off;
    u8 res[32];
    if (w == 0) {
      std::memset(res, 0, 32);
    } else {
      u64 acc = 0;
      u32 nacc = 0;
      u32 qi = 0;
      for (u32 k = 0; k < 32; ++k) {
        while (nacc < w) {
          acc |= ((u64)q[qi++]) << nacc;
          nacc += 8;
        }
        res[k] = (u8)(acc & ((1u << w) - 1));
        acc >>= w;
        nacc -= w;
      }
    }
    for (u32 k = 0; k < 32; ++k) {
      u32 i = g * 32 + k;
      if (i >= v.nbytes) break;
      u8 pred;
      if (v.is_key) {
        pred = (k == 0) ? 128 : cur[i - 1];
      } else {
        pred = prev[i];
      }
      cur[i] = (u8)(pred + unzigzag(res[k]));
    }
  }
}

}  // namespace

SvcPacketView svc_parse_packet(const u8* pkt, size_t size) {
  SCA_CHECK(size >= 20, "svc packet too small");
  SvcPacketView v;
  u32 magic;
  std::memcpy(&magic, pkt, 4);
  SCA_CHECK(magic == kMagic, "bad svc packet magic");
  v.is_key = pkt[4] == 0;
  std::memcpy(&v.nbytes, pkt + 8, 4);
  std::memcpy(&v.ngroups, pkt + 12, 4);
  std::memcpy(&v.nsuper, pkt + 16, 4);
  // geometry invariants in 64-bit: a corrupt header must not be able to
  // wrap the size check (e.g. nsuper=2^30 makes nsuper*4 tiny in u32) or
  // claim more groups than nbytes implies
  SCA_CHECK(v.ngroups == ((u64)v.nbytes + 31) / 32,
            "svc packet: ngroups inconsistent with nbytes");
  SCA_CHECK(v.nsuper == ((u64)v.ngroups + 127) / 128,
            "svc packet: nsuper inconsistent with ngroups");
  u64 header = 20;
  u64 widths_padded = ((u64)v.ngroups + 3) / 4 * 4;
  u64 need = header + (u64)v.nsuper * 4 + widths_padded;
  SCA_CHECK((u64)size >= need, "svc packet truncated");
  v.super_off = reinterpret_cast<const u32*>(pkt + header);
  v.widths = pkt + header + (size_t)v.nsuper * 4;
  v.packed = v.widths + widths_padded;
  v.packed_size = size - (size_t)need;
  return v;
}

void svc_encode_cpu(const u8* frames, i64 n, i32 h, i32 w, i32 c, i32 gop,
                    std::vector<u8>& stream, VideoMetadata& vm) {
  u32 nbytes = (u32)((i64)h * w * c);
  vm.codec = "svc";
  vm.width = w;
  vm.height = h;
  vm.channels = c;
  vm.frame_type = FrameType::U8;
  vm.num_frames = n;
  vm.keyframe_indices.clear();
  vm.sample_offsets.clear();
  vm.sample_sizes.clear();
  for (i64 f = 0; f < n; ++f) {
    bool key = (f % gop) == 0;
    if (key) vm.keyframe_indices.push_back(f);
    size_t before = stream.size();
    encode_frame(frames + (size_t)f * nbytes,
                 f > 0 ? frames + (size_t)(f - 1) * nbytes : nullptr, nbytes,
                 key, stream);
    vm.sample_offsets.push_back(before);
    vm.sample_sizes.push_back(stream.size() - before);
  }
}

std::vector<i64> svc_decode_span(const VideoMetadata& vm,
                                 const std::vector<i64>& want) {
  std::vector<i64> out;
  if (want.empty()) return out;
  const auto& kf = vm.keyframe_indices;
  i64 pos = -1;
  for (i64 f : want) {
    // keyframe at or before f
    auto it = std::upper_bound(kf.begin(), kf.end(), f);
    SCA_CHECK(it != kf.begin(), "no keyframe before frame");
    i64 k = *std::prev(it);
    i64 start = std::max(pos + 1, k);
    // if we already decoded past k continuously, just continue from pos+1
    if (pos >= k && pos < f) start = pos + 1;
    else if (pos >= f) continue;  // already decoded
    else start = k;
    for (i64 i = start; i <= f; ++i) out.push_back(i);
    pos = f;
  }
  return out;
}

void svc_decode_cpu(const u8* stream, size_t size, const VideoMetadata& vm,
                    const std::vector<i64>& want,
                    std::vector<std::vector<u8>>& out, u64 stream_offset) {
  u32 nbytes = (u32)((i64)vm.height * vm.width * vm.channels);
  std::vector<i64> span = svc_decode_span(vm, want);
  std::vector<u8> prev(nbytes), cur(nbytes);
  size_t wi = 0;
  i64 last = -2;
  for (i64 f : span) {
    SCA_CHECK(f < (i64)vm.sample_offsets.size(), "frame beyond stream");
    SCA_CHECK(vm.sample_offsets[f] >= stream_offset &&
                  vm.sample_offsets[f] + vm.sample_sizes[f] <=
                      stream_offset + size,
              "svc stream range does not cover decode span");
    const u8* pkt = stream + (vm.sample_offsets[f] - stream_offset);
    SvcPacketView v = svc_parse_packet(pkt, vm.sample_sizes[f]);
    // the output buffers are sized from the TABLE metadata; a corrupt
    // packet must not be able to claim a larger frame and write past them
    SCA_CHECK(v.nbytes == nbytes, "svc packet: frame size mismatch");
    SCA_CHECK(v.is_key || f == last + 1,
              "svc decode: non-contiguous delta frame");
    decode_frame(v, prev.data(), cur.data());
    last = f;
    if (wi < want.size() && want[wi] == f) {
      out.push_back(cur);
      ++wi;
    }
    std::swap(prev, cur);
  }
  SCA_CHECK(wi == want.size(), "svc decode: not all wanted frames produced");
}

}  // namespace sca
