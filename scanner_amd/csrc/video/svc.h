// SVC — the scanner_amd video codec.
//
// Context: this image ships no codec libraries at all (no FFmpeg/libav, no
// rocDecode, no OpenCV), so H.264 cannot be decoded here by any route. The
// video layer therefore defines a pluggable codec interface whose
// first-party implementation is SVC: a GOP-structured, block-parallel,
// losslessly-compressed format designed to be decoded by HIP kernels
// straight into HBM — preserving every structural property the engine needs
// from a real codec (keyframes + delta frames, per-frame byte ranges,
// keyframe-aligned seeks, GOP-span decode) with a GPU-native bit layout.
// rocDecode/VCN H.264 slots in behind the same DecoderAutomata interface
// when present (capability parity: scanner/video/decoder_automata.h +
// nvidia_video_decoder.cpp, which the reference shipped disabled).
//
// Format (one packet per frame):
//   u32 magic 'SVC1'
//   u8  type (0=key, 1=delta), u8[3] pad
//   u32 nbytes   — raw payload bytes of the frame (H*W*C)
//   u32 ngroups  — ceil(nbytes/32); each group = 32 bytes of payload
//   u32 nsuper   — ceil(ngroups/128)
//   u32 super_off[nsuper] — byte offset of each supergroup's packed data,
//                           relative to the packed region start
//   u8  width[ngroups]    — bits per value in the group (0..8)
//   (pad to 4-byte alignment)
//   packed region: per group, 32 values * width bits = 4*width bytes —
//   every group payload is 4-byte aligned, so the GPU decoder unpacks with
//   u32 loads
//
// Residuals: delta frames predict from the previous frame byte; key frames
// predict from the previous byte within the group (first byte from 128).
// Residuals are zigzag-mapped (i8 -> u8) then bit-packed LSB-first.
// A group's payload is exactly 4*width bytes, so offsets derive from a
// prefix sum over widths — the GPU decoder does one LDS scan per
// supergroup (128 groups = 4 KiB of payload).
#pragma once

#include "../element.h"
#include "../metadata.h"

namespace sca {

// Encode n frames (contiguous raw HWC bytes, frame_size = h*w*c) into a
// packet stream. Fills vm: codec="svc", num_frames, keyframe_indices,
// sample_offsets/sizes.
void svc_encode_cpu(const u8* frames, i64 n, i32 h, i32 w, i32 c, i32 gop,
                    std::vector<u8>& stream, VideoMetadata& vm);

// Decode the given (sorted, item-local) frame indices. out[i] = raw frame.
// `stream` holds bytes [stream_offset, stream_offset+size) of the item —
// callers read only the GOP-span byte range (reference analogue:
// keyframe-aligned encoded-range reads, column_source.cpp:209).
void svc_decode_cpu(const u8* stream, size_t size, const VideoMetadata& vm,
                    const std::vector<i64>& want,
                    std::vector<std::vector<u8>>& out, u64 stream_offset = 0);

// Given wanted frames, the minimal list of frames that must be decoded
// (keyframe-aligned spans; reference analogue: DecodeArgs GOP spans).
std::vector<i64> svc_decode_span(const VideoMetadata& vm,
                                 const std::vector<i64>& want);

// GPU decode (kernels/svc_codec.hip): uploads the needed GOP spans once,
// then one kernel launch per frame on the calling thread's HIP stream;
// decoded frames land in HBM as engine elements (device-resident), so a
// downstream GPU op consumes them with zero copies.
std::vector<Element> svc_decode_gpu(const u8* stream_host, size_t size,
                                    const VideoMetadata& vm,
                                    const std::vector<i64>& want,
                                    DeviceHandle dev, u64 stream_offset = 0);

// Decode from an already device-resident encoded stream (the HBM span
// cache path — video/span_cache.h): `stream_dev` holds the item's bytes
// starting at stream offset `dev_lo` and must cover the keyframe-aligned
// span of `want`. No host access, no H2D.
std::vector<Element> svc_decode_gpu_dev(const u8* stream_dev, u64 dev_lo,
                                        const VideoMetadata& vm,
                                        const std::vector<i64>& want,
                                        DeviceHandle dev);

struct SvcPacketView {
  bool is_key = false;
  u32 nbytes = 0;
  u32 ngroups = 0;
  u32 nsuper = 0;
  const u32* super_off = nullptr;
  const u8* widths = nullptr;
  const u8* packed = nullptr;
  size_t packed_size = 0;  // bytes available in the packed region
};

// Parse + validate a packet header. All size arithmetic is 64-bit and the
// geometry invariants (ngroups = ceil(nbytes/32), nsuper = ceil(ngroups/
// 128)) are enforced, so a corrupt header cannot make the derived
// pointers overrun `size` (tests/cpp/asan_parsers.cpp fuzzes this under
// AddressSanitizer). Per-group width/offset payloads are validated lazily
// by the CPU decoder as it walks them.
SvcPacketView svc_parse_packet(const u8* pkt, size_t size);

// debug: launch the decode kernel with a diagnostic dump (tests only)
std::vector<u32> svc_gpu_debug_dump(const std::vector<u8>& stream,
                                    const VideoMetadata& vm);

}  // namespace sca
