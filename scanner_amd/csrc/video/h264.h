// H.264 Annex-B bitstream indexing — pure parsing, no codec libraries.
//
// Capability parity: the reference ingests real video by demuxing mp4 via
// FFmpeg, converting to Annex-B and building a keyframe byte-offset index
// (scanner/engine/ingest.cpp:175-380,
//  scanner/video/h264_byte_stream_index_creator.cpp:60-200). This image
// ships no codec libraries, but *indexing* needs none: NAL scanning, RBSP
// un-escaping and Exp-Golomb SPS/slice-header parsing are plain C++. The
// produced index is exactly what a rocDecode/VCN decoder automaton needs to
// drop in behind the codec interface (decode itself requires the hardware
// decoder and fails loudly until the image provides one).
#pragma once

#include <vector>

#include "../common.h"

namespace sca {

struct H264Sps {
  i32 profile_idc = 0;
  i32 level_idc = 0;
  i32 sps_id = 0;
  i32 chroma_format_idc = 1;  // 4:2:0 default
  bool frame_mbs_only = true;
  i32 width = 0, height = 0;  // display size after cropping
  std::vector<u8> raw;        // NAL bytes incl. the 1-byte header
};

struct H264Pps {
  i32 pps_id = 0;
  i32 sps_id = 0;
  std::vector<u8> raw;
};

// Per-access-unit index over an Annex-B stream. Offsets are byte positions
// of the AU's first start code (parameter/SEI NALs preceding a slice attach
// to that slice's AU, so seeking to a keyframe offset replays its SPS/PPS).
struct H264Index {
  i32 width = 0, height = 0;
  i64 num_frames = 0;
  std::vector<u64> sample_offsets;
  std::vector<u64> sample_sizes;
  std::vector<i64> keyframe_indices;  // IDR access units
  std::vector<u8> sps;                // active SPS NAL (raw bytes)
  std::vector<u8> pps;                // active PPS NAL (raw bytes)
};

// Scan an Annex-B stream into an access-unit index. Throws ScannerError on
// malformed input (no start code, slice before SPS/PPS, truncated NAL).
H264Index h264_index_annexb(const u8* data, size_t size);

// Parse one SPS NAL (input = NAL bytes including the header byte; handles
// emulation-prevention un-escaping). Throws on malformed bits.
H264Sps h264_parse_sps(const u8* nal, size_t size);

// Exp-Golomb bit reader over an un-escaped RBSP (exposed for tests).
class BitReader {
 public:
  BitReader(const u8* data, size_t size) : d_(data), n_(size) {}
  u32 u(int bits);   // fixed-width
  u32 ue();          // unsigned Exp-Golomb
  i32 se();          // signed Exp-Golomb
  bool eof() const { return pos_ >= n_ * 8; }

 private:
  const u8* d_;
  size_t n_;
  size_t pos_ = 0;  // bit position
};

// Strip emulation_prevention_three_byte (00 00 03 -> 00 00).
std::vector<u8> rbsp_unescape(const u8* data, size_t size);
// Insert emulation prevention (inverse; used by the mp4 writer for
// synthesized parameter sets in tests).
std::vector<u8> rbsp_escape(const u8* data, size_t size);

}  // namespace sca
