// Minimal ISO-BMFF (mp4) demuxer and remuxer — pure parsing, no codec or
// container libraries.
//
// Capability parity: the reference demuxes mp4 through FFmpeg AVIO during
// ingest (scanner/engine/ingest.cpp:82-155) and indexes mp4s in place via
// the external hwang parser (ingest.cpp:382); its save_mp4 export
// (python/scannerpy/storage.py:353-374) also rides on FFmpeg. Here the
// sample tables (stsd/avcC, stts, stss, stsc, stsz, stco/co64) are parsed
// directly, giving per-sample absolute byte offsets + keyframes, and the
// writer remuxes an indexed H.264 elementary stream back into a playable
// .mp4 — no transcode, so no codec needed.
#pragma once

#include <string>
#include <vector>

#include "../common.h"

namespace sca {

struct Mp4Track {
  i32 width = 0, height = 0;        // from the SPS (authoritative)
  i32 length_size = 4;              // AVCC NAL length prefix bytes
  std::vector<std::vector<u8>> sps; // raw NAL bytes (no start code)
  std::vector<std::vector<u8>> pps;
  std::vector<u64> sample_offsets;  // absolute file offsets
  std::vector<u64> sample_sizes;
  std::vector<i64> keyframe_indices;
  u32 timescale = 0;
  std::vector<u32> sample_deltas;   // per-sample duration (expanded stts)
};

// Parse the first AVC video track of an mp4 in memory. Throws ScannerError
// on anything malformed or missing.
Mp4Track mp4_parse(const u8* data, size_t size);

// Remux: wrap H.264 access units (Annex-B byte ranges of `stream` given by
// offsets/sizes) into an mp4. `sps`/`pps` are raw NAL bytes. `fps` sets
// stts. Returns the complete file contents.
std::vector<u8> mp4_write(const u8* stream, size_t stream_size,
                          const std::vector<u64>& sample_offsets,
                          const std::vector<u64>& sample_sizes,
                          const std::vector<i64>& keyframe_indices,
                          const std::vector<u8>& sps,
                          const std::vector<u8>& pps, i32 width, i32 height,
                          double fps = 30.0);

}  // namespace sca
