// Real-video ingest + mp4 export.
//
// Capability parity: scanner/engine/ingest.cpp:175-380 (demux to Annex-B,
// keyframe byte-offset index, VideoDescriptor write) and storage.py:353-374
// (save_mp4) — implemented with the first-party mp4/h264 parsers (no
// FFmpeg in this image; decode of the ingested H.264 requires the
// rocDecode/VCN hardware path and fails loudly until the image provides
// it, mirroring how the reference shipped its NVDEC path present-but-
// disabled, evaluate_worker.cpp:90-93).
#pragma once

#include <string>

#include "../metadata.h"

namespace sca {

struct IngestResult {
  i64 num_frames = 0;
  i32 width = 0, height = 0;
  std::string codec;
};

// Ingest a video file (.mp4 with an AVC track, or a raw Annex-B .h264
// elementary stream) into `table_name` as one H.264-indexed video column:
// the stream is stored Annex-B with per-access-unit byte offsets +
// keyframe indices in the VideoMetadata, exactly the index a
// rocDecode-based decoder automaton consumes. Throws ScannerError on
// malformed input.
IngestResult ingest_video_file(Database& db, const std::string& table_name,
                               const std::string& column,
                               const std::string& path);

// Remux an ingested H.264 table back into a playable .mp4 (no transcode).
void export_mp4(Database& db, const std::string& table_name,
                const std::string& column, const std::string& out_path,
                double fps = 30.0);

}  // namespace sca
