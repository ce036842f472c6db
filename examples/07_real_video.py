"""Tutorial 07: real-video ingest and mp4 export.

The framework ingests real video files with pure parsing — an mp4 demuxer
(sample tables + avcC) and an H.264 Annex-B access-unit indexer — no
FFmpeg, no codec library (parity: the reference's ingest.cpp +
h264_byte_stream_index_creator.cpp). The result is a table whose
VideoMetadata carries per-frame byte offsets and keyframe indices: exactly
what a hardware (rocDecode/VCN) decoder automaton consumes. Decoding
H.264 *content* needs that hardware decoder, so this tutorial synthesizes
a tiny spec-conformant stream, ingests it, inspects the index, and remuxes
it back to a playable .mp4 (no transcode).
"""
import os
import sys
import tempfile

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "tests"))

import scanner_amd as sp
from test_video_ingest import make_annexb  # synthesized H.264 stream

tmp = tempfile.mkdtemp(prefix="scanner_tut07_")
sc = sp.Client(db_path=os.path.join(tmp, "db"))

# 1) synthesize a 12-frame H.264 elementary stream (3 GOPs of 4)
stream, au_offsets, keyframes = make_annexb(gops=3, frames_per_gop=4)
src = os.path.join(tmp, "clip.h264")
with open(src, "wb") as f:
    f.write(stream)

# 2) ingest: demux + keyframe byte-offset index
info = sc.ingest_video_file(src, "clip")
print(f"ingested: {info['num_frames']} frames, "
      f"{info['width']}x{info['height']}, codec={info['codec']}")
assert info["num_frames"] == 12
assert sc.table_info("clip")["num_rows"] == 12

# 3) export: remux to .mp4 (ftyp/mdat/moov with full sample tables)
out = os.path.join(tmp, "clip_out.mp4")
sp.NamedVideoStream(sc, "clip").save_mp4(out, fps=24)
print(f"exported {os.path.getsize(out)} bytes ->", out)

# 4) the export is a valid mp4: re-ingest it
info2 = sc.ingest_video_file(out, "clip_roundtrip")
assert info2["num_frames"] == 12
print("round-trip OK")
