#include "ingest.h"

#include <fstream>
#include <set>

#include "../engine/table_io.h"
#include "h264.h"
#include "mp4.h"

namespace sca {

namespace {

std::vector<u8> read_file(const std::string& path) {
  std::ifstream f(path, std::ios::binary | std::ios::ate);
  SCA_CHECK(f.good(), "cannot open video file: " + path);
  std::streamsize n = f.tellg();
  SCA_CHECK(n > 0, "empty video file: " + path);
  std::vector<u8> buf((size_t)n);
  f.seekg(0);
  f.read((char*)buf.data(), n);
  SCA_CHECK(f.good(), "short read on video file: " + path);
  return buf;
}

void append_sc(std::vector<u8>& out, const std::vector<u8>& nal) {
  const u8 sc[4] = {0, 0, 0, 1};
  out.insert(out.end(), sc, sc + 4);
  out.insert(out.end(), nal.begin(), nal.end());
}

}  // namespace

IngestResult ingest_video_file(Database& db, const std::string& table_name,
                               const std::string& column,
                               const std::string& path) {
  std::vector<u8> file = read_file(path);

  std::vector<u8> stream;  // Annex-B with in-band SPS/PPS at keyframes
  VideoMetadata vm;
  vm.codec = "h264";
  vm.channels = 3;
  vm.frame_type = FrameType::U8;

  bool is_mp4 = file.size() >= 8 && file[4] == 'f' && file[5] == 't' &&
                file[6] == 'y' && file[7] == 'p';
  if (is_mp4) {
    Mp4Track t = mp4_parse(file.data(), file.size());
    vm.width = t.width;
    vm.height = t.height;
    vm.num_frames = (i64)t.sample_sizes.size();
    vm.keyframe_indices = t.keyframe_indices;
    std::set<i64> kf(t.keyframe_indices.begin(), t.keyframe_indices.end());
    stream.reserve(file.size());
    for (size_t s = 0; s < t.sample_sizes.size(); ++s) {
      u64 au_start = stream.size();
      if (kf.count((i64)s)) {
        // replay parameter sets so keyframe-aligned seeks are
        // self-contained (reference writes metadata packets the same way,
        // ingest.cpp:310-380)
        for (auto& sp : t.sps) append_sc(stream, sp);
        for (auto& pp : t.pps) append_sc(stream, pp);
      }
      // AVCC -> Annex-B: length-prefixed NALs become start codes
      const u8* p = file.data() + t.sample_offsets[s];
      u64 left = t.sample_sizes[s];
      while (left > 0) {
        SCA_CHECK(left >= (u64)t.length_size, "mp4: truncated AVCC NAL");
        u64 nl = 0;
        for (i32 k = 0; k < t.length_size; ++k) nl = (nl << 8) | p[k];
        p += t.length_size;
        left -= t.length_size;
        SCA_CHECK(nl > 0 && nl <= left, "mp4: AVCC NAL length out of range");
        const u8 sc[4] = {0, 0, 0, 1};
        stream.insert(stream.end(), sc, sc + 4);
        stream.insert(stream.end(), p, p + nl);
        p += nl;
        left -= nl;
      }
      vm.sample_offsets.push_back(au_start);
      vm.sample_sizes.push_back(stream.size() - au_start);
    }
  } else {
    // Annex-B elementary stream
    H264Index idx = h264_index_annexb(file.data(), file.size());
    vm.width = idx.width;
    vm.height = idx.height;
    vm.num_frames = idx.num_frames;
    vm.keyframe_indices = idx.keyframe_indices;
    vm.sample_offsets = idx.sample_offsets;
    vm.sample_sizes = idx.sample_sizes;
    stream = std::move(file);
  }

  TableMetadata t =
      db.new_table(table_name, {column}, {ColumnType::Video}, true);
  write_video_item(db, t, column, 0, stream, vm);
  t.end_rows = {vm.num_frames};
  db.update_table(t);
  db.commit_table(t.id);

  IngestResult r;
  r.num_frames = vm.num_frames;
  r.width = vm.width;
  r.height = vm.height;
  r.codec = "h264";
  return r;
}

void export_mp4(Database& db, const std::string& table_name,
                const std::string& column, const std::string& out_path,
                double fps) {
  TableMetadata t = db.get_table(table_name);
  SCA_CHECK(!t.end_rows.empty(), "export_mp4: empty table");
  // Concatenate item streams (usually one item for ingested video).
  std::vector<u8> stream;
  std::vector<u64> offsets, sizes;
  std::vector<i64> keyframes;
  i64 row_base = 0;
  for (i32 item = 0; item < (i32)t.end_rows.size(); ++item) {
    VideoMetadata vm = read_video_metadata(db, t, column, item);
    SCA_CHECK(vm.codec == "h264",
              "export_mp4: column is '" + vm.codec +
                  "', only h264-indexed tables can be remuxed");
    std::vector<u8> item_bytes =
        db.storage()->read_all(db.paths().item(t.id, t.column_id(column),
                                               item));
    u64 base = stream.size();
    stream.insert(stream.end(), item_bytes.begin(), item_bytes.end());
    for (size_t s = 0; s < vm.sample_offsets.size(); ++s) {
      offsets.push_back(base + vm.sample_offsets[s]);
      sizes.push_back(vm.sample_sizes[s]);
    }
    for (i64 k : vm.keyframe_indices) keyframes.push_back(row_base + k);
    row_base += vm.num_frames;
  }
  // parameter sets are in-band — pull them from the stream's own index
  H264Index idx = h264_index_annexb(stream.data(), stream.size());
  std::vector<u8> mp4 =
      mp4_write(stream.data(), stream.size(), offsets, sizes, keyframes,
                idx.sps, idx.pps, idx.width, idx.height, fps);
  std::ofstream f(out_path, std::ios::binary | std::ios::trunc);
  SCA_CHECK(f.good(), "export_mp4: cannot open " + out_path);
  f.write((const char*)mp4.data(), (std::streamsize)mp4.size());
  SCA_CHECK(f.good(), "export_mp4: short write to " + out_path);
}

}  // namespace sca
