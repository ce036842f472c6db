// Database / table / video metadata records + CRUD with commit semantics.
// Capability parity: scanner/engine/metadata.{h,cpp} + metadata.proto
// (DatabaseMetadata, TableMetadata, VideoMetadata, commit flags, recovery).
#pragma once

#include <mutex>
#include <set>
#include <unordered_map>

#include "storage.h"

namespace sca {

enum class ColumnType : i32 {
  Bytes = 0,  // opaque serialized elements
  Video = 1,  // frame column (raw or codec-compressed)
};

enum class FrameType : i32 { U8 = 0, U16 = 1, F32 = 2, F64 = 3 };

inline size_t frame_type_size(FrameType t) {
  switch (t) {
    case FrameType::U8: return 1;
    case FrameType::U16: return 2;
    case FrameType::F32: return 4;
    case FrameType::F64: return 8;
  }
  return 1;
}

struct ColumnMeta {
  i32 id = 0;
  std::string name;
  ColumnType type = ColumnType::Bytes;
};

struct TableMetadata {
  i32 id = -1;
  std::string name;
  std::vector<ColumnMeta> columns;
  // end_rows[i] = total rows after item i (items are the on-disk files; one
  // item per task by default). Empty => 0 rows.
  std::vector<i64> end_rows;

  i64 num_rows() const { return end_rows.empty() ? 0 : end_rows.back(); }
  i32 num_items() const { return (i32)end_rows.size(); }
  i32 column_id(const std::string& name) const;
  bool has_column(const std::string& name) const;
  ColumnType column_type(const std::string& name) const;

  std::vector<u8> serialize() const;
  static TableMetadata deserialize(const std::vector<u8>& buf);
};

// Per-(video column, item) descriptor: frame geometry + codec + per-frame
// byte ranges + keyframe index, so tasks decode only their GOP spans
// (reference: VideoDescriptor/VideoIndexEntry).
struct VideoMetadata {
  i32 width = 0, height = 0, channels = 0;
  FrameType frame_type = FrameType::U8;
  // "raw" = uncompressed frames; "svc" = scanner-amd GPU codec (GOP-based).
  std::string codec = "raw";
  i64 num_frames = 0;
  std::vector<i64> keyframe_indices;  // frame index of each keyframe
  std::vector<u64> sample_offsets;    // byte offset of each frame's packet
  std::vector<u64> sample_sizes;      // byte size of each frame's packet

  std::vector<u8> serialize() const;
  static VideoMetadata deserialize(const std::vector<u8>& buf);
};

struct DatabaseMetadata {
  i32 next_table_id = 0;
  i32 next_job_id = 0;
  std::unordered_map<std::string, i32> table_ids;       // name -> id
  std::set<i32> committed_tables;
  std::unordered_map<std::string, i32> job_ids;
  std::set<i32> committed_jobs;

  std::vector<u8> serialize() const;
  static DatabaseMetadata deserialize(const std::vector<u8>& buf);
};

// Thread-safe database facade over a storage backend. All mutations persist
// db_metadata.bin (temp+rename atomic).
class Database {
 public:
  Database(std::shared_ptr<StorageBackend> storage, const std::string& db_path);

  // Startup recovery: drop uncommitted tables/jobs (reference:
  // recover_and_init_database master.cpp:1311).
  void recover();

  TableMetadata new_table(const std::string& name,
                          const std::vector<std::string>& column_names,
                          const std::vector<ColumnType>& column_types,
                          bool overwrite = false);
  void commit_table(i32 table_id);
  bool table_committed(i32 table_id);
  void update_table(const TableMetadata& meta);  // rewrite descriptor
  void delete_table(const std::string& name);
  bool has_table(const std::string& name);
  TableMetadata get_table(const std::string& name);
  TableMetadata get_table(i32 id);
  std::vector<std::string> table_names();

  i32 new_job(const std::string& name);
  void commit_job(i32 job_id);

  // Table megafile (reference: write_table_megafile metadata.cpp:441-530,
  // a cloud-storage round-trip optimization): batch every COMMITTED
  // table's descriptor into one object. On open, a present megafile
  // pre-warms the descriptor cache so get_table() needs zero per-table
  // reads; committed tables are immutable, so cached entries cannot go
  // stale (deletes are filtered by the name->id map).
  void write_megafile();

  StorageBackend* storage() { return storage_.get(); }
  const DatabasePaths& paths() const { return paths_; }

 private:
  void persist();  // caller holds mu_
  // Re-read db_metadata.bin from storage (caller holds mu_). Mutators call
  // this under the inter-process file lock so concurrent writers (master
  // job executors, ingesting clients, workers) merge instead of losing
  // updates.
  void refresh();
  std::shared_ptr<StorageBackend> storage_;
  DatabasePaths paths_;
  std::mutex mu_;
  DatabaseMetadata meta_;
  std::unordered_map<i32, TableMetadata> table_cache_;
};

}  // namespace sca
