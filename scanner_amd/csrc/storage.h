// Storage backend + database path scheme.
//
// Capability parity: reference layer 0 ("storehouse" StorageBackend with
// RandomReadFile/WriteFile) + scanner/engine/metadata.h:37-86 path scheme.
// POSIX is the first-party backend; the interface is small enough that an
// object-store backend slots in behind it.
#pragma once

#include <memory>

#include "common.h"

namespace sca {

class StorageBackend {
 public:
  virtual ~StorageBackend() = default;

  virtual std::vector<u8> read_all(const std::string& path) = 0;
  // Random-access read of [offset, offset+size).
  virtual void read_range(const std::string& path, u64 offset, u64 size,
                          u8* out) = 0;
  virtual u64 file_size(const std::string& path) = 0;
  virtual void write_all(const std::string& path,
                         const u8* data, size_t size) = 0;
  virtual bool exists(const std::string& path) = 0;
  virtual void remove(const std::string& path) = 0;
  virtual void remove_tree(const std::string& path) = 0;
  virtual void make_dirs(const std::string& path) = 0;
  virtual std::vector<std::string> list_dir(const std::string& path) = 0;

  static std::unique_ptr<StorageBackend> make_posix();
  // S3-semantics object store (flat keyspace, whole-object PUT, range GET,
  // prefix listing) emulated over a local bucket directory; a networked
  // S3/GCS client implements the same class (reference: storehouse GCS/S3
  // configs, scannerpy config.py:75-89).
  static std::unique_ptr<StorageBackend> make_object_store(
      const std::string& bucket_dir);
};

// Path scheme (mirrors metadata.h:37-86):
//   <db>/db_metadata.bin
//   <db>/tables/<table_id>/descriptor.bin
//   <db>/tables/<table_id>/<column_id>_<item_id>.bin
//   <db>/tables/<table_id>/<column_id>_<item_id>_metadata.bin
//   <db>/tables/<table_id>/<column_id>_<item_id>_video_metadata.bin
//   <db>/jobs/<job_id>/profile_<node>.bin
struct DatabasePaths {
  std::string root;
  explicit DatabasePaths(std::string r) : root(std::move(r)) {}
  std::string db_metadata() const { return root + "/db_metadata.bin"; }
  std::string megafile() const { return root + "/table_megafile.bin"; }
  std::string table_dir(i32 table_id) const {
    return root + "/tables/" + std::to_string(table_id);
  }
  std::string table_descriptor(i32 table_id) const {
    return table_dir(table_id) + "/descriptor.bin";
  }
  std::string item(i32 table_id, i32 column_id, i32 item_id) const {
    return table_dir(table_id) + "/" + std::to_string(column_id) + "_" +
           std::to_string(item_id) + ".bin";
  }
  std::string item_metadata(i32 table_id, i32 column_id, i32 item_id) const {
    return table_dir(table_id) + "/" + std::to_string(column_id) + "_" +
           std::to_string(item_id) + "_metadata.bin";
  }
  std::string video_metadata(i32 table_id, i32 column_id, i32 item_id) const {
    return table_dir(table_id) + "/" + std::to_string(column_id) + "_" +
           std::to_string(item_id) + "_video_metadata.bin";
  }
  std::string job_dir(i32 job_id) const {
    return root + "/jobs/" + std::to_string(job_id);
  }
  std::string job_profile(i32 job_id, i32 node) const {
    return job_dir(job_id) + "/profile_" + std::to_string(node) + ".bin";
  }
};

}  // namespace sca
