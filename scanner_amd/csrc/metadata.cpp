#include "metadata.h"

#include <fcntl.h>
#include <sys/file.h>
#include <unistd.h>

#include "serialize.h"

namespace sca {

namespace {
constexpr u32 kTableMagic = 0x53435442;  // "SCTB"
constexpr u32 kVideoMagic = 0x53435644;  // "SCVD"
constexpr u32 kDbMagic = 0x53434442;     // "SCDB"
constexpr u32 kVersion = 1;

void header(BinWriter& w, u32 magic) {
  w.u32v(magic);
  w.u32v(kVersion);
}
void check_header(BinReader& r, u32 magic, const char* what) {
  u32 m = r.u32v(), v = r.u32v();
  if (m != magic || v != kVersion)
    throw ScannerError(std::string("bad metadata header for ") + what);
}
}  // namespace

i32 TableMetadata::column_id(const std::string& n) const {
  for (auto& c : columns)
    if (c.name == n) return c.id;
  throw ScannerError("no column '" + n + "' in table '" + name + "'");
}

bool TableMetadata::has_column(const std::string& n) const {
  for (auto& c : columns)
    if (c.name == n) return true;
  return false;
}

ColumnType TableMetadata::column_type(const std::string& n) const {
  for (auto& c : columns)
    if (c.name == n) return c.type;
  throw ScannerError("no column '" + n + "' in table '" + name + "'");
}

std::vector<u8> TableMetadata::serialize() const {
  BinWriter w;
  header(w, kTableMagic);
  w.i32v(id);
  w.str(name);
  w.u64v(columns.size());
  for (auto& c : columns) {
    w.i32v(c.id);
    w.str(c.name);
    w.i32v((i32)c.type);
  }
  w.vec_pod(end_rows);
  return w.take();
}

TableMetadata TableMetadata::deserialize(const std::vector<u8>& buf) {
  BinReader r(buf);
  check_header(r, kTableMagic, "table");
  TableMetadata m;
  m.id = r.i32v();
  m.name = r.str();
  u64 nc = r.u64v();
  for (u64 i = 0; i < nc; ++i) {
    ColumnMeta c;
    c.id = r.i32v();
    c.name = r.str();
    c.type = (ColumnType)r.i32v();
    m.columns.push_back(c);
  }
  m.end_rows = r.vec_pod<i64>();
  return m;
}

std::vector<u8> VideoMetadata::serialize() const {
  BinWriter w;
  header(w, kVideoMagic);
  w.i32v(width);
  w.i32v(height);
  w.i32v(channels);
  w.i32v((i32)frame_type);
  w.str(codec);
  w.i64v(num_frames);
  w.vec_pod(keyframe_indices);
  w.vec_pod(sample_offsets);
  w.vec_pod(sample_sizes);
  return w.take();
}

VideoMetadata VideoMetadata::deserialize(const std::vector<u8>& buf) {
  BinReader r(buf);
  check_header(r, kVideoMagic, "video");
  VideoMetadata m;
  m.width = r.i32v();
  m.height = r.i32v();
  m.channels = r.i32v();
  m.frame_type = (FrameType)r.i32v();
  m.codec = r.str();
  m.num_frames = r.i64v();
  m.keyframe_indices = r.vec_pod<i64>();
  m.sample_offsets = r.vec_pod<u64>();
  m.sample_sizes = r.vec_pod<u64>();
  return m;
}

std::vector<u8> DatabaseMetadata::serialize() const {
  BinWriter w;
  header(w, kDbMagic);
  w.i32v(next_table_id);
  w.i32v(next_job_id);
  w.u64v(table_ids.size());
  for (auto& kv : table_ids) {
    w.str(kv.first);
    w.i32v(kv.second);
  }
  std::vector<i32> ct(committed_tables.begin(), committed_tables.end());
  w.vec_pod(ct);
  w.u64v(job_ids.size());
  for (auto& kv : job_ids) {
    w.str(kv.first);
    w.i32v(kv.second);
  }
  std::vector<i32> cj(committed_jobs.begin(), committed_jobs.end());
  w.vec_pod(cj);
  return w.take();
}

DatabaseMetadata DatabaseMetadata::deserialize(const std::vector<u8>& buf) {
  BinReader r(buf);
  check_header(r, kDbMagic, "db");
  DatabaseMetadata m;
  m.next_table_id = r.i32v();
  m.next_job_id = r.i32v();
  u64 nt = r.u64v();
  for (u64 i = 0; i < nt; ++i) {
    std::string k = r.str();
    m.table_ids[k] = r.i32v();
  }
  for (i32 t : r.vec_pod<i32>()) m.committed_tables.insert(t);
  u64 nj = r.u64v();
  for (u64 i = 0; i < nj; ++i) {
    std::string k = r.str();
    m.job_ids[k] = r.i32v();
  }
  for (i32 j : r.vec_pod<i32>()) m.committed_jobs.insert(j);
  return m;
}

// ---------------- Database ----------------

Database::Database(std::shared_ptr<StorageBackend> storage,
                   const std::string& db_path)
    : storage_(std::move(storage)), paths_(db_path) {
  storage_->make_dirs(db_path + "/tables");
  storage_->make_dirs(db_path + "/jobs");
  if (storage_->exists(paths_.db_metadata())) {
    meta_ = DatabaseMetadata::deserialize(storage_->read_all(paths_.db_metadata()));
  } else {
    std::lock_guard<std::mutex> l(mu_);
    persist();
  }
  // Megafile pre-warm: one read covers every committed table's descriptor
  // (committed tables are immutable; deleted ids are unreachable through
  // the name map). A corrupt/stale megafile degrades to per-table reads.
  if (storage_->exists(paths_.megafile())) {
    try {
      auto buf = storage_->read_all(paths_.megafile());
      BinReader r(buf);
      u32 magic = r.u32v();
      u32 ver = r.u32v();
      if (magic == 0x5343544d && ver == kVersion) {  // "SCTM"
        u64 n = r.u64v();
        std::lock_guard<std::mutex> l(mu_);
        for (u64 i = 0; i < n; ++i) {
          auto blob = r.bytes();
          TableMetadata t = TableMetadata::deserialize(blob);
          if (meta_.committed_tables.count(t.id)) {
            table_cache_.emplace(t.id, std::move(t));
          }
        }
      }
    } catch (const std::exception&) {
      // degrade silently to per-table descriptor reads
    }
  }
}


void Database::persist() {
  auto buf = meta_.serialize();
  storage_->write_all(paths_.db_metadata(), buf.data(), buf.size());
}

void Database::refresh() {
  if (storage_->exists(paths_.db_metadata())) {
    meta_ = DatabaseMetadata::deserialize(
        storage_->read_all(paths_.db_metadata()));
  }
}

namespace {
// Inter-process exclusive lock on <db>/db.lock for metadata
// read-modify-write cycles. Multiple processes mutate the shared db
// (master job executors, ingesting clients); without this, concurrent
// persist() calls lose each other's tables. flock is posix-only, like the
// shipped storage backend; an object-store backend would route metadata
// mutations through the master instead (reference model).
class ScopedDbLock {
 public:
  explicit ScopedDbLock(const std::string& db_path) {
    fd_ = ::open((db_path + "/db.lock").c_str(), O_CREAT | O_RDWR, 0644);
    if (fd_ >= 0) (void)::flock(fd_, LOCK_EX);
  }
  ~ScopedDbLock() {
    if (fd_ >= 0) {
      (void)::flock(fd_, LOCK_UN);
      (void)::close(fd_);
    }
  }

 private:
  int fd_ = -1;
};
}  // namespace

void Database::recover() {
  std::lock_guard<std::mutex> l(mu_);
  std::vector<std::string> dead;
  for (auto& kv : meta_.table_ids) {
    if (!meta_.committed_tables.count(kv.second)) dead.push_back(kv.first);
  }
  for (auto& name : dead) {
    i32 id = meta_.table_ids[name];
    storage_->remove_tree(paths_.table_dir(id));
    meta_.table_ids.erase(name);
    table_cache_.erase(id);
  }
  persist();
}

TableMetadata Database::new_table(const std::string& name,
                                  const std::vector<std::string>& column_names,
                                  const std::vector<ColumnType>& column_types,
                                  bool overwrite) {
  SCA_CHECK(column_names.size() == column_types.size(), "column spec mismatch");
  ScopedDbLock dbl(paths_.root);
  std::lock_guard<std::mutex> l(mu_);
  refresh();
  auto it = meta_.table_ids.find(name);
  if (it != meta_.table_ids.end()) {
    if (!overwrite)
      throw ScannerError("table '" + name + "' already exists");
    i32 old_id = it->second;
    storage_->remove_tree(paths_.table_dir(old_id));
    meta_.table_ids.erase(it);
    meta_.committed_tables.erase(old_id);
    table_cache_.erase(old_id);
  }
  TableMetadata t;
  t.id = meta_.next_table_id++;
  t.name = name;
  for (size_t i = 0; i < column_names.size(); ++i) {
    t.columns.push_back(ColumnMeta{(i32)i, column_names[i], column_types[i]});
  }
  meta_.table_ids[name] = t.id;
  storage_->make_dirs(paths_.table_dir(t.id));
  auto buf = t.serialize();
  storage_->write_all(paths_.table_descriptor(t.id), buf.data(), buf.size());
  table_cache_[t.id] = t;
  persist();
  return t;
}

void Database::commit_table(i32 table_id) {
  ScopedDbLock dbl(paths_.root);
  std::lock_guard<std::mutex> l(mu_);
  refresh();
  meta_.committed_tables.insert(table_id);
  persist();
}

void Database::write_megafile() {
  ScopedDbLock dbl(paths_.root);
  std::lock_guard<std::mutex> l(mu_);
  refresh();
  BinWriter w;
  w.u32v(0x5343544d);  // "SCTM"
  w.u32v(kVersion);
  std::vector<std::vector<u8>> blobs;
  for (i32 id : meta_.committed_tables) {
    auto it = table_cache_.find(id);
    if (it != table_cache_.end()) {
      blobs.push_back(it->second.serialize());
      continue;
    }
    if (!storage_->exists(paths_.table_descriptor(id))) continue;
    blobs.push_back(storage_->read_all(paths_.table_descriptor(id)));
  }
  w.u64v(blobs.size());
  for (auto& b : blobs) w.bytes(b);
  auto buf = w.take();
  storage_->write_all(paths_.megafile(), buf.data(), buf.size());
}

bool Database::table_committed(i32 table_id) {
  std::lock_guard<std::mutex> l(mu_);
  return meta_.committed_tables.count(table_id) > 0;
}

void Database::update_table(const TableMetadata& t) {
  std::lock_guard<std::mutex> l(mu_);
  auto buf = t.serialize();
  storage_->write_all(paths_.table_descriptor(t.id), buf.data(), buf.size());
  table_cache_[t.id] = t;
}

void Database::delete_table(const std::string& name) {
  ScopedDbLock dbl(paths_.root);
  std::lock_guard<std::mutex> l(mu_);
  refresh();
  auto it = meta_.table_ids.find(name);
  if (it == meta_.table_ids.end()) return;
  i32 id = it->second;
  storage_->remove_tree(paths_.table_dir(id));
  meta_.table_ids.erase(it);
  meta_.committed_tables.erase(id);
  table_cache_.erase(id);
  persist();
}

bool Database::has_table(const std::string& name) {
  std::lock_guard<std::mutex> l(mu_);
  if (meta_.table_ids.count(name)) return true;
  refresh();
  return meta_.table_ids.count(name) > 0;
}

TableMetadata Database::get_table(const std::string& name) {
  i32 id;
  {
    std::lock_guard<std::mutex> l(mu_);
    auto it = meta_.table_ids.find(name);
    if (it == meta_.table_ids.end()) {
      // another process may have created it since we loaded metadata
      refresh();
      it = meta_.table_ids.find(name);
      if (it == meta_.table_ids.end())
        throw ScannerError("no table '" + name + "'");
    }
    id = it->second;
  }
  return get_table(id);
}

TableMetadata Database::get_table(i32 id) {
  {
    std::lock_guard<std::mutex> l(mu_);
    auto it = table_cache_.find(id);
    if (it != table_cache_.end()) return it->second;
  }
  auto t = TableMetadata::deserialize(
      storage_->read_all(paths_.table_descriptor(id)));
  std::lock_guard<std::mutex> l(mu_);
  table_cache_[id] = t;
  return t;
}

std::vector<std::string> Database::table_names() {
  std::lock_guard<std::mutex> l(mu_);
  std::vector<std::string> names;
  for (auto& kv : meta_.table_ids) names.push_back(kv.first);
  return names;
}

i32 Database::new_job(const std::string& name) {
  ScopedDbLock dbl(paths_.root);
  std::lock_guard<std::mutex> l(mu_);
  refresh();
  i32 id = meta_.next_job_id++;
  meta_.job_ids[name + "#" + std::to_string(id)] = id;
  storage_->make_dirs(paths_.job_dir(id));
  persist();
  return id;
}

void Database::commit_job(i32 job_id) {
  ScopedDbLock dbl(paths_.root);
  std::lock_guard<std::mutex> l(mu_);
  refresh();
  meta_.committed_jobs.insert(job_id);
  persist();
}

}  // namespace sca
