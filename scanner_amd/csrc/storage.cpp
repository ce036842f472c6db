#include "storage.h"

#include <dirent.h>
#include <fcntl.h>
#include <sys/stat.h>
#include <sys/types.h>
#include <unistd.h>

#include <atomic>
#include <cerrno>
#include <cstdio>
#include <cstring>
#include <set>
#include <thread>

namespace sca {

namespace {

class PosixStorage : public StorageBackend {
 public:
  std::vector<u8> read_all(const std::string& path) override {
    int fd = ::open(path.c_str(), O_RDONLY);
    if (fd < 0) throw ScannerError("open failed: " + path + ": " + strerror(errno));
    struct stat st;
    if (fstat(fd, &st) != 0) {
      ::close(fd);
      throw ScannerError("fstat failed: " + path);
    }
    std::vector<u8> buf(st.st_size);
    size_t off = 0;
    while (off < buf.size()) {
      ssize_t n = ::read(fd, buf.data() + off, buf.size() - off);
      if (n <= 0) {
        ::close(fd);
        throw ScannerError("read failed: " + path);
      }
      off += n;
    }
    ::close(fd);
    return buf;
  }

  void read_range(const std::string& path, u64 offset, u64 size,
                  u8* out) override {
    int fd = ::open(path.c_str(), O_RDONLY);
    if (fd < 0) throw ScannerError("open failed: " + path + ": " + strerror(errno));
    // One pread from page cache is a single-thread memcpy (~10 GB/s); the
    // decode stage eats multi-GB video spans per step, so fan large reads
    // out over a few threads (the reference reads items on a thread pool
    // too — load_worker + storehouse).
    constexpr u64 kParallelMin = 16 << 20;
    constexpr u64 kChunk = 8 << 20;
    std::atomic<bool> fail{false};  // written from reader threads
    auto read_span = [&](u64 lo, u64 hi) {
      u64 off = lo;
      while (off < hi) {
        ssize_t n = ::pread(fd, out + off, hi - off, offset + off);
        if (n <= 0) {
          fail.store(true, std::memory_order_relaxed);
          return;
        }
        off += n;
      }
    };
    if (size < kParallelMin) {
      read_span(0, size);
    } else {
      int nthreads = (int)std::min<u64>(8, (size + kChunk - 1) / kChunk);
      u64 per = (size + nthreads - 1) / nthreads;
      std::vector<std::thread> ts;
      for (int i = 1; i < nthreads; ++i) {
        u64 lo = i * per;
        if (lo >= size) break;
        ts.emplace_back(read_span, lo, std::min(size, lo + per));
      }
      read_span(0, std::min(size, per));
      for (auto& th : ts) th.join();
    }
    ::close(fd);
    if (fail) throw ScannerError("pread failed: " + path);
  }

  u64 file_size(const std::string& path) override {
    struct stat st;
    if (stat(path.c_str(), &st) != 0)
      throw ScannerError("stat failed: " + path);
    return st.st_size;
  }

  void write_all(const std::string& path, const u8* data,
                 size_t size) override {
    // Write to temp + rename for atomicity (commit semantics depend on
    // it). The temp name must be unique PER WRITER: two processes
    // writing the same path (e.g. both creating a fresh db's metadata)
    // collided on a shared ".tmp" — the first rename stole the second
    // writer's file and its rename failed with ENOENT (found by
    // test_concurrent_clients_table_creation).
    static std::atomic<u64> seq{0};
    std::string tmp = path + ".tmp." + std::to_string((u64)::getpid()) +
                      "." + std::to_string(seq.fetch_add(1));
    int fd = ::open(tmp.c_str(), O_WRONLY | O_CREAT | O_TRUNC, 0644);
    if (fd < 0) throw ScannerError("open for write failed: " + tmp + ": " + strerror(errno));
    size_t off = 0;
    while (off < size) {
      ssize_t n = ::write(fd, data + off, size - off);
      if (n < 0) {
        ::close(fd);
        throw ScannerError("write failed: " + tmp);
      }
      off += n;
    }
    if (fsync(fd) != 0 || ::close(fd) != 0)
      throw ScannerError("fsync/close failed: " + tmp);
    if (::rename(tmp.c_str(), path.c_str()) != 0)
      throw ScannerError("rename failed: " + path + ": " +
                         strerror(errno));
  }

  bool exists(const std::string& path) override {
    struct stat st;
    return stat(path.c_str(), &st) == 0;
  }

  void remove(const std::string& path) override { ::unlink(path.c_str()); }

  void remove_tree(const std::string& path) override {
    DIR* d = opendir(path.c_str());
    if (!d) {
      ::unlink(path.c_str());
      return;
    }
    struct dirent* e;
    while ((e = readdir(d)) != nullptr) {
      std::string name = e->d_name;
      if (name == "." || name == "..") continue;
      remove_tree(path + "/" + name);
    }
    closedir(d);
    ::rmdir(path.c_str());
  }

  void make_dirs(const std::string& path) override {
    std::string cur;
    size_t i = 0;
    while (i < path.size()) {
      size_t j = path.find('/', i + 1);
      if (j == std::string::npos) j = path.size();
      cur = path.substr(0, j);
      if (!cur.empty() && cur != "/") {
        if (mkdir(cur.c_str(), 0755) != 0 && errno != EEXIST) {
          throw ScannerError("mkdir failed: " + cur + ": " + strerror(errno));
        }
      }
      i = j;
    }
  }

  std::vector<std::string> list_dir(const std::string& path) override {
    std::vector<std::string> out;
    DIR* d = opendir(path.c_str());
    if (!d) return out;
    struct dirent* e;
    while ((e = readdir(d)) != nullptr) {
      std::string name = e->d_name;
      if (name == "." || name == "..") continue;
      out.push_back(name);
    }
    closedir(d);
    return out;
  }
};

// S3-semantics emulation over a local "bucket" directory (capability
// parity: the reference's GCS/S3 storehouse configs, scannerpy
// config.py:75-89). Contract differences from POSIX that callers must not
// assume away — this backend enforces them:
//   * flat keyspace: keys map to url-encoded filenames in ONE directory;
//     there are no directories, make_dirs is a no-op
//   * whole-object PUT: write_all replaces the object atomically (no
//     rename contract exposed, no partial writes visible)
//   * range GET: read_range serves [offset, offset+size)
//   * prefix scan: list_dir/remove_tree operate on key prefixes
// A networked S3/GCS client slots in behind the same class by swapping the
// file operations for HTTP calls.
class ObjectStorage : public StorageBackend {
 public:
  explicit ObjectStorage(std::string bucket) : bucket_(std::move(bucket)) {
    // create the bucket itself (the only real directory)
    std::string cur;
    size_t i = 0;
    while (i < bucket_.size()) {
      size_t j = bucket_.find('/', i + 1);
      if (j == std::string::npos) j = bucket_.size();
      cur = bucket_.substr(0, j);
      if (!cur.empty() && cur != "/") (void)mkdir(cur.c_str(), 0755);
      i = j;
    }
  }

  std::vector<u8> read_all(const std::string& key) override {
    return posix_.read_all(keyfile(key));
  }
  void read_range(const std::string& key, u64 offset, u64 size,
                  u8* out) override {
    posix_.read_range(keyfile(key), offset, size, out);
  }
  u64 file_size(const std::string& key) override {
    return posix_.file_size(keyfile(key));
  }
  void write_all(const std::string& key, const u8* data,
                 size_t size) override {
    posix_.write_all(keyfile(key), data, size);  // PUT (atomic replace)
  }
  bool exists(const std::string& key) override {
    return posix_.exists(keyfile(key));
  }
  void remove(const std::string& key) override {
    posix_.remove(keyfile(key));
  }
  void remove_tree(const std::string& prefix) override {
    for (auto& name : bucket_keys()) {
      std::string key = decode(name);
      if (key == prefix || key.rfind(prefix + "/", 0) == 0) {
        posix_.remove(bucket_ + "/" + name);
      }
    }
  }
  void make_dirs(const std::string&) override {}  // no directories
  std::vector<std::string> list_dir(const std::string& prefix) override {
    // next path segment under prefix/, deduped (S3 delimiter listing)
    std::set<std::string> segs;
    std::string p = prefix + "/";
    for (auto& name : bucket_keys()) {
      std::string key = decode(name);
      if (key.rfind(p, 0) != 0) continue;
      std::string rest = key.substr(p.size());
      size_t cut = rest.find('/');
      segs.insert(cut == std::string::npos ? rest : rest.substr(0, cut));
    }
    return std::vector<std::string>(segs.begin(), segs.end());
  }

 private:
  std::string keyfile(const std::string& key) const {
    std::string out;
    out.reserve(key.size());
    for (char c : key) {
      if (c == '/') {
        out += "%2F";
      } else if (c == '%') {
        out += "%25";
      } else {
        out.push_back(c);
      }
    }
    return bucket_ + "/" + out;
  }
  static std::string decode(const std::string& name) {
    std::string out;
    for (size_t i = 0; i < name.size(); ++i) {
      if (name[i] == '%' && i + 2 < name.size()) {
        if (name.compare(i, 3, "%2F") == 0) {
          out.push_back('/');
          i += 2;
          continue;
        }
        if (name.compare(i, 3, "%25") == 0) {
          out.push_back('%');
          i += 2;
          continue;
        }
      }
      out.push_back(name[i]);
    }
    return out;
  }
  std::vector<std::string> bucket_keys() {
    return posix_.list_dir(bucket_);
  }

  std::string bucket_;
  PosixStorage posix_;
};

}  // namespace

std::unique_ptr<StorageBackend> StorageBackend::make_posix() {
  return std::make_unique<PosixStorage>();
}

std::unique_ptr<StorageBackend> StorageBackend::make_object_store(
    const std::string& bucket_dir) {
  return std::make_unique<ObjectStorage>(bucket_dir);
}

}  // namespace sca
