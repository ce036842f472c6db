// Registries for user sources/sinks + the built-in Files source/sink.
//
// FilesSource: one row per file path (args: {"paths": [...]}) — the proof
// that the engine can be fed from something that is not a table (reference
// analogue: the Files source in scannertools / source.h SDK usage).
// FilesSink: writes each row of each column to
// <dir>/<column>_<row>.<ext> (args: {"dir": ..., "ext": ...}).
#include <fstream>

#include "../memory.h"
#include "../msgpack.h"
#include "source.h"

namespace sca {

void SourceRegistry::add(SourceFactory f) {
  SCA_CHECK(!f.name.empty() && f.make && f.make_enumerator,
            "incomplete SourceFactory");
  factories_[f.name] = std::move(f);
}
bool SourceRegistry::has(const std::string& name) const {
  return factories_.count(name) > 0;
}
const SourceFactory& SourceRegistry::get(const std::string& name) const {
  auto it = factories_.find(name);
  SCA_CHECK(it != factories_.end(), "unknown source '" + name + "'");
  return it->second;
}
std::vector<std::string> SourceRegistry::names() const {
  std::vector<std::string> out;
  for (auto& kv : factories_) out.push_back(kv.first);
  return out;
}

void SinkRegistry::add(SinkFactory f) {
  SCA_CHECK(!f.name.empty() && f.make, "incomplete SinkFactory");
  factories_[f.name] = std::move(f);
}
bool SinkRegistry::has(const std::string& name) const {
  return factories_.count(name) > 0;
}
const SinkFactory& SinkRegistry::get(const std::string& name) const {
  auto it = factories_.find(name);
  SCA_CHECK(it != factories_.end(), "unknown sink '" + name + "'");
  return it->second;
}
std::vector<std::string> SinkRegistry::names() const {
  std::vector<std::string> out;
  for (auto& kv : factories_) out.push_back(kv.first);
  return out;
}

SourceRegistry& source_registry() {
  static SourceRegistry r;
  return r;
}
SinkRegistry& sink_registry() {
  static SinkRegistry r;
  return r;
}

SourceRegistrar::SourceRegistrar(SourceFactory f) {
  source_registry().add(std::move(f));
}
SinkRegistrar::SinkRegistrar(SinkFactory f) {
  sink_registry().add(std::move(f));
}

namespace {

std::vector<std::string> paths_from_args(const std::vector<u8>& args) {
  SCA_CHECK(!args.empty(), "Files source needs {'paths': [...]} args");
  auto v = mp::decode(args);
  std::vector<std::string> paths;
  for (auto& p : v.as_map().at("paths").as_array())
    paths.push_back(p.as_str());
  return paths;
}

class FilesSource : public Source {
 public:
  explicit FilesSource(const SourceConfig& c)
      : Source(c), paths_(paths_from_args(c.args)) {}

  void read(const std::vector<i64>& rows, ElementVector& out) override {
    for (i64 r : rows) {
      SCA_CHECK(r >= 0 && r < (i64)paths_.size(),
                "FilesSource row out of range");
      std::ifstream f(paths_[r], std::ios::binary | std::ios::ate);
      SCA_CHECK(f.good(), "FilesSource: cannot open " + paths_[r]);
      std::streamsize n = f.tellg();
      f.seekg(0);
      Element e;
      e.size = (size_t)n;
      e.buffer = new_buffer(CPU_DEVICE, e.size);
      f.read((char*)e.buffer, n);
      SCA_CHECK(f.good() || n == 0, "FilesSource: short read " + paths_[r]);
      e.index = r;
      out.push_back(e);
      if (config_.profiler)
        config_.profiler->increment("io_read_bytes", (i64)n);
    }
  }

 private:
  std::vector<std::string> paths_;
};

class FilesEnumerator : public Enumerator {
 public:
  explicit FilesEnumerator(const SourceConfig& c)
      : Enumerator(c), n_((i64)paths_from_args(c.args).size()) {}
  i64 total_elements() override { return n_; }

 private:
  i64 n_;
};

class FilesSink : public Sink {
 public:
  explicit FilesSink(const SinkConfig& c) : Sink(c) {
    SCA_CHECK(!c.args.empty(), "Files sink needs {'dir': ...} args");
    auto v = mp::decode(c.args);
    dir_ = v.as_map().at("dir").as_str();
    ext_ = v.get_str("ext", "bin");
  }

  void write(const std::vector<ElementVector>& columns) override {
    for (size_t c = 0; c < columns.size(); ++c) {
      for (auto& e : columns[c]) {
        if (e.is_null) continue;
        std::string path = dir_ + "/c" + std::to_string(c) + "_" +
                           std::to_string(e.index) + "." + ext_;
        std::ofstream f(path, std::ios::binary | std::ios::trunc);
        SCA_CHECK(f.good(), "FilesSink: cannot open " + path);
        f.write((const char*)e.buffer, (std::streamsize)e.size);
        SCA_CHECK(f.good(), "FilesSink: short write " + path);
        if (config_.profiler)
          config_.profiler->increment("io_write_bytes", (i64)e.size);
      }
    }
  }

 private:
  std::string dir_;
  std::string ext_;
};

}  // namespace

void register_files_source_sink() {
  SourceFactory sf;
  sf.name = "Files";
  sf.output_type = ColumnType::Bytes;
  sf.make = [](const SourceConfig& c) -> std::unique_ptr<Source> {
    return std::make_unique<FilesSource>(c);
  };
  sf.make_enumerator =
      [](const SourceConfig& c) -> std::unique_ptr<Enumerator> {
    return std::make_unique<FilesEnumerator>(c);
  };
  source_registry().add(std::move(sf));

  SinkFactory kf;
  kf.name = "Files";
  kf.make = [](const SinkConfig& c) -> std::unique_ptr<Sink> {
    return std::make_unique<FilesSink>(c);
  };
  sink_registry().add(std::move(kf));
}

}  // namespace sca
