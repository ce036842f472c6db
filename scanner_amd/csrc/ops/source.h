// Source / Sink / Enumerator SDK — feed the engine from (and write results
// to) places that are not database tables.
//
// Capability parity: scanner/api/source.h:68 (Source::read),
// sink.h (Sink::new_task/write/finished), enumerator.h:49-55
// (total_elements), and their REGISTER_* macros + registries
// (engine/source_registry.h etc.). The built-in column/video source and
// sink remain engine-internal fast paths; registered sources plug into the
// same Input/Output op slots via per-stream args (JobBinding.sources[i]
// .source / .sink).
#pragma once

#include <functional>
#include <map>
#include <memory>

#include "../element.h"
#include "../profiler.h"

namespace sca {

struct SourceConfig {
  std::vector<u8> args;  // per-stream source args (msgpack)
  Profiler* profiler = nullptr;
};

// Reads elements for explicit global row ids (the engine's task plans name
// exact rows — samplers/stencils included). Elements must be CPU-resident;
// the engine moves them to kernel devices as needed.
class Source {
 public:
  explicit Source(const SourceConfig& config) : config_(config) {}
  virtual ~Source() = default;
  virtual void read(const std::vector<i64>& rows, ElementVector& out) = 0;

 protected:
  SourceConfig config_;
};

// Names the stream's domain: how many rows exist (reference
// Enumerator::total_elements; element_args_at is subsumed by Source::read
// taking row ids directly).
class Enumerator {
 public:
  explicit Enumerator(const SourceConfig& config) : config_(config) {}
  virtual ~Enumerator() = default;
  virtual i64 total_elements() = 0;

 protected:
  SourceConfig config_;
};

struct SinkConfig {
  std::vector<u8> args;  // per-stream sink args (msgpack)
  Profiler* profiler = nullptr;
};

class Sink {
 public:
  explicit Sink(const SinkConfig& config) : config_(config) {}
  virtual ~Sink() = default;
  virtual void new_task(i32 task_id) {}
  // columns[c][r]: CPU-resident elements of output column c for this
  // task's rows (element .index = global row id; .is_null for nulls).
  virtual void write(const std::vector<ElementVector>& columns) = 0;
  virtual void finished() {}

 protected:
  SinkConfig config_;
};

struct SourceFactory {
  std::string name;
  ColumnType output_type = ColumnType::Bytes;
  std::function<std::unique_ptr<Source>(const SourceConfig&)> make;
  std::function<std::unique_ptr<Enumerator>(const SourceConfig&)>
      make_enumerator;
};

struct SinkFactory {
  std::string name;
  std::function<std::unique_ptr<Sink>(const SinkConfig&)> make;
};

class SourceRegistry {
 public:
  void add(SourceFactory f);
  bool has(const std::string& name) const;
  const SourceFactory& get(const std::string& name) const;
  std::vector<std::string> names() const;

 private:
  std::map<std::string, SourceFactory> factories_;
};

class SinkRegistry {
 public:
  void add(SinkFactory f);
  bool has(const std::string& name) const;
  const SinkFactory& get(const std::string& name) const;
  std::vector<std::string> names() const;

 private:
  std::map<std::string, SinkFactory> factories_;
};

SourceRegistry& source_registry();
SinkRegistry& sink_registry();

struct SourceRegistrar {
  explicit SourceRegistrar(SourceFactory f);
};
struct SinkRegistrar {
  explicit SinkRegistrar(SinkFactory f);
};

#define SCA_REGISTER_SOURCE(var, ...) \
  static ::sca::SourceRegistrar source_registrar_##var(__VA_ARGS__)
#define SCA_REGISTER_SINK(var, ...) \
  static ::sca::SinkRegistrar sink_registrar_##var(__VA_ARGS__)

// Built-in Files source/sink (blobs from / to a directory); registered at
// module init alongside the op stdlib.
void register_files_source_sink();

}  // namespace sca
