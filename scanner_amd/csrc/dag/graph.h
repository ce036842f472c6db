// Op-graph representation, per-stream job bindings, and the task-time
// analysis results. Capability parity: scanner/engine/dag_analysis.{h,cpp}
// (validation, slice levels, liveness inputs, derive_stencil_requirements).
//
// Design difference vs the reference: instead of packed per-op column
// vectors with index remapping, ops exchange elements through row-id-keyed
// maps and every builtin (Sample/Space/Slice/Unslice) reduces to a
// remap vector computed here, so the executor is mechanical.
#pragma once

#include <functional>
#include <map>

#include "../metadata.h"
#include "../msgpack.h"
#include "sampler.h"

namespace sca {

struct OpEdge {
  i32 op = -1;          // parent op index
  std::string column;   // parent output column name
};

struct OpNode {
  std::string name;              // registry name or builtin
  std::vector<OpEdge> inputs;
  std::vector<u8> args;          // msgpack op args (same across streams)
  DeviceType device = DeviceType::CPU;
  i32 device_id = 0;             // resolved at execution time
  i32 batch = 0;                 // 0 = factory default
  std::vector<i32> stencil;      // empty = registry default
  i32 warmup = -1;               // -1 = registry default

  bool is(const char* n) const { return name == n; }
};

// Builtin op names. *Frame variants behave identically; the column type
// differs (tracked by the type checker).
inline bool is_input_op(const std::string& n) { return n == "Input"; }
inline bool is_output_op(const std::string& n) { return n == "Output"; }
inline bool is_sample_op(const std::string& n) {
  return n == "Sample" || n == "SampleFrame" || n == "Space" || n == "SpaceFrame";
}
inline bool is_slice_op(const std::string& n) {
  return n == "Slice" || n == "SliceFrame";
}
inline bool is_unslice_op(const std::string& n) {
  return n == "Unslice" || n == "UnsliceFrame";
}
inline bool is_builtin_op(const std::string& n) {
  return is_input_op(n) || is_output_op(n) || is_sample_op(n) ||
         is_slice_op(n) || is_unslice_op(n);
}

struct JobGraph {
  std::vector<OpNode> ops;  // topological order; exactly one Output, last.

  i32 output_op() const { return (i32)ops.size() - 1; }
  mp::Value to_msgpack() const;
  static JobGraph from_msgpack(const mp::Value& v);
};

// ---- per-stream bindings (one "job" = one output stream) ----

struct SourceArgsC {
  std::string table;
  std::string column;
  // Registered user source (ops/source.h): when non-empty, `source` names
  // a SourceFactory and `args` are its per-stream msgpack args; the
  // table/column fields are unused.
  std::string source;
  std::vector<u8> args;
};

struct SinkArgsC {
  std::string table;
  // Registered user sink: when non-empty, output bypasses table storage.
  std::string sink;
  std::vector<u8> args;
};

struct JobBinding {
  std::map<i32, SourceArgsC> sources;       // Input op -> table/column
  std::map<i32, SamplingArgs> sampling;     // Sample/Space/Slice op -> args
  std::map<i32, std::vector<u8>> op_args;   // user op -> per-stream args
  SinkArgsC sink;

  mp::Value to_msgpack() const;
  static JobBinding from_msgpack(const mp::Value& v);
};

// ---- analysis ----

struct OpDomain {
  i64 num_rows = 0;
  i32 slice_level = 0;
  // Slice-group boundaries in this op's OUTPUT row domain, ascending,
  // starting 0 and ending num_rows. Level 0 => {0, num_rows}.
  std::vector<i64> group_starts;
  // For Slice ops: upstream offset of each group.
  std::vector<i64> slice_group_offsets;
};

// Per-op static info resolved from the registry + node overrides.
struct OpStaticInfo {
  bool is_builtin = false;
  std::vector<i32> stencil = {0};
  i32 warmup = 0;
  bool bounded_state = false;
  bool unbounded_state = false;
  i32 batch = 1;
  // (op, column) consumers for liveness: number of consumer edges per
  // output column name.
  std::map<std::string, i32> column_consumers;
  std::vector<std::string> output_columns;   // resolved output column names
  std::vector<ColumnType> output_types;
};

struct JobAnalysis {
  std::vector<OpDomain> domains;
  std::vector<OpStaticInfo> info;
  i64 output_rows = 0;  // rows of the sink
};

// Execution plan of one op for one task.
struct OpTaskPlan {
  // Rows (op output domain) downstream actually consumes, ascending.
  std::vector<i64> required_rows;
  // Kernel ops: output rows to execute (ascending; includes warmup rows).
  std::vector<i64> compute_rows;
  // Kernel ops: per compute row, the input-domain rows fed (stencil window,
  // clamped to slice-group bounds; size == |stencil|).
  std::vector<std::vector<i64>> windows;
  // Kernel ops: 1 if the kernel must reset() before computing this row
  // (slice-group start).
  std::vector<u8> reset_before;
  // Builtin remap ops: per required row, the parent row (-1 => null element).
  std::vector<i64> remap;
};

struct TaskPlan {
  std::vector<OpTaskPlan> ops;
  std::map<i32, std::vector<i64>> load_rows;  // Input op -> rows to read
};

// Row-count resolver for Input ops (table lookups are injected so analysis
// is testable without storage).
using SourceRowsFn = std::function<i64(const SourceArgsC&)>;

// Validate graph structure against the registry; throws ScannerError.
void validate_graph(const JobGraph& graph);

// Forward pass: per-op row domains + slice levels for one stream.
JobAnalysis analyze_job(const JobGraph& graph, const JobBinding& binding,
                        const SourceRowsFn& source_rows);

// Backward pass (derive_stencil_requirements parity): exact per-op row sets
// for an output-row range [task_start, task_end).
TaskPlan derive_task_plan(const JobGraph& graph, const JobAnalysis& analysis,
                          const JobBinding& binding, i64 task_start,
                          i64 task_end);

}  // namespace sca
