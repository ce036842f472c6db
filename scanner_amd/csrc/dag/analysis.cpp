#include <algorithm>
#include <set>

#include "../ops/kernel.h"
#include "graph.h"

namespace sca {

// ---------------- msgpack (de)serialization ----------------

mp::Value JobGraph::to_msgpack() const {
  mp::Array ops_arr;
  for (auto& op : ops) {
    mp::Map m;
    m["name"] = op.name;
    mp::Array ins;
    for (auto& e : op.inputs) {
      mp::Map em;
      em["op"] = (i64)e.op;
      em["column"] = e.column;
      ins.push_back(mp::Value(std::move(em)));
    }
    m["inputs"] = std::move(ins);
    m["args"] = op.args;
    m["device"] = (i64)op.device;
    m["batch"] = (i64)op.batch;
    mp::Array st;
    for (i32 s : op.stencil) st.push_back(mp::Value((i64)s));
    m["stencil"] = std::move(st);
    m["warmup"] = (i64)op.warmup;
    ops_arr.push_back(mp::Value(std::move(m)));
  }
  mp::Map top;
  top["ops"] = std::move(ops_arr);
  return mp::Value(std::move(top));
}

JobGraph JobGraph::from_msgpack(const mp::Value& v) {
  JobGraph g;
  for (auto& opv : v.as_map().at("ops").as_array()) {
    OpNode op;
    op.name = opv.get_str("name", "");
    for (auto& ev : opv.as_map().at("inputs").as_array()) {
      OpEdge e;
      e.op = (i32)ev.get_int("op", -1);
      e.column = ev.get_str("column", "");
      op.inputs.push_back(e);
    }
    op.args = opv.as_map().at("args").as_bin();
    op.device = (DeviceType)opv.get_int("device", 0);
    op.batch = (i32)opv.get_int("batch", 0);
    for (auto& sv : opv.as_map().at("stencil").as_array())
      op.stencil.push_back((i32)sv.as_int());
    op.warmup = (i32)opv.get_int("warmup", -1);
    g.ops.push_back(std::move(op));
  }
  return g;
}

mp::Value JobBinding::to_msgpack() const {
  mp::Map m;
  mp::Map srcs;
  for (auto& kv : sources) {
    mp::Map sm;
    sm["table"] = kv.second.table;
    sm["column"] = kv.second.column;
    if (!kv.second.source.empty()) {
      sm["source"] = kv.second.source;
      sm["args"] = kv.second.args;
    }
    srcs[std::to_string(kv.first)] = mp::Value(std::move(sm));
  }
  m["sources"] = std::move(srcs);
  mp::Map smp;
  for (auto& kv : sampling)
    smp[std::to_string(kv.first)] = kv.second.to_msgpack();
  m["sampling"] = std::move(smp);
  mp::Map oa;
  for (auto& kv : op_args) oa[std::to_string(kv.first)] = kv.second;
  m["op_args"] = std::move(oa);
  m["sink_table"] = sink.table;
  if (!sink.sink.empty()) {
    m["sink_name"] = sink.sink;
    m["sink_args"] = sink.args;
  }
  return mp::Value(std::move(m));
}

JobBinding JobBinding::from_msgpack(const mp::Value& v) {
  JobBinding b;
  for (auto& kv : v.as_map().at("sources").as_map()) {
    SourceArgsC s;
    s.table = kv.second.get_str("table", "");
    s.column = kv.second.get_str("column", "");
    s.source = kv.second.get_str("source", "");
    auto& sm = kv.second.as_map();
    if (sm.count("args")) s.args = sm.at("args").as_bin();
    b.sources[std::stoi(kv.first)] = s;
  }
  for (auto& kv : v.as_map().at("sampling").as_map()) {
    b.sampling[std::stoi(kv.first)] = SamplingArgs::from_msgpack(kv.second);
  }
  for (auto& kv : v.as_map().at("op_args").as_map()) {
    b.op_args[std::stoi(kv.first)] = kv.second.as_bin();
  }
  b.sink.table = v.get_str("sink_table", "");
  b.sink.sink = v.get_str("sink_name", "");
  if (v.as_map().count("sink_args")) {
    b.sink.args = v.as_map().at("sink_args").as_bin();
  }
  return b;
}

// ---------------- static info resolution ----------------

namespace {

OpStaticInfo resolve_static(const JobGraph& graph, i32 op_idx) {
  const OpNode& op = graph.ops[op_idx];
  OpStaticInfo si;
  if (is_builtin_op(op.name)) {
    si.is_builtin = true;
    if (is_input_op(op.name)) {
      auto args = mp::decode(op.args);
      bool is_frame =
          args.has("is_frame") ? args.as_map().at("is_frame").as_bool() : false;
      std::string col = args.get_str("column", is_frame ? "frame" : "col");
      si.output_columns = {col};
      si.output_types = {is_frame ? ColumnType::Video : ColumnType::Bytes};
    } else if (!is_output_op(op.name)) {
      // Sample/Space/Slice/Unslice: pass-through single column.
      SCA_CHECK(op.inputs.size() == 1,
                "builtin op '" + op.name + "' takes exactly one input");
      si.output_columns = {op.inputs[0].column};
      bool frame = op.name.find("Frame") != std::string::npos;
      si.output_types = {frame ? ColumnType::Video : ColumnType::Bytes};
    }
    return si;
  }
  const OpInfo& info = op_registry().get(op.name);
  si.stencil = op.stencil.empty() ? info.stencil : op.stencil;
  std::sort(si.stencil.begin(), si.stencil.end());
  si.bounded_state = info.has_bounded_state;
  si.unbounded_state = info.has_unbounded_state;
  si.warmup = op.warmup >= 0 ? op.warmup : info.warmup;
  const KernelFactory& kf = kernel_registry().get(op.name, op.device);
  si.batch = op.batch > 0 ? op.batch : kf.preferred_batch;
  for (auto& c : info.output_columns) {
    si.output_columns.push_back(c.name);
    si.output_types.push_back(c.type);
  }
  return si;
}

}  // namespace

void validate_graph(const JobGraph& graph) {
  SCA_CHECK(!graph.ops.empty(), "empty graph");
  SCA_CHECK(is_output_op(graph.ops.back().name), "last op must be Output");
  i32 n_out = 0;
  for (size_t i = 0; i < graph.ops.size(); ++i) {
    const OpNode& op = graph.ops[i];
    if (is_output_op(op.name)) n_out++;
    for (auto& e : op.inputs) {
      SCA_CHECK(e.op >= 0 && e.op < (i32)i,
                "op '" + op.name + "' input references op " +
                    std::to_string(e.op) + " (not topologically earlier)");
    }
    if (is_input_op(op.name)) {
      SCA_CHECK(op.inputs.empty(), "Input op takes no inputs");
    } else {
      SCA_CHECK(!op.inputs.empty(), "op '" + op.name + "' has no inputs");
    }
    if (!is_builtin_op(op.name)) {
      SCA_CHECK(op_registry().has(op.name), "unregistered op '" + op.name + "'");
      const OpInfo& info = op_registry().get(op.name);
      SCA_CHECK(kernel_registry().has(op.name, op.device),
                "no kernel for op '" + op.name + "' on requested device");
      if (!info.variadic_inputs) {
        SCA_CHECK(op.inputs.size() == info.input_columns.size(),
                  "op '" + op.name + "' wants " +
                      std::to_string(info.input_columns.size()) + " inputs, got " +
                      std::to_string(op.inputs.size()));
      }
    }
  }
  SCA_CHECK(n_out == 1, "graph must have exactly one Output op");
  // Validate column references + slice levels.
  std::vector<i32> level(graph.ops.size(), 0);
  for (size_t i = 0; i < graph.ops.size(); ++i) {
    const OpNode& op = graph.ops[i];
    if (op.inputs.empty()) continue;
    i32 l0 = level[op.inputs[0].op];
    for (auto& e : op.inputs) {
      SCA_CHECK(level[e.op] == level[op.inputs[0].op],
                "op '" + op.name + "' mixes inputs at different slice levels");
    }
    level[i] = l0;
    if (is_slice_op(op.name)) level[i]++;
    if (is_unslice_op(op.name)) {
      SCA_CHECK(level[i] > 0, "Unslice without matching Slice");
      level[i]--;
    }
    if (is_slice_op(op.name)) {
      SCA_CHECK(level[i] == 1, "nested Slice is not supported");
    }
  }
  SCA_CHECK(level[graph.ops.size() - 1] == 0,
            "graph output is still inside a Slice (missing Unslice)");
}

JobAnalysis analyze_job(const JobGraph& graph, const JobBinding& binding,
                        const SourceRowsFn& source_rows) {
  JobAnalysis ja;
  size_t n = graph.ops.size();
  ja.domains.resize(n);
  ja.info.resize(n);
  for (size_t i = 0; i < n; ++i) ja.info[i] = resolve_static(graph, (i32)i);

  // Column-consumer counts for liveness.
  for (size_t i = 0; i < n; ++i) {
    for (auto& e : graph.ops[i].inputs) {
      ja.info[e.op].column_consumers[e.column]++;
    }
  }

  for (size_t i = 0; i < n; ++i) {
    const OpNode& op = graph.ops[i];
    OpDomain& d = ja.domains[i];
    if (is_input_op(op.name)) {
      auto it = binding.sources.find((i32)i);
      SCA_CHECK(it != binding.sources.end(),
                "job missing source binding for Input op " + std::to_string(i));
      d.num_rows = source_rows(it->second);
      d.group_starts = {0, d.num_rows};
      continue;
    }
    const OpDomain& p = ja.domains[op.inputs[0].op];
    for (auto& e : op.inputs) {
      SCA_CHECK(ja.domains[e.op].num_rows == p.num_rows,
                "op '" + op.name + "' inputs have mismatched row counts (" +
                    std::to_string(ja.domains[e.op].num_rows) + " vs " +
                    std::to_string(p.num_rows) + ")");
    }
    d.slice_level = p.slice_level;
    if (is_output_op(op.name)) {
      d.num_rows = p.num_rows;
      d.group_starts = p.group_starts;
    } else if (is_sample_op(op.name)) {
      auto it = binding.sampling.find((i32)i);
      SCA_CHECK(it != binding.sampling.end(),
                "job missing sampling args for op " + std::to_string(i));
      // Group-aware: sample each slice group independently (level 0 is one
      // group). Per-group args supported via kind=="PerGroup".
      size_t ngroups = p.group_starts.size() - 1;
      d.group_starts = {0};
      for (size_t g = 0; g < ngroups; ++g) {
        i64 up = p.group_starts[g + 1] - p.group_starts[g];
        auto sampler = make_domain_sampler(it->second.for_group(g));
        d.group_starts.push_back(d.group_starts.back() +
                                 sampler->num_downstream(up));
      }
      d.num_rows = d.group_starts.back();
      if (d.slice_level == 0) d.group_starts = {0, d.num_rows};
    } else if (is_slice_op(op.name)) {
      auto it = binding.sampling.find((i32)i);
      SCA_CHECK(it != binding.sampling.end(),
                "job missing partitioner args for Slice op " + std::to_string(i));
      auto part = make_partitioner(it->second);
      i64 ng = part->num_groups(p.num_rows);
      d.slice_level = p.slice_level + 1;
      d.group_starts = {0};
      for (i64 g = 0; g < ng; ++g) {
        i64 sz = part->group_size(g, p.num_rows);
        SCA_CHECK(sz >= 0, "negative slice group size");
        d.slice_group_offsets.push_back(part->group_offset(g, p.num_rows));
        d.group_starts.push_back(d.group_starts.back() + sz);
      }
      d.num_rows = d.group_starts.back();
    } else if (is_unslice_op(op.name)) {
      d.slice_level = p.slice_level - 1;
      d.num_rows = p.num_rows;
      d.group_starts = {0, d.num_rows};
    } else {
      // kernel op: 1:1, inherits group structure
      d.num_rows = p.num_rows;
      d.group_starts = p.group_starts;
    }
  }
  ja.output_rows = ja.domains[n - 1].num_rows;
  return ja;
}

namespace {

inline void sort_unique(std::vector<i64>& v) {
  std::sort(v.begin(), v.end());
  v.erase(std::unique(v.begin(), v.end()), v.end());
}

// Group index containing row r given boundary vector.
inline size_t group_of(const std::vector<i64>& starts, i64 r) {
  auto it = std::upper_bound(starts.begin(), starts.end(), r);
  SCA_CHECK(it != starts.begin(), "row before first group");
  return (size_t)(it - starts.begin()) - 1;
}

}  // namespace

TaskPlan derive_task_plan(const JobGraph& graph, const JobAnalysis& ja,
                          const JobBinding& binding, i64 task_start,
                          i64 task_end) {
  size_t n = graph.ops.size();
  TaskPlan plan;
  plan.ops.resize(n);
  std::vector<std::vector<i64>> required(n);

  // Seed at the sink.
  for (i64 r = task_start; r < task_end; ++r)
    required[n - 1].push_back(r);

  for (i64 i = (i64)n - 1; i >= 0; --i) {
    const OpNode& op = graph.ops[i];
    const OpStaticInfo& si = ja.info[i];
    const OpDomain& dom = ja.domains[i];
    OpTaskPlan& otp = plan.ops[i];
    sort_unique(required[i]);
    otp.required_rows = required[i];
    if (otp.required_rows.empty()) continue;
    SCA_CHECK(otp.required_rows.front() >= 0 &&
                  otp.required_rows.back() < dom.num_rows,
              "required rows out of op domain for op '" + op.name + "'");

    if (is_input_op(op.name)) {
      plan.load_rows[(i32)i] = otp.required_rows;
      continue;
    }

    if (is_output_op(op.name)) {
      for (auto& e : op.inputs) {
        for (i64 r : otp.required_rows) required[e.op].push_back(r);
      }
      continue;
    }

    if (is_sample_op(op.name)) {
      const SamplingArgs& sargs = binding.sampling.at((i32)i);
      const OpDomain& pdom = ja.domains[op.inputs[0].op];
      for (i64 r : otp.required_rows) {
        size_t g = group_of(dom.group_starts, r);
        i64 local = r - dom.group_starts[g];
        auto sampler = make_domain_sampler(sargs.for_group(g));
        i64 up_local = sampler->upstream_row(local);
        i64 up = up_local < 0 ? -1 : pdom.group_starts[g] + up_local;
        otp.remap.push_back(up);
        if (up >= 0) required[op.inputs[0].op].push_back(up);
      }
      continue;
    }

    if (is_slice_op(op.name)) {
      for (i64 r : otp.required_rows) {
        size_t g = group_of(dom.group_starts, r);
        i64 local = r - dom.group_starts[g];
        i64 up = dom.slice_group_offsets[g] + local;
        otp.remap.push_back(up);
        required[op.inputs[0].op].push_back(up);
      }
      continue;
    }

    if (is_unslice_op(op.name)) {
      // Concatenated-group domain == flattened domain for contiguous
      // non-overlapping partitions (validated by the Python client).
      for (i64 r : otp.required_rows) {
        otp.remap.push_back(r);
        required[op.inputs[0].op].push_back(r);
      }
      continue;
    }

    // ---- kernel op ----
    // 1) compute rows: required plus state warmup, contiguous per group for
    //    stateful ops.
    if (si.bounded_state || si.unbounded_state) {
      // Partition required rows by slice group; compute contiguous spans.
      size_t idx = 0;
      while (idx < otp.required_rows.size()) {
        i64 first = otp.required_rows[idx];
        size_t g = group_of(dom.group_starts, first);
        i64 ge = dom.group_starts[g + 1];
        size_t j = idx;
        i64 last = first;
        while (j < otp.required_rows.size() && otp.required_rows[j] < ge) {
          last = otp.required_rows[j];
          ++j;
        }
        i64 gs = dom.group_starts[g];
        i64 lo = si.unbounded_state ? gs
                                    : std::max(gs, first - (i64)si.warmup);
        for (i64 r = lo; r <= last; ++r) otp.compute_rows.push_back(r);
        idx = j;
      }
    } else {
      otp.compute_rows = otp.required_rows;
    }

    // 2) stencil windows, clamped to group bounds; reset markers at group
    //    starts.
    size_t cur_group = SIZE_MAX;
    for (i64 c : otp.compute_rows) {
      size_t g = group_of(dom.group_starts, c);
      otp.reset_before.push_back(g != cur_group ? 1 : 0);
      cur_group = g;
      i64 gs = dom.group_starts[g];
      i64 ge = dom.group_starts[g + 1];
      std::vector<i64> win;
      win.reserve(si.stencil.size());
      for (i32 s : si.stencil) {
        i64 r = std::min(std::max(c + (i64)s, gs), ge - 1);  // REPEAT_EDGE
        win.push_back(r);
      }
      otp.windows.push_back(std::move(win));
    }

    // 3) propagate the union of windows to every parent edge.
    std::vector<i64> need;
    for (auto& w : otp.windows) need.insert(need.end(), w.begin(), w.end());
    sort_unique(need);
    for (auto& e : op.inputs) {
      required[e.op].insert(required[e.op].end(), need.begin(), need.end());
    }
  }

  return plan;
}

}  // namespace sca
