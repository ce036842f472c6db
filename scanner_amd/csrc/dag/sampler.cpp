#include "sampler.h"

#include <algorithm>

namespace sca {

SamplingArgs SamplingArgs::from_msgpack(const mp::Value& v) {
  SamplingArgs a;
  a.kind = v.get_str("kind", "All");
  a.stride = v.get_int("stride", 1);
  a.spacing = v.get_int("spacing", 1);
  a.starts = v.get_int_vec("starts");
  a.ends = v.get_int_vec("ends");
  a.rows = v.get_int_vec("rows");
  if (v.has("groups")) {
    for (auto& gv : v.as_map().at("groups").as_array()) {
      a.groups.push_back(SamplingArgs::from_msgpack(gv));
    }
  }
  return a;
}

mp::Value SamplingArgs::to_msgpack() const {
  mp::Map m;
  m["kind"] = kind;
  m["stride"] = stride;
  m["spacing"] = spacing;
  mp::Array s, e, r;
  for (i64 x : starts) s.push_back(mp::Value(x));
  for (i64 x : ends) e.push_back(mp::Value(x));
  for (i64 x : rows) r.push_back(mp::Value(x));
  m["starts"] = std::move(s);
  m["ends"] = std::move(e);
  m["rows"] = std::move(r);
  if (!groups.empty()) {
    mp::Array g;
    for (auto& ga : groups) g.push_back(ga.to_msgpack());
    m["groups"] = std::move(g);
  }
  return mp::Value(std::move(m));
}

namespace {

// All: identity (reference DefaultDomainSampler sampler.cpp:33)
struct AllSampler : DomainSampler {
  i64 num_downstream(i64 n) const override { return n; }
  i64 upstream_row(i64 d) const override { return d; }
};

// Strided over optional ranges (covers Strided / Range / StridedRange /
// StridedRanges; reference sampler.cpp:78,140)
struct StridedRangesSampler : DomainSampler {
  i64 stride;
  std::vector<i64> starts, ends;
  StridedRangesSampler(i64 s, std::vector<i64> st, std::vector<i64> en)
      : stride(std::max<i64>(1, s)), starts(std::move(st)), ends(std::move(en)) {}

  i64 rows_in_range(size_t i) const {
    return (ends[i] - starts[i] + stride - 1) / stride;
  }
  i64 num_downstream(i64 n) const override {
    if (starts.empty()) return (n + stride - 1) / stride;
    i64 total = 0;
    for (size_t i = 0; i < starts.size(); ++i) total += rows_in_range(i);
    return total;
  }
  i64 upstream_row(i64 d) const override {
    if (starts.empty()) return d * stride;
    for (size_t i = 0; i < starts.size(); ++i) {
      i64 in_range = rows_in_range(i);
      if (d < in_range) return starts[i] + d * stride;
      d -= in_range;
    }
    throw ScannerError("sampler: downstream row out of range");
  }
};

struct GatherSampler : DomainSampler {
  std::vector<i64> rows;
  explicit GatherSampler(std::vector<i64> r) : rows(std::move(r)) {}
  i64 num_downstream(i64) const override { return (i64)rows.size(); }
  i64 upstream_row(i64 d) const override {
    SCA_CHECK(d >= 0 && d < (i64)rows.size(), "gather row out of range");
    return rows[d];
  }
};

// SpaceNull: upstream row u lands at downstream u*spacing; gaps are null
// elements (reference sampler.cpp:337).
struct SpaceNullSampler : DomainSampler {
  i64 spacing;
  explicit SpaceNullSampler(i64 s) : spacing(std::max<i64>(1, s)) {}
  i64 num_downstream(i64 n) const override { return n * spacing; }
  i64 upstream_row(i64 d) const override {
    return d % spacing == 0 ? d / spacing : -1;
  }
};

// SpaceRepeat: each upstream row repeated `spacing` times (sampler.cpp:400).
struct SpaceRepeatSampler : DomainSampler {
  i64 spacing;
  explicit SpaceRepeatSampler(i64 s) : spacing(std::max<i64>(1, s)) {}
  i64 num_downstream(i64 n) const override { return n * spacing; }
  i64 upstream_row(i64 d) const override { return d / spacing; }
};

// ---- partitioners ----

struct AllPartitioner : Partitioner {
  // one group covering everything (used when Slice has "All" args with a
  // group size); here: single group
  i64 num_groups(i64) const override { return 1; }
  i64 group_offset(i64, i64) const override { return 0; }
  i64 group_size(i64, i64 n) const override { return n; }
};

// Fixed-size contiguous groups (reference StridedPartitioner with stride ==
// group size; "stride" arg = items per group).
struct StridedPartitioner : Partitioner {
  i64 group;
  explicit StridedPartitioner(i64 g) : group(std::max<i64>(1, g)) {}
  i64 num_groups(i64 n) const override { return (n + group - 1) / group; }
  i64 group_offset(i64 g_, i64) const override { return g_ * group; }
  i64 group_size(i64 g_, i64 n) const override {
    return std::min(group, n - g_ * group);
  }
};

// Explicit (possibly overlapping) ranges (reference StridedRangePartitioner).
struct RangesPartitioner : Partitioner {
  std::vector<i64> starts, ends;
  RangesPartitioner(std::vector<i64> s, std::vector<i64> e)
      : starts(std::move(s)), ends(std::move(e)) {}
  i64 num_groups(i64) const override { return (i64)starts.size(); }
  i64 group_offset(i64 g, i64) const override { return starts[g]; }
  i64 group_size(i64 g, i64) const override { return ends[g] - starts[g]; }
};

}  // namespace

std::unique_ptr<DomainSampler> make_domain_sampler(const SamplingArgs& a) {
  if (a.kind == "All") return std::make_unique<AllSampler>();
  if (a.kind == "Strided")
    return std::make_unique<StridedRangesSampler>(a.stride, a.starts, a.ends);
  if (a.kind == "Range" || a.kind == "StridedRange" || a.kind == "StridedRanges")
    return std::make_unique<StridedRangesSampler>(a.stride, a.starts, a.ends);
  if (a.kind == "Gather") return std::make_unique<GatherSampler>(a.rows);
  if (a.kind == "RepeatNull") return std::make_unique<SpaceNullSampler>(a.spacing);
  if (a.kind == "Repeat") return std::make_unique<SpaceRepeatSampler>(a.spacing);
  throw ScannerError("unknown sampler kind '" + a.kind + "'");
}

std::unique_ptr<Partitioner> make_partitioner(const SamplingArgs& a) {
  if (a.kind == "All") return std::make_unique<AllPartitioner>();
  if (a.kind == "Strided")
    return std::make_unique<StridedPartitioner>(a.stride);
  if (a.kind == "Range" || a.kind == "StridedRange" || a.kind == "StridedRanges" ||
      a.kind == "Ranges")
    return std::make_unique<RangesPartitioner>(a.starts, a.ends);
  throw ScannerError("unknown partitioner kind '" + a.kind + "'");
}

}  // namespace sca
