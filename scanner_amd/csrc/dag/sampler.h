// Domain samplers and stream partitioners.
// Capability parity: scanner/engine/sampler.{h,cpp} — DomainSamplers remap a
// downstream row space to the upstream row space (All/Strided/StridedRanges/
// Gather/SpaceNull/SpaceRepeat); Partitioners split a stream into slice
// groups (All/Strided/StridedRanges/Gather).
#pragma once

#include <memory>

#include "../common.h"
#include "../msgpack.h"

namespace sca {

// Args for both samplers and partitioners (a tagged union; matches the
// Python streams DSL in scanner_amd/streams.py).
struct SamplingArgs {
  std::string kind = "All";  // All|Strided|Range|StridedRange|StridedRanges|
                             // Gather|RepeatNull|Repeat|PerGroup
  i64 stride = 1;
  i64 spacing = 1;              // for RepeatNull / Repeat
  std::vector<i64> starts, ends;  // for (Strided)Range(s)
  std::vector<i64> rows;          // for Gather
  // kind == "PerGroup": one sampler per slice group (reference: SliceList
  // per-slice sampling args, py_test.py:364-371).
  std::vector<SamplingArgs> groups;

  // Args for slice group g (per-group if present, else shared).
  const SamplingArgs& for_group(size_t g) const {
    if (groups.empty()) return *this;
    if (g >= groups.size())
      throw ScannerError("fewer per-group sampling args than slice groups");
    return groups[g];
  }

  static SamplingArgs from_msgpack(const mp::Value& v);
  mp::Value to_msgpack() const;
};

class DomainSampler {
 public:
  virtual ~DomainSampler() = default;
  // Number of downstream rows produced from num_upstream upstream rows.
  virtual i64 num_downstream(i64 num_upstream) const = 0;
  // The upstream row a downstream row reads, or -1 for a null element.
  virtual i64 upstream_row(i64 downstream_row) const = 0;
};

class Partitioner {
 public:
  virtual ~Partitioner() = default;
  virtual i64 num_groups(i64 num_upstream) const = 0;
  // [offset, offset+size) of group g in the upstream domain.
  virtual i64 group_offset(i64 g, i64 num_upstream) const = 0;
  virtual i64 group_size(i64 g, i64 num_upstream) const = 0;
};

std::unique_ptr<DomainSampler> make_domain_sampler(const SamplingArgs& args);
std::unique_ptr<Partitioner> make_partitioner(const SamplingArgs& args);

}  // namespace sca
