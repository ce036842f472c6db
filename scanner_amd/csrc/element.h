// Element/Frame data model (capability parity: scanner/api/kernel.h:28-66 +
// scanner/api/frame.h). An Element is one row of one column: either an
// opaque byte blob or a Frame (shaped, typed buffer). Frame geometry is kept
// inline in the element (no separate heap Frame object — simpler ownership
// than the reference's Frame*).
#pragma once

#include "common.h"
#include "metadata.h"  // FrameType

namespace sca {

struct FrameInfo {
  i32 shape[3] = {0, 0, 0};  // e.g. H, W, C
  FrameType type = FrameType::U8;

  i64 count() const {
    return (i64)shape[0] * (shape[1] ? shape[1] : 1) * (shape[2] ? shape[2] : 1);
  }
  size_t size() const { return count() * frame_type_size(type); }
  bool operator==(const FrameInfo& o) const {
    return shape[0] == o.shape[0] && shape[1] == o.shape[1] &&
           shape[2] == o.shape[2] && type == o.type;
  }
};

struct Element {
  u8* buffer = nullptr;
  size_t size = 0;
  DeviceHandle device;   // where buffer lives
  bool is_frame = false;
  FrameInfo frame_info;  // valid iff is_frame
  i64 index = 0;         // global row id this element corresponds to
  bool is_null = false;  // RepeatNull gap marker (reference: NullElement)
};

// One column's worth of rows.
using ElementVector = std::vector<Element>;
// Per-column rows.
using BatchedElements = std::vector<ElementVector>;
// Per-column, per-row, per-stencil-offset.
using StenciledElements = std::vector<std::vector<ElementVector>>;

}  // namespace sca
