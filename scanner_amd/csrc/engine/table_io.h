// Column item readers/writers (capability parity: scanner/engine/
// column_source.cpp + column_sink.cpp). A table's rows are partitioned into
// items; each (column, item) pair is one data file plus one metadata file
// (element sizes). Video columns additionally carry a VideoMetadata record
// (geometry + codec + per-frame byte ranges + keyframe index).
#pragma once

#include "../element.h"
#include "../metadata.h"

namespace sca {

struct ItemRef {
  i32 item = 0;
  i64 row_start = 0;  // global row of first row in item
  i64 row_end = 0;
};

// Map sorted global rows onto items.
std::vector<ItemRef> items_for_rows(const TableMetadata& table,
                                    const std::vector<i64>& rows);

// Read the given (sorted, global) rows of a Bytes column (or a raw/encoded
// video column's packets) into CPU elements. Uses a dense whole-item read
// when the request covers >= 1/sparsity of the item's rows, else per-row
// range reads (reference: load_sparsity_threshold, column_source.cpp:303).
// Elements are allocated from the CPU block allocator; caller owns one ref
// each. element.index = global row.
ElementVector read_column_rows(Database& db, const TableMetadata& table,
                               const std::string& column,
                               const std::vector<i64>& rows,
                               i32 sparsity_threshold = 8);

// Read a video column's VideoMetadata for one item.
VideoMetadata read_video_metadata(Database& db, const TableMetadata& table,
                                  const std::string& column, i32 item);

// Write one item of a column. Elements must be CPU-resident. For video
// columns pass the VideoMetadata describing what was written.
void write_column_item(Database& db, const TableMetadata& table,
                       const std::string& column, i32 item,
                       const std::vector<Element>& elements);

void write_video_item(Database& db, const TableMetadata& table,
                      const std::string& column, i32 item,
                      const std::vector<u8>& stream,
                      const VideoMetadata& meta);

}  // namespace sca
