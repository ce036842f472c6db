#include "table_io.h"

#include <algorithm>

#include "../memory.h"
#include "../serialize.h"

namespace sca {

std::vector<ItemRef> items_for_rows(const TableMetadata& table,
                                    const std::vector<i64>& rows) {
  std::vector<ItemRef> out;
  if (rows.empty()) return out;
  const auto& er = table.end_rows;
  for (i64 r : rows) {
    auto it = std::upper_bound(er.begin(), er.end(), r);
    SCA_CHECK(it != er.end(), "row " + std::to_string(r) + " out of table");
    i32 item = (i32)(it - er.begin());
    i64 start = item == 0 ? 0 : er[item - 1];
    if (out.empty() || out.back().item != item) {
      out.push_back(ItemRef{item, start, er[item]});
    }
  }
  return out;
}

namespace {

// Item metadata file: element byte sizes.
std::vector<u64> read_item_sizes(Database& db, i32 table_id, i32 col_id,
                                 i32 item) {
  auto buf = db.storage()->read_all(
      db.paths().item_metadata(table_id, col_id, item));
  BinReader r(buf);
  return r.vec_pod<u64>();
}

}  // namespace

ElementVector read_column_rows(Database& db, const TableMetadata& table,
                               const std::string& column,
                               const std::vector<i64>& rows,
                               i32 sparsity_threshold) {
  ElementVector out;
  i32 col_id = table.column_id(column);
  auto items = items_for_rows(table, rows);
  size_t ri = 0;
  for (const ItemRef& ir : items) {
    // rows belonging to this item
    std::vector<i64> local;
    while (ri < rows.size() && rows[ri] < ir.row_end) {
      SCA_CHECK(rows[ri] >= ir.row_start, "rows not sorted");
      local.push_back(rows[ri] - ir.row_start);
      ++ri;
    }
    auto sizes = read_item_sizes(db, table.id, col_id, ir.item);
    std::vector<u64> offsets(sizes.size() + 1, 0);
    for (size_t i = 0; i < sizes.size(); ++i)
      offsets[i + 1] = offsets[i] + sizes[i];
    std::string path = db.paths().item(table.id, col_id, ir.item);

    i64 item_rows = ir.row_end - ir.row_start;
    bool dense = (i64)local.size() * sparsity_threshold >= item_rows;
    if (dense) {
      auto data = db.storage()->read_all(path);
      for (i64 lr : local) {
        SCA_CHECK((size_t)lr < sizes.size(), "row beyond item contents");
        size_t sz = sizes[lr];
        u8* buf = new_buffer(CPU_DEVICE, sz);
        std::memcpy(buf, data.data() + offsets[lr], sz);
        Element e;
        e.buffer = buf;
        e.size = sz;
        e.index = ir.row_start + lr;
        out.push_back(e);
      }
    } else {
      for (i64 lr : local) {
        SCA_CHECK((size_t)lr < sizes.size(), "row beyond item contents");
        size_t sz = sizes[lr];
        u8* buf = new_buffer(CPU_DEVICE, sz);
        db.storage()->read_range(path, offsets[lr], sz, buf);
        Element e;
        e.buffer = buf;
        e.size = sz;
        e.index = ir.row_start + lr;
        out.push_back(e);
      }
    }
  }
  SCA_CHECK(ri == rows.size(), "some rows not covered by table items");
  return out;
}

VideoMetadata read_video_metadata(Database& db, const TableMetadata& table,
                                  const std::string& column, i32 item) {
  i32 col_id = table.column_id(column);
  auto buf = db.storage()->read_all(
      db.paths().video_metadata(table.id, col_id, item));
  return VideoMetadata::deserialize(buf);
}

void write_column_item(Database& db, const TableMetadata& table,
                       const std::string& column, i32 item,
                       const std::vector<Element>& elements) {
  i32 col_id = table.column_id(column);
  std::vector<u64> sizes;
  size_t total = 0;
  for (auto& e : elements) {
    sizes.push_back(e.size);
    total += e.size;
  }
  std::vector<u8> data;
  data.reserve(total);
  for (auto& e : elements) {
    SCA_CHECK(!e.device.is_gpu(), "write_column_item needs CPU elements");
    data.insert(data.end(), e.buffer, e.buffer + e.size);
  }
  db.storage()->write_all(db.paths().item(table.id, col_id, item), data.data(),
                          data.size());
  BinWriter w;
  w.vec_pod(sizes);
  auto meta = w.take();
  db.storage()->write_all(db.paths().item_metadata(table.id, col_id, item),
                          meta.data(), meta.size());
}

void write_video_item(Database& db, const TableMetadata& table,
                      const std::string& column, i32 item,
                      const std::vector<u8>& stream, const VideoMetadata& meta) {
  i32 col_id = table.column_id(column);
  db.storage()->write_all(db.paths().item(table.id, col_id, item),
                          stream.data(), stream.size());
  // sample sizes double as the per-element size list
  BinWriter w;
  w.vec_pod(meta.sample_sizes);
  auto m = w.take();
  db.storage()->write_all(db.paths().item_metadata(table.id, col_id, item),
                          m.data(), m.size());
  auto v = meta.serialize();
  db.storage()->write_all(db.paths().video_metadata(table.id, col_id, item),
                          v.data(), v.size());
}

}  // namespace sca
