// Tiny versioned binary serialization for metadata records. The reference
// uses protobuf for db/table/video descriptors (metadata.proto); this image
// has no protoc/C++ protobuf, so we use an explicit little-endian
// writer/reader pair. Format: [magic u32][version u32][payload].
#pragma once

#include <cstring>
#include <map>
#include <string>
#include <unordered_map>
#include <vector>

#include "common.h"

namespace sca {

class BinWriter {
 public:
  void u32v(u32 v) { raw(&v, 4); }
  void u64v(u64 v) { raw(&v, 8); }
  void i32v(i32 v) { raw(&v, 4); }
  void i64v(i64 v) { raw(&v, 8); }
  void f64v(f64 v) { raw(&v, 8); }
  void b(bool v) { u8 x = v ? 1 : 0; raw(&x, 1); }
  void str(const std::string& s) {
    u64v(s.size());
    raw(s.data(), s.size());
  }
  void bytes(const std::vector<u8>& v) {
    u64v(v.size());
    raw(v.data(), v.size());
  }
  template <typename T>
  void vec_pod(const std::vector<T>& v) {
    static_assert(std::is_trivially_copyable<T>::value, "pod only");
    u64v(v.size());
    raw(v.data(), v.size() * sizeof(T));
  }
  void vec_str(const std::vector<std::string>& v) {
    u64v(v.size());
    for (auto& s : v) str(s);
  }
  const std::vector<u8>& data() const { return buf_; }
  std::vector<u8> take() { return std::move(buf_); }

 private:
  void raw(const void* p, size_t n) {
    const u8* b = static_cast<const u8*>(p);
    buf_.insert(buf_.end(), b, b + n);
  }
  std::vector<u8> buf_;
};

class BinReader {
 public:
  BinReader(const u8* data, size_t size) : p_(data), end_(data + size) {}
  explicit BinReader(const std::vector<u8>& v) : BinReader(v.data(), v.size()) {}

  u32 u32v() { return pod<u32>(); }
  u64 u64v() { return pod<u64>(); }
  i32 i32v() { return pod<i32>(); }
  i64 i64v() { return pod<i64>(); }
  f64 f64v() { return pod<f64>(); }
  bool b() { return pod<u8>() != 0; }
  std::string str() {
    u64 n = u64v();
    check(n);
    std::string s(reinterpret_cast<const char*>(p_), n);
    p_ += n;
    return s;
  }
  std::vector<u8> bytes() {
    u64 n = u64v();
    check(n);
    std::vector<u8> v(p_, p_ + n);
    p_ += n;
    return v;
  }
  template <typename T>
  std::vector<T> vec_pod() {
    u64 n = u64v();
    // divide, don't multiply: n * sizeof(T) can wrap for a corrupt
    // 64-bit count, passing the check and then allocating n elements
    if (n > remaining() / sizeof(T))
      throw ScannerError("binary deserialize out of bounds");
    std::vector<T> v((size_t)n);
    if (n) std::memcpy(v.data(), p_, (size_t)n * sizeof(T));
    p_ += (size_t)n * sizeof(T);
    return v;
  }
  std::vector<std::string> vec_str() {
    u64 n = u64v();
    // every string needs at least its 8-byte length; a count beyond the
    // remaining bytes is corrupt — reject before reserving (same
    // trusted-count DoS the msgpack decoder had; ASan fuzz corpus in
    // tests/cpp/asan_parsers.cpp)
    if (n > remaining() / 8)
      throw ScannerError("binary deserialize out of bounds");
    std::vector<std::string> v;
    v.reserve((size_t)n);
    for (u64 i = 0; i < n; ++i) v.push_back(str());
    return v;
  }
  bool at_end() const { return p_ == end_; }

 private:
  template <typename T>
  T pod() {
    check(sizeof(T));
    T v;
    std::memcpy(&v, p_, sizeof(T));
    p_ += sizeof(T);
    return v;
  }
  u64 remaining() const { return (u64)(end_ - p_); }
  void check(u64 n) {
    // compare against the remaining length — `p_ + n` could overflow the
    // pointer (UB) for hostile 64-bit counts
    if (n > remaining())
      throw ScannerError("binary deserialize out of bounds");
  }
  const u8* p_;
  const u8* end_;
};

}  // namespace sca
