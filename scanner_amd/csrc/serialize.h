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
    check(n * sizeof(T));
    std::vector<T> v(n);
    std::memcpy(v.data(), p_, n * sizeof(T));
    p_ += n * sizeof(T);
    return v;
  }
  std::vector<std::string> vec_str() {
    u64 n = u64v();
    std::vector<std::string> v;
    v.reserve(n);
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
  void check(u64 n) {
    if (p_ + n > end_) throw ScannerError("binary deserialize out of bounds");
  }
  const u8* p_;
  const u8* end_;
};

}  // namespace sca
