// Minimal msgpack subset codec (nil/bool/int/float64/str/bin/array/map).
// Op arguments travel Python<->C++ as msgpack bytes (the Python side uses
// the installed msgpack wheel); the reference used protobuf op args, which
// this image cannot compile (no protoc).
#pragma once

#include <map>
#include <memory>
#include <variant>

#include "common.h"

namespace sca {
namespace mp {

struct Value;
using Array = std::vector<Value>;
using Map = std::map<std::string, Value>;

struct Value {
  std::variant<std::monostate, bool, i64, f64, std::string, std::vector<u8>,
               Array, Map>
      v;

  Value() = default;
  Value(bool b) : v(b) {}
  Value(i64 i) : v(i) {}
  Value(int i) : v((i64)i) {}
  Value(f64 d) : v(d) {}
  Value(const char* s) : v(std::string(s)) {}
  Value(std::string s) : v(std::move(s)) {}
  Value(std::vector<u8> b) : v(std::move(b)) {}
  Value(Array a) : v(std::move(a)) {}
  Value(Map m) : v(std::move(m)) {}

  bool is_nil() const { return std::holds_alternative<std::monostate>(v); }
  bool as_bool() const { return std::get<bool>(v); }
  i64 as_int() const {
    if (std::holds_alternative<f64>(v)) return (i64)std::get<f64>(v);
    return std::get<i64>(v);
  }
  f64 as_float() const {
    if (std::holds_alternative<i64>(v)) return (f64)std::get<i64>(v);
    return std::get<f64>(v);
  }
  const std::string& as_str() const { return std::get<std::string>(v); }
  const std::vector<u8>& as_bin() const { return std::get<std::vector<u8>>(v); }
  const Array& as_array() const { return std::get<Array>(v); }
  const Map& as_map() const { return std::get<Map>(v); }

  // Map convenience with defaults.
  i64 get_int(const std::string& k, i64 dflt) const {
    auto& m = as_map();
    auto it = m.find(k);
    return it == m.end() || it->second.is_nil() ? dflt : it->second.as_int();
  }
  f64 get_float(const std::string& k, f64 dflt) const {
    auto& m = as_map();
    auto it = m.find(k);
    return it == m.end() || it->second.is_nil() ? dflt : it->second.as_float();
  }
  std::string get_str(const std::string& k, const std::string& dflt) const {
    auto& m = as_map();
    auto it = m.find(k);
    return it == m.end() || it->second.is_nil() ? dflt : it->second.as_str();
  }
  bool has(const std::string& k) const {
    auto& m = as_map();
    auto it = m.find(k);
    return it != m.end() && !it->second.is_nil();
  }
  std::vector<i64> get_int_vec(const std::string& k) const {
    std::vector<i64> out;
    if (!has(k)) return out;
    for (auto& e : as_map().at(k).as_array()) out.push_back(e.as_int());
    return out;
  }
};

// ---------------- encode ----------------

inline void encode_into(const Value& val, std::vector<u8>& out);

inline void put(std::vector<u8>& o, u8 b) { o.push_back(b); }
inline void put_be(std::vector<u8>& o, u64 v, int n) {
  for (int i = n - 1; i >= 0; --i) o.push_back((u8)((v >> (8 * i)) & 0xff));
}

inline void encode_into(const Value& val, std::vector<u8>& out) {
  struct V {
    std::vector<u8>& o;
    void operator()(std::monostate) { put(o, 0xc0); }
    void operator()(bool b) { put(o, b ? 0xc3 : 0xc2); }
    void operator()(i64 i) {
      if (i >= 0) {
        if (i < 128) put(o, (u8)i);
        else if (i <= 0xff) { put(o, 0xcc); put(o, (u8)i); }
        else if (i <= 0xffff) { put(o, 0xcd); put_be(o, i, 2); }
        else if (i <= 0xffffffffLL) { put(o, 0xce); put_be(o, i, 4); }
        else { put(o, 0xcf); put_be(o, i, 8); }
      } else {
        if (i >= -32) put(o, (u8)(0xe0 | (i + 32)));
        else if (i >= -128) { put(o, 0xd0); put(o, (u8)i); }
        else if (i >= -32768) { put(o, 0xd1); put_be(o, (u16)i, 2); }
        else if (i >= -2147483648LL) { put(o, 0xd2); put_be(o, (u32)i, 4); }
        else { put(o, 0xd3); put_be(o, (u64)i, 8); }
      }
    }
    void operator()(f64 d) {
      put(o, 0xcb);
      u64 bits;
      std::memcpy(&bits, &d, 8);
      put_be(o, bits, 8);
    }
    void operator()(const std::string& s) {
      size_t n = s.size();
      if (n < 32) put(o, (u8)(0xa0 | n));
      else if (n <= 0xff) { put(o, 0xd9); put(o, (u8)n); }
      else if (n <= 0xffff) { put(o, 0xda); put_be(o, n, 2); }
      else { put(o, 0xdb); put_be(o, n, 4); }
      o.insert(o.end(), s.begin(), s.end());
    }
    void operator()(const std::vector<u8>& b) {
      size_t n = b.size();
      if (n <= 0xff) { put(o, 0xc4); put(o, (u8)n); }
      else if (n <= 0xffff) { put(o, 0xc5); put_be(o, n, 2); }
      else { put(o, 0xc6); put_be(o, n, 4); }
      o.insert(o.end(), b.begin(), b.end());
    }
    void operator()(const Array& a) {
      size_t n = a.size();
      if (n < 16) put(o, (u8)(0x90 | n));
      else if (n <= 0xffff) { put(o, 0xdc); put_be(o, n, 2); }
      else { put(o, 0xdd); put_be(o, n, 4); }
      for (auto& e : a) encode_into(e, o);
    }
    void operator()(const Map& m) {
      size_t n = m.size();
      if (n < 16) put(o, (u8)(0x80 | n));
      else if (n <= 0xffff) { put(o, 0xde); put_be(o, n, 2); }
      else { put(o, 0xdf); put_be(o, n, 4); }
      for (auto& kv : m) {
        encode_into(Value(kv.first), o);
        encode_into(kv.second, o);
      }
    }
  } vis{out};
  std::visit(vis, val.v);
}

inline std::vector<u8> encode(const Value& v) {
  std::vector<u8> out;
  encode_into(v, out);
  return out;
}

// ---------------- decode ----------------

class Decoder {
 public:
  Decoder(const u8* p, size_t n) : p_(p), end_(p + n) {}
  Value decode() {
    u8 t = next();
    if (t < 0x80) return Value((i64)t);
    if (t >= 0xe0) return Value((i64)(i8)t);
    if ((t & 0xf0) == 0x80) return map(t & 0x0f);
    if ((t & 0xf0) == 0x90) return array(t & 0x0f);
    if ((t & 0xe0) == 0xa0) return str(t & 0x1f);
    switch (t) {
      case 0xc0: return Value();
      case 0xc2: return Value(false);
      case 0xc3: return Value(true);
      case 0xc4: return bin(be(1));
      case 0xc5: return bin(be(2));
      case 0xc6: return bin(be(4));
      case 0xca: {  // float32
        u32 bits = (u32)be(4);
        float f;
        std::memcpy(&f, &bits, 4);
        return Value((f64)f);
      }
      case 0xcb: {
        u64 bits = be(8);
        f64 d;
        std::memcpy(&d, &bits, 8);
        return Value(d);
      }
      case 0xcc: return Value((i64)be(1));
      case 0xcd: return Value((i64)be(2));
      case 0xce: return Value((i64)be(4));
      case 0xcf: return Value((i64)be(8));
      case 0xd0: return Value((i64)(i8)be(1));
      case 0xd1: return Value((i64)(i16)be(2));
      case 0xd2: return Value((i64)(i32)be(4));
      case 0xd3: return Value((i64)be(8));
      case 0xd9: return str(be(1));
      case 0xda: return str(be(2));
      case 0xdb: return str(be(4));
      case 0xdc: return array(be(2));
      case 0xdd: return array(be(4));
      case 0xde: return map(be(2));
      case 0xdf: return map(be(4));
    }
    throw ScannerError("msgpack: unsupported type byte " + std::to_string(t));
  }

 private:
  u8 next() {
    if (p_ >= end_) throw ScannerError("msgpack: truncated");
    return *p_++;
  }
  u64 be(int n) {
    u64 v = 0;
    for (int i = 0; i < n; ++i) v = (v << 8) | next();
    return v;
  }
  Value str(u64 n) {
    check(n);
    std::string s((const char*)p_, n);
    p_ += n;
    return Value(std::move(s));
  }
  Value bin(u64 n) {
    check(n);
    std::vector<u8> b(p_, p_ + n);
    p_ += n;
    return Value(std::move(b));
  }
  Value array(u64 n) {
    // every element takes >= 1 byte, so a count beyond the remaining
    // buffer is corrupt — reject BEFORE reserving (a trusted 4-byte
    // count let one bad message attempt a ~100 GB allocation; caught by
    // the ASan fuzz, tests/cpp/asan_parsers.cpp)
    check(n);
    Array a;
    a.reserve((size_t)n);
    for (u64 i = 0; i < n; ++i) a.push_back(decode());
    return Value(std::move(a));
  }
  Value map(u64 n) {
    check(n);  // >= 1 byte per entry minimum
    Map m;
    for (u64 i = 0; i < n; ++i) {
      Value k = decode();
      m[k.as_str()] = decode();
    }
    return Value(std::move(m));
  }
  void check(u64 n) {
    // compare against the remaining length — `p_ + n` could overflow the
    // pointer (UB) for hostile 64-bit counts
    if (n > (u64)(end_ - p_)) throw ScannerError("msgpack: truncated");
  }
  const u8* p_;
  const u8* end_;
};

inline Value decode(const std::vector<u8>& buf) {
  if (buf.empty()) return Value(Map{});
  return Decoder(buf.data(), buf.size()).decode();
}
inline Value decode(const u8* p, size_t n) {
  if (n == 0) return Value(Map{});
  return Decoder(p, n).decode();
}

}  // namespace mp
}  // namespace sca
