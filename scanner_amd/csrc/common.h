// Core types shared across the scanner_amd C++ engine.
//
// Role parity: scanner/engine/runtime.h + scanner/util/common.h in the
// reference (DeviceType/DeviceHandle/Result). Re-designed for MI355X: the
// only device type besides CPU is an AMD GPU addressed by HIP device id.
#pragma once

#include <cstdint>
#include <cstring>
#include <memory>
#include <stdexcept>
#include <string>
#include <vector>

namespace sca {

using u8 = uint8_t;
using u16 = uint16_t;
using u32 = uint32_t;
using u64 = uint64_t;
using i8 = int8_t;
using i16 = int16_t;
using i32 = int32_t;
using i64 = int64_t;
using f32 = float;
using f64 = double;

enum class DeviceType : i32 {
  CPU = 0,
  GPU = 1,  // MI355X via HIP
};

struct DeviceHandle {
  DeviceType type = DeviceType::CPU;
  i32 id = 0;

  bool operator==(const DeviceHandle& o) const {
    return type == o.type && id == o.id;
  }
  bool operator!=(const DeviceHandle& o) const { return !(*this == o); }
  bool is_gpu() const { return type == DeviceType::GPU; }
  std::string to_string() const {
    return (type == DeviceType::CPU ? std::string("CPU") : std::string("GPU")) +
           ":" + std::to_string(id);
  }
};

inline const DeviceHandle CPU_DEVICE{DeviceType::CPU, 0};

// Fatal-less error propagation (reference uses proto::Result; we use a plain
// struct since the control plane serializes with msgpack on the Python side).
struct Result {
  bool success = true;
  std::string msg;
  static Result Ok() { return Result{true, ""}; }
  static Result Err(std::string m) { return Result{false, std::move(m)}; }
};

class ScannerError : public std::runtime_error {
 public:
  explicit ScannerError(const std::string& m) : std::runtime_error(m) {}
};

#define SCA_CHECK(cond, msg)                                 \
  do {                                                       \
    if (!(cond)) {                                           \
      throw ::sca::ScannerError(std::string("check failed: ") + #cond + \
                                ": " + (msg));               \
    }                                                        \
  } while (0)

}  // namespace sca
