// HIP helpers for the MI355X engine. Unlike the reference's cuda.h (which
// compiles GPU support out via HAVE_CUDA), we always compile against HIP —
// the ROCm runtime is present on every box; GPU *presence* is a runtime
// question answered by hipGetDeviceCount.
#pragma once

#include <hip/hip_runtime.h>

#include "common.h"

namespace sca {

#define HIP_CHECK(expr)                                                   \
  do {                                                                    \
    hipError_t _e = (expr);                                               \
    if (_e != hipSuccess) {                                               \
      throw ::sca::ScannerError(std::string("HIP error: ") +              \
                                hipGetErrorString(_e) + " at " __FILE__   \
                                ":" + std::to_string(__LINE__) +          \
                                " in " #expr);                            \
    }                                                                     \
  } while (0)

inline int gpu_device_count() {
  int n = 0;
  hipError_t e = hipGetDeviceCount(&n);
  if (e != hipSuccess) return 0;
  return n;
}

inline bool have_gpu() { return gpu_device_count() > 0; }

// RAII device guard
class DeviceGuard {
 public:
  explicit DeviceGuard(i32 dev) {
    HIP_CHECK(hipGetDevice(&prev_));
    if (prev_ != dev) HIP_CHECK(hipSetDevice(dev));
    dev_ = dev;
  }
  ~DeviceGuard() {
    if (prev_ != dev_) (void)hipSetDevice(prev_);
  }

 private:
  int prev_ = 0, dev_ = 0;
};

}  // namespace sca
