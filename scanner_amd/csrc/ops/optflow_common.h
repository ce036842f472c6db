// Shared parameters of the dense optical-flow op (pyramidal Lucas-Kanade).
// The CPU kernel (stdlib_cpu.cpp) and the CDNA4 kernel (kernels/optflow.hip)
// implement the SAME algorithm with these constants so numerics tests can
// compare them directly.
//
// Capability parity: the reference's OpticalFlow test op wraps OpenCV
// Farneback (tests/test_ops.cpp:63-113, stencil [0,1], dense H x W x 2 f32
// flow output). We use pyramidal LK: same I/O contract, better fit for a
// hand-written data-parallel kernel.
#pragma once

namespace sca {
namespace optflow {

constexpr int kDefaultRadius = 3;   // window = (2R+1)^2 = 7x7
constexpr int kDefaultIters = 2;    // LK iterations per pyramid level
constexpr int kDefaultMaxLevels = 4;
constexpr float kDetEps = 1e-4f;    // singular-system guard

// Number of pyramid levels: halve until min dim < 24 or maxl reached.
inline int num_levels(int h, int w, int maxl) {
  int l = 1;
  while (l < maxl && (h >> l) >= 24 && (w >> l) >= 24) ++l;
  return l;
}

}  // namespace optflow
}  // namespace sca
