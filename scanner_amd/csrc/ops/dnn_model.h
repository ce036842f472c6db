// Shared host-side model plumbing for the DNN inference ops (ResNet-50,
// Pose). A model is a list of ConvSpecs; weights are He-init random (no
// network in this environment) or loaded from a TNSR tensor file for
// numerics tests. Device layout: one bf16 weight blob ([out][kp] per conv,
// concatenated, k padded to 64 for the MFMA GEMM) + one f32 scale/bias
// blob. Models are cached per (name, device, weights_file, seed) and shared
// across pipeline instances (reference analogue: fetch_resources +
// setup_with_resources barrier, evaluate_worker.cpp:493-550).
#pragma once

#include <cmath>
#include <cstring>
#include <map>
#include <memory>
#include <mutex>
#include <random>
#include <string>
#include <vector>

#include "../memory.h"
#include "../serialize.h"
#include "../storage.h"
#include "kernel.h"

namespace sca {
namespace dnn {

struct ConvSpec {
  std::string name;
  int in_c, out_c, r, s, stride, pad;
  bool relu;
  int kp() const { return (r * s * in_c + 63) / 64 * 64; }
  int np() const { return (out_c + 63) / 64 * 64; }
};

struct Tensors {
  std::map<std::string, std::vector<f32>> t;
  bool has(const std::string& n) const { return t.count(n); }
  std::vector<f32>& operator[](const std::string& n) { return t[n]; }
};

// Deterministic random weights: He-normal conv weights, BN folded to
// scale ~ U(0.7, 1.3), bias ~ N(0, 0.05).
inline void random_init(Tensors& ts, const ConvSpec& sp, std::mt19937& rng) {
  std::normal_distribution<f32> nd(
      0.f, std::sqrt(2.f / ((f32)sp.in_c * sp.r * sp.s)));
  auto& w = ts[sp.name + ".weight"];
  w.resize((size_t)sp.out_c * sp.r * sp.s * sp.in_c);
  for (auto& v : w) v = nd(rng);
  std::uniform_real_distribution<f32> su(0.7f, 1.3f);
  std::normal_distribution<f32> bn(0.f, 0.05f);
  auto& sc = ts[sp.name + ".scale"];
  auto& bi = ts[sp.name + ".bias"];
  sc.resize(sp.out_c);
  bi.resize(sp.out_c);
  for (auto& v : sc) v = su(rng);
  for (auto& v : bi) v = bn(rng);
}

inline Tensors load_tensor_file(const std::string& path) {
  auto storage = StorageBackend::make_posix();
  auto buf = storage->read_all(path);
  BinReader r(buf);
  u32 magic = r.u32v();
  SCA_CHECK(magic == 0x52534E54, "bad tensor file magic");  // 'TNSR'
  u32 n = r.u32v();
  Tensors ts;
  for (u32 i = 0; i < n; ++i) {
    std::string name = r.str();
    ts[name] = r.vec_pod<f32>();
  }
  return ts;
}

inline u16 f32_to_bf16_host(f32 v) {
  u32 bits;
  std::memcpy(&bits, &v, 4);
  u32 lsb = (bits >> 16) & 1;  // round-to-nearest-even
  bits += 0x7fff + lsb;
  return (u16)(bits >> 16);
}

struct DeviceModel {
  DeviceHandle dev;
  u8* weights = nullptr;    // bf16
  u8* scalebias = nullptr;  // f32: per conv scale then bias (padded np)
  struct Entry {
    ConvSpec spec;
    size_t w_off;   // bf16 elements
    size_t sb_off;  // f32 elements (scale at sb_off, bias at sb_off+np)
  };
  std::vector<Entry> convs;
  std::map<std::string, int> by_name;
  float* mean = nullptr;  // 3 floats mean + 3 std (imagenet normalization)
  u64 generation = 0;     // memory_generation() at build time
  ~DeviceModel() {
    // Free only into the allocator generation that produced these buffers
    // (skip at interpreter exit after teardown, and after a
    // destroy+re-init cycle where the pointers would be stale).
    if (!memory_initialized() || generation != memory_generation()) return;
    if (weights) delete_buffer(dev, weights);
    if (scalebias) delete_buffer(dev, scalebias);
    if (mean) delete_buffer(dev, (u8*)mean);
  }
};

// Build the device model from a spec list. `ts` may be pre-populated (from
// a tensor file); any conv without weights gets random init from `rng`.
inline std::shared_ptr<DeviceModel> build_device_model(
    DeviceHandle dev, const std::vector<ConvSpec>& specs, Tensors ts,
    u64 seed) {
  std::mt19937 rng((u32)seed);
  for (const auto& sp : specs) {
    if (!ts.has(sp.name + ".weight")) random_init(ts, sp, rng);
  }
  auto model = std::make_shared<DeviceModel>();
  model->dev = dev;
  model->generation = memory_generation();
  size_t w_elems = 0, sb_elems = 0;
  for (const auto& sp : specs) {
    model->by_name[sp.name] = (int)model->convs.size();
    model->convs.push_back({sp, w_elems, sb_elems});
    w_elems += (size_t)sp.np() * sp.kp();
    sb_elems += 2 * (size_t)sp.np();
  }
  std::vector<u16> wh(w_elems, 0);
  std::vector<f32> sbh(sb_elems, 0.f);
  for (auto& e : model->convs) {
    const ConvSpec& sp = e.spec;
    auto& w = ts[sp.name + ".weight"];
    SCA_CHECK((i64)w.size() == (i64)sp.out_c * sp.r * sp.s * sp.in_c,
              "weight size mismatch for " + sp.name);
    int krs = sp.r * sp.s * sp.in_c;
    // tensor layout: [out][r][s][in] — same k ordering as im2col
    for (int o = 0; o < sp.out_c; ++o) {
      for (int k = 0; k < krs; ++k) {
        wh[e.w_off + (size_t)o * sp.kp() + k] =
            f32_to_bf16_host(w[(size_t)o * krs + k]);
      }
    }
    auto& sc = ts[sp.name + ".scale"];
    auto& bi = ts[sp.name + ".bias"];
    SCA_CHECK((i64)sc.size() == sp.out_c && (i64)bi.size() == sp.out_c,
              "scale/bias size mismatch for " + sp.name);
    for (int o = 0; o < sp.out_c; ++o) {
      sbh[e.sb_off + o] = sc[o];
      sbh[e.sb_off + sp.np() + o] = bi[o];
    }
  }
  model->weights = new_buffer(dev, w_elems * 2);
  memcpy_buffer(model->weights, dev, (const u8*)wh.data(), CPU_DEVICE,
                w_elems * 2);
  model->scalebias = new_buffer(dev, sb_elems * 4);
  memcpy_buffer(model->scalebias, dev, (const u8*)sbh.data(), CPU_DEVICE,
                sb_elems * 4);
  f32 ms[6] = {0.485f, 0.456f, 0.406f, 0.229f, 0.224f, 0.225f};
  model->mean = (float*)new_buffer(dev, 6 * 4);
  memcpy_buffer((u8*)model->mean, dev, (const u8*)ms, CPU_DEVICE, 24);
  return model;
}

// Global cache keyed by (model name, device, weights file, seed). The
// cache registers a memory-teardown callback so destroy_memory_allocators
// frees the cached device models while the allocators still exist (and a
// later re-init rebuilds models instead of using stale slab pointers).
inline std::shared_ptr<DeviceModel> get_model(
    const std::string& model_name, DeviceHandle dev,
    const std::string& weights_file, u64 seed,
    const std::function<std::shared_ptr<DeviceModel>()>& build) {
  static std::mutex mu;
  static std::map<std::string, std::shared_ptr<DeviceModel>> cache;
  static bool registered = []() {
    register_memory_teardown_callback([]() {
      std::lock_guard<std::mutex> l(mu);
      cache.clear();
    });
    return true;
  }();
  (void)registered;
  std::string key = model_name + "|" + dev.to_string() + "|" + weights_file +
                    "|" + std::to_string(seed);
  std::lock_guard<std::mutex> l(mu);
  auto it = cache.find(key);
  if (it != cache.end()) return it->second;
  auto m = build();
  cache[key] = m;
  return m;
}

}  // namespace dnn
}  // namespace sca
