// Detector op: single-shot anchor-based object detector (capability
// parity: the reference ecosystem's scannertools detection ops — the third
// DNN family next to classification (resnet50.cpp) and pose (pose.cpp)).
// bf16 inference on the MFMA GEMM via the same implicit-conv path;
// random-init weights (no network in this environment) or a TNSR file.
//
// Topology (input resized to 320x320, feature stride 8):
//   backbone: 6 x 3x3 convs (stride 2 at b2/b4/b6) -> 40 x 40 x 128
//   heads: 3x3 conv -> per-anchor box deltas (A*4) and class scores (A*C),
//          A=4 anchor sizes, C=8 classes
// Post-process (host, per frame): sigmoid scores, decode center-form
// deltas against the anchor grid, greedy best-NMS (util/bbox.cpp parity,
// same semantics as scanner_amd.types.nms_best), emit a BoundingBoxList
// blob (u32 count + per box {x1,y1,x2,y2,score f32; label i32}) readable
// by scanner_amd.types.unpack_bboxes.
#include <algorithm>
#include <cmath>

#include "../../kernels/dnn.h"
#include "../memory.h"
#include "../msgpack.h"
#include "dnn_model.h"
#include "kernel.h"

namespace sca {

namespace {

using dnn::ConvSpec;
using dnn::DeviceModel;
using dnn::Tensors;

constexpr int kInHW = 320;
constexpr int kFeat = 40;   // 320 / 8
constexpr int kA = 4;       // anchors per cell
constexpr int kC = 8;       // classes
const float kAnchorSizes[kA] = {16.f, 32.f, 64.f, 128.f};

std::vector<ConvSpec> detector_specs() {
  std::vector<ConvSpec> sp;
  auto c3 = [&](const std::string& n, int ic, int oc, int stride,
                bool relu = true) {
    sp.push_back({n, ic, oc, 3, 3, stride, 1, relu});
  };
  // b1 consumes 8-channel zero-padded input (implicit-GEMM entry conv)
  c3("b1", 8, 64, 1);
  c3("b2", 64, 64, 2);
  c3("b3", 64, 128, 1);
  c3("b4", 128, 128, 2);
  c3("b5", 128, 128, 1);
  c3("b6", 128, 128, 2);
  c3("head", 128, 128, 1);
  c3("loc", 128, kA * 4, 1, /*relu=*/false);   // np -> 64
  c3("cls", 128, kA * kC, 1, /*relu=*/false);  // np -> 64
  return sp;
}

struct Box {
  float x1, y1, x2, y2, score;
  i32 label;
};

float iou(const Box& a, const Box& b) {
  float ix = std::max(0.f, std::min(a.x2, b.x2) - std::max(a.x1, b.x1));
  float iy = std::max(0.f, std::min(a.y2, b.y2) - std::max(a.y1, b.y1));
  float inter = ix * iy;
  float ua = (a.x2 - a.x1) * (a.y2 - a.y1) +
             (b.x2 - b.x1) * (b.y2 - b.y1) - inter;
  return ua > 1e-9f ? inter / ua : 0.f;
}

// Greedy best-first NMS (reference: bbox.cpp best_nms; same semantics as
// scanner_amd.types.nms_best).
std::vector<Box> nms_best(std::vector<Box> boxes, float thresh) {
  std::sort(boxes.begin(), boxes.end(),
            [](const Box& a, const Box& b) { return a.score > b.score; });
  std::vector<Box> keep;
  std::vector<bool> alive(boxes.size(), true);
  for (size_t i = 0; i < boxes.size(); ++i) {
    if (!alive[i]) continue;
    keep.push_back(boxes[i]);
    for (size_t j = i + 1; j < boxes.size(); ++j) {
      if (alive[j] && iou(boxes[i], boxes[j]) >= thresh) alive[j] = false;
    }
  }
  return keep;
}

class DetectorKernelGPU : public BatchedKernel {
 public:
  ~DetectorKernelGPU() override {
    if (bufs_[0] && memory_initialized() && gen_ == memory_generation()) {
      for (u8* b : bufs_) delete_buffer(config_.device, b);
    }
  }

  explicit DetectorKernelGPU(const KernelConfig& cfg) : BatchedKernel(cfg) {
    auto a = mp::decode(cfg.args);
    weights_file_ = a.get_str("weights_file", "");
    seed_ = (u64)a.get_int("seed", 777);
    score_thresh_ = 0.5f;
    nms_thresh_ = 0.5f;
    model_ = dnn::get_model("detector", cfg.device, weights_file_, seed_,
                            [&]() {
                              Tensors ts;
                              if (!weights_file_.empty())
                                ts = dnn::load_tensor_file(weights_file_);
                              if (ts.has("b1.weight")) {
                                auto& w = ts["b1.weight"];
                                if ((i64)w.size() == 64LL * 9 * 3) {
                                  std::vector<f32> e(64LL * 9 * 8, 0.f);
                                  for (i64 o = 0; o < 64 * 9; ++o)
                                    for (i64 c = 0; c < 3; ++c)
                                      e[o * 8 + c] = w[o * 3 + c];
                                  w = std::move(e);
                                }
                              }
                              return dnn::build_device_model(
                                  cfg.device, detector_specs(),
                                  std::move(ts), seed_);
                            });
  }

  void execute_batch(const BatchedElements& in, BatchedElements& out) override {
    void* s = per_thread_hip_stream();
    DeviceHandle dev = config_.device;
    int n = (int)in[0].size();
    if (n == 0) return;
    const Element& f0 = in[0][0];
    SCA_CHECK(f0.is_frame && f0.device.is_gpu(), "Detector needs GPU frames");
    int ih = f0.frame_info.shape[0], iw = f0.frame_info.shape[1],
        ic = f0.frame_info.shape[2];
    // Geometry is taken from frame 0 for the whole batch (preprocess letterbox
    // + box rescale), so every element must match it (same check
    // the OpticalFlow kernel does).
    for (int k = 1; k < n; ++k) {
      const Element& fk = in[0][k];
      SCA_CHECK(fk.is_frame && fk.frame_info.shape[0] == ih &&
                    fk.frame_info.shape[1] == iw &&
                    fk.frame_info.shape[2] == ic,
                "Detector batch has mixed frame geometries");
    }

    // Persistent workspace sized for max_batch.
    int nb = std::max(n, std::max(1, config_.max_batch));
    size_t hw0 = (size_t)kInHW * kInHW;
    if (!bufs_[0]) {
      size_t fpix_b = (size_t)nb * kFeat * kFeat;
      bufs_[0] = new_buffer(dev, (size_t)nb * sizeof(u8*));   // d_ptrs
      bufs_[1] = new_buffer(dev, (size_t)nb * hw0 * 8 * 2);   // pre (c=8)
      bufs_[2] = new_buffer(dev, (size_t)nb * hw0 * 64 * 2);  // actA
      bufs_[3] = new_buffer(dev, (size_t)nb * hw0 * 64 * 2);  // actB
      bufs_[4] = new_buffer(dev, (size_t)nb * hw0 * 64 * 2);  // colbuf (b1)
      bufs_[5] = new_buffer(dev, fpix_b * 64 * 2);            // loc
      bufs_[6] = new_buffer(dev, fpix_b * 64 * 2);            // cls
      bufs_[7] = new_buffer(dev, fpix_b * 64 * 4);            // loc_f
      bufs_[8] = new_buffer(dev, fpix_b * 64 * 4);            // cls_f
      gen_ = memory_generation();
    }
    u8* d_ptrs = bufs_[0];
    u8* pre = bufs_[1];
    u8* actA = bufs_[2];
    u8* actB = bufs_[3];
    u8* colbuf = bufs_[4];
    u8* loc = bufs_[5];
    u8* cls = bufs_[6];
    std::vector<const u8*> ptrs(n);
    for (int i = 0; i < n; ++i) ptrs[i] = in[0][i].buffer;
    memcpy_buffer(d_ptrs, dev, (const u8*)ptrs.data(), CPU_DEVICE,
                  n * sizeof(u8*));

    auto conv = [&](const std::string& name, const u8* x, int h, int w,
                    u8* y, int& oh, int& ow) {
      const auto& e = model_->convs[model_->by_name.at(name)];
      const ConvSpec& sp = e.spec;
      oh = (h + 2 * sp.pad - sp.r) / sp.stride + 1;
      ow = (w + 2 * sp.pad - sp.s) / sp.stride + 1;
      GemmArgs g;
      g.B = model_->weights + e.w_off * 2;
      g.C = y;
      g.M = n * oh * ow;
      g.N = sp.np();
      g.K = sp.kp();
      g.scale = (const float*)model_->scalebias + e.sb_off;
      g.bias = (const float*)model_->scalebias + e.sb_off + sp.np();
      g.relu = sp.relu;
      if (sp.in_c % 8 == 0) {
        g.A = x;
        ConvDesc d{n, h, w, sp.in_c, sp.r, sp.s, sp.stride, sp.pad, oh, ow};
        conv_gemm_bf16(g, d, s);
      } else {
        im2col_bf16(x, n, h, w, sp.in_c, sp.r, sp.s, sp.stride, sp.pad,
                    colbuf, oh, ow, sp.kp(), s);
        g.A = colbuf;
        gemm_bf16(g, s);
      }
    };

    f32* mean = model_->mean;
    preprocess_frames_bf16(d_ptrs, n, ih, iw, ic, kInHW, pre, mean, mean + 3,
                           s, /*out_c=*/8);
    int h = kInHW, w = kInHW, oh, ow;
    u8* x = pre;
    u8* bufs[2] = {actA, actB};
    int cur = 0;
    for (const char* name : {"b1", "b2", "b3", "b4", "b5", "b6", "head"}) {
      conv(name, x, h, w, bufs[cur], oh, ow);
      x = bufs[cur];
      cur ^= 1;
      h = oh;
      w = ow;
    }
    SCA_CHECK(h == kFeat && w == kFeat, "detector feature size mismatch");
    conv("loc", x, h, w, loc, oh, ow);
    conv("cls", x, h, w, cls, oh, ow);

    // Small maps: bring both heads to the host as f32 and post-process.
    i64 cells = (i64)n * kFeat * kFeat;
    u8* loc_f = bufs_[7];
    u8* cls_f = bufs_[8];
    bf16_rows_to_f32(loc, cells, 64, 64, loc_f, s);
    bf16_rows_to_f32(cls, cells, 64, 64, cls_f, s);
    std::vector<f32> loc_h(cells * 64), cls_h(cells * 64);
    sync_per_thread_stream();
    memcpy_buffer((u8*)loc_h.data(), CPU_DEVICE, loc_f, dev,
                  loc_h.size() * 4);
    memcpy_buffer((u8*)cls_h.data(), CPU_DEVICE, cls_f, dev,
                  cls_h.size() * 4);

    // Per frame: sigmoid scores, decode anchors, NMS, pack.
    float sx = (float)iw / kInHW, sy = (float)ih / kInHW;
    for (int f = 0; f < n; ++f) {
      std::vector<Box> cand;
      const f32* lf = loc_h.data() + (size_t)f * kFeat * kFeat * 64;
      const f32* cf = cls_h.data() + (size_t)f * kFeat * kFeat * 64;
      for (int cell = 0; cell < kFeat * kFeat; ++cell) {
        float cxg = (cell % kFeat + 0.5f) * 8.f;
        float cyg = (cell / kFeat + 0.5f) * 8.f;
        for (int a = 0; a < kA; ++a) {
          // best class for this anchor
          int best_c = 0;
          float best_s = -1e30f;
          for (int c = 0; c < kC; ++c) {
            float v = cf[(size_t)cell * 64 + a * kC + c];
            if (v > best_s) {
              best_s = v;
              best_c = c;
            }
          }
          float score = 1.f / (1.f + std::exp(-best_s));
          if (score < score_thresh_) continue;
          const f32* d = lf + (size_t)cell * 64 + a * 4;
          float aw = kAnchorSizes[a];
          float cx = cxg + std::tanh(d[0]) * aw;
          float cy = cyg + std::tanh(d[1]) * aw;
          float bw = aw * std::exp(std::min(2.f, d[2]));
          float bh = aw * std::exp(std::min(2.f, d[3]));
          Box b;
          b.x1 = std::max(0.f, (cx - bw / 2) * sx);
          b.y1 = std::max(0.f, (cy - bh / 2) * sy);
          b.x2 = std::min((float)iw, (cx + bw / 2) * sx);
          b.y2 = std::min((float)ih, (cy + bh / 2) * sy);
          if (b.x2 <= b.x1 || b.y2 <= b.y1) continue;
          b.score = score;
          b.label = best_c;
          cand.push_back(b);
        }
      }
      // standard SSD pre-NMS top-k: bounds the O(k^2) NMS on dense
      // candidate sets (random-init heads score half the anchor grid)
      constexpr size_t kPreNmsTopK = 1000;
      if (cand.size() > kPreNmsTopK) {
        std::partial_sort(cand.begin(), cand.begin() + kPreNmsTopK,
                          cand.end(), [](const Box& a, const Box& b) {
                            return a.score > b.score;
                          });
        cand.resize(kPreNmsTopK);
      }
      auto kept = nms_best(std::move(cand), nms_thresh_);
      // BoundingBoxList blob (types.py format: '<I' count + '<5fi' boxes)
      size_t blob = 4 + kept.size() * 24;
      Element e;
      e.size = blob;
      e.buffer = new_buffer(CPU_DEVICE, blob);
      e.device = CPU_DEVICE;
      u32 cnt = (u32)kept.size();
      std::memcpy(e.buffer, &cnt, 4);
      u8* p = e.buffer + 4;
      for (auto& b : kept) {
        std::memcpy(p, &b, 24);
        p += 24;
      }
      out[0].push_back(e);
    }
  }

 private:
  std::string weights_file_;
  u64 seed_;
  float score_thresh_, nms_thresh_;
  std::shared_ptr<DeviceModel> model_;
  u8* bufs_[9] = {};
  u64 gen_ = 0;
};

}  // namespace

void register_detector_op() {
  static bool done = false;
  if (done) return;
  done = true;
  OpInfo o;
  o.name = "Detector";
  o.input_columns = {{"frame", ColumnType::Video}};
  o.output_columns = {{"boxes", ColumnType::Bytes}};
  op_registry().add(o);
  KernelFactory f;
  f.op_name = "Detector";
  f.device_type = DeviceType::GPU;
  // box lists are assembled host-side after the D2H of the head maps
  f.output_device_type = (i32)DeviceType::CPU;
  f.preferred_batch = 16;
  f.make = [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
    return std::make_unique<DetectorKernelGPU>(c);
  };
  kernel_registry().add(f);
}

}  // namespace sca
