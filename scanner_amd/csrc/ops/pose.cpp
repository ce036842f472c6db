// Pose-estimation op: OpenPose-style two-branch multi-stage CNN
// (capability parity: the reference ecosystem's scannertools pose op /
// BASELINE.json config 5 "OpenPose-style multi-DNN pose-detection graph").
// bf16 inference on the hand-written MFMA GEMM (kernels/gemm_mfma.hip),
// implicit-GEMM for every conv with c % 8 == 0 (the 185-ch stage concat is
// zero-padded to 192 for exactly this reason), like ResNet-50
// (resnet50.cpp); weights are He-init random (no network in this
// environment) or loaded from a TNSR file.
//
// Topology (input resized to 368x368, feature stride 8 like OpenPose):
//   backbone: 8 x 3x3 convs (stride-2 at b2/b4/b6 — stride convs instead
//     of pools keep everything on the GEMM path) -> F: 46 x 46 x 128
//   stage 1 (per branch L=38 PAF ch, S=19 heatmap ch):
//     3x(3x3 128) + 1x1 128->512 + 1x1 512->out
//   stages 2..T (default T=3): input concat(F, L, S) = 185 ch:
//     3x(3x3 ->128) + 1x1 128->128 + 1x1 128->out
//   output: per-channel spatial argmax of the final heatmaps ->
//     19 keypoints {x, y, score} f32 per frame (228 B blob).
// The two branches are independent GEMM chains — a genuinely multi-DNN
// graph per frame, matching the reference pipeline's shape.
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

constexpr int kInHW = 368;
constexpr int kFeatHW = 46;  // 368 / 8
constexpr int kPafC = 38;    // L branch (part affinity fields)
constexpr int kHeatC = 19;   // S branch (keypoint heatmaps)

std::vector<ConvSpec> pose_specs(int stages) {
  std::vector<ConvSpec> sp;
  auto c3 = [&](const std::string& n, int ic, int oc, int stride) {
    sp.push_back({n, ic, oc, 3, 3, stride, 1, true});
  };
  // b1 consumes 8-channel input: preprocess zero-pads RGB so the entry
  // conv takes the implicit-GEMM path (no explicit im2col round trip);
  // 3-channel weight tensors are expanded at load.
  c3("b1", 8, 64, 1);
  c3("b2", 64, 64, 2);
  c3("b3", 64, 128, 1);
  c3("b4", 128, 128, 2);
  c3("b5", 128, 256, 1);
  c3("b6", 256, 256, 2);
  c3("b7", 256, 256, 1);
  c3("b8", 256, 128, 1);
  for (int t = 1; t <= stages; ++t) {
    for (const char* br : {"L", "S"}) {
      std::string p = "s" + std::to_string(t) + br;
      int out_c = br[0] == 'L' ? kPafC : kHeatC;
      // concat(F,L,S) = 185 ch, zero-padded to 192 so the stage convs run
      // on the implicit-GEMM path (c % 8 == 0)
      int in_c = t == 1 ? 128 : 192;
      c3(p + "_1", in_c, 128, 1);
      c3(p + "_2", 128, 128, 1);
      c3(p + "_3", 128, 128, 1);
      int mid = t == 1 ? 512 : 128;
      sp.push_back({p + "_4", 128, mid, 1, 1, 1, 0, true});
      sp.push_back({p + "_5", mid, out_c, 1, 1, 1, 0, false});
    }
  }
  return sp;
}

class PoseKernelGPU : public BatchedKernel {
 public:
  explicit PoseKernelGPU(const KernelConfig& cfg) : BatchedKernel(cfg) {
    auto a = mp::decode(cfg.args);
    weights_file_ = a.get_str("weights_file", "");
    seed_ = (u64)a.get_int("seed", 4321);
    stages_ = (int)a.get_int("stages", 3);
    SCA_CHECK(stages_ >= 1 && stages_ <= 6, "Pose stages must be 1..6");
    model_ = dnn::get_model("pose" + std::to_string(stages_), cfg.device,
                            weights_file_, seed_, [&]() {
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
                                  cfg.device, pose_specs(stages_),
                                  std::move(ts), seed_);
                            });
  }

  ~PoseKernelGPU() override {
    if (bufs_[0] && memory_initialized() && gen_ == memory_generation()) {
      for (u8* b : bufs_) delete_buffer(config_.device, b);
    }
  }

  void execute_batch(const BatchedElements& in, BatchedElements& out) override {
    void* s = per_thread_hip_stream();
    DeviceHandle dev = config_.device;
    int n = (int)in[0].size();
    if (n == 0) return;
    const Element& f0 = in[0][0];
    SCA_CHECK(f0.is_frame && f0.device.is_gpu(), "Pose needs GPU frames");
    int ih = f0.frame_info.shape[0], iw = f0.frame_info.shape[1],
        ic = f0.frame_info.shape[2];

    // Persistent workspace sized for max_batch (no per-execute pool
    // traffic). Peak activation: b1 output n x 368^2 x 64; peak im2col:
    // stage concat convs.
    int nb = std::max(n, std::max(1, config_.max_batch));
    size_t hw0 = (size_t)kInHW * kInHW;
    if (!bufs_[0]) {
      size_t fpix_b = (size_t)nb * kFeatHW * kFeatHW;
      bufs_[0] = new_buffer(dev, (size_t)nb * sizeof(u8*));       // d_ptrs
      bufs_[1] = new_buffer(dev, (size_t)nb * hw0 * 8 * 2);       // pre (c=8)
      bufs_[2] = new_buffer(dev, (size_t)nb * hw0 * 64 * 2);      // actA
      bufs_[3] = new_buffer(dev, (size_t)nb * hw0 * 64 * 2);      // actB
      bufs_[4] = new_buffer(dev, (size_t)nb * (hw0 / 4) * 640 * 2);
      bufs_[5] = new_buffer(dev, fpix_b * 128 * 2);               // feat
      bufs_[6] = new_buffer(dev, fpix_b * 64 * 2);                // brL
      bufs_[7] = new_buffer(dev, fpix_b * 64 * 2);                // brS
      bufs_[8] = new_buffer(dev, fpix_b * 512 * 2);               // brT
      bufs_[9] = new_buffer(dev, fpix_b * 512 * 2);               // brU
      bufs_[10] = new_buffer(dev, fpix_b * 192 * 2);              // cat
      gen_ = memory_generation();
    }
    u8* d_ptrs = bufs_[0];
    u8* pre = bufs_[1];
    u8* actA = bufs_[2];
    u8* actB = bufs_[3];
    u8* colbuf = bufs_[4];
    u8* feat = bufs_[5];
    u8* brL = bufs_[6];
    u8* brS = bufs_[7];
    u8* brT = bufs_[8];
    u8* brU = bufs_[9];
    u8* cat = bufs_[10];
    size_t featpix = (size_t)n * kFeatHW * kFeatHW;

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
      const u8* A = x;
      bool direct = sp.r == 1 && sp.s == 1 && sp.stride == 1 && sp.pad == 0;
      // Implicit GEMM when c % 8 == 0 (everything except b1's c=3): the
      // GEMM stages im2col rows straight from the activations.
      bool implicit = !direct && sp.in_c % 8 == 0;
      if (!direct && !implicit) {
        im2col_bf16(x, n, h, w, sp.in_c, sp.r, sp.s, sp.stride, sp.pad,
                    colbuf, oh, ow, sp.kp(), s);
        A = colbuf;
      }
      GemmArgs g;
      g.A = A;
      g.B = model_->weights + e.w_off * 2;
      g.C = y;
      g.M = n * oh * ow;
      g.N = sp.np();
      g.K = sp.kp();
      g.scale = (const float*)model_->scalebias + e.sb_off;
      g.bias = (const float*)model_->scalebias + e.sb_off + sp.np();
      g.relu = sp.relu;
      if (implicit) {
        ConvDesc d{n, h, w, sp.in_c, sp.r, sp.s, sp.stride, sp.pad, oh, ow};
        conv_gemm_bf16(g, d, s);
      } else {
        gemm_bf16(g, s);
      }
    };

    // ---- backbone ----
    f32* mean = model_->mean;
    preprocess_frames_bf16(d_ptrs, n, ih, iw, ic, kInHW, pre, mean, mean + 3,
                           s, /*out_c=*/8);
    int h = kInHW, w = kInHW, oh, ow;
    const char* bb[] = {"b1", "b2", "b3", "b4", "b5", "b6", "b7"};
    u8* x = pre;
    u8* bufs[2] = {actA, actB};
    int cur = 0;
    for (const char* name : bb) {
      conv(name, x, h, w, bufs[cur], oh, ow);
      x = bufs[cur];
      cur ^= 1;
      h = oh;
      w = ow;
    }
    conv("b8", x, h, w, feat, oh, ow);  // F: n x 46 x 46 x 128
    SCA_CHECK(oh == kFeatHW && ow == kFeatHW, "pose feature size mismatch");

    // ---- stages ----
    for (int t = 1; t <= stages_; ++t) {
      const u8* stage_in;
      int sh = kFeatHW, sw = kFeatHW;
      if (t == 1) {
        stage_in = feat;
      } else {
        concat3_bf16(feat, 128, 128, brL, kPafC, 64, brS, kHeatC, 64,
                     (i64)featpix, cat, 192, s);
        stage_in = cat;
      }
      for (const char* br : {"L", "S"}) {
        std::string p = "s" + std::to_string(t) + br;
        u8* outbuf = br[0] == 'L' ? brL : brS;
        conv(p + "_1", stage_in, sh, sw, brT, oh, ow);
        conv(p + "_2", brT, sh, sw, brU, oh, ow);
        conv(p + "_3", brU, sh, sw, brT, oh, ow);
        conv(p + "_4", brT, sh, sw, brU, oh, ow);
        conv(p + "_5", brU, sh, sw, outbuf, oh, ow);
      }
    }

    // ---- keypoints: argmax over the final heatmaps ----
    size_t out_bytes = (size_t)kHeatC * 3 * 4;
    u8* out_block = new_block_buffer(dev, (size_t)n * out_bytes, n);
    heatmap_argmax(brS, n, kFeatHW, kFeatHW, 64, kHeatC, out_block, s);

    sync_per_thread_stream();

    for (int i = 0; i < n; ++i) {
      Element e;
      e.buffer = out_block + (size_t)i * out_bytes;
      e.size = out_bytes;
      e.device = dev;
      out[0].push_back(e);
    }
  }

 private:
  std::string weights_file_;
  u64 seed_;
  int stages_;
  std::shared_ptr<DeviceModel> model_;
  u8* bufs_[11] = {};
  u64 gen_ = 0;
};

}  // namespace

void register_pose_op() {
  static bool done = false;
  if (done) return;
  done = true;
  OpInfo o;
  o.name = "Pose";
  o.input_columns = {{"frame", ColumnType::Video}};
  o.output_columns = {{"pose", ColumnType::Bytes}};
  op_registry().add(o);
  KernelFactory f;
  f.op_name = "Pose";
  f.device_type = DeviceType::GPU;
  f.preferred_batch = 16;
  f.make = [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
    return std::make_unique<PoseKernelGPU>(c);
  };
  kernel_registry().add(f);
}

}  // namespace sca
