// ImageEncoder op: u8 frames -> PNG blobs (capability parity:
// scanner/util/image_encoder.cpp, which registers an ImageEncoder op
// backed by lodepng/OpenCV). Here: a from-scratch PNG writer over the
// system zlib (deflate of filter-0 scanlines), supporting gray (1ch) and
// RGB (3ch) u8 frames. CPU-only, like the reference.
#include <zlib.h>

#include <cstring>
#include <vector>

#include "../memory.h"
#include "../msgpack.h"
#include "kernel.h"

namespace sca {

namespace {

void put_be32(std::vector<u8>& v, u32 x) {
  v.push_back((u8)(x >> 24));
  v.push_back((u8)(x >> 16));
  v.push_back((u8)(x >> 8));
  v.push_back((u8)x);
}

void put_chunk(std::vector<u8>& png, const char type[4], const u8* data,
               size_t n) {
  put_be32(png, (u32)n);
  size_t type_off = png.size();
  png.insert(png.end(), type, type + 4);
  png.insert(png.end(), data, data + n);
  u32 crc = crc32(0L, Z_NULL, 0);
  crc = crc32(crc, png.data() + type_off, (uInt)(4 + n));
  put_be32(png, crc);
}

std::vector<u8> encode_png(const u8* pix, int h, int w, int c) {
  SCA_CHECK(c == 1 || c == 3, "PNG encoder supports 1- or 3-channel u8");
  // filter byte 0 before every scanline
  std::vector<u8> raw((size_t)h * (1 + (size_t)w * c));
  for (int y = 0; y < h; ++y) {
    u8* row = raw.data() + (size_t)y * (1 + (size_t)w * c);
    row[0] = 0;
    std::memcpy(row + 1, pix + (size_t)y * w * c, (size_t)w * c);
  }
  uLongf bound = compressBound((uLong)raw.size());
  std::vector<u8> comp(bound);
  int rc = compress2(comp.data(), &bound, raw.data(), (uLong)raw.size(), 6);
  SCA_CHECK(rc == Z_OK, "zlib compress failed in PNG encoder");
  comp.resize(bound);

  std::vector<u8> png = {0x89, 'P', 'N', 'G', 0x0d, 0x0a, 0x1a, 0x0a};
  u8 ihdr[13];
  ihdr[0] = (u8)(w >> 24);
  ihdr[1] = (u8)(w >> 16);
  ihdr[2] = (u8)(w >> 8);
  ihdr[3] = (u8)w;
  ihdr[4] = (u8)(h >> 24);
  ihdr[5] = (u8)(h >> 16);
  ihdr[6] = (u8)(h >> 8);
  ihdr[7] = (u8)h;
  ihdr[8] = 8;                       // bit depth
  ihdr[9] = c == 3 ? 2 : 0;          // color type: RGB / gray
  ihdr[10] = ihdr[11] = ihdr[12] = 0;  // deflate, filter 0, no interlace
  put_chunk(png, "IHDR", ihdr, 13);
  put_chunk(png, "IDAT", comp.data(), comp.size());
  put_chunk(png, "IEND", nullptr, 0);
  return png;
}

class ImageEncoderKernelCPU : public BatchedKernel {
 public:
  explicit ImageEncoderKernelCPU(const KernelConfig& cfg)
      : BatchedKernel(cfg) {
    auto a = mp::decode(cfg.args);
    format_ = a.get_str("format", "png");
    SCA_CHECK(format_ == "png",
              "ImageEncoder supports format=png (no OpenCV/libjpeg in "
              "this build)");
  }
  void execute_batch(const BatchedElements& in, BatchedElements& out) override {
    for (const Element& f : in[0]) {
      SCA_CHECK(f.is_frame && f.frame_info.type == FrameType::U8,
                "ImageEncoder needs u8 frames");
      auto png = encode_png(f.buffer, f.frame_info.shape[0],
                            f.frame_info.shape[1], f.frame_info.shape[2]);
      Element e;
      e.size = png.size();
      e.buffer = new_buffer(config_.device, e.size);
      e.device = config_.device;
      std::memcpy(e.buffer, png.data(), png.size());
      out[0].push_back(e);
    }
  }

 private:
  std::string format_;
};

}  // namespace

void register_image_encoder_op() {
  static bool done = false;
  if (done) return;
  done = true;
  OpInfo o;
  o.name = "ImageEncoder";
  o.input_columns = {{"frame", ColumnType::Video}};
  o.output_columns = {{"img", ColumnType::Bytes}};
  op_registry().add(o);
  KernelFactory f;
  f.op_name = "ImageEncoder";
  f.device_type = DeviceType::CPU;
  f.preferred_batch = 8;
  f.make = [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
    return std::make_unique<ImageEncoderKernelCPU>(c);
  };
  kernel_registry().add(f);
}

}  // namespace sca
