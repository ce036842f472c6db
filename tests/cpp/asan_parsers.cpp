// AddressSanitizer/UBSan-verified mutation fuzz of the pure-parsing video
// layer (csrc/video/mp4.cpp, h264.cpp — no HIP, compiles with plain g++).
// The reference trusts ffmpeg/libav for all demux/parse (SURVEY section 2:
// scanner/video/*); this build's from-scratch ISO-BMFF walker and Annex-B
// /Exp-Golomb parsers must therefore carry their own memory-safety
// evidence: every iteration feeds a truncated/mutated valid file and the
// parser must either return or throw — any OOB read/write is an ASAN
// report, any crash a nonzero exit. tests/test_fuzz.py compiles this with
// -fsanitize=address,undefined and runs a fixed-seed pass.
//
// Usage: asan_parsers <mp4file> <annexbfile> <spsfile> <iters> [seed]
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <fstream>
#include <stdexcept>
#include <string>
#include <vector>

#include "../../scanner_amd/csrc/metadata.h"
#include "../../scanner_amd/csrc/msgpack.h"
#include "../../scanner_amd/csrc/video/h264.h"
#include "../../scanner_amd/csrc/video/mp4.h"
#include "../../scanner_amd/csrc/video/svc.h"

using sca::u8;

namespace {

uint64_t rng_state;
uint64_t xorshift() {
  uint64_t x = rng_state;
  x ^= x << 13;
  x ^= x >> 7;
  x ^= x << 17;
  return rng_state = x;
}

std::vector<u8> read_file(const char* path) {
  std::ifstream f(path, std::ios::binary);
  if (!f) throw std::runtime_error(std::string("cannot open ") + path);
  return std::vector<u8>(std::istreambuf_iterator<char>(f),
                         std::istreambuf_iterator<char>());
}

template <typename Parse>
void fuzz(const char* name, const std::vector<u8>& base, int iters,
          Parse parse) {
  int ok = 0, raised = 0;
  for (int i = 0; i < iters; ++i) {
    std::vector<u8> data = base;
    int mode = (int)(xorshift() % 3);
    if (mode == 0 || mode == 2) {
      data.resize(xorshift() % (data.size() + 1));
    }
    if ((mode == 1 || mode == 2) && !data.empty()) {
      int nmut = 1 + (int)(xorshift() % 8);
      for (int m = 0; m < nmut; ++m)
        data[xorshift() % data.size()] = (u8)(xorshift() & 0xff);
    }
    // heap-allocate the exact size so ASAN redzones catch any read past
    // the end of the parser's input
    std::vector<u8> exact(data);
    try {
      parse(exact.data(), exact.size());
      ++ok;
    } catch (const std::exception&) {
      ++raised;
    }
  }
  std::printf("%s: %d iters, %d parsed, %d raised\n", name, iters, ok,
              raised);
}

}  // namespace

int main(int argc, char** argv) {
  if (argc < 5) {
    std::fprintf(stderr,
                 "usage: %s <mp4> <annexb> <sps> <iters> [seed]\n", argv[0]);
    return 2;
  }
  int iters = std::atoi(argv[4]);
  rng_state = argc > 5 ? (uint64_t)std::atoll(argv[5]) | 1 : 0x9e3779b9ull;
  auto mp4 = read_file(argv[1]);
  auto annexb = read_file(argv[2]);
  auto sps = read_file(argv[3]);
  fuzz("mp4", mp4, iters,
       [](const u8* p, size_t n) { (void)sca::mp4_parse(p, n); });
  fuzz("annexb", annexb, iters,
       [](const u8* p, size_t n) { (void)sca::h264_index_annexb(p, n); });
  fuzz("sps", sps, iters,
       [](const u8* p, size_t n) { (void)sca::h264_parse_sps(p, n); });

  // SVC codec: encode a small synthetic clip in-process, then decode
  // mutated/truncated streams with the TRUSTED VideoMetadata (the
  // realistic corruption model: storage bytes rot, table metadata does
  // not). The decoder must reject every corruption via ScannerError —
  // pre-hardening, packet-controlled nbytes/ngroups/widths/super_off
  // could drive OOB heap reads and writes here.
  {
    int h = 24, w = 32, c = 3;
    sca::i64 n = 6;
    std::vector<u8> frames((size_t)n * h * w * c);
    for (size_t i = 0; i < frames.size(); ++i)
      frames[i] = (u8)((i * 7 + (i / 997)) & 0xff);
    std::vector<u8> stream;
    sca::VideoMetadata vm;
    sca::svc_encode_cpu(frames.data(), n, h, w, c, 3, stream, vm);
    std::vector<sca::i64> want{0, 2, 4, 5};
    fuzz("svc", stream, iters, [&](const u8* p, size_t sz) {
      std::vector<std::vector<u8>> out;
      sca::svc_decode_cpu(p, sz, vm, want, out);
    });
    // sanity: the unmutated stream still round-trips
    std::vector<std::vector<u8>> out;
    sca::svc_decode_cpu(stream.data(), stream.size(), vm, want, out);
    for (size_t k = 0; k < want.size(); ++k) {
      if (std::memcmp(out[k].data(),
                      frames.data() + (size_t)want[k] * h * w * c,
                      (size_t)h * w * c) != 0) {
        std::fprintf(stderr, "svc roundtrip mismatch frame %lld\n",
                     (long long)want[k]);
        return 1;
      }
    }
  }
  // msgpack codec: job specs/op args reach the C++ engine as msgpack
  // bytes over the RPC — decode of mutated/truncated payloads must
  // reject, not read out of bounds. Corpus: a representative job-shaped
  // value round-tripped through our own encoder.
  {
    sca::mp::Map m;
    m["name"] = std::string("Histogram");
    m["stride"] = (sca::i64)7;
    m["flag"] = true;
    m["rate"] = 29.97;
    sca::mp::Array arr;
    for (int i = 0; i < 20; ++i) arr.push_back(sca::mp::Value((sca::i64)i));
    m["rows"] = std::move(arr);
    m["blob"] = std::vector<u8>{1, 2, 3, 4, 5, 6, 7, 8};
    sca::mp::Map inner;
    inner["kind"] = std::string("Gather");
    m["sampling"] = sca::mp::Value(std::move(inner));
    std::vector<u8> enc = sca::mp::encode(sca::mp::Value(std::move(m)));
    fuzz("msgpack", enc, iters,
         [](const u8* p, size_t n) { (void)sca::mp::decode(p, n); });
  }
  // metadata records (BinWriter/BinReader, csrc/serialize.h): table and
  // video descriptors are read back from storage bytes — the same
  // corruption model as SVC packets. BinReader's count handling had the
  // same trusted-prefix/overflow patterns as the msgpack decoder.
  {
    sca::VideoMetadata vm;
    vm.width = 1920;
    vm.height = 1080;
    vm.channels = 3;
    vm.codec = "svc";
    vm.num_frames = 64;
    for (int i = 0; i < 64; i += 16) vm.keyframe_indices.push_back(i);
    for (int i = 0; i < 64; ++i) {
      vm.sample_offsets.push_back((sca::u64)i * 1000);
      vm.sample_sizes.push_back(1000);
    }
    std::vector<u8> enc = vm.serialize();
    fuzz("video_meta", enc, iters, [](const u8* p, size_t n) {
      (void)sca::VideoMetadata::deserialize(std::vector<u8>(p, p + n));
    });

    sca::TableMetadata tm;
    tm.id = 7;
    tm.name = "clip_table";
    for (int i = 1; i <= 5; ++i) tm.end_rows.push_back(i * 128);
    for (int c = 0; c < 3; ++c)
      tm.columns.push_back(sca::ColumnMeta{
          c, "col" + std::to_string(c),
          c ? sca::ColumnType::Bytes : sca::ColumnType::Video});
    std::vector<u8> enc2 = tm.serialize();
    fuzz("table_meta", enc2, iters, [](const u8* p, size_t n) {
      (void)sca::TableMetadata::deserialize(std::vector<u8>(p, p + n));
    });
  }
  std::printf("asan parser fuzz: OK\n");
  return 0;
}
