#include "mp4.h"

#include <cstring>

#include "h264.h"

namespace sca {

namespace {

// ---- reading ----

struct Cursor {
  const u8* d;
  size_t n;
  size_t p = 0;

  size_t left() const { return n - p; }
  void need(size_t k, const char* what) {
    SCA_CHECK(p + k <= n, std::string("mp4: truncated ") + what);
  }
  u8 u8v() { need(1, "u8"); return d[p++]; }
  u16 u16v() {
    need(2, "u16");
    u16 v = ((u16)d[p] << 8) | d[p + 1];
    p += 2;
    return v;
  }
  u32 u32v() {
    need(4, "u32");
    u32 v = ((u32)d[p] << 24) | ((u32)d[p + 1] << 16) | ((u32)d[p + 2] << 8) |
            d[p + 3];
    p += 4;
    return v;
  }
  u64 u64v() {
    u64 hi = u32v();
    return (hi << 32) | u32v();
  }
  void skip(size_t k, const char* what) { need(k, what); p += k; }
};

struct Box {
  u32 type = 0;
  size_t body_off = 0;  // absolute offset of box body
  size_t body_size = 0;
};

constexpr u32 fourcc(const char* s) {
  return ((u32)(u8)s[0] << 24) | ((u32)(u8)s[1] << 16) | ((u32)(u8)s[2] << 8) |
         (u32)(u8)s[3];
}

// Iterate child boxes within [off, off+size); returns them in order.
std::vector<Box> children(const u8* d, size_t off, size_t size) {
  std::vector<Box> out;
  size_t p = off;
  size_t end = off + size;
  while (p + 8 <= end) {
    u32 sz = ((u32)d[p] << 24) | ((u32)d[p + 1] << 16) | ((u32)d[p + 2] << 8) |
             d[p + 3];
    u32 ty = ((u32)d[p + 4] << 24) | ((u32)d[p + 5] << 16) |
             ((u32)d[p + 6] << 8) | d[p + 7];
    size_t hdr = 8;
    u64 full = sz;
    if (sz == 1) {
      SCA_CHECK(p + 16 <= end, "mp4: truncated largesize box");
      full = 0;
      for (int k = 0; k < 8; ++k) full = (full << 8) | d[p + 8 + k];
      hdr = 16;
    } else if (sz == 0) {
      full = end - p;  // extends to end of enclosing box
    }
    SCA_CHECK(full >= hdr && p + full <= end, "mp4: box overruns container");
    out.push_back(Box{ty, p + hdr, (size_t)(full - hdr)});
    p += full;
  }
  return out;
}

const Box* find(const std::vector<Box>& boxes, u32 type) {
  for (auto& b : boxes)
    if (b.type == type) return &b;
  return nullptr;
}

}  // namespace

Mp4Track mp4_parse(const u8* data, size_t size) {
  SCA_CHECK(data && size >= 16, "mp4: file too short");
  auto top = children(data, 0, size);
  const Box* moov = find(top, fourcc("moov"));
  SCA_CHECK(moov, "mp4: no moov box");

  auto moov_kids = children(data, moov->body_off, moov->body_size);
  for (auto& trak : moov_kids) {
    if (trak.type != fourcc("trak")) continue;
    auto trak_kids = children(data, trak.body_off, trak.body_size);
    const Box* mdia = find(trak_kids, fourcc("mdia"));
    if (!mdia) continue;
    auto mdia_kids = children(data, mdia->body_off, mdia->body_size);
    const Box* hdlr = find(mdia_kids, fourcc("hdlr"));
    if (!hdlr || hdlr->body_size < 12) continue;
    u32 handler = ((u32)data[hdlr->body_off + 8] << 24) |
                  ((u32)data[hdlr->body_off + 9] << 16) |
                  ((u32)data[hdlr->body_off + 10] << 8) |
                  data[hdlr->body_off + 11];
    if (handler != fourcc("vide")) continue;

    Mp4Track t;
    const Box* mdhd = find(mdia_kids, fourcc("mdhd"));
    if (mdhd && mdhd->body_size >= 16) {
      Cursor c{data + mdhd->body_off, mdhd->body_size};
      u8 version = c.u8v();
      c.skip(3, "mdhd flags");
      c.skip(version == 1 ? 16 : 8, "mdhd times");
      t.timescale = c.u32v();
    }
    const Box* minf = find(mdia_kids, fourcc("minf"));
    SCA_CHECK(minf, "mp4: video trak missing minf");
    auto minf_kids = children(data, minf->body_off, minf->body_size);
    const Box* stbl = find(minf_kids, fourcc("stbl"));
    SCA_CHECK(stbl, "mp4: video trak missing stbl");
    auto stbl_kids = children(data, stbl->body_off, stbl->body_size);

    // ---- stsd / avc1 / avcC ----
    const Box* stsd = find(stbl_kids, fourcc("stsd"));
    SCA_CHECK(stsd, "mp4: missing stsd");
    {
      Cursor c{data + stsd->body_off, stsd->body_size};
      c.skip(4, "stsd header");
      u32 count = c.u32v();
      SCA_CHECK(count >= 1, "mp4: empty stsd");
      // first sample entry
      size_t entry_off = stsd->body_off + c.p;
      c.need(8, "sample entry");
      u32 esz = c.u32v();
      u32 ety = c.u32v();
      SCA_CHECK(ety == fourcc("avc1") || ety == fourcc("avc3"),
                "mp4: video track is not AVC (avc1/avc3)");
      // VisualSampleEntry: 6 reserved + 2 data_ref_index + 16 predefined/
      // reserved + 2 width + 2 height + 4+4 dpi + 4 reserved + 2 frame_count
      // + 32 compressor + 2 depth + 2 predefined = 78 bytes
      c.skip(6 + 2 + 16, "sample entry fields");
      c.u16v();  // coded width (SPS is authoritative)
      c.u16v();  // coded height
      c.skip(4 + 4 + 4 + 2 + 32 + 2 + 2, "sample entry fields");
      // child boxes of the sample entry up to esz
      SCA_CHECK(esz >= 8 && entry_off + esz <= stsd->body_off + stsd->body_size,
                "mp4: sample entry overrun");
      auto entry_kids =
          children(data, stsd->body_off + c.p, esz - (c.p - (entry_off - stsd->body_off)));
      const Box* avcc = find(entry_kids, fourcc("avcC"));
      SCA_CHECK(avcc, "mp4: avc1 entry missing avcC");
      Cursor a{data + avcc->body_off, avcc->body_size};
      u8 cfg_ver = a.u8v();
      SCA_CHECK(cfg_ver == 1, "mp4: bad avcC configurationVersion");
      a.skip(3, "avcC profile/level");
      t.length_size = (a.u8v() & 0x3) + 1;
      u32 nsps = a.u8v() & 0x1f;
      SCA_CHECK(nsps >= 1, "mp4: avcC has no SPS");
      for (u32 i = 0; i < nsps; ++i) {
        u16 len = a.u16v();
        a.need(len, "avcC SPS");
        t.sps.emplace_back(a.d + a.p, a.d + a.p + len);
        a.p += len;
      }
      u32 npps = a.u8v();
      SCA_CHECK(npps >= 1, "mp4: avcC has no PPS");
      for (u32 i = 0; i < npps; ++i) {
        u16 len = a.u16v();
        a.need(len, "avcC PPS");
        t.pps.emplace_back(a.d + a.p, a.d + a.p + len);
        a.p += len;
      }
      H264Sps s = h264_parse_sps(t.sps[0].data(), t.sps[0].size());
      t.width = s.width;
      t.height = s.height;
    }

    // ---- stts: total sample count + per-sample durations ----
    const Box* stts = find(stbl_kids, fourcc("stts"));
    SCA_CHECK(stts, "mp4: missing stts");
    u64 n_samples_stts = 0;
    {
      Cursor c{data + stts->body_off, stts->body_size};
      c.skip(4, "stts header");
      u32 entries = c.u32v();
      for (u32 i = 0; i < entries; ++i) {
        u32 count = c.u32v();
        u32 delta = c.u32v();
        n_samples_stts += count;
        SCA_CHECK(n_samples_stts <= (1u << 30), "mp4: absurd sample count");
        for (u32 k = 0; k < count; ++k) t.sample_deltas.push_back(delta);
      }
    }

    // ---- stsz ----
    const Box* stsz = find(stbl_kids, fourcc("stsz"));
    SCA_CHECK(stsz, "mp4: missing stsz");
    {
      Cursor c{data + stsz->body_off, stsz->body_size};
      c.skip(4, "stsz header");
      u32 fixed = c.u32v();
      u32 count = c.u32v();
      SCA_CHECK(count == n_samples_stts, "mp4: stsz/stts sample count mismatch");
      t.sample_sizes.reserve(count);
      if (fixed != 0) {
        for (u32 i = 0; i < count; ++i) t.sample_sizes.push_back(fixed);
      } else {
        for (u32 i = 0; i < count; ++i) t.sample_sizes.push_back(c.u32v());
      }
    }

    // ---- chunk offsets ----
    std::vector<u64> chunk_offsets;
    if (const Box* stco = find(stbl_kids, fourcc("stco"))) {
      Cursor c{data + stco->body_off, stco->body_size};
      c.skip(4, "stco header");
      u32 count = c.u32v();
      for (u32 i = 0; i < count; ++i) chunk_offsets.push_back(c.u32v());
    } else if (const Box* co64 = find(stbl_kids, fourcc("co64"))) {
      Cursor c{data + co64->body_off, co64->body_size};
      c.skip(4, "co64 header");
      u32 count = c.u32v();
      for (u32 i = 0; i < count; ++i) chunk_offsets.push_back(c.u64v());
    } else {
      throw ScannerError("mp4: missing stco/co64");
    }

    // ---- stsc: expand samples over chunks -> absolute offsets ----
    const Box* stsc = find(stbl_kids, fourcc("stsc"));
    SCA_CHECK(stsc, "mp4: missing stsc");
    {
      Cursor c{data + stsc->body_off, stsc->body_size};
      c.skip(4, "stsc header");
      u32 entries = c.u32v();
      struct Run {
        u32 first_chunk, per_chunk;
      };
      std::vector<Run> runs;
      for (u32 i = 0; i < entries; ++i) {
        u32 first = c.u32v();
        u32 per = c.u32v();
        c.u32v();  // sample description index
        runs.push_back({first, per});
      }
      SCA_CHECK(!runs.empty(), "mp4: empty stsc");
      size_t si = 0;
      for (size_t ri = 0; ri < runs.size() && si < t.sample_sizes.size();
           ++ri) {
        u32 last_chunk = ri + 1 < runs.size() ? runs[ri + 1].first_chunk
                                              : (u32)chunk_offsets.size() + 1;
        for (u32 ch = runs[ri].first_chunk;
             ch < last_chunk && si < t.sample_sizes.size(); ++ch) {
          SCA_CHECK(ch >= 1 && ch <= chunk_offsets.size(),
                    "mp4: stsc chunk out of range");
          u64 off = chunk_offsets[ch - 1];
          for (u32 k = 0; k < runs[ri].per_chunk && si < t.sample_sizes.size();
               ++k, ++si) {
            SCA_CHECK(off + t.sample_sizes[si] <= size,
                      "mp4: sample extends past end of file");
            t.sample_offsets.push_back(off);
            off += t.sample_sizes[si];
          }
        }
      }
      SCA_CHECK(t.sample_offsets.size() == t.sample_sizes.size(),
                "mp4: stsc does not cover all samples");
    }

    // ---- stss (sync samples); absent => every sample is a keyframe ----
    if (const Box* stss = find(stbl_kids, fourcc("stss"))) {
      Cursor c{data + stss->body_off, stss->body_size};
      c.skip(4, "stss header");
      u32 count = c.u32v();
      for (u32 i = 0; i < count; ++i) {
        u32 sn = c.u32v();  // 1-based
        SCA_CHECK(sn >= 1 && sn <= t.sample_sizes.size(),
                  "mp4: stss sample out of range");
        t.keyframe_indices.push_back((i64)sn - 1);
      }
    } else {
      for (size_t i = 0; i < t.sample_sizes.size(); ++i)
        t.keyframe_indices.push_back((i64)i);
    }
    SCA_CHECK(!t.keyframe_indices.empty() && t.keyframe_indices[0] == 0,
              "mp4: first sample is not a sync sample");
    return t;
  }
  throw ScannerError("mp4: no AVC video track found");
}

// ---- writing ----

namespace {

struct W {
  std::vector<u8> b;
  void u8v(u8 v) { b.push_back(v); }
  void u16v(u16 v) {
    b.push_back(v >> 8);
    b.push_back(v & 0xff);
  }
  void u32v(u32 v) {
    b.push_back(v >> 24);
    b.push_back((v >> 16) & 0xff);
    b.push_back((v >> 8) & 0xff);
    b.push_back(v & 0xff);
  }
  void raw(const void* p, size_t n) {
    const u8* q = (const u8*)p;
    b.insert(b.end(), q, q + n);
  }
  void tag(const char* s) { raw(s, 4); }
  // open a box, returns patch position for its size
  size_t open(const char* type) {
    size_t at = b.size();
    u32v(0);
    tag(type);
    return at;
  }
  void close(size_t at) {
    u32 sz = (u32)(b.size() - at);
    b[at] = sz >> 24;
    b[at + 1] = (sz >> 16) & 0xff;
    b[at + 2] = (sz >> 8) & 0xff;
    b[at + 3] = sz & 0xff;
  }
};

}  // namespace

std::vector<u8> mp4_write(const u8* stream, size_t stream_size,
                          const std::vector<u64>& sample_offsets,
                          const std::vector<u64>& sample_sizes,
                          const std::vector<i64>& keyframe_indices,
                          const std::vector<u8>& sps,
                          const std::vector<u8>& pps, i32 width, i32 height,
                          double fps) {
  SCA_CHECK(!sample_offsets.empty() &&
                sample_offsets.size() == sample_sizes.size(),
            "mp4_write: empty or mismatched sample index");
  SCA_CHECK(!sps.empty() && !pps.empty(), "mp4_write: missing SPS/PPS");
  u32 n = (u32)sample_offsets.size();
  u32 timescale = 90000;
  u32 delta = (u32)(timescale / (fps > 0 ? fps : 30.0));
  u64 duration = (u64)n * delta;

  // Convert each Annex-B access unit to AVCC (4-byte length prefixes),
  // stripping SPS/PPS/AUD NALs (they live in avcC).
  std::vector<std::vector<u8>> avcc(n);
  for (u32 s = 0; s < n; ++s) {
    const u8* au = stream + sample_offsets[s];
    size_t len = sample_sizes[s];
    SCA_CHECK(sample_offsets[s] + len <= stream_size,
              "mp4_write: sample range outside stream");
    size_t p = 0;
    auto sc_at = [&](size_t q) {
      return q + 3 <= len && au[q] == 0 && au[q + 1] == 0 && au[q + 2] == 1;
    };
    while (p + 3 <= len && !sc_at(p)) ++p;
    while (p + 3 <= len) {
      size_t hdr = p + 3;
      size_t q = hdr + 1;
      while (q + 3 <= len && !sc_at(q)) ++q;
      size_t end = q + 3 <= len ? q : len;
      while (end > hdr + 1 && au[end - 1] == 0 && q + 3 <= len) --end;
      u8 type = au[hdr] & 0x1f;
      if (type != 7 && type != 8 && type != 9) {
        u32 nl = (u32)(end - hdr);
        avcc[s].push_back(nl >> 24);
        avcc[s].push_back((nl >> 16) & 0xff);
        avcc[s].push_back((nl >> 8) & 0xff);
        avcc[s].push_back(nl & 0xff);
        avcc[s].insert(avcc[s].end(), au + hdr, au + end);
      }
      if (q + 3 > len) break;
      p = q;
    }
    SCA_CHECK(!avcc[s].empty(), "mp4_write: sample has no slice NALs");
  }

  W f;
  // ftyp
  {
    size_t at = f.open("ftyp");
    f.tag("isom");
    f.u32v(0x200);
    f.tag("isom");
    f.tag("avc1");
    f.close(at);
  }
  // mdat
  std::vector<u64> out_offsets(n);
  {
    size_t at = f.open("mdat");
    for (u32 s = 0; s < n; ++s) {
      out_offsets[s] = f.b.size();
      f.raw(avcc[s].data(), avcc[s].size());
    }
    f.close(at);
  }
  // moov
  size_t moov = f.open("moov");
  {
    size_t at = f.open("mvhd");
    f.u32v(0);            // version/flags
    f.u32v(0);            // creation
    f.u32v(0);            // modification
    f.u32v(timescale);
    f.u32v((u32)duration);
    f.u32v(0x00010000);   // rate
    f.u16v(0x0100);       // volume
    f.u16v(0);
    f.u32v(0);
    f.u32v(0);
    const u32 matrix[9] = {0x10000, 0, 0, 0, 0x10000, 0, 0, 0, 0x40000000};
    for (u32 m : matrix) f.u32v(m);
    for (int i = 0; i < 6; ++i) f.u32v(0);  // predefined
    f.u32v(2);            // next track id
    f.close(at);
  }
  size_t trak = f.open("trak");
  {
    size_t at = f.open("tkhd");
    f.u32v(0x7);  // version 0, flags: enabled|in_movie|in_preview
    f.u32v(0);
    f.u32v(0);
    f.u32v(1);    // track id
    f.u32v(0);
    f.u32v((u32)duration);
    f.u32v(0);
    f.u32v(0);
    f.u16v(0);    // layer
    f.u16v(0);    // alternate group
    f.u16v(0);    // volume
    f.u16v(0);
    const u32 matrix[9] = {0x10000, 0, 0, 0, 0x10000, 0, 0, 0, 0x40000000};
    for (u32 m : matrix) f.u32v(m);
    f.u32v((u32)width << 16);
    f.u32v((u32)height << 16);
    f.close(at);
  }
  size_t mdia = f.open("mdia");
  {
    size_t at = f.open("mdhd");
    f.u32v(0);
    f.u32v(0);
    f.u32v(0);
    f.u32v(timescale);
    f.u32v((u32)duration);
    f.u16v(0x55c4);  // language 'und'
    f.u16v(0);
    f.close(at);
  }
  {
    size_t at = f.open("hdlr");
    f.u32v(0);
    f.u32v(0);
    f.tag("vide");
    for (int i = 0; i < 3; ++i) f.u32v(0);
    f.raw("VideoHandler", 13);  // includes NUL
    f.close(at);
  }
  size_t minf = f.open("minf");
  {
    size_t at = f.open("vmhd");
    f.u32v(1);  // version 0, flags 1
    f.u16v(0);
    f.u16v(0);
    f.u16v(0);
    f.u16v(0);
    f.close(at);
  }
  {
    size_t dinf = f.open("dinf");
    size_t dref = f.open("dref");
    f.u32v(0);
    f.u32v(1);
    size_t url = f.open("url ");
    f.u32v(1);  // self-contained
    f.close(url);
    f.close(dref);
    f.close(dinf);
  }
  size_t stbl = f.open("stbl");
  {
    size_t stsd = f.open("stsd");
    f.u32v(0);
    f.u32v(1);
    size_t avc1 = f.open("avc1");
    for (int i = 0; i < 6; ++i) f.u8v(0);  // reserved
    f.u16v(1);                             // data ref index
    f.u16v(0);
    f.u16v(0);
    for (int i = 0; i < 3; ++i) f.u32v(0);
    f.u16v((u16)width);
    f.u16v((u16)height);
    f.u32v(0x00480000);  // 72 dpi
    f.u32v(0x00480000);
    f.u32v(0);
    f.u16v(1);  // frame count
    for (int i = 0; i < 32; ++i) f.u8v(0);  // compressor name
    f.u16v(0x18);    // depth
    f.u16v(0xffff);  // predefined
    {
      size_t avcc_at = f.open("avcC");
      f.u8v(1);                      // configurationVersion
      f.u8v(sps.size() > 1 ? sps[1] : 66);  // profile
      f.u8v(sps.size() > 2 ? sps[2] : 0);   // compat
      f.u8v(sps.size() > 3 ? sps[3] : 30);  // level
      f.u8v(0xfc | 3);               // 4-byte lengths
      f.u8v(0xe0 | 1);               // 1 SPS
      f.u16v((u16)sps.size());
      f.raw(sps.data(), sps.size());
      f.u8v(1);                      // 1 PPS
      f.u16v((u16)pps.size());
      f.raw(pps.data(), pps.size());
      f.close(avcc_at);
    }
    f.close(avc1);
    f.close(stsd);
  }
  {
    size_t at = f.open("stts");
    f.u32v(0);
    f.u32v(1);
    f.u32v(n);
    f.u32v(delta);
    f.close(at);
  }
  {
    size_t at = f.open("stss");
    f.u32v(0);
    f.u32v((u32)keyframe_indices.size());
    for (i64 k : keyframe_indices) f.u32v((u32)k + 1);
    f.close(at);
  }
  {
    size_t at = f.open("stsc");
    f.u32v(0);
    f.u32v(1);
    f.u32v(1);  // first chunk
    f.u32v(n);  // samples per chunk (single chunk)
    f.u32v(1);  // sample description
    f.close(at);
  }
  {
    size_t at = f.open("stsz");
    f.u32v(0);
    f.u32v(0);  // variable sizes
    f.u32v(n);
    for (u32 s = 0; s < n; ++s) f.u32v((u32)avcc[s].size());
    f.close(at);
  }
  {
    size_t at = f.open("stco");
    f.u32v(0);
    f.u32v(1);
    f.u32v((u32)out_offsets[0]);
    f.close(at);
  }
  f.close(stbl);
  f.close(minf);
  f.close(mdia);
  f.close(trak);
  f.close(moov);
  return std::move(f.b);
}

}  // namespace sca
