#include "h264.h"

#include <cstring>

namespace sca {

u32 BitReader::u(int bits) {
  u32 v = 0;
  for (int i = 0; i < bits; ++i) {
    SCA_CHECK(pos_ < n_ * 8, "h264: bitstream truncated");
    u32 bit = (d_[pos_ >> 3] >> (7 - (pos_ & 7))) & 1;
    v = (v << 1) | bit;
    ++pos_;
  }
  return v;
}

u32 BitReader::ue() {
  int zeros = 0;
  while (true) {
    SCA_CHECK(pos_ < n_ * 8, "h264: bitstream truncated in ue()");
    u32 bit = (d_[pos_ >> 3] >> (7 - (pos_ & 7))) & 1;
    ++pos_;
    if (bit) break;
    ++zeros;
    SCA_CHECK(zeros <= 31, "h264: ue() exceeds 32 bits");
  }
  u32 v = (1u << zeros) - 1 + (zeros ? u(zeros) : 0);
  return v;
}

i32 BitReader::se() {
  u32 k = ue();
  // 0,1,2,3,... -> 0,1,-1,2,-2,...
  return (k & 1) ? (i32)((k + 1) / 2) : -(i32)(k / 2);
}

std::vector<u8> rbsp_unescape(const u8* data, size_t size) {
  std::vector<u8> out;
  out.reserve(size);
  int zeros = 0;
  for (size_t i = 0; i < size; ++i) {
    if (zeros >= 2 && data[i] == 0x03) {
      zeros = 0;
      continue;  // emulation prevention byte
    }
    zeros = data[i] == 0 ? zeros + 1 : 0;
    out.push_back(data[i]);
  }
  return out;
}

std::vector<u8> rbsp_escape(const u8* data, size_t size) {
  std::vector<u8> out;
  out.reserve(size + size / 64);
  int zeros = 0;
  for (size_t i = 0; i < size; ++i) {
    if (zeros >= 2 && data[i] <= 0x03) {
      out.push_back(0x03);
      zeros = 0;
    }
    zeros = data[i] == 0 ? zeros + 1 : 0;
    out.push_back(data[i]);
  }
  return out;
}

namespace {

// Skip a seq_scaling_list as per the spec's decoding process (the values
// are irrelevant for indexing but the bit consumption must be exact).
void skip_scaling_list(BitReader& r, int size) {
  i32 last_scale = 8, next_scale = 8;
  for (int j = 0; j < size; ++j) {
    if (next_scale != 0) {
      i32 delta = r.se();
      next_scale = (last_scale + delta + 256) % 256;
    }
    last_scale = next_scale == 0 ? last_scale : next_scale;
  }
}

}  // namespace

H264Sps h264_parse_sps(const u8* nal, size_t size) {
  SCA_CHECK(size >= 4, "h264: SPS NAL too short");
  SCA_CHECK((nal[0] & 0x1f) == 7, "h264: not an SPS NAL");
  std::vector<u8> rbsp = rbsp_unescape(nal + 1, size - 1);
  BitReader r(rbsp.data(), rbsp.size());
  H264Sps s;
  s.raw.assign(nal, nal + size);
  s.profile_idc = r.u(8);
  r.u(8);  // constraint flags + reserved
  s.level_idc = r.u(8);
  s.sps_id = r.ue();
  bool high_profile = false;
  switch (s.profile_idc) {
    case 100: case 110: case 122: case 244: case 44: case 83:
    case 86: case 118: case 128: case 138: case 139: case 134: case 135:
      high_profile = true;
      break;
    default:
      break;
  }
  bool separate_colour_plane = false;
  if (high_profile) {
    s.chroma_format_idc = r.ue();
    if (s.chroma_format_idc == 3) separate_colour_plane = r.u(1);
    r.ue();  // bit_depth_luma_minus8
    r.ue();  // bit_depth_chroma_minus8
    r.u(1);  // qpprime_y_zero_transform_bypass
    if (r.u(1)) {  // seq_scaling_matrix_present
      int lists = s.chroma_format_idc != 3 ? 8 : 12;
      for (int i = 0; i < lists; ++i) {
        if (r.u(1)) skip_scaling_list(r, i < 6 ? 16 : 64);
      }
    }
  }
  r.ue();  // log2_max_frame_num_minus4
  u32 poc_type = r.ue();
  if (poc_type == 0) {
    r.ue();  // log2_max_pic_order_cnt_lsb_minus4
  } else if (poc_type == 1) {
    r.u(1);  // delta_pic_order_always_zero
    r.se();  // offset_for_non_ref_pic
    r.se();  // offset_for_top_to_bottom_field
    u32 cycles = r.ue();
    for (u32 i = 0; i < cycles; ++i) r.se();
  }
  r.ue();  // max_num_ref_frames
  r.u(1);  // gaps_in_frame_num_value_allowed
  u32 pw_mbs = r.ue() + 1;
  u32 ph_map = r.ue() + 1;
  s.frame_mbs_only = r.u(1);
  if (!s.frame_mbs_only) r.u(1);  // mb_adaptive_frame_field
  r.u(1);                         // direct_8x8_inference
  u32 crop_l = 0, crop_r = 0, crop_t = 0, crop_b = 0;
  if (r.u(1)) {  // frame_cropping_flag
    crop_l = r.ue();
    crop_r = r.ue();
    crop_t = r.ue();
    crop_b = r.ue();
  }
  // Crop units per the spec: ChromaArrayType 0 (mono / separate planes)
  // crops in luma samples, else in chroma sample units.
  int chroma_array = separate_colour_plane ? 0 : s.chroma_format_idc;
  int sub_w = (chroma_array == 1 || chroma_array == 2) ? 2 : 1;
  int sub_h = (chroma_array == 1) ? 2 : 1;
  int crop_x = chroma_array == 0 ? 1 : sub_w;
  int crop_y = (chroma_array == 0 ? 1 : sub_h) * (2 - (int)s.frame_mbs_only);
  i64 w = (i64)pw_mbs * 16 - (i64)crop_x * (crop_l + crop_r);
  i64 h = (i64)ph_map * 16 * (2 - (int)s.frame_mbs_only) -
          (i64)crop_y * (crop_t + crop_b);
  SCA_CHECK(w > 0 && h > 0 && w <= 16384 && h <= 16384,
            "h264: SPS yields invalid dimensions");
  s.width = (i32)w;
  s.height = (i32)h;
  return s;
}

H264Index h264_index_annexb(const u8* data, size_t size) {
  SCA_CHECK(data && size >= 5, "h264: stream too short");
  H264Index idx;

  // Collect NAL (offset includes its start code) positions.
  struct Nal {
    u64 sc_off;   // offset of the start code
    u64 off;      // offset of the NAL header byte
    u64 end;      // one past the last payload byte
    u8 type;
  };
  std::vector<Nal> nals;
  size_t i = 0;
  // find first start code
  auto is_sc3 = [&](size_t p) {
    return p + 3 <= size && data[p] == 0 && data[p + 1] == 0 &&
           data[p + 2] == 1;
  };
  while (i + 3 <= size && !is_sc3(i)) ++i;
  SCA_CHECK(i + 3 <= size, "h264: no start code found");
  while (i + 3 <= size) {
    size_t sc = i;
    // tolerate 4-byte start codes: back the sc offset up over leading zeros
    size_t sc_begin = sc;
    while (sc_begin > 0 && data[sc_begin - 1] == 0) --sc_begin;
    size_t hdr = sc + 3;
    SCA_CHECK(hdr < size, "h264: truncated NAL header");
    size_t j = hdr + 1;
    while (j + 3 <= size && !is_sc3(j)) ++j;
    size_t end = (j + 3 <= size) ? j : size;
    // trim trailing zeros that belong to the next start code
    while (end > hdr + 1 && data[end - 1] == 0 && j + 3 <= size) --end;
    Nal n;
    n.sc_off = sc_begin;
    n.off = hdr;
    n.end = end;
    n.type = data[hdr] & 0x1f;
    nals.push_back(n);
    if (j + 3 > size) break;
    i = j;
  }
  SCA_CHECK(!nals.empty(), "h264: no NAL units");

  // Walk NALs into access units: a VCL NAL (slice, type 1/5) with
  // first_mb_in_slice == 0 starts a new AU; non-VCL NALs since the last
  // slice attach to the upcoming AU.
  i64 au = -1;
  u64 pending_start = (u64)-1;
  u64 au_start = 0;
  bool have_sps = false, have_pps = false;
  auto close_au = [&](u64 end_off) {
    if (au < 0) return;
    idx.sample_sizes.push_back(end_off - idx.sample_offsets.back());
  };
  for (auto& n : nals) {
    switch (n.type) {
      case 7: {  // SPS
        H264Sps s = h264_parse_sps(data + n.off, n.end - n.off);
        if (!have_sps) {
          idx.width = s.width;
          idx.height = s.height;
          idx.sps = s.raw;
        } else {
          SCA_CHECK(s.width == idx.width && s.height == idx.height,
                    "h264: mid-stream resolution change unsupported");
        }
        have_sps = true;
        if (pending_start == (u64)-1) pending_start = n.sc_off;
        break;
      }
      case 8:  // PPS
        if (!have_pps) idx.pps.assign(data + n.sc_off + (n.off - n.sc_off),
                                      data + n.end);
        have_pps = true;
        if (pending_start == (u64)-1) pending_start = n.sc_off;
        break;
      case 6:   // SEI
      case 9:   // AUD
      case 15:  // sub-SPS etc.
        if (pending_start == (u64)-1) pending_start = n.sc_off;
        break;
      case 1:
      case 5: {  // slice
        SCA_CHECK(have_sps && have_pps,
                  "h264: slice before SPS/PPS");
        std::vector<u8> rb =
            rbsp_unescape(data + n.off + 1,
                          std::min<size_t>(n.end - n.off - 1, 16));
        BitReader r(rb.data(), rb.size());
        u32 first_mb = r.ue();
        if (first_mb == 0) {
          u64 start = pending_start != (u64)-1 ? pending_start : n.sc_off;
          close_au(start);
          ++au;
          au_start = start;
          (void)au_start;
          idx.sample_offsets.push_back(start);
          if (n.type == 5) idx.keyframe_indices.push_back(au);
        }
        pending_start = (u64)-1;
        break;
      }
      default:
        // attach unknown NALs to the following AU
        if (pending_start == (u64)-1) pending_start = n.sc_off;
        break;
    }
  }
  close_au(size);
  idx.num_frames = au + 1;
  SCA_CHECK(idx.num_frames > 0, "h264: no access units found");
  SCA_CHECK(!idx.keyframe_indices.empty() && idx.keyframe_indices[0] == 0,
            "h264: stream does not start with an IDR access unit");
  return idx;
}

}  // namespace sca
