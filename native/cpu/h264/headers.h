// SPS/PPS/slice-header writers shared by the CPU encoder and the
// GPU-pipeline entropy stage. Bitstream layout documented in encoder.h.
#pragma once

#include "bitwriter.h"

namespace hipflux {
namespace h264 {

inline int level_idc_for(int mbw, int mbh) {
  int fs = mbw * mbh;
  if (fs <= 1620) return 31;
  if (fs <= 3600) return 32;
  if (fs <= 8192) return 42;   // 1080p60
  if (fs <= 22080) return 51;  // 4K
  return 52;
}

inline void write_sps_nal(std::vector<uint8_t>& out, int mbw, int mbh,
                          int width, int height) {
  BitWriter b;
  b.u(66, 8);   // profile_idc: Baseline
  b.u(1, 1);    // constraint_set0
  b.u(1, 1);    // constraint_set1 (constrained baseline)
  b.u(0, 6);
  b.u(level_idc_for(mbw, mbh), 8);
  b.ue(0);      // sps id
  b.ue(12);     // log2_max_frame_num_minus4 -> 16 bits
  b.ue(2);      // pic_order_cnt_type
  b.ue(1);      // max_num_ref_frames
  b.u(0, 1);
  b.ue(mbw - 1);
  b.ue(mbh - 1);
  b.u(1, 1);    // frame_mbs_only
  b.u(1, 1);    // direct_8x8_inference
  int crop_r = (mbw * 16 - width) / 2, crop_b = (mbh * 16 - height) / 2;
  if (crop_r || crop_b) {
    b.u(1, 1);
    b.ue(0);
    b.ue(crop_r);
    b.ue(0);
    b.ue(crop_b);
  } else {
    b.u(0, 1);
  }
  // VUI: BT.601 full range (matches our CSC)
  b.u(1, 1);
  b.u(0, 1);
  b.u(0, 1);
  b.u(1, 1);
  b.u(5, 3);
  b.u(1, 1);
  b.u(1, 1);
  b.u(6, 8);
  b.u(6, 8);
  b.u(6, 8);
  b.u(0, 1);
  b.u(0, 1);
  b.u(0, 1);
  b.u(0, 1);
  b.u(0, 1);
  b.u(0, 1);
  b.rbsp_trailing();
  b.emit_nal(out, 3, 7);
}

inline void write_pps_nal(std::vector<uint8_t>& out) {
  BitWriter b;
  b.ue(0);
  b.ue(0);
  b.u(0, 1);    // CAVLC
  b.u(0, 1);
  b.ue(0);
  b.ue(0);
  b.ue(0);
  b.u(0, 1);
  b.u(0, 2);
  b.se(0);      // pic_init_qp_minus26
  b.se(0);
  b.se(0);      // chroma_qp_index_offset
  b.u(1, 1);    // deblocking_filter_control_present
  b.u(0, 1);
  b.u(0, 1);
  b.rbsp_trailing();
  b.emit_nal(out, 3, 8);
}

inline void write_slice_header_bits(BitWriter& b, bool idr, int first_mb,
                                    uint32_t frame_num, uint32_t idr_pic_id,
                                    int qp, int deblock_idc = 1) {
  b.ue(first_mb);
  b.ue(idr ? 7 : 5);
  b.ue(0);
  b.u(frame_num & 0xFFFF, 16);
  if (idr) b.ue(idr_pic_id);
  if (!idr) {
    b.u(0, 1);
    b.u(0, 1);
  }
  if (idr) {
    b.u(0, 1);
    b.u(0, 1);
  } else {
    b.u(0, 1);
  }
  b.se(qp - 26);
  // 1 = deblocking off; 2 = filter within the slice only (our slices are
  // single MB rows / row segments, so no cross-row dependency)
  b.ue(deblock_idc);
}

}  // namespace h264
}  // namespace hipflux
