// HEVC parameter-set / slice-header writers (§7.3). Shared by the CPU
// encoder and the GPU pipeline's bitstream assembly.
#pragma once

#include "../h264/bitwriter.h"

namespace hipflux {
namespace hevc {

using h264::BitWriter;

enum NalType { kNalIdr = 19, kNalVps = 32, kNalSps = 33, kNalPps = 34 };

// Append one HEVC NAL (2-byte header, §7.3.1.2) with start code and
// emulation prevention.
inline void emit_nal(const BitWriter& rbsp, std::vector<uint8_t>& out,
                     int nal_type) {
  out.push_back(0);
  out.push_back(0);
  out.push_back(0);
  out.push_back(1);
  out.push_back(static_cast<uint8_t>(nal_type << 1));  // layer id 0
  out.push_back(1);                                    // temporal id +1
  int zeros = 0;
  for (uint8_t b : rbsp.bytes()) {
    if (zeros >= 2 && b <= 3) {
      out.push_back(3);
      zeros = 0;
    }
    out.push_back(b);
    zeros = (b == 0) ? zeros + 1 : 0;
  }
}

inline int level_idc_for(int w, int h) {
  long ps = static_cast<long>(w) * h;
  if (ps <= 2228224) return 123;   // 4.1 (1080p60)
  if (ps <= 8912896) return 153;   // 5.1 (4K60)
  return 183;                      // 6.1 (8K60)
}

// profile_tier_level for Main profile (§7.3.3).
inline void write_ptl(BitWriter& b, int level_idc) {
  b.u(0, 2);            // general_profile_space
  b.u(0, 1);            // general_tier_flag
  b.u(1, 5);            // general_profile_idc: Main
  b.u(0x60000000, 32);  // compatibility flags: bits 1 (Main) + 2 (Main 10)
  b.u(1, 1);            // general_progressive_source_flag
  b.u(0, 1);            // general_interlaced_source_flag
  b.u(1, 1);            // general_non_packed_constraint_flag
  b.u(1, 1);            // general_frame_only_constraint_flag
  b.u(0, 22);           // general_reserved_zero_44bits
  b.u(0, 22);
  b.u(level_idc, 8);
}

inline void write_vps_nal(std::vector<uint8_t>& out, int level_idc) {
  BitWriter b;
  b.u(0, 4);   // vps_video_parameter_set_id
  b.u(1, 1);   // vps_base_layer_internal_flag
  b.u(1, 1);   // vps_base_layer_available_flag
  b.u(0, 6);   // vps_max_layers_minus1
  b.u(0, 3);   // vps_max_sub_layers_minus1
  b.u(1, 1);   // vps_temporal_id_nesting_flag
  b.u(0xFFFF, 16);  // vps_reserved_0xffff_16bits
  write_ptl(b, level_idc);
  b.u(1, 1);   // vps_sub_layer_ordering_info_present (code layer 0)
  b.ue(0);     // vps_max_dec_pic_buffering_minus1
  b.ue(0);     // vps_max_num_reorder_pics
  b.ue(0);     // vps_max_latency_increase_plus1
  b.u(0, 6);   // vps_max_layer_id
  b.ue(0);     // vps_num_layer_sets_minus1
  b.u(0, 1);   // vps_timing_info_present_flag
  b.u(0, 1);   // vps_extension_flag
  b.rbsp_trailing();
  emit_nal(b, out, kNalVps);
}

// Coded dims are 16-aligned; visible dims signal via conformance window.
inline void write_sps_nal(std::vector<uint8_t>& out, int coded_w, int coded_h,
                          int width, int height) {
  BitWriter b;
  b.u(0, 4);   // sps_video_parameter_set_id
  b.u(0, 3);   // sps_max_sub_layers_minus1
  b.u(1, 1);   // sps_temporal_id_nesting_flag
  write_ptl(b, level_idc_for(coded_w, coded_h));
  b.ue(0);     // sps_seq_parameter_set_id
  b.ue(1);     // chroma_format_idc: 4:2:0
  b.ue(coded_w);
  b.ue(coded_h);
  int crop_r = (coded_w - width) / 2, crop_b = (coded_h - height) / 2;
  if (crop_r || crop_b) {
    b.u(1, 1);         // conformance_window_flag
    b.ue(0);           // left (units of 2 luma samples)
    b.ue(crop_r);
    b.ue(0);
    b.ue(crop_b);
  } else {
    b.u(0, 1);
  }
  b.ue(0);     // bit_depth_luma_minus8
  b.ue(0);     // bit_depth_chroma_minus8
  b.ue(4);     // log2_max_pic_order_cnt_lsb_minus4
  b.u(1, 1);   // sps_sub_layer_ordering_info_present_flag
  b.ue(0);     // sps_max_dec_pic_buffering_minus1
  b.ue(0);     // sps_max_num_reorder_pics
  b.ue(0);     // sps_max_latency_increase_plus1
  b.ue(0);     // log2_min_luma_coding_block_size_minus3 (8)
  b.ue(1);     // log2_diff_max_min_luma_coding_block_size (CTU 16)
  b.ue(0);     // log2_min_luma_transform_block_size_minus2 (4)
  b.ue(2);     // log2_diff_max_min_luma_transform_block_size (16)
  b.ue(0);     // max_transform_hierarchy_depth_inter
  b.ue(0);     // max_transform_hierarchy_depth_intra
  b.u(0, 1);   // scaling_list_enabled_flag
  b.u(0, 1);   // amp_enabled_flag
  b.u(0, 1);   // sample_adaptive_offset_enabled_flag
  b.u(0, 1);   // pcm_enabled_flag
  b.ue(0);     // num_short_term_ref_pic_sets
  b.u(0, 1);   // long_term_ref_pics_present_flag
  b.u(0, 1);   // sps_temporal_mvp_enabled_flag
  b.u(0, 1);   // strong_intra_smoothing_enabled_flag
  b.u(0, 1);   // vui_parameters_present_flag
  b.u(0, 1);   // sps_extension_present_flag
  b.rbsp_trailing();
  emit_nal(b, out, kNalSps);
}

inline void write_pps_nal(std::vector<uint8_t>& out) {
  BitWriter b;
  b.ue(0);     // pps_pic_parameter_set_id
  b.ue(0);     // pps_seq_parameter_set_id
  b.u(0, 1);   // dependent_slice_segments_enabled_flag
  b.u(0, 1);   // output_flag_present_flag
  b.u(0, 3);   // num_extra_slice_header_bits
  b.u(0, 1);   // sign_data_hiding_enabled_flag
  b.u(0, 1);   // cabac_init_present_flag
  b.ue(0);     // num_ref_idx_l0_default_active_minus1
  b.ue(0);     // num_ref_idx_l1_default_active_minus1
  b.se(0);     // init_qp_minus26
  b.u(0, 1);   // constrained_intra_pred_flag
  b.u(0, 1);   // transform_skip_enabled_flag
  b.u(0, 1);   // cu_qp_delta_enabled_flag
  b.se(0);     // pps_cb_qp_offset
  b.se(0);     // pps_cr_qp_offset
  b.u(0, 1);   // pps_slice_chroma_qp_offsets_present_flag
  b.u(0, 1);   // weighted_pred_flag
  b.u(0, 1);   // weighted_bipred_flag
  b.u(0, 1);   // transquant_bypass_enabled_flag
  b.u(0, 1);   // tiles_enabled_flag
  b.u(0, 1);   // entropy_coding_sync_enabled_flag
  b.u(0, 1);   // pps_loop_filter_across_slices_enabled_flag
  b.u(1, 1);   // deblocking_filter_control_present_flag
  b.u(0, 1);   // deblocking_filter_override_enabled_flag
  b.u(1, 1);   // pps_deblocking_filter_disabled_flag
  b.u(0, 1);   // pps_scaling_list_data_present_flag
  b.u(0, 1);   // lists_modification_present_flag
  b.ue(0);     // log2_parallel_merge_level_minus2
  b.u(0, 1);   // slice_segment_header_extension_present_flag
  b.u(0, 1);   // pps_extension_present_flag
  b.rbsp_trailing();
  emit_nal(b, out, kNalPps);
}

// IDR slice segment header (§7.3.6.1); returns with byte_alignment done so
// CABAC data can be appended directly.
inline void write_slice_header(BitWriter& b, bool first_slice,
                               int slice_address, int addr_bits, int qp) {
  b.u(first_slice ? 1 : 0, 1);  // first_slice_segment_in_pic_flag
  b.u(0, 1);                    // no_output_of_prior_pics_flag (IRAP)
  b.ue(0);                      // slice_pic_parameter_set_id
  if (!first_slice) b.u(slice_address, addr_bits);
  b.ue(2);                      // slice_type: I
  b.se(qp - 26);                // slice_qp_delta (init_qp is 26)
  // deblocking: pps disabled + no override -> nothing to code
  // byte_alignment()
  b.put_bit(1);
  while (b.bit_count() % 8) b.put_bit(0);
}

}  // namespace hevc
}  // namespace hipflux
