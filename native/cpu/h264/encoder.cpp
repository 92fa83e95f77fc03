#include "encoder.h"

#include <algorithm>
#include <cassert>
#include <cmath>
#include <cstring>

#include "bitwriter.h"
#include "cavlc.h"
#include "deblock.h"
#include "transform.h"

namespace hipflux {
namespace h264 {
namespace {

inline uint8_t clip8(int v) {
  return static_cast<uint8_t>(v < 0 ? 0 : (v > 255 ? 255 : v));
}

// blkIdx (Z-order) -> 4x4 block coords within MB
inline void blk_xy(int blk, int& bx, int& by) {
  bx = 2 * ((blk >> 2) & 1) + (blk & 1);
  by = 2 * (blk >> 3) + ((blk >> 1) & 1);
}

struct Plane {
  std::vector<uint8_t> data;
  int pitch = 0;
  void alloc(int w, int h) {
    pitch = w;
    data.assign(static_cast<size_t>(w) * h, 0);
  }
  uint8_t* row(int y) { return data.data() + static_cast<size_t>(y) * pitch; }
  const uint8_t* row(int y) const {
    return data.data() + static_cast<size_t>(y) * pitch;
  }
};

struct Frame {
  Plane y, cb, cr;
  void alloc(int yw, int yh) {
    y.alloc(yw, yh);
    cb.alloc(yw / 2, yh / 2);
    cr.alloc(yw / 2, yh / 2);
  }
};

// ---- per-row entropy/pred context ----------------------------------------
struct RowCtx {
  bool have_left = false;
  bool left_is_inter = false;
  int left_mvx = 0, left_mvy = 0;  // quarter-pel
  uint8_t left_luma_nc[4] = {0, 0, 0, 0};   // totals of left MB col-3 blocks
  uint8_t left_cb_nc[2] = {0, 0}, left_cr_nc[2] = {0, 0};
  int skip_run = 0;
};

struct MbTotals {
  uint8_t luma[16] = {};
  uint8_t cb[4] = {}, cr[4] = {};
};

// coded_block_pattern me(v) codeNum for Inter prediction (Table 9-4,
// ChromaArrayType=1). We only ever emit cbp_luma in {0,15} and
// cbp_chroma in {0,1,2}, so only these six of the 48 entries are
// reachable; the subset decoder (tests/h264_ref_decoder.py) carries the
// same six and asserts nothing else appears.
inline int inter_cbp_codenum(int cbp) {
  switch (cbp) {
    case 0: return 0;
    case 16: return 1;
    case 32: return 6;
    case 15: return 11;
    case 47: return 12;
    case 31: return 19;
  }
  return -1;  // unreachable
}

}  // namespace

struct StripeEncoder::Impl {
  int w, h, mbw, mbh, yw, yh;
  bool deblock;          // in-loop deblocking (idc=2: within-slice only)
  // >= 0: this instance codes ONE colour plane of a Hi444
  // separate_colour_plane stream as monochrome (ChromaArrayType == 0):
  // no chroma syntax at all, and inter residual falls back to I16x16
  // (whose luma cbp lives in mb_type) so the only coded_block_pattern
  // ever emitted is cbp == 0 -> codeNum 0, identical in every Table 9-4
  // column -- no monochrome cbp table has to be transcribed.
  int colour_plane = -1;
  bool mono() const { return colour_plane >= 0; }
  Frame ref, cur;        // reference (prev recon) and current recon
  Frame src;             // padded source planes
  uint32_t frame_num = 0;
  uint32_t idr_pic_id = 0;
  bool need_idr = true;
  int level_idc = 0;
  std::vector<int> prev_mv;   // per-MB quarter-pel mv of the last frame
  std::vector<uint8_t> prev_bad;  // MB was hopeless-intra last frame

  Impl(int width, int height, bool deblock_on, int plane = -1)
      : w(width), h(height), deblock(deblock_on), colour_plane(plane) {
    mbw = (w + 15) / 16;
    mbh = (h + 15) / 16;
    yw = mbw * 16;
    yh = mbh * 16;
    ref.alloc(yw, yh);
    cur.alloc(yw, yh);
    src.alloc(yw, yh);
    prev_mv.assign(static_cast<size_t>(mbw) * mbh, 0);
    prev_bad.assign(static_cast<size_t>(mbw) * mbh, 0);
    // level from MB count (MaxFS) and 60 fps MaxMBPS, generous
    int fs = mbw * mbh;
    if (fs <= 1620) level_idc = 31;        // <= 480p
    else if (fs <= 3600) level_idc = 32;
    else if (fs <= 8192) level_idc = 42;   // 1080p60
    else if (fs <= 22080) level_idc = 51;  // 4K
    else level_idc = 52;
  }

  // ---- headers ------------------------------------------------------------
  void write_sps(std::vector<uint8_t>& out) {
    BitWriter b;
    b.u(66, 8);   // profile_idc: Baseline
    b.u(1, 1);    // constraint_set0
    b.u(1, 1);    // constraint_set1 (constrained baseline)
    b.u(0, 6);    // constraint_set2..5 + reserved
    b.u(level_idc, 8);
    b.ue(0);      // sps id
    b.ue(12);     // log2_max_frame_num_minus4 -> 16-bit frame_num
    b.ue(2);      // pic_order_cnt_type = 2
    b.ue(1);      // max_num_ref_frames
    b.u(0, 1);    // gaps_in_frame_num_value_allowed
    b.ue(mbw - 1);
    b.ue(mbh - 1);
    b.u(1, 1);    // frame_mbs_only
    b.u(1, 1);    // direct_8x8_inference
    int crop_r = (yw - w) / 2, crop_b = (yh - h) / 2;
    if (crop_r || crop_b) {
      b.u(1, 1);
      b.ue(0);
      b.ue(crop_r);
      b.ue(0);
      b.ue(crop_b);
    } else {
      b.u(0, 1);
    }
    // VUI: declare BT.601 full-range to match our CSC
    b.u(1, 1);    // vui_parameters_present
    b.u(0, 1);    // aspect_ratio_info_present
    b.u(0, 1);    // overscan_info_present
    b.u(1, 1);    // video_signal_type_present
    b.u(5, 3);    // video_format: unspecified
    b.u(1, 1);    // video_full_range_flag
    b.u(1, 1);    // colour_description_present
    b.u(6, 8);    // colour_primaries: SMPTE 170M
    b.u(6, 8);    // transfer_characteristics
    b.u(6, 8);    // matrix_coefficients: BT.601
    b.u(0, 1);    // chroma_loc_info_present
    b.u(0, 1);    // timing_info_present
    b.u(0, 1);    // nal_hrd_parameters_present
    b.u(0, 1);    // vcl_hrd_parameters_present
    b.u(0, 1);    // pic_struct_present
    b.u(0, 1);    // bitstream_restriction
    b.rbsp_trailing();
    b.emit_nal(out, 3, 7);
  }

  void write_pps(std::vector<uint8_t>& out) {
    BitWriter b;
    b.ue(0);      // pps id
    b.ue(0);      // sps id
    b.u(0, 1);    // entropy_coding_mode: CAVLC
    b.u(0, 1);    // bottom_field_pic_order_in_frame_present
    b.ue(0);      // num_slice_groups_minus1
    b.ue(0);      // num_ref_idx_l0_default_active_minus1
    b.ue(0);      // num_ref_idx_l1_default_active_minus1
    b.u(0, 1);    // weighted_pred
    b.u(0, 2);    // weighted_bipred_idc
    b.se(0);      // pic_init_qp_minus26
    b.se(0);      // pic_init_qs_minus26
    b.se(0);      // chroma_qp_index_offset
    b.u(1, 1);    // deblocking_filter_control_present
    b.u(0, 1);    // constrained_intra_pred
    b.u(0, 1);    // redundant_pic_cnt_present
    b.rbsp_trailing();
    b.emit_nal(out, 3, 8);
  }

  void write_slice_header(BitWriter& b, bool idr, int mb_row, int qp) {
    b.ue(mb_row * mbw);          // first_mb_in_slice
    b.ue(idr ? 7 : 5);           // slice_type: I / P (all-slices-same form)
    b.ue(0);                     // pps id
    if (mono()) b.u(colour_plane, 2);   // colour_plane_id
    b.u(frame_num & 0xFFFF, 16); // frame_num
    if (idr) b.ue(idr_pic_id);
    if (!idr) {
      b.u(0, 1);                 // num_ref_idx_active_override
      b.u(0, 1);                 // ref_pic_list_modification_flag_l0
    }
    // dec_ref_pic_marking (nal_ref_idc != 0)
    if (idr) {
      b.u(0, 1);                 // no_output_of_prior_pics
      b.u(0, 1);                 // long_term_reference
    } else {
      b.u(0, 1);                 // adaptive_ref_pic_marking_mode
    }
    b.se(qp - 26);               // slice_qp_delta
    b.ue(deblock ? 2 : 1);       // disable_deblocking_filter_idc
  }

  // ---- intra 16x16 macroblock ---------------------------------------------
  // Encodes src MB at (mbx, mby) as I16x16 into bw; reconstructs into cur.
  // Returns totals for nC bookkeeping. p_slice: emit P-slice mb_type offset.
  void encode_i16(BitWriter& bw, int mbx, int mby, int qp, bool p_slice,
                  RowCtx& ctx, MbTotals& tot) {
    const int x0 = mbx * 16, y0 = mby * 16;
    const int cx0 = mbx * 8, cy0 = mby * 8;
    const bool left = ctx.have_left;

    // ----- luma prediction: H (mode 1) vs DC (mode 2)
    uint8_t predH[256], predDC[256];
    int dc = 128;
    if (left) {
      int s = 0;
      for (int r = 0; r < 16; ++r) s += cur.y.row(y0 + r)[x0 - 1];
      dc = (s + 8) >> 4;
    }
    std::memset(predDC, dc, 256);
    long costH = 1 << 30, costDC = 0;
    if (left) {
      costH = 0;
      for (int r = 0; r < 16; ++r) {
        uint8_t v = cur.y.row(y0 + r)[x0 - 1];
        std::memset(predH + 16 * r, v, 16);
      }
      for (int r = 0; r < 16; ++r) {
        const uint8_t* s = src.y.row(y0 + r) + x0;
        for (int c = 0; c < 16; ++c) {
          costH += std::abs(int(s[c]) - int(predH[16 * r + c]));
        }
      }
    }
    for (int r = 0; r < 16; ++r) {
      const uint8_t* s = src.y.row(y0 + r) + x0;
      for (int c = 0; c < 16; ++c) costDC += std::abs(int(s[c]) - dc);
    }
    const int luma_mode = (left && costH < costDC) ? 1 : 2;
    const uint8_t* pred = luma_mode == 1 ? predH : predDC;

    // ----- luma transform: 16 blocks, DC Hadamard path
    int coef[16][16];  // raster coeffs per block
    int dcs[16];       // dc per block, raster (by*4+bx)
    for (int by = 0; by < 4; ++by)
      for (int bx = 0; bx < 4; ++bx) {
        int resid[16];
        for (int r = 0; r < 4; ++r)
          for (int c = 0; c < 4; ++c) {
            int yy = by * 4 + r, xx = bx * 4 + c;
            resid[4 * r + c] =
                int(src.y.row(y0 + yy)[x0 + xx]) - int(pred[16 * yy + xx]);
          }
        fdct4x4(resid, coef[by * 4 + bx]);
        dcs[by * 4 + bx] = coef[by * 4 + bx][0];
      }
    int had[16];
    hadamard4x4_fwd(dcs, had);
    int qdc_r[16];  // quantized DC, raster
    for (int i = 0; i < 16; ++i) qdc_r[i] = quant_dc(had[i], qp, true);
    int zz_dc[16];
    for (int i = 0; i < 16; ++i) zz_dc[i] = qdc_r[kZigzag4[i]];
    cap_coeffs(zz_dc, 16);
    for (int i = 0; i < 16; ++i) qdc_r[kZigzag4[i]] = zz_dc[i];

    int zz_ac[16][15];
    bool any_ac = false;
    for (int b = 0; b < 16; ++b) {
      for (int i = 1; i < 16; ++i) {
        int pos = kZigzag4[i];
        zz_ac[b][i - 1] = quant_coeff(coef[b][pos], qp,
                                      coeff_class(pos >> 2, pos & 3), true);
      }
      cap_coeffs(zz_ac[b], 15);
      for (int i = 0; i < 15; ++i) any_ac |= zz_ac[b][i] != 0;
    }
    const int cbp_luma = any_ac ? 15 : 0;

    // ----- luma reconstruction
    int ihad[16], dcrec[16];
    hadamard4x4_inv(qdc_r, ihad);
    for (int i = 0; i < 16; ++i) dcrec[i] = dequant_luma_dc(ihad[i], qp);
    for (int by = 0; by < 4; ++by)
      for (int bx = 0; bx < 4; ++bx) {
        int b = by * 4 + bx;
        int dq[16] = {};
        dq[0] = dcrec[b];
        if (cbp_luma) {
          for (int i = 1; i < 16; ++i) {
            int pos = kZigzag4[i];
            dq[pos] = dequant_coeff(zz_ac[b][i - 1], qp,
                                    coeff_class(pos >> 2, pos & 3));
          }
        }
        int rec[16];
        idct4x4(dq, rec);
        for (int r = 0; r < 4; ++r)
          for (int c = 0; c < 4; ++c) {
            int yy = by * 4 + r, xx = bx * 4 + c;
            cur.y.row(y0 + yy)[x0 + xx] =
                clip8(rec[4 * r + c] + pred[16 * yy + xx]);
          }
      }

    // ----- chroma: mode DC (0) vs H (1); shared by Cb/Cr
    const int qpc = chroma_qp(qp);
    int cbp_chroma = 0;
    int chroma_mode = 0;
    int cqdc[2][4] = {};
    int czz_ac[2][4][15] = {};
    if (!mono()) {
    uint8_t cpred[2][64];
    long ccostH = 1 << 30, ccostDC = 0;
    uint8_t cpredH[2][64], cpredDC[2][64];
    for (int comp = 0; comp < 2; ++comp) {
      Plane& rc = comp ? cur.cr : cur.cb;
      Plane& sp = comp ? src.cr : src.cb;
      for (int sub = 0; sub < 4; ++sub) {
        int scx = (sub & 1) * 4, scy = (sub >> 1) * 4;
        int d = 128;
        if (left) {
          int s = 0;
          for (int r = 0; r < 4; ++r) s += rc.row(cy0 + scy + r)[cx0 - 1];
          d = (s + 2) >> 2;
        }
        for (int r = 0; r < 4; ++r)
          std::memset(cpredDC[comp] + 8 * (scy + r) + scx, d, 4);
      }
      if (left)
        for (int r = 0; r < 8; ++r)
          std::memset(cpredH[comp] + 8 * r, rc.row(cy0 + r)[cx0 - 1], 8);
      for (int r = 0; r < 8; ++r)
        for (int c = 0; c < 8; ++c) {
          int s = sp.row(cy0 + r)[cx0 + c];
          ccostDC += std::abs(s - int(cpredDC[comp][8 * r + c]));
          if (left) {
            if (comp == 0 && r == 0 && c == 0) ccostH = 0;
            ccostH += std::abs(s - int(cpredH[comp][8 * r + c]));
          }
        }
    }
    chroma_mode = (left && ccostH < ccostDC) ? 1 : 0;
    std::memcpy(cpred[0], chroma_mode ? cpredH[0] : cpredDC[0], 64);
    std::memcpy(cpred[1], chroma_mode ? cpredH[1] : cpredDC[1], 64);

    int ccoef[2][4][16], cdc[2][4];
    bool c_any_ac = false;
    for (int comp = 0; comp < 2; ++comp) {
      Plane& sp = comp ? src.cr : src.cb;
      for (int sub = 0; sub < 4; ++sub) {
        int scx = (sub & 1) * 4, scy = (sub >> 1) * 4;
        int resid[16];
        for (int r = 0; r < 4; ++r)
          for (int c = 0; c < 4; ++c)
            resid[4 * r + c] = int(sp.row(cy0 + scy + r)[cx0 + scx + c]) -
                               int(cpred[comp][8 * (scy + r) + scx + c]);
        fdct4x4(resid, ccoef[comp][sub]);
        cdc[comp][sub] = ccoef[comp][sub][0];
        for (int i = 1; i < 16; ++i) {
          int pos = kZigzag4[i];
          czz_ac[comp][sub][i - 1] = quant_coeff(
              ccoef[comp][sub][pos], qpc, coeff_class(pos >> 2, pos & 3), true);
        }
        cap_coeffs(czz_ac[comp][sub], 15);
        for (int i = 0; i < 15; ++i) c_any_ac |= czz_ac[comp][sub][i] != 0;
      }
    }
    // 2x2 Hadamard + quant of chroma DC
    bool c_any_dc = false;
    for (int comp = 0; comp < 2; ++comp) {
      int* d = cdc[comp];
      int w0 = d[0] + d[1] + d[2] + d[3];
      int w1 = d[0] - d[1] + d[2] - d[3];
      int w2 = d[0] + d[1] - d[2] - d[3];
      int w3 = d[0] - d[1] - d[2] + d[3];
      cqdc[comp][0] = quant_dc(w0, qpc, true);
      cqdc[comp][1] = quant_dc(w1, qpc, true);
      cqdc[comp][2] = quant_dc(w2, qpc, true);
      cqdc[comp][3] = quant_dc(w3, qpc, true);
      for (int i = 0; i < 4; ++i) c_any_dc |= cqdc[comp][i] != 0;
    }
    cbp_chroma = c_any_ac ? 2 : (c_any_dc ? 1 : 0);

    // ----- chroma recon
    for (int comp = 0; comp < 2; ++comp) {
      Plane& rc = comp ? cur.cr : cur.cb;
      int dcq[4] = {0, 0, 0, 0};
      if (cbp_chroma >= 1) {
        int* q = cqdc[comp];
        int w0 = q[0] + q[1] + q[2] + q[3];
        int w1 = q[0] - q[1] + q[2] - q[3];
        int w2 = q[0] + q[1] - q[2] - q[3];
        int w3 = q[0] - q[1] - q[2] + q[3];
        dcq[0] = dequant_chroma_dc(w0, qpc);
        dcq[1] = dequant_chroma_dc(w1, qpc);
        dcq[2] = dequant_chroma_dc(w2, qpc);
        dcq[3] = dequant_chroma_dc(w3, qpc);
      }
      for (int sub = 0; sub < 4; ++sub) {
        int scx = (sub & 1) * 4, scy = (sub >> 1) * 4;
        int dq[16] = {};
        dq[0] = dcq[sub];
        if (cbp_chroma == 2)
          for (int i = 1; i < 16; ++i) {
            int pos = kZigzag4[i];
            dq[pos] = dequant_coeff(czz_ac[comp][sub][i - 1], qpc,
                                    coeff_class(pos >> 2, pos & 3));
          }
        int rec[16];
        idct4x4(dq, rec);
        for (int r = 0; r < 4; ++r)
          for (int c = 0; c < 4; ++c)
            rc.row(cy0 + scy + r)[cx0 + scx + c] = clip8(
                rec[4 * r + c] + cpred[comp][8 * (scy + r) + scx + c]);
      }
    }
    }  // !mono()

    // ----- entropy
    // I-slice mb_type for I16x16: 1 + predMode + 4*cbp_chroma + 12*cbp_luma01
    // (predMode = Intra16x16PredMode: ours is 1=H or 2=DC).
    // In P slices intra mb_types are offset by 5.
    const int i16_type =
        1 + luma_mode + 4 * cbp_chroma + 12 * (cbp_luma ? 1 : 0);
    bw.ue(p_slice ? 5 + i16_type : i16_type);
    if (!mono()) bw.ue(chroma_mode);   // intra_chroma_pred_mode
    bw.se(0);             // mb_qp_delta
    // DC block: nC from neighboring 4x4 blk(0,0)
    int nC_dc = blk_nc(ctx, tot, 0, true);
    int tc_dc = cavlc_residual(bw, zz_dc, 16, nC_dc);
    (void)tc_dc;
    MbTotals newtot;
    if (cbp_luma) {
      // bitstream block order is Z-order; zz_ac is stored raster (by*4+bx)
      for (int blk = 0; blk < 16; ++blk) {
        int bx, by;
        blk_xy(blk, bx, by);
        int r = by * 4 + bx;
        int nC = luma_nc_for(ctx, newtot, bx, by);
        int tc = cavlc_residual(bw, zz_ac[r], 15, nC);
        newtot.luma[r] = static_cast<uint8_t>(tc);
      }
    }
    if (cbp_chroma > 0) {
      int zz[4];
      for (int comp = 0; comp < 2; ++comp) {
        for (int i = 0; i < 4; ++i) zz[i] = cqdc[comp][i];
        cavlc_residual(bw, zz, 4, -1);
      }
    }
    if (cbp_chroma == 2) {
      for (int comp = 0; comp < 2; ++comp) {
        uint8_t* t = comp ? newtot.cr : newtot.cb;
        for (int sub = 0; sub < 4; ++sub) {
          int cx = sub & 1, cy = sub >> 1;
          int nC = chroma_nc_for(ctx, t, comp, cx, cy);
          int tc = cavlc_residual(bw, czz_ac[comp][sub], 15, nC);
          t[cy * 2 + cx] = static_cast<uint8_t>(tc);
        }
      }
    }
    tot = newtot;
  }

  // nC for luma blk (bx,by): top row always unavailable (slice-per-row);
  // left within MB, or col-3 of left MB.
  int luma_nc_for(const RowCtx& ctx, const MbTotals& cur_tot, int bx,
                  int by) const {
    if (bx > 0) return cur_tot.luma[by * 4 + (bx - 1)];
    if (ctx.have_left) return ctx.left_luma_nc[by];
    return 0;
  }
  int blk_nc(const RowCtx& ctx, const MbTotals& cur_tot, int blk,
             bool /*dc*/) const {
    int bx, by;
    blk_xy(blk, bx, by);
    return luma_nc_for(ctx, cur_tot, bx, by);
  }
  int chroma_nc_for(const RowCtx& ctx, const uint8_t* t, int comp, int cx,
                    int cy) const {
    if (cx > 0) return t[cy * 2];
    if (ctx.have_left)
      return comp ? ctx.left_cr_nc[cy] : ctx.left_cb_nc[cy];
    return 0;
  }

  // ---- P_L0_16x16 with coded residual -------------------------------------
  // Luma residual = src - MC pred, coded as 16 full 4x4 blocks (no DC
  // Hadamard for inter); chroma as 2x2-Hadamard DC + AC. Inter quant
  // rounding (f = 2^qbits/6). MVs are even integers so the chroma MC is
  // an integer copy (mv/2). Mirrors the GPU row kernel's inter path.
  // Returns false (nothing written) when mono() and the quantized luma
  // residual is nonzero -- the caller re-codes the MB as I16x16 so no
  // monochrome coded_block_pattern codeNum other than 0 is ever needed.
  bool encode_p16(BitWriter& bw, int mbx, int mby, int qp, int mvq_x,
                  int mvq_y, RowCtx& ctx, MbTotals& tot) {
    const int x0 = mbx * 16, y0 = mby * 16;
    const int cx0 = mbx * 8, cy0 = mby * 8;
    const int ix = mvq_x >> 2, iy = mvq_y >> 2;
    const int fx = mvq_x & 3, fy = mvq_y & 3;
    const int cix = mvq_x >> 3, ciy = mvq_y >> 3;
    const int cdx = mvq_x & 7, cdy = mvq_y & 7;
    // prediction planes (interpolated once, used for residual AND recon)
    uint8_t pred_y[256];
    uint8_t pred_c[2][64];
    for (int r = 0; r < 16; ++r)
      for (int c = 0; c < 16; ++c)
        pred_y[16 * r + c] = static_cast<uint8_t>(
            luma_pred_px(ref.y, x0 + ix + c, y0 + iy + r, fx, fy));
    if (!mono())
      for (int comp = 0; comp < 2; ++comp) {
        const Plane& rp = comp ? ref.cr : ref.cb;
        for (int r = 0; r < 8; ++r)
          for (int c = 0; c < 8; ++c)
            pred_c[comp][8 * r + c] = static_cast<uint8_t>(
                chroma_pred_px(rp, cx0 + cix + c, cy0 + ciy + r, cdx, cdy));
      }

    // ----- luma: 16 blocks, full 16-coeff zigzag, inter quant
    int zz[16][16];
    bool any_l = false;
    for (int by = 0; by < 4; ++by)
      for (int bx = 0; bx < 4; ++bx) {
        int b = by * 4 + bx;
        int resid[16], coef[16];
        for (int r = 0; r < 4; ++r)
          for (int c = 0; c < 4; ++c) {
            int yy = by * 4 + r, xx = bx * 4 + c;
            resid[4 * r + c] = int(src.y.row(y0 + yy)[x0 + xx]) -
                               int(pred_y[16 * yy + xx]);
          }
        fdct4x4(resid, coef);
        for (int i = 0; i < 16; ++i) {
          int pos = kZigzag4[i];
          zz[b][i] = quant_coeff(coef[pos], qp,
                                 coeff_class(pos >> 2, pos & 3), false);
        }
        cap_coeffs(zz[b], 16);
        for (int i = 0; i < 16; ++i) any_l |= zz[b][i] != 0;
      }
    if (mono() && any_l) return false;
    const int cbp_luma = any_l ? 15 : 0;

    // ----- chroma: DC 2x2 Hadamard + AC, inter quant
    const int qpc = chroma_qp(qp);
    int cdc[2][4], czz[2][4][15], cqdc[2][4] = {};
    bool c_any_ac = false, c_any_dc = false;
    if (!mono())
    for (int comp = 0; comp < 2; ++comp) {
      Plane& sp = comp ? src.cr : src.cb;
      for (int sub = 0; sub < 4; ++sub) {
        int scx = (sub & 1) * 4, scy = (sub >> 1) * 4;
        int resid[16], coef[16];
        for (int r = 0; r < 4; ++r)
          for (int c = 0; c < 4; ++c) {
            int yy = scy + r, xx = scx + c;
            resid[4 * r + c] = int(sp.row(cy0 + yy)[cx0 + xx]) -
                               int(pred_c[comp][8 * yy + xx]);
          }
        fdct4x4(resid, coef);
        cdc[comp][sub] = coef[0];
        for (int i = 1; i < 16; ++i) {
          int pos = kZigzag4[i];
          czz[comp][sub][i - 1] = quant_coeff(
              coef[pos], qpc, coeff_class(pos >> 2, pos & 3), false);
        }
        cap_coeffs(czz[comp][sub], 15);
        for (int i = 0; i < 15; ++i) c_any_ac |= czz[comp][sub][i] != 0;
      }
      int* d = cdc[comp];
      int w0 = d[0] + d[1] + d[2] + d[3], w1 = d[0] - d[1] + d[2] - d[3];
      int w2 = d[0] + d[1] - d[2] - d[3], w3 = d[0] - d[1] - d[2] + d[3];
      cqdc[comp][0] = quant_dc(w0, qpc, false);
      cqdc[comp][1] = quant_dc(w1, qpc, false);
      cqdc[comp][2] = quant_dc(w2, qpc, false);
      cqdc[comp][3] = quant_dc(w3, qpc, false);
      for (int i = 0; i < 4; ++i) c_any_dc |= cqdc[comp][i] != 0;
    }
    const int cbp_chroma = mono() ? 0 : (c_any_ac ? 2 : (c_any_dc ? 1 : 0));
    const int cbp = (cbp_chroma << 4) | cbp_luma;

    // ----- bitstream
    bw.ue(0);  // mb_type P_L0_16x16
    int mvpx = ctx.have_left && ctx.left_is_inter ? ctx.left_mvx : 0;
    int mvpy = ctx.have_left && ctx.left_is_inter ? ctx.left_mvy : 0;
    bw.se(mvq_x - mvpx);
    bw.se(mvq_y - mvpy);
    bw.ue(inter_cbp_codenum(cbp));
    MbTotals newtot;
    if (cbp) {
      bw.se(0);  // mb_qp_delta
      if (cbp_luma) {
        for (int blk = 0; blk < 16; ++blk) {
          int bx, by;
          blk_xy(blk, bx, by);
          int r = by * 4 + bx;
          int nC = luma_nc_for(ctx, newtot, bx, by);
          int tc = cavlc_residual(bw, zz[r], 16, nC);
          newtot.luma[r] = static_cast<uint8_t>(tc);
        }
      }
      if (cbp_chroma > 0) {
        int zdc[4];
        for (int comp = 0; comp < 2; ++comp) {
          for (int i = 0; i < 4; ++i) zdc[i] = cqdc[comp][i];
          cavlc_residual(bw, zdc, 4, -1);
        }
      }
      if (cbp_chroma == 2) {
        for (int comp = 0; comp < 2; ++comp) {
          uint8_t* t = comp ? newtot.cr : newtot.cb;
          for (int sub = 0; sub < 4; ++sub) {
            int cx = sub & 1, cy = sub >> 1;
            int nC = chroma_nc_for(ctx, t, comp, cx, cy);
            int tc = cavlc_residual(bw, czz[comp][sub], 15, nC);
            t[cy * 2 + cx] = static_cast<uint8_t>(tc);
          }
        }
      }
    }
    tot = newtot;

    // ----- recon: MC pred + idct(dequant)
    for (int by = 0; by < 4; ++by)
      for (int bx = 0; bx < 4; ++bx) {
        int b = by * 4 + bx;
        if (!cbp_luma) {
          for (int r = 0; r < 4; ++r) {
            int yy = by * 4 + r, xx = bx * 4;
            std::memcpy(cur.y.row(y0 + yy) + x0 + xx,
                        pred_y + 16 * yy + xx, 4);
          }
          continue;
        }
        int dqb[16] = {};
        for (int i = 0; i < 16; ++i) {
          int pos = kZigzag4[i];
          dqb[pos] =
              dequant_coeff(zz[b][i], qp, coeff_class(pos >> 2, pos & 3));
        }
        int rec[16];
        idct4x4(dqb, rec);
        for (int r = 0; r < 4; ++r)
          for (int c = 0; c < 4; ++c) {
            int yy = by * 4 + r, xx = bx * 4 + c;
            cur.y.row(y0 + yy)[x0 + xx] =
                clip8(rec[4 * r + c] + int(pred_y[16 * yy + xx]));
          }
      }
    if (!mono())
    for (int comp = 0; comp < 2; ++comp) {
      Plane& rc = comp ? cur.cr : cur.cb;
      int dcq[4] = {0, 0, 0, 0};
      if (cbp_chroma >= 1) {
        int* q = cqdc[comp];
        int w0 = q[0] + q[1] + q[2] + q[3], w1 = q[0] - q[1] + q[2] - q[3];
        int w2 = q[0] + q[1] - q[2] - q[3], w3 = q[0] - q[1] - q[2] + q[3];
        dcq[0] = dequant_chroma_dc(w0, qpc);
        dcq[1] = dequant_chroma_dc(w1, qpc);
        dcq[2] = dequant_chroma_dc(w2, qpc);
        dcq[3] = dequant_chroma_dc(w3, qpc);
      }
      for (int sub = 0; sub < 4; ++sub) {
        int scx = (sub & 1) * 4, scy = (sub >> 1) * 4;
        if (cbp_chroma == 0) {
          for (int r = 0; r < 4; ++r) {
            int yy = scy + r, xx = scx;
            std::memcpy(rc.row(cy0 + yy) + cx0 + xx,
                        pred_c[comp] + 8 * yy + xx, 4);
          }
          continue;
        }
        int dqb[16] = {};
        dqb[0] = dcq[sub];
        if (cbp_chroma == 2)
          for (int i = 1; i < 16; ++i) {
            int pos = kZigzag4[i];
            dqb[pos] = dequant_coeff(czz[comp][sub][i - 1], qpc,
                                     coeff_class(pos >> 2, pos & 3));
          }
        int rec[16];
        idct4x4(dqb, rec);
        for (int r = 0; r < 4; ++r)
          for (int c = 0; c < 4; ++c) {
            int yy = scy + r, xx = scx + c;
            rc.row(cy0 + yy)[cx0 + xx] =
                clip8(rec[4 * r + c] + int(pred_c[comp][8 * yy + xx]));
          }
      }
    }
    return true;
  }

  // ---- sub-pel motion compensation (8.4.2.2.1/8.4.2.2.2) ------------------
  // MVs live on the full QUARTER-pel grid: half samples from the 6-tap
  // (1,-5,20,20,-5,1) filter, quarter samples as the spec's rounded
  // averages of the two nearest half/integer samples (Table 8-12);
  // chroma uses the bilinear with eighth-pel weights.
  static inline int tap6(int a, int b, int c, int d, int e, int f) {
    return a - 5 * b + 20 * c + 20 * d - 5 * e + f;
  }

  // unclipped horizontal 6-tap sum at integer (x,y), half offset in x
  inline int hsum6(const Plane& p, int x, int y) const {
    const uint8_t* r = p.row(y);
    return tap6(r[x - 2], r[x - 1], r[x], r[x + 1], r[x + 2], r[x + 3]);
  }

  // half samples at integer base (x,y)
  inline int half_b(const Plane& p, int x, int y) const {   // (x+1/2, y)
    return clip8((hsum6(p, x, y) + 16) >> 5);
  }
  inline int half_h(const Plane& p, int x, int y) const {   // (x, y+1/2)
    int v = tap6(p.row(y - 2)[x], p.row(y - 1)[x], p.row(y)[x],
                 p.row(y + 1)[x], p.row(y + 2)[x], p.row(y + 3)[x]);
    return clip8((v + 16) >> 5);
  }
  inline int half_j(const Plane& p, int x, int y) const {   // (x+1/2,y+1/2)
    int v = tap6(hsum6(p, x, y - 2), hsum6(p, x, y - 1), hsum6(p, x, y),
                 hsum6(p, x, y + 1), hsum6(p, x, y + 2),
                 hsum6(p, x, y + 3));
    return clip8((v + 512) >> 10);
  }

  // predicted luma sample at integer base (x,y), frac (fx,fy) in 0..3
  // (Table 8-12: quarter samples average the two nearest half/integer
  // samples with upward rounding)
  inline int luma_pred_px(const Plane& p, int x, int y, int fx,
                          int fy) const {
    if (fx == 0 && fy == 0) return p.row(y)[x];
    if (fy == 0) {
      if (fx == 2) return half_b(p, x, y);
      int b = half_b(p, x, y);
      int g = fx == 1 ? p.row(y)[x] : p.row(y)[x + 1];
      return (g + b + 1) >> 1;
    }
    if (fx == 0) {
      if (fy == 2) return half_h(p, x, y);
      int hh = half_h(p, x, y);
      int g = fy == 1 ? p.row(y)[x] : p.row(y + 1)[x];
      return (g + hh + 1) >> 1;
    }
    if (fx == 2 && fy == 2) return half_j(p, x, y);
    if (fx == 2) {            // f (fy=1) / q (fy=3): avg of b and j
      int j = half_j(p, x, y);
      int b = fy == 1 ? half_b(p, x, y) : half_b(p, x, y + 1);
      return (b + j + 1) >> 1;
    }
    if (fy == 2) {            // i (fx=1) / k (fx=3): avg of h and j
      int j = half_j(p, x, y);
      int hh = fx == 1 ? half_h(p, x, y) : half_h(p, x + 1, y);
      return (hh + j + 1) >> 1;
    }
    // diagonal quarters: avg of nearest b (above/below) and h (left/right)
    int b = fy == 1 ? half_b(p, x, y) : half_b(p, x, y + 1);
    int hh = fx == 1 ? half_h(p, x, y) : half_h(p, x + 1, y);
    return (b + hh + 1) >> 1;
  }

  // predicted chroma sample: integer base (cx,cy), eighth-pel (dx,dy)
  inline int chroma_pred_px(const Plane& p, int cx, int cy, int dx,
                            int dy) const {
    if (dx == 0 && dy == 0) return p.row(cy)[cx];
    int a = p.row(cy)[cx], b = p.row(cy)[cx + 1];
    int c = p.row(cy + 1)[cx], d = p.row(cy + 1)[cx + 1];
    return ((8 - dx) * (8 - dy) * a + dx * (8 - dy) * b +
            (8 - dx) * dy * c + dx * dy * d + 32) >> 6;
  }

  // A candidate MV (quarter-pel) is usable iff the full interpolation
  // window stays inside the padded plane (uniform margin rule, applied
  // to integer candidates too so refinements around them stay legal).
  inline bool mv_window_ok(int x0, int y0, int mvq_x, int mvq_y) const {
    int ix = mvq_x >> 2, iy = mvq_y >> 2;
    // +20: quarter positions may also read the x+1 / y+1 half sample
    return x0 + ix - 2 >= 0 && x0 + ix + 20 <= yw && y0 + iy - 2 >= 0 &&
           y0 + iy + 20 <= yh;
  }

  // interpolated-SAD of the 16x16 at (x0,y0) for quarter-pel mv
  long sad16_q(int x0, int y0, int mvq_x, int mvq_y) const {
    int ix = mvq_x >> 2, iy = mvq_y >> 2;
    int fx = mvq_x & 3, fy = mvq_y & 3;
    long s = 0;
    for (int r = 0; r < 16; ++r) {
      const uint8_t* sp = src.y.row(y0 + r) + x0;
      for (int c = 0; c < 16; ++c)
        s += std::abs(int(sp[c]) -
                      luma_pred_px(ref.y, x0 + ix + c, y0 + iy + r, fx,
                                   fy));
    }
    return s;
  }

  // ---- P macroblock helpers -----------------------------------------------
  long sad16(const uint8_t* a, int ap, const uint8_t* b, int bp) const {
    long s = 0;
    for (int r = 0; r < 16; ++r) {
      const uint8_t* pa = a + static_cast<size_t>(r) * ap;
      const uint8_t* pb = b + static_cast<size_t>(r) * bp;
      for (int c = 0; c < 16; ++c) s += std::abs(int(pa[c]) - int(pb[c]));
    }
    return s;
  }

  void copy_mb_from_ref(int mbx, int mby, int mvx, int mvy) {
    int x0 = mbx * 16, y0 = mby * 16;
    for (int r = 0; r < 16; ++r)
      std::memcpy(cur.y.row(y0 + r) + x0,
                  ref.y.row(y0 + r + mvy) + x0 + mvx, 16);
    if (mono()) return;
    int cx0 = mbx * 8, cy0 = mby * 8, cmx = mvx / 2, cmy = mvy / 2;
    for (int r = 0; r < 8; ++r) {
      std::memcpy(cur.cb.row(cy0 + r) + cx0,
                  ref.cb.row(cy0 + r + cmy) + cx0 + cmx, 8);
      std::memcpy(cur.cr.row(cy0 + r) + cx0,
                  ref.cr.row(cy0 + r + cmy) + cx0 + cmx, 8);
    }
  }

  // ---- frame encode --------------------------------------------------------
  void encode(const uint8_t* sy, int syp, const uint8_t* scb,
              const uint8_t* scr, int scp, int qp, bool force_idr,
              std::vector<uint8_t>& out, EncodeStats* stats) {
    qp = std::clamp(qp, 0, 51);
    // pad source into aligned planes
    for (int r = 0; r < yh; ++r) {
      int sr = std::min(r, h - 1);
      std::memcpy(src.y.row(r), sy + static_cast<size_t>(sr) * syp, w);
      uint8_t e = src.y.row(r)[w - 1];
      std::memset(src.y.row(r) + w, e, yw - w);
    }
    if (!mono()) {
      int cw = (w + 1) / 2, ch = (h + 1) / 2, cwp = yw / 2, chp = yh / 2;
      for (int r = 0; r < chp; ++r) {
        int sr = std::min(r, ch - 1);
        std::memcpy(src.cb.row(r), scb + static_cast<size_t>(sr) * scp, cw);
        std::memcpy(src.cr.row(r), scr + static_cast<size_t>(sr) * scp, cw);
        std::memset(src.cb.row(r) + cw, src.cb.row(r)[cw - 1], cwp - cw);
        std::memset(src.cr.row(r) + cw, src.cr.row(r)[cw - 1], cwp - cw);
      }
    }

    bool idr = force_idr || need_idr;
    if (idr) {
      frame_num = 0;
      if (!mono()) {             // Hi444 wrapper writes the shared headers
        write_sps(out);
        write_pps(out);
      }
      ++idr_pic_id;
      need_idr = false;
    }
    if (stats) {
      *stats = EncodeStats{};
      stats->frame_qp = qp;
      stats->is_idr = idr;
    }

    const long skip_thresh = 48L << (qp / 6);       // tuned-for-screen default
    // With coded inter residuals the MC only has to be a decent predictor,
    // not a near-match: fall back to intra only when even the best MV
    // leaves mostly-uncorrelated content (fresh content / occlusion).
    const long inter_thresh = 6 * skip_thresh;

    std::vector<DbMb> rowmb(mbw);
    for (int mb_row = 0; mb_row < mbh; ++mb_row) {
      BitWriter b;
      write_slice_header(b, idr, mb_row, qp);
      RowCtx ctx;
      MbTotals tot;
      for (int mbx = 0; mbx < mbw; ++mbx) {
        rowmb[mbx] = DbMb{};
        if (idr) {
          rowmb[mbx].intra = 1;
          encode_i16(b, mbx, mb_row, qp, false, ctx, tot);
          ctx.have_left = true;
          ctx.left_is_inter = false;
          for (int by = 0; by < 4; ++by)
            ctx.left_luma_nc[by] = tot.luma[by * 4 + 3];
          for (int cy = 0; cy < 2; ++cy) {
            ctx.left_cb_nc[cy] = tot.cb[cy * 2 + 1];
            ctx.left_cr_nc[cy] = tot.cr[cy * 2 + 1];
          }
          if (stats) ++stats->mb_intra;
          continue;
        }
        // ---- P slice decisions
        int x0 = mbx * 16, y0 = mb_row * 16;
        long sad0 = sad16(src.y.row(y0) + x0, src.y.pitch,
                          ref.y.row(y0) + x0, ref.y.pitch);
        if (sad0 <= skip_thresh) {
          // P_Skip (skip MV is always (0,0) here: top neighbor is in
          // another slice, which forces the zero-MV rule)
          prev_mv[static_cast<size_t>(mb_row) * mbw + mbx] = 0;
          copy_mb_from_ref(mbx, mb_row, 0, 0);
          ++ctx.skip_run;
          ctx.have_left = true;
          ctx.left_is_inter = true;
          ctx.left_mvx = 0;
          ctx.left_mvy = 0;
          std::memset(ctx.left_luma_nc, 0, 4);
          std::memset(ctx.left_cb_nc, 0, 2);
          std::memset(ctx.left_cr_nc, 0, 2);
          if (stats) ++stats->mb_skip;
          continue;
        }
        // integer diamond search seeded at (0,0) and at the previous
        // frame's MV for this MB (tracks sustained motion); a coarse
        // +-16 step-4 scan acquires fast new motion when the local walk
        // fails, followed by another walk
        int best_mvx = 0, best_mvy = 0;
        long best = sad0;
        const size_t mb_idx = static_cast<size_t>(mb_row) * mbw + mbx;
        {
          int pm = prev_mv[mb_idx];
          int pvx = (static_cast<int16_t>(pm & 0xFFFF)) >> 2;
          int pvy = (pm >> 16) >> 2;
          pvx = std::clamp(pvx, -48, 48);
          pvy = std::clamp(pvy, -48, 48);
          if ((pvx | pvy) != 0 && !prev_bad[mb_idx]) {
            // fine grid around the hint (mirrors the MFMA kernel's
            // predictor pass): acquisition beyond the local walk
            for (int dy = -8; dy <= 8; dy += 2)
              for (int dx = -8; dx <= 8; dx += 2) {
                int cx2 = pvx + dx, cy2 = pvy + dy;
                if (!mv_window_ok(x0, y0, cx2 * 4, cy2 * 4)) continue;
                long s = sad16(src.y.row(y0) + x0, src.y.pitch,
                               ref.y.row(y0 + cy2) + x0 + cx2,
                               ref.y.pitch);
                if (s < best) {
                  best = s;
                  best_mvx = cx2;
                  best_mvy = cy2;
                }
              }
          }
        }
        static const int pat[8][2] = {{-1, 0}, {1, 0},  {0, -1}, {0, 1},
                                      {-1, -1}, {1, 1}, {-1, 1}, {1, -1}};
        auto walk = [&]() {
          for (int iter = 0; iter < 16; ++iter) {
            int bmx = best_mvx, bmy = best_mvy;
            bool improved = false;
            for (auto& p : pat) {
              int mx = bmx + p[0], my = bmy + p[1];
              if (std::abs(mx) > 48 || std::abs(my) > 48) continue;
              if (!(mx == 0 && my == 0) &&
                  !mv_window_ok(x0, y0, mx * 4, my * 4))
                continue;
              long s = sad16(src.y.row(y0) + x0, src.y.pitch,
                             ref.y.row(y0 + my) + x0 + mx, ref.y.pitch);
              if (s < best) {
                best = s;
                best_mvx = mx;
                best_mvy = my;
                improved = true;
              }
            }
            if (!improved) break;
          }
        };
        walk();
        if (best > 2 * skip_thresh &&
            !(prev_bad[mb_idx] && (frame_num & 7) != 0)) {
          // coarse acquisition over +-16 whenever the local match is
          // mediocre (a permissive accept must not mask real motion),
          // then walk from the best cell
          for (int cy = -16; cy <= 16; cy += 4)
            for (int cx = -16; cx <= 16; cx += 4) {
              if ((cx | cy) == 0) continue;
              if (!mv_window_ok(x0, y0, cx * 4, cy * 4)) continue;
              long s = sad16(src.y.row(y0) + x0, src.y.pitch,
                             ref.y.row(y0 + cy) + x0 + cx, ref.y.pitch);
              if (s < best) {
                best = s;
                best_mvx = cx;
                best_mvy = cy;
              }
            }
          walk();
        }
        // half- then quarter-pel refinement rings; skipped when the MB
        // is headed to intra regardless
        int best_q_x = best_mvx * 4, best_q_y = best_mvy * 4;
        if (best <= 2 * inter_thresh) {
          for (int step = 2; step >= 1; --step) {
            int cqx = best_q_x, cqy = best_q_y;
            for (auto& p : pat) {
              int qx = cqx + step * p[0], qy = cqy + step * p[1];
              if (!mv_window_ok(x0, y0, qx, qy)) continue;
              long s = sad16_q(x0, y0, qx, qy);
              if (s < best) {
                best = s;
                best_q_x = qx;
                best_q_y = qy;
              }
            }
          }
        }
        prev_bad[mb_idx] = best > 3 * inter_thresh ? 1 : 0;
        bool wrote_inter = false;
        if (best <= inter_thresh) {
          // P_L0_16x16 with coded residual (cbp may still come out 0);
          // in mono() a nonzero residual falls through to I16x16
          prev_mv[mb_idx] =
              (best_q_x & 0xFFFF) |
              static_cast<int>(static_cast<uint32_t>(best_q_y) << 16);
          flush_skip_run(b, ctx);
          wrote_inter =
              encode_p16(b, mbx, mb_row, qp, best_q_x, best_q_y, ctx, tot);
          if (wrote_inter) {
            rowmb[mbx].mvx = static_cast<int16_t>(best_q_x);
            rowmb[mbx].mvy = static_cast<int16_t>(best_q_y);
            for (int bi = 0; bi < 16; ++bi)
              if (tot.luma[bi]) rowmb[mbx].nz |= 1u << bi;
            ctx.have_left = true;
            ctx.left_is_inter = true;
            ctx.left_mvx = best_q_x;
            ctx.left_mvy = best_q_y;
            for (int by = 0; by < 4; ++by)
              ctx.left_luma_nc[by] = tot.luma[by * 4 + 3];
            for (int cy = 0; cy < 2; ++cy) {
              ctx.left_cb_nc[cy] = tot.cb[cy * 2 + 1];
              ctx.left_cr_nc[cy] = tot.cr[cy * 2 + 1];
            }
            if (stats) ++stats->mb_inter;
          }
        }
        if (!wrote_inter) {
          if (best > inter_thresh) {
            // keep the best-found mv as a tracking hint across the intra
            // fallback (fast new motion locks on within a frame or two)
            prev_mv[mb_idx] = ((best_mvx * 4) & 0xFFFF) |
                              static_cast<int>(
                                  static_cast<uint32_t>(best_mvy * 4) << 16);
            flush_skip_run(b, ctx);
          }
          rowmb[mbx].intra = 1;
          encode_i16(b, mbx, mb_row, qp, true, ctx, tot);
          ctx.have_left = true;
          ctx.left_is_inter = false;
          for (int by = 0; by < 4; ++by)
            ctx.left_luma_nc[by] = tot.luma[by * 4 + 3];
          for (int cy = 0; cy < 2; ++cy) {
            ctx.left_cb_nc[cy] = tot.cb[cy * 2 + 1];
            ctx.left_cr_nc[cy] = tot.cr[cy * 2 + 1];
          }
          if (stats) ++stats->mb_intra;
        }
      }
      if (!idr && ctx.skip_run > 0) {
        b.ue(ctx.skip_run);  // trailing skip run
        ctx.skip_run = 0;
      }
      b.rbsp_trailing();
      b.emit_nal(out, idr ? 3 : 2, idr ? 5 : 1, mb_row == 0);
      if (deblock)
        deblock_segment(cur.y.data.data(), cur.y.pitch,
                        mono() ? nullptr : cur.cb.data.data(),
                        mono() ? nullptr : cur.cr.data.data(), cur.cb.pitch,
                        mb_row, 0, mbw, rowmb.data(), qp, chroma_qp(qp));
    }

    std::swap(ref, cur);
    ++frame_num;
    if (stats) stats->bytes = out.size();
  }

  void flush_skip_run(BitWriter& b, RowCtx& ctx) {
    b.ue(ctx.skip_run);
    ctx.skip_run = 0;
  }
};

namespace {

// SPS for High 4:4:4 Predictive with separate colour planes: each plane
// decodes as a monochrome picture (ChromaArrayType == 0), so CropUnitX/Y
// are 1 (no /2 on the crop amounts, unlike 4:2:0).
void write_sps_444(std::vector<uint8_t>& out, int mbw, int mbh, int w,
                   int h, int level_idc) {
  BitWriter b;
  b.u(244, 8);  // profile_idc: High 4:4:4 Predictive
  b.u(0, 8);    // no constraint flags
  b.u(level_idc, 8);
  b.ue(0);      // sps id
  b.ue(3);      // chroma_format_idc: 4:4:4
  b.u(1, 1);    // separate_colour_plane_flag
  b.ue(0);      // bit_depth_luma_minus8
  b.ue(0);      // bit_depth_chroma_minus8
  b.u(0, 1);    // qpprime_y_zero_transform_bypass
  b.u(0, 1);    // seq_scaling_matrix_present
  b.ue(12);     // log2_max_frame_num_minus4 -> 16-bit frame_num
  b.ue(2);      // pic_order_cnt_type = 2
  b.ue(1);      // max_num_ref_frames
  b.u(0, 1);    // gaps_in_frame_num_value_allowed
  b.ue(mbw - 1);
  b.ue(mbh - 1);
  b.u(1, 1);    // frame_mbs_only
  b.u(1, 1);    // direct_8x8_inference
  int crop_r = mbw * 16 - w, crop_b = mbh * 16 - h;
  if (crop_r || crop_b) {
    b.u(1, 1);
    b.ue(0);
    b.ue(crop_r);
    b.ue(0);
    b.ue(crop_b);
  } else {
    b.u(0, 1);
  }
  // VUI: BT.601 full range, as in the 4:2:0 SPS
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

}  // namespace

StripeEncoder::StripeEncoder(int width, int height, bool deblock,
                             bool fullcolor)
    : impl_(new Impl(width, height, deblock, fullcolor ? 0 : -1)),
      width_(width), height_(height), fullcolor_(fullcolor) {
  if (fullcolor) {
    impl_cb_.reset(new Impl(width, height, deblock, 1));
    impl_cr_.reset(new Impl(width, height, deblock, 2));
  }
}
StripeEncoder::~StripeEncoder() = default;

void StripeEncoder::encode_frame(const uint8_t* y, int ypitch,
                                 const uint8_t* cb, const uint8_t* cr,
                                 int cpitch, int qp, bool force_idr,
                                 std::vector<uint8_t>& out,
                                 EncodeStats* stats) {
  if (!fullcolor_) {
    impl_->encode(y, ypitch, cb, cr, cpitch, qp, force_idr, out, stats);
    return;
  }
  // Hi444 separate planes: shared SPS/PPS, then the three planes'
  // slices (plane-major) forming ONE access unit per frame.
  bool idr = force_idr || impl_->need_idr;
  if (idr) {
    write_sps_444(out, impl_->mbw, impl_->mbh, width_, height_,
                  impl_->level_idc);
    impl_->write_pps(out);
  }
  EncodeStats s0, s1, s2;
  impl_->encode(y, ypitch, nullptr, nullptr, 0, qp, idr, out, &s0);
  impl_cb_->encode(cb, cpitch, nullptr, nullptr, 0, qp, idr, out, &s1);
  impl_cr_->encode(cr, cpitch, nullptr, nullptr, 0, qp, idr, out, &s2);
  if (stats) {
    *stats = s0;
    stats->mb_intra = s0.mb_intra + s1.mb_intra + s2.mb_intra;
    stats->mb_inter = s0.mb_inter + s1.mb_inter + s2.mb_inter;
    stats->mb_skip = s0.mb_skip + s1.mb_skip + s2.mb_skip;
    stats->bytes = out.size();
  }
}

const uint8_t* StripeEncoder::recon_y() const { return impl_->ref.y.data.data(); }
const uint8_t* StripeEncoder::recon_cb() const {
  return fullcolor_ ? impl_cb_->ref.y.data.data()
                    : impl_->ref.cb.data.data();
}
const uint8_t* StripeEncoder::recon_cr() const {
  return fullcolor_ ? impl_cr_->ref.y.data.data()
                    : impl_->ref.cr.data.data();
}
int StripeEncoder::recon_ypitch() const { return impl_->ref.y.pitch; }
int StripeEncoder::recon_cpitch() const {
  return fullcolor_ ? impl_cb_->ref.y.pitch : impl_->ref.cb.pitch;
}

void bgrx_to_yuv420(const uint8_t* bgrx, int stride, int width, int height,
                    uint8_t* y, int ypitch, uint8_t* cb, uint8_t* cr,
                    int cpitch) {
  auto clampf = [](float v) {
    return static_cast<uint8_t>(std::lrintf(std::min(std::max(v, 0.f), 255.f)));
  };
  int cw = (width + 1) / 2, chh = (height + 1) / 2;
  for (int cy = 0; cy < chh; ++cy)
    for (int cx = 0; cx < cw; ++cx) {
      float cbs = 0, crs = 0;
      for (int dy = 0; dy < 2; ++dy)
        for (int dx = 0; dx < 2; ++dx) {
          int xx = std::min(cx * 2 + dx, width - 1);
          int yy = std::min(cy * 2 + dy, height - 1);
          const uint8_t* p = bgrx + static_cast<size_t>(yy) * stride + xx * 4;
          float r = p[2], g = p[1], b = p[0];
          float yv = 0.299f * r + 0.587f * g + 0.114f * b;
          cbs += -0.168736f * r - 0.331264f * g + 0.5f * b + 128.f;
          crs += 0.5f * r - 0.418688f * g - 0.081312f * b + 128.f;
          if (cy * 2 + dy < height && cx * 2 + dx < width)
            y[static_cast<size_t>(cy * 2 + dy) * ypitch + cx * 2 + dx] =
                clampf(yv);
        }
      cb[static_cast<size_t>(cy) * cpitch + cx] = clampf(cbs * 0.25f);
      cr[static_cast<size_t>(cy) * cpitch + cx] = clampf(crs * 0.25f);
    }
}

void bgrx_to_yuv444(const uint8_t* bgrx, int stride, int width, int height,
                    uint8_t* y, uint8_t* cb, uint8_t* cr, int pitch) {
  auto clampf = [](float v) {
    return static_cast<uint8_t>(std::lrintf(std::min(std::max(v, 0.f), 255.f)));
  };
  for (int yy = 0; yy < height; ++yy) {
    const uint8_t* rowp = bgrx + static_cast<size_t>(yy) * stride;
    uint8_t* yr = y + static_cast<size_t>(yy) * pitch;
    uint8_t* cbr = cb + static_cast<size_t>(yy) * pitch;
    uint8_t* crr = cr + static_cast<size_t>(yy) * pitch;
    for (int xx = 0; xx < width; ++xx) {
      const uint8_t* p = rowp + xx * 4;
      float r = p[2], g = p[1], b = p[0];
      yr[xx] = clampf(0.299f * r + 0.587f * g + 0.114f * b);
      cbr[xx] = clampf(-0.168736f * r - 0.331264f * g + 0.5f * b + 128.f);
      crr[xx] = clampf(0.5f * r - 0.418688f * g - 0.081312f * b + 128.f);
    }
  }
}

}  // namespace h264
}  // namespace hipflux
