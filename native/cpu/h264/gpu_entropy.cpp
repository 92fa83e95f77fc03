#include "gpu_entropy.h"

#include <cstring>

#include "../../hip/h264_gpu_layout.h"
#include "bitwriter.h"
#include "cavlc.h"
#include "headers.h"
#include "transform.h"

namespace hipflux {
namespace h264 {
namespace {

using h264gpu::kChromaAcOff;
using h264gpu::kChromaDcOff;
using h264gpu::kLevelsPerMb;
using h264gpu::kLumaAcOff;
using h264gpu::kLumaDcOff;
using h264gpu::kMetaPerMb;

inline void blk_xy(int blk, int& bx, int& by) {
  bx = 2 * ((blk >> 2) & 1) + (blk & 1);
  by = 2 * (blk >> 3) + ((blk >> 1) & 1);
}

struct RowCtx {
  bool have_left = false;
  bool left_is_inter = false;
  int left_mvx = 0, left_mvy = 0;
  uint8_t left_luma_nc[4] = {};
  uint8_t left_cb_nc[2] = {};
  uint8_t left_cr_nc[2] = {};
  int skip_run = 0;
};

// Encode one intra MB from the GPU level buffer. Mirrors
// encoder.cpp::encode_i16's entropy section exactly.
void entropy_i16(BitWriter& bw, const int16_t* L, int m0, int m1, int qp,
                 bool p_slice, RowCtx& ctx) {
  const int luma_mode = (m0 >> 2) & 7;
  // chroma pipeline reports its mode via meta word 1 (unused for intra)
  const int chroma_mode = m1 & 7;

  // gather zigzag blocks + cbp from levels
  int zz_dc[16];
  for (int i = 0; i < 16; ++i) zz_dc[i] = L[kLumaDcOff + kZigzag4[i]];
  int zz_ac[16][15];
  bool any_ac = false;
  for (int r = 0; r < 16; ++r) {
    for (int i = 1; i < 16; ++i) {
      zz_ac[r][i - 1] = L[kLumaAcOff + r * 16 + kZigzag4[i]];
      any_ac |= zz_ac[r][i - 1] != 0;
    }
  }
  const int cbp_luma = any_ac ? 15 : 0;
  int cdc[2][4];
  bool any_cdc = false;
  for (int comp = 0; comp < 2; ++comp)
    for (int i = 0; i < 4; ++i) {
      cdc[comp][i] = L[kChromaDcOff + comp * 4 + i];
      any_cdc |= cdc[comp][i] != 0;
    }
  int czz[8][15];
  bool any_cac = false;
  for (int b = 0; b < 8; ++b)
    for (int i = 1; i < 16; ++i) {
      czz[b][i - 1] = L[kChromaAcOff + b * 16 + kZigzag4[i]];
      any_cac |= czz[b][i - 1] != 0;
    }
  const int cbp_chroma = any_cac ? 2 : (any_cdc ? 1 : 0);

  const int i16_type =
      1 + luma_mode + 4 * cbp_chroma + 12 * (cbp_luma ? 1 : 0);
  bw.ue(p_slice ? 5 + i16_type : i16_type);
  bw.ue(chroma_mode);
  bw.se(0);  // mb_qp_delta

  uint8_t new_luma[16] = {};
  uint8_t new_cb[4] = {}, new_cr[4] = {};
  // DC block nC = blk(0,0) context
  int nC = ctx.have_left ? ctx.left_luma_nc[0] : 0;
  cavlc_residual(bw, zz_dc, 16, nC);
  if (cbp_luma) {
    for (int blk = 0; blk < 16; ++blk) {
      int bx, by;
      blk_xy(blk, bx, by);
      int r = by * 4 + bx;
      int n;
      if (bx > 0)
        n = new_luma[by * 4 + bx - 1];
      else
        n = ctx.have_left ? ctx.left_luma_nc[by] : 0;
      int tc = cavlc_residual(bw, zz_ac[r], 15, n);
      new_luma[r] = static_cast<uint8_t>(tc);
    }
  }
  if (cbp_chroma > 0) {
    cavlc_residual(bw, cdc[0], 4, -1);
    cavlc_residual(bw, cdc[1], 4, -1);
  }
  if (cbp_chroma == 2) {
    for (int comp = 0; comp < 2; ++comp) {
      uint8_t* t = comp ? new_cr : new_cb;
      const uint8_t* lt = comp ? ctx.left_cr_nc : ctx.left_cb_nc;
      for (int sub = 0; sub < 4; ++sub) {
        int cx = sub & 1, cy = sub >> 1;
        int n = cx > 0 ? t[cy * 2] : (ctx.have_left ? lt[cy] : 0);
        int tc = cavlc_residual(bw, czz[comp * 4 + sub], 15, n);
        t[cy * 2 + cx] = static_cast<uint8_t>(tc);
      }
    }
  }
  ctx.have_left = true;
  ctx.left_is_inter = false;
  for (int by = 0; by < 4; ++by) ctx.left_luma_nc[by] = new_luma[by * 4 + 3];
  for (int cy = 0; cy < 2; ++cy) {
    ctx.left_cb_nc[cy] = new_cb[cy * 2 + 1];
    ctx.left_cr_nc[cy] = new_cr[cy * 2 + 1];
  }
}

// Inter column of Table 9-4 (six reachable values only).
inline int inter_cbp_codenum(int cbp) {
  switch (cbp) {
    case 0: return 0;
    case 16: return 1;
    case 32: return 6;
    case 15: return 11;
    case 47: return 12;
    default: return 19;  // 31
  }
}

// Encode one P_L0_16x16 MB (with residual) from the GPU level buffer.
// Inter luma blocks are full 16-coeff zigzags (no DC Hadamard); mirrors
// encoder.cpp::encode_p16's entropy section exactly.
void entropy_p16(BitWriter& bw, const int16_t* L, int mvx, int mvy,
                 RowCtx& ctx) {
  int zz[16][16];
  bool any_l = false;
  for (int r = 0; r < 16; ++r)
    for (int i = 0; i < 16; ++i) {
      zz[r][i] = L[kLumaAcOff + r * 16 + kZigzag4[i]];
      any_l |= zz[r][i] != 0;
    }
  const int cbp_luma = any_l ? 15 : 0;
  int cdc[2][4];
  bool any_cdc = false;
  for (int comp = 0; comp < 2; ++comp)
    for (int i = 0; i < 4; ++i) {
      cdc[comp][i] = L[kChromaDcOff + comp * 4 + i];
      any_cdc |= cdc[comp][i] != 0;
    }
  int czz[8][15];
  bool any_cac = false;
  for (int b = 0; b < 8; ++b)
    for (int i = 1; i < 16; ++i) {
      czz[b][i - 1] = L[kChromaAcOff + b * 16 + kZigzag4[i]];
      any_cac |= czz[b][i - 1] != 0;
    }
  const int cbp_chroma = any_cac ? 2 : (any_cdc ? 1 : 0);
  const int cbp = (cbp_chroma << 4) | cbp_luma;

  bw.ue(0);  // P_L0_16x16
  int mvpx = (ctx.have_left && ctx.left_is_inter) ? ctx.left_mvx : 0;
  int mvpy = (ctx.have_left && ctx.left_is_inter) ? ctx.left_mvy : 0;
  bw.se(mvx - mvpx);
  bw.se(mvy - mvpy);
  bw.ue(inter_cbp_codenum(cbp));

  uint8_t new_luma[16] = {};
  uint8_t new_cb[4] = {}, new_cr[4] = {};
  if (cbp) {
    bw.se(0);  // mb_qp_delta
    if (cbp_luma) {
      for (int blk = 0; blk < 16; ++blk) {
        int bx, by;
        blk_xy(blk, bx, by);
        int r = by * 4 + bx;
        int n;
        if (bx > 0)
          n = new_luma[by * 4 + bx - 1];
        else
          n = ctx.have_left ? ctx.left_luma_nc[by] : 0;
        int tc = cavlc_residual(bw, zz[r], 16, n);
        new_luma[r] = static_cast<uint8_t>(tc);
      }
    }
    if (cbp_chroma > 0) {
      cavlc_residual(bw, cdc[0], 4, -1);
      cavlc_residual(bw, cdc[1], 4, -1);
    }
    if (cbp_chroma == 2) {
      for (int comp = 0; comp < 2; ++comp) {
        uint8_t* t = comp ? new_cr : new_cb;
        const uint8_t* lt = comp ? ctx.left_cr_nc : ctx.left_cb_nc;
        for (int sub = 0; sub < 4; ++sub) {
          int cx = sub & 1, cy = sub >> 1;
          int n = cx > 0 ? t[cy * 2] : (ctx.have_left ? lt[cy] : 0);
          int tc = cavlc_residual(bw, czz[comp * 4 + sub], 15, n);
          t[cy * 2 + cx] = static_cast<uint8_t>(tc);
        }
      }
    }
  }
  ctx.have_left = true;
  ctx.left_is_inter = true;
  ctx.left_mvx = mvx;
  ctx.left_mvy = mvy;
  for (int by = 0; by < 4; ++by) ctx.left_luma_nc[by] = new_luma[by * 4 + 3];
  for (int cy = 0; cy < 2; ++cy) {
    ctx.left_cb_nc[cy] = new_cb[cy * 2 + 1];
    ctx.left_cr_nc[cy] = new_cr[cy * 2 + 1];
  }
}

}  // namespace

void encode_stripe_from_gpu(const GpuStripeParams& p,
                            std::vector<uint8_t>& out) {
  const int mbh = p.n_mb_rows;
  const int mbw_stripe = (p.width + 15) / 16;
  if (p.idr) {
    write_sps_nal(out, mbw_stripe, mbh, p.width, p.height);
    write_pps_nal(out);
  }
  for (int row = 0; row < mbh; ++row) encode_row_nal_from_gpu(p, row, out);
}

void encode_row_nal_from_gpu(const GpuStripeParams& p, int row,
                             std::vector<uint8_t>& out) {
  encode_seg_nal_from_gpu(p, row, 0, (p.width + 15) / 16, row == 0, out);
}

void encode_seg_nal_from_gpu(const GpuStripeParams& p, int row, int mbx0,
                             int seg_mbw, bool long_startcode,
                             std::vector<uint8_t>& out) {
  const int mbw_stripe = (p.width + 15) / 16;
  {
    BitWriter b;
    write_slice_header_bits(b, p.idr, row * mbw_stripe + mbx0, p.frame_num,
                            p.idr_pic_id, p.qp, p.deblock ? 2 : 1);
    RowCtx ctx;
    const size_t mb_base = (size_t)(p.mb_row0 + row) * p.mbw;
    for (int mbx = mbx0; mbx < mbx0 + seg_mbw; ++mbx) {
      const int16_t* L = p.levels + (mb_base + mbx) * kLevelsPerMb;
      const int* M = p.meta + (mb_base + mbx) * kMetaPerMb;
      int m0 = M[0];
      int m1 = M[1];
      int mode = p.idr ? h264gpu::kIntra : (m0 & 3);
      if (!p.idr && mode == h264gpu::kSkip) {
        ++ctx.skip_run;
        ctx.have_left = true;
        ctx.left_is_inter = true;
        ctx.left_mvx = 0;
        ctx.left_mvy = 0;
        std::fill(ctx.left_luma_nc, ctx.left_luma_nc + 4, 0);
        ctx.left_cb_nc[0] = ctx.left_cb_nc[1] = 0;
        ctx.left_cr_nc[0] = ctx.left_cr_nc[1] = 0;
        continue;
      }
      if (!p.idr) {
        b.ue(ctx.skip_run);
        ctx.skip_run = 0;
      }
      if (!p.idr && mode == h264gpu::kInter) {
        int mvx = (int16_t)(m1 & 0xFFFF);
        int mvy = m1 >> 16;
        entropy_p16(b, L, mvx, mvy, ctx);
        continue;
      }
      entropy_i16(b, L, m0, m1, p.qp, !p.idr, ctx);
    }
    if (!p.idr && ctx.skip_run > 0) b.ue(ctx.skip_run);
    b.rbsp_trailing();
    b.emit_nal(out, p.idr ? 3 : 2, p.idr ? 5 : 1, long_startcode);
  }
}

void assemble_gpu_row_nal(const uint32_t* words, int bits, bool idr,
                          bool long_startcode, std::vector<uint8_t>& out) {
  // big-endian unpack word-wise (the GPU packs MSB-first u32 words).
  // thread_local scratch: a fresh vector here cost a malloc + value-init
  // memset per NAL (~10 KB each, 136 NALs/frame at 1080p)
  int nbytes = (bits + 7) / 8;
  int nwords = nbytes / 4 + 1;
  static thread_local std::vector<uint32_t> swapped;
  if (swapped.size() < static_cast<size_t>(nwords) + 1)
    swapped.resize(nwords + 1);
  swapped[nwords] = 0;
  for (int i = 0; i < nwords; ++i)
    swapped[i] = __builtin_bswap32(words[i]);
  uint8_t* rbsp = reinterpret_cast<uint8_t*>(swapped.data());
  // rbsp stop bit (trailing bits beyond `bits` are zero by construction)
  int stop_byte = bits / 8;
  rbsp[stop_byte] |= 0x80 >> (bits % 8);
  size_t total = static_cast<size_t>(stop_byte) + 1;

  out.reserve(out.size() + total + total / 64 + 8);
  if (long_startcode) out.push_back(0);
  out.push_back(0);
  out.push_back(0);
  out.push_back(1);
  out.push_back(static_cast<uint8_t>(
      idr ? ((3 << 5) | 5) : ((2 << 5) | 1)));
  // emulation prevention: bulk-copy runs between zero bytes (memchr),
  // handle the 00 00 0x escape rule only at the zeros themselves
  size_t p = 0;
  int zeros = 0;
  while (p < total) {
    if (zeros >= 2 && rbsp[p] <= 3) {
      out.push_back(3);
      zeros = 0;
    }
    const void* z = memchr(rbsp + p, 0, total - p);
    size_t zi = z ? static_cast<const uint8_t*>(z) - rbsp : total;
    if (zi > p) {
      out.insert(out.end(), rbsp + p, rbsp + zi);
      zeros = 0;
      p = zi;
      continue;
    }
    out.push_back(0);   // rbsp[p] == 0
    ++zeros;
    ++p;
  }
}

}  // namespace h264
}  // namespace hipflux
