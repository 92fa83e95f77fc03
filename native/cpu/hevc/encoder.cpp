// HEVC stripe encoder — CPU reference (see encoder.h for the design).
#include "encoder.h"

#include <algorithm>
#include <climits>
#include <cstring>

#include "cabac.h"
#include "headers.h"
#include "intra.h"
#include "tables.h"
#include "transform.h"

namespace hipflux {
namespace hevc {

namespace {

inline int ceil_div(int a, int b) { return (a + b - 1) / b; }

inline uint8_t clip8(int v) {
  return static_cast<uint8_t>(v < 0 ? 0 : v > 255 ? 255 : v);
}

// One transform block's coded data.
template <int N>
struct TbData {
  int16_t level[N * N];
  bool cbf = false;
};

// ---- residual_coding (§7.3.8.11) for one TB ------------------------------
// N = 16 (luma) or 8 (chroma). Levels indexed [y*N + x].
template <int N>
void code_residual(CabacEncoder& cab, ContextBank& bank,
                   const int16_t* level, int cidx) {
  constexpr int kSbW = N / 4;  // sub-blocks per side
  const int log2n = N == 8 ? 3 : 4;
  const ScanPos* sb_scan = (N == 8) ? kDiagScan2 : kDiagScan4;
  const int n_sb = kSbW * kSbW;

  // locate last significant coefficient in forward scan order
  int last_sb = -1, last_pos = -1, last_x = 0, last_y = 0;
  for (int i = n_sb - 1; i >= 0 && last_sb < 0; --i) {
    int sx = sb_scan[i].x * 4, sy = sb_scan[i].y * 4;
    for (int n = 15; n >= 0; --n) {
      int x = sx + kDiagScan4[n].x, y = sy + kDiagScan4[n].y;
      if (level[y * N + x]) {
        last_sb = i;
        last_pos = n;
        last_x = x;
        last_y = y;
        break;
      }
    }
  }
  // caller guarantees cbf => at least one nonzero
  // ---- last_sig_coeff_{x,y}_prefix / suffix ----
  int ctx_off, ctx_shift;
  if (cidx == 0) {
    ctx_off = 3 * (log2n - 2) + ((log2n - 1) >> 2);
    ctx_shift = (log2n + 1) >> 2;
  } else {
    ctx_off = 15;
    ctx_shift = log2n - 2;
  }
  const int g_max = (log2n << 1) - 1;
  // groupIdx maps a coordinate to its TR prefix (inverse of
  // minCoord(prefix) = prefix < 4 ? prefix
  //                    : (2 + (prefix & 1)) << ((prefix >> 1) - 1))
  auto group_idx = [](int v) {
    int g = 0;
    while (true) {
      int base = g < 4 ? g : (2 + (g & 1)) << ((g >> 1) - 1);
      int next = (g + 1) < 4 ? (g + 1) : (2 + ((g + 1) & 1)) << (((g + 1) >> 1) - 1);
      if (v >= base && v < next) return g;
      ++g;
    }
  };
  int px = group_idx(last_x), py = group_idx(last_y);
  for (int b = 0; b < std::min(px, g_max); ++b)
    cab.encode_bin(bank.ctx[kCtxLastSigX + ctx_off + (b >> ctx_shift)], 1);
  if (px < g_max)
    cab.encode_bin(bank.ctx[kCtxLastSigX + ctx_off + (px >> ctx_shift)], 0);
  for (int b = 0; b < std::min(py, g_max); ++b)
    cab.encode_bin(bank.ctx[kCtxLastSigY + ctx_off + (b >> ctx_shift)], 1);
  if (py < g_max)
    cab.encode_bin(bank.ctx[kCtxLastSigY + ctx_off + (py >> ctx_shift)], 0);
  if (px > 3) {
    int nbits = (px >> 1) - 1;
    int base = (2 + (px & 1)) << nbits;
    cab.encode_bypass_bins(last_x - base, nbits);
  }
  if (py > 3) {
    int nbits = (py >> 1) - 1;
    int base = (2 + (py & 1)) << nbits;
    cab.encode_bypass_bins(last_y - base, nbits);
  }

  // sub-block significance map
  bool csbf[16] = {false};
  for (int i = 0; i <= last_sb; ++i) {
    int sx = sb_scan[i].x * 4, sy = sb_scan[i].y * 4;
    for (int n = 0; n < 16; ++n) {
      int x = sx + kDiagScan4[n].x, y = sy + kDiagScan4[n].y;
      if (level[y * N + x]) {
        csbf[i] = true;
        break;
      }
    }
  }

  int prev_g1_zero = -1;  // greater1Ctx at end of previously coded subset
  for (int i = last_sb; i >= 0; --i) {
    const int sx = sb_scan[i].x * 4, sy = sb_scan[i].y * 4;
    bool coded_flag_explicit = false;
    if (i < last_sb && i > 0) {
      // csbf context: right / below neighbor sub-blocks
      int right = 0, below = 0;
      for (int j = 0; j < n_sb; ++j) {
        if (sb_scan[j].x == sb_scan[i].x + 1 && sb_scan[j].y == sb_scan[i].y)
          right = csbf[j];
        if (sb_scan[j].x == sb_scan[i].x && sb_scan[j].y == sb_scan[i].y + 1)
          below = csbf[j];
      }
      int ctx = std::min(1, right + below) + (cidx ? 2 : 0);
      cab.encode_bin(bank.ctx[kCtxCodedSubBlock + ctx], csbf[i] ? 1 : 0);
      coded_flag_explicit = true;
    }
    if (!csbf[i] && i != last_sb && i != 0) continue;
    bool infer_dc_sig = coded_flag_explicit;

    // significance flags (reverse scan inside the sub-block)
    int sig_pos[16], n_sig = 0;  // scan positions with nonzero, rev order
    int start = (i == last_sb) ? last_pos - 1 : 15;
    if (i == last_sb) sig_pos[n_sig++] = last_pos;
    for (int n = start; n >= 0; --n) {
      int x = sx + kDiagScan4[n].x, y = sy + kDiagScan4[n].y;
      int sig = level[y * N + x] != 0;
      if (n > 0 || !infer_dc_sig) {
        int sig_ctx;
        if (x == 0 && y == 0 && sx == 0 && sy == 0) {
          sig_ctx = 0;
        } else {
          int right = 0, below = 0;
          for (int j = 0; j < n_sb; ++j) {
            if (sb_scan[j].x == sb_scan[i].x + 1 &&
                sb_scan[j].y == sb_scan[i].y)
              right = csbf[j];
            if (sb_scan[j].x == sb_scan[i].x &&
                sb_scan[j].y == sb_scan[i].y + 1)
              below = csbf[j];
          }
          int prev = right + (below << 1);
          int xp = x & 3, yp = y & 3;
          if (prev == 0)
            sig_ctx = (xp + yp == 0) ? 2 : (xp + yp < 3) ? 1 : 0;
          else if (prev == 1)
            sig_ctx = (yp == 0) ? 2 : (yp == 1) ? 1 : 0;
          else if (prev == 2)
            sig_ctx = (xp == 0) ? 2 : (xp == 1) ? 1 : 0;
          else
            sig_ctx = 2;
          if (cidx == 0) {
            if ((x >> 2) + (y >> 2) > 0) sig_ctx += 3;
            sig_ctx += (log2n == 3) ? 9 : 21;
          } else {
            sig_ctx += (log2n == 3) ? 9 : 12;
          }
        }
        int ctx = (cidx == 0 ? 0 : 27) + sig_ctx;
        cab.encode_bin(bank.ctx[kCtxSigCoeff + ctx], sig);
        if (sig) infer_dc_sig = false;
      }
      if (sig) sig_pos[n_sig++] = n;
    }
    if (n_sig == 0) continue;

    // greater1 / greater2 flags
    int ctx_set = (i == 0 || cidx > 0) ? 0 : 2;
    if (i != last_sb && prev_g1_zero == 1) ctx_set += 1;
    int g1_ctx = 1;
    int first_g1 = -1;  // scan pos of first coeff (rev order) with g1 = 1
    int abs_lvl[16];
    for (int k = 0; k < n_sig; ++k) {
      int n = sig_pos[k];
      int x = sx + kDiagScan4[n].x, y = sy + kDiagScan4[n].y;
      int l = level[y * N + x];
      abs_lvl[k] = l < 0 ? -l : l;
    }
    for (int k = 0; k < n_sig && k < 8; ++k) {
      int g1 = abs_lvl[k] > 1;
      int ctx = ctx_set * 4 + std::min(3, g1_ctx) + (cidx ? 16 : 0);
      cab.encode_bin(bank.ctx[kCtxGreater1 + ctx], g1);
      if (g1) {
        g1_ctx = 0;
        if (first_g1 < 0) first_g1 = k;
      } else if (g1_ctx > 0 && g1_ctx < 3) {
        ++g1_ctx;
      }
    }
    prev_g1_zero = (g1_ctx == 0) ? 1 : 0;
    if (first_g1 >= 0) {
      int g2 = abs_lvl[first_g1] > 2;
      cab.encode_bin(bank.ctx[kCtxGreater2 + ctx_set + (cidx ? 4 : 0)], g2);
    }
    // signs (no sign data hiding)
    for (int k = 0; k < n_sig; ++k) {
      int n = sig_pos[k];
      int x = sx + kDiagScan4[n].x, y = sy + kDiagScan4[n].y;
      cab.encode_bypass(level[y * N + x] < 0 ? 1 : 0);
    }
    // remaining levels (Rice / exp-Golomb, §9.3.3.13)
    int rice = 0;
    for (int k = 0; k < n_sig; ++k) {
      int base = (k < 8) ? ((k == first_g1) ? 3 : 2) : 1;
      if (abs_lvl[k] >= base) {
        uint32_t rem = abs_lvl[k] - base;
        if (rem < (3u << rice)) {
          int len = rem >> rice;
          for (int b = 0; b < len; ++b) cab.encode_bypass(1);
          cab.encode_bypass(0);
          if (rice) cab.encode_bypass_bins(rem & ((1 << rice) - 1), rice);
        } else {
          int len = rice;
          uint32_t v = rem - (3u << rice);
          while (v >= (1u << len)) {
            v -= 1u << len;
            ++len;
          }
          for (int b = 0; b < 3 + len - rice; ++b) cab.encode_bypass(1);
          cab.encode_bypass(0);
          cab.encode_bypass_bins(v, len);
        }
        if (abs_lvl[k] > (3 << rice) && rice < 4) ++rice;
      }
    }
  }
}

}  // namespace

// ---- encoder impl --------------------------------------------------------

struct StripeEncoder::Impl {
  int w, h;            // visible
  int cw, ch;          // coded (16-aligned)
  int ctb_w, ctb_h;    // CTUs per row / rows
  int slices_per_row;
  int addr_bits;
  std::vector<uint8_t> ry, rcb, rcr;        // recon planes (coded dims)
  std::vector<uint8_t> sy, scb, scr;        // padded source planes
  std::vector<uint8_t> luma_mode;           // per-CTU chosen mode
  std::vector<uint8_t> ps_cache;            // VPS/SPS/PPS bytes
  int ypitch, cpitch;

  Impl(int width, int height, int spr)
      : w(width), h(height), cw((width + 15) & ~15), ch((height + 15) & ~15),
        ctb_w(cw / 16), ctb_h(ch / 16),
        slices_per_row(std::max(1, std::min(spr, cw / 16))) {
    int n_ctb = ctb_w * ctb_h;
    addr_bits = 1;
    while ((1 << addr_bits) < n_ctb) ++addr_bits;
    ypitch = cw;
    cpitch = cw / 2;
    ry.resize(static_cast<size_t>(ypitch) * ch);
    rcb.resize(static_cast<size_t>(cpitch) * ch / 2);
    rcr.resize(static_cast<size_t>(cpitch) * ch / 2);
    sy = ry;
    scb = rcb;
    scr = rcr;
    luma_mode.resize(n_ctb);
    write_vps_nal(ps_cache, level_idc_for(cw, ch));
    write_sps_nal(ps_cache, cw, ch, w, h);
    write_pps_nal(ps_cache);
  }

  // pad source into coded-size planes with edge replication
  void load_source(const uint8_t* y, int ypitch_in, const uint8_t* cb,
                   const uint8_t* cr, int cpitch_in) {
    for (int r = 0; r < ch; ++r) {
      int sr = std::min(r, h - 1);
      std::memcpy(&sy[static_cast<size_t>(r) * ypitch],
                  y + static_cast<size_t>(sr) * ypitch_in, w);
      for (int x = w; x < cw; ++x)
        sy[static_cast<size_t>(r) * ypitch + x] = sy[r * ypitch + w - 1];
    }
    int cwh = cw / 2, chh = ch / 2, wh = (w + 1) / 2, hh = (h + 1) / 2;
    for (int r = 0; r < chh; ++r) {
      int sr = std::min(r, hh - 1);
      std::memcpy(&scb[static_cast<size_t>(r) * cpitch],
                  cb + static_cast<size_t>(sr) * cpitch_in, wh);
      std::memcpy(&scr[static_cast<size_t>(r) * cpitch],
                  cr + static_cast<size_t>(sr) * cpitch_in, wh);
      for (int x = wh; x < cwh; ++x) {
        scb[static_cast<size_t>(r) * cpitch + x] = scb[r * cpitch + wh - 1];
        scr[static_cast<size_t>(r) * cpitch + x] = scr[r * cpitch + wh - 1];
      }
    }
  }

  // Encode one 16x16 CTU: choose mode, transform, reconstruct, and emit
  // CABAC bins. left_avail reflects the slice-segment boundary.
  void encode_ctu(CabacEncoder& cab, ContextBank& bank, int cx, int cy,
                  bool left_avail, int qp, uint8_t left_mode) {
    const int x0 = cx * 16, y0 = cy * 16;
    // --- luma mode decision + transform
    Avail av;  // one slice per CTU row: only left neighbors exist
    av.left = left_avail;
    av.corner = false;
    av.top = av.top_right = av.below_left = false;

    uint8_t ref[65], reff[65];
    build_refs<16>(ry.data(), ypitch, x0, y0, av, ref);
    filter_refs<16>(ref, reff);

    static const int kModes[4] = {kDc, kHor, kVer, kPlanar};
    uint8_t pred[4][256];
    int best = 0;
    long best_sad = LONG_MAX;
    for (int m = 0; m < 4; ++m) {
      predict<16>(kModes[m], kModes[m] == kPlanar ? reff : ref, 0, pred[m]);
      long sad = 0;
      for (int yy = 0; yy < 16; ++yy)
        for (int xx = 0; xx < 16; ++xx) {
          int d = sy[static_cast<size_t>(y0 + yy) * ypitch + x0 + xx] -
                  pred[m][yy * 16 + xx];
          sad += d < 0 ? -d : d;
        }
      if (sad < best_sad) {
        best_sad = sad;
        best = m;
      }
    }
    const int mode = kModes[best];
    luma_mode[cy * ctb_w + cx] = static_cast<uint8_t>(mode);

    int16_t res[256], coef[256];
    TbData<16> yb;
    for (int yy = 0; yy < 16; ++yy)
      for (int xx = 0; xx < 16; ++xx)
        res[yy * 16 + xx] =
            static_cast<int16_t>(
                sy[static_cast<size_t>(y0 + yy) * ypitch + x0 + xx] -
                pred[best][yy * 16 + xx]);
    fwd_transform<16>(res, 16, coef);
    yb.cbf = quantize<16>(coef, qp, yb.level) != 0;
    if (yb.cbf) {
      int16_t deq[256], rres[256];
      dequantize<16>(yb.level, qp, deq);
      inv_transform<16>(deq, rres, 16);
      for (int yy = 0; yy < 16; ++yy)
        for (int xx = 0; xx < 16; ++xx)
          ry[static_cast<size_t>(y0 + yy) * ypitch + x0 + xx] =
              clip8(pred[best][yy * 16 + xx] + rres[yy * 16 + xx]);
    } else {
      for (int yy = 0; yy < 16; ++yy)
        std::memcpy(&ry[static_cast<size_t>(y0 + yy) * ypitch + x0],
                    &pred[best][yy * 16], 16);
    }

    // --- chroma (DM mode), 8x8 TBs
    const int qpc = chroma_qp(qp);
    TbData<8> cbb, crb;
    uint8_t cpred[2][64];
    const int cx0 = x0 / 2, cy0 = y0 / 2;
    for (int pl = 0; pl < 2; ++pl) {
      uint8_t* rp = pl ? rcr.data() : rcb.data();
      const uint8_t* sp = pl ? scr.data() : scb.data();
      TbData<8>& tb = pl ? crb : cbb;
      uint8_t cref[33];
      build_refs<8>(rp, cpitch, cx0, cy0, av, cref);
      predict<8>(mode, cref, 1, cpred[pl]);
      int16_t cres[64], ccoef[64];
      for (int yy = 0; yy < 8; ++yy)
        for (int xx = 0; xx < 8; ++xx)
          cres[yy * 8 + xx] = static_cast<int16_t>(
              sp[static_cast<size_t>(cy0 + yy) * cpitch + cx0 + xx] -
              cpred[pl][yy * 8 + xx]);
      fwd_transform<8>(cres, 8, ccoef);
      tb.cbf = quantize<8>(ccoef, qpc, tb.level) != 0;
      if (tb.cbf) {
        int16_t deq[64], rres[64];
        dequantize<8>(tb.level, qpc, deq);
        inv_transform<8>(deq, rres, 8);
        for (int yy = 0; yy < 8; ++yy)
          for (int xx = 0; xx < 8; ++xx)
            rp[static_cast<size_t>(cy0 + yy) * cpitch + cx0 + xx] =
                clip8(cpred[pl][yy * 8 + xx] + rres[yy * 8 + xx]);
      } else {
        for (int yy = 0; yy < 8; ++yy)
          std::memcpy(&rp[static_cast<size_t>(cy0 + yy) * cpitch + cx0],
                      &cpred[pl][yy * 8], 8);
      }
    }

    // --- CABAC bins for the CTU (coding_quadtree -> coding_unit)
    cab.encode_bin(bank.ctx[kCtxSplitCu + 0], 0);  // split_cu_flag
    // MPM derivation (§8.4.2): above PU is outside the CTU row -> DC.
    int cand_a = left_avail ? left_mode : kDc;
    int cand_b = kDc;
    int list[3];
    if (cand_a == cand_b) {
      if (cand_a < 2) {
        list[0] = kPlanar;
        list[1] = kDc;
        list[2] = kVer;
      } else {
        list[0] = cand_a;
        list[1] = 2 + ((cand_a + 29) % 32);
        list[2] = 2 + ((cand_a - 2 + 1) % 32);
      }
    } else {
      list[0] = cand_a;
      list[1] = cand_b;
      list[2] = (cand_a != kPlanar && cand_b != kPlanar) ? kPlanar
                : (cand_a != kDc && cand_b != kDc)       ? kDc
                                                         : kVer;
    }
    int mpm_idx = -1;
    for (int i = 0; i < 3; ++i)
      if (list[i] == mode) {
        mpm_idx = i;
        break;
      }
    cab.encode_bin(bank.ctx[kCtxPrevIntraLuma], mpm_idx >= 0 ? 1 : 0);
    if (mpm_idx >= 0) {
      cab.encode_bypass(mpm_idx > 0 ? 1 : 0);
      if (mpm_idx > 0) cab.encode_bypass(mpm_idx - 1);
    } else {
      int srt[3] = {list[0], list[1], list[2]};
      std::sort(srt, srt + 3);
      int rem = mode;
      for (int i = 2; i >= 0; --i)
        if (mode > srt[i]) --rem;
      cab.encode_bypass_bins(rem, 5);
    }
    cab.encode_bin(bank.ctx[kCtxIntraChroma], 0);  // DM chroma mode
    // transform_tree: cbf_cb, cbf_cr (trafoDepth 0), cbf_luma (ctx 1)
    cab.encode_bin(bank.ctx[kCtxCbfChroma + 0], cbb.cbf ? 1 : 0);
    cab.encode_bin(bank.ctx[kCtxCbfChroma + 0], crb.cbf ? 1 : 0);
    cab.encode_bin(bank.ctx[kCtxCbfLuma + 1], yb.cbf ? 1 : 0);
    if (yb.cbf) code_residual<16>(cab, bank, yb.level, 0);
    if (cbb.cbf) code_residual<8>(cab, bank, cbb.level, 1);
    if (crb.cbf) code_residual<8>(cab, bank, crb.level, 2);
  }
};

StripeEncoder::StripeEncoder(int width, int height, int slices_per_row)
    : impl_(new Impl(width, height, slices_per_row)),
      width_(width),
      height_(height) {}

StripeEncoder::~StripeEncoder() = default;

const uint8_t* StripeEncoder::recon_y() const { return impl_->ry.data(); }
const uint8_t* StripeEncoder::recon_cb() const { return impl_->rcb.data(); }
const uint8_t* StripeEncoder::recon_cr() const { return impl_->rcr.data(); }
int StripeEncoder::recon_ypitch() const { return impl_->ypitch; }
int StripeEncoder::recon_cpitch() const { return impl_->cpitch; }

void StripeEncoder::encode_frame(const uint8_t* y, int ypitch,
                                 const uint8_t* cb, const uint8_t* cr,
                                 int cpitch, int qp,
                                 std::vector<uint8_t>& out,
                                 EncodeStats* stats) {
  Impl& im = *impl_;
  qp = std::clamp(qp, 0, 51);
  im.load_source(y, ypitch, cb, cr, cpitch);
  out.insert(out.end(), im.ps_cache.begin(), im.ps_cache.end());

  // one slice segment per CTU row (split into slices_per_row segments)
  for (int cy = 0; cy < im.ctb_h; ++cy) {
    int per = ceil_div(im.ctb_w, im.slices_per_row);
    for (int s0 = 0; s0 < im.ctb_w; s0 += per) {
      int s1 = std::min(s0 + per, im.ctb_w);
      BitWriter bw;
      bool first = (cy == 0 && s0 == 0);
      write_slice_header(bw, first, cy * im.ctb_w + s0, im.addr_bits, qp);
      std::vector<uint8_t> cabac_bytes;
      CabacEncoder cab(cabac_bytes);
      ContextBank bank;
      bank.init(qp);
      uint8_t left_mode = kDc;
      for (int cx = s0; cx < s1; ++cx) {
        im.encode_ctu(cab, bank, cx, cy, cx > s0, qp, left_mode);
        left_mode = im.luma_mode[cy * im.ctb_w + cx];
        cab.encode_terminate(cx == s1 - 1 ? 1 : 0);
      }
      auto tail = cab.finish();
      for (uint8_t b : cabac_bytes) bw.u(b, 8);
      bw.u(tail.bits, tail.nbits);
      // rbsp_slice_segment_trailing_bits: stop bit + alignment
      bw.rbsp_trailing();
      emit_nal(bw, out, kNalIdr);
    }
  }
  if (stats) {
    stats->frame_qp = qp;
    stats->ctu_count = im.ctb_w * im.ctb_h;
    stats->bytes = out.size();
  }
}

}  // namespace hevc
}  // namespace hipflux
