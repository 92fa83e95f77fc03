// HEVC CABAC syntax emission for one CTU / one TB, shared by the CPU
// encoder and the GPU pipeline's host entropy fallback. The GPU CABAC
// kernel (native/hip/hevc_kernels.hip) transliterates exactly this
// logic; tests assert byte equality between all three paths.
#pragma once

#include <algorithm>
#include <cstdint>

#include "cabac.h"
#include "intra.h"
#include "tables.h"

namespace hipflux {
namespace hevc {

// ---- residual_coding (§7.3.8.11) for one TB ------------------------------
// N = 16 (luma) or 8 (chroma). Levels indexed [y*N + x].
template <int N>
inline void code_residual(CabacEncoder& cab, ContextBank& bank,
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

// All CABAC bins of one CTU (coding_quadtree -> coding_unit ->
// transform_tree): split flag, intra mode w/ MPM, chroma DM, cbfs,
// residuals. left_mode < 0 means the left CTU is unavailable (slice
// start). Levels are raster [y*N + x] arrays; cbf flags must match
// their content (cbf set <=> any nonzero).
inline void code_ctu_syntax(CabacEncoder& cab, ContextBank& bank, int mode,
                            int left_mode, bool cbf_y, bool cbf_cb,
                            bool cbf_cr, const int16_t* ylv,
                            const int16_t* cblv, const int16_t* crlv) {
  cab.encode_bin(bank.ctx[kCtxSplitCu + 0], 0);  // split_cu_flag
  // MPM derivation (§8.4.2): above PU is outside the CTU row -> DC.
  int cand_a = left_mode >= 0 ? left_mode : kDc;
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
  cab.encode_bin(bank.ctx[kCtxCbfChroma + 0], cbf_cb ? 1 : 0);
  cab.encode_bin(bank.ctx[kCtxCbfChroma + 0], cbf_cr ? 1 : 0);
  cab.encode_bin(bank.ctx[kCtxCbfLuma + 1], cbf_y ? 1 : 0);
  if (cbf_y) code_residual<16>(cab, bank, ylv, 0);
  if (cbf_cb) code_residual<8>(cab, bank, cblv, 1);
  if (cbf_cr) code_residual<8>(cab, bank, crlv, 2);
}

}  // namespace hevc
}  // namespace hipflux
