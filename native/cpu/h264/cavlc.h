// CAVLC residual encoder (ITU-T H.264 9.2). Tables generated+verified by
// tools/gen_cavlc_tables.py (Kraft/prefix checks + Richardson worked
// example). TotalCoeff is capped at kMaxCoeffs=12 upstream (see the table
// generator note) so the synthetic tc>=13 region of the 4<=nC<8 table is
// never emitted.
#pragma once

#include <cstdlib>

#include "bitwriter.h"
#include "cavlc_tables.h"

namespace hipflux {
namespace h264 {

constexpr int kMaxCoeffs = 12;

// Cap the number of nonzero coefficients at kMaxCoeffs by zeroing the
// highest-frequency nonzeros. Must run BEFORE dequant/reconstruction so the
// encoder's recon matches what a decoder will see. zz: zigzag order.
inline void cap_coeffs(int* zz, int n) {
  int tc = 0;
  for (int i = 0; i < n; ++i)
    if (zz[i]) ++tc;
  for (int i = n - 1; i >= 0 && tc > kMaxCoeffs; --i)
    if (zz[i]) {
      zz[i] = 0;
      --tc;
    }
}

// Encode one residual block. zz: zigzag-ordered coefficients, n of them
// (16 = full block, 15 = AC-only, 4 = chroma DC). nC: CAVLC context
// (-1 for chroma DC). Returns TotalCoeff (for neighbor nC bookkeeping).
inline int cavlc_residual(BitWriter& bw, const int* zz, int n, int nC) {
  int coeffs[16], pos[16], tc = 0;
  for (int i = 0; i < n; ++i)
    if (zz[i]) {
      coeffs[tc] = zz[i];
      pos[tc] = i;
      ++tc;
    }
  int t1 = 0;
  for (int k = tc - 1; k >= 0 && t1 < 3; --k) {
    if (std::abs(coeffs[k]) == 1)
      ++t1;
    else
      break;
  }
  // coeff_token
  if (nC == -1) {
    const Vlc& v = kCoeffTokenCDC[tc][t1];
    bw.u(v.bits, v.len);
  } else if (nC < 2) {
    const Vlc& v = kCoeffToken0[tc][t1];
    bw.u(v.bits, v.len);
  } else if (nC < 4) {
    const Vlc& v = kCoeffToken1[tc][t1];
    bw.u(v.bits, v.len);
  } else if (nC < 8) {
    const Vlc& v = kCoeffToken2[tc][t1];
    bw.u(v.bits, v.len);
  } else {
    bw.u(tc == 0 ? 3 : ((tc - 1) << 2) | t1, 6);
  }
  if (tc == 0) return 0;

  // trailing one signs, high frequency first
  for (int k = tc - 1; k >= tc - t1; --k) bw.put_bit(coeffs[k] > 0 ? 0 : 1);

  // remaining levels, high frequency first
  int suffix_len = (tc > 10 && t1 < 3) ? 1 : 0;
  bool first = true;
  for (int k = tc - t1 - 1; k >= 0; --k) {
    const int true_level = coeffs[k];  // suffix growth uses the TRUE value
    int level = coeffs[k];
    if (first && t1 < 3) level += level > 0 ? -1 : 1;
    first = false;
    int code = level > 0 ? 2 * level - 2 : -2 * level - 1;
    if (suffix_len == 0) {
      if (code < 14) {
        bw.u(1, code + 1);
      } else if (code < 30) {
        bw.u(1, 15);
        bw.u(code - 14, 4);
      } else {
        bw.u(1, 16);
        bw.u(code - 30, 12);
      }
    } else {
      int prefix = code >> suffix_len;
      if (prefix < 15) {
        bw.u(1, prefix + 1);
        bw.u(code & ((1 << suffix_len) - 1), suffix_len);
      } else {
        bw.u(1, 16);
        bw.u(code - (15 << suffix_len), 12);
      }
    }
    if (suffix_len == 0) suffix_len = 1;
    if (std::abs(true_level) > (3 << (suffix_len - 1)) && suffix_len < 6)
      ++suffix_len;
  }

  // total_zeros
  int total_zeros = pos[tc - 1] + 1 - tc;
  if (tc < n) {
    if (nC == -1) {
      const Vlc& v = kTotalZerosCDC[tc][total_zeros];
      bw.u(v.bits, v.len);
    } else {
      const Vlc& v = kTotalZeros[tc][total_zeros];
      bw.u(v.bits, v.len);
    }
  }

  // run_before, high frequency first
  int zeros_left = total_zeros;
  for (int k = tc - 1; k > 0 && zeros_left > 0; --k) {
    int run = pos[k] - pos[k - 1] - 1;
    const Vlc& v = kRunBefore[zeros_left < 7 ? zeros_left : 7][run];
    bw.u(v.bits, v.len);
    zeros_left -= run;
  }
  return tc;
}

}  // namespace h264
}  // namespace hipflux
