// CDNA4 (gfx950) H.264 encode kernels.
//
// Parallel structure (see cpu/h264/encoder.h for the slice design):
//  * one slice per MB row -> every MB row is independent; k_h264_rows runs
//    ONE WAVE (64 lanes) per scheduled row, walking its MBs left-to-right
//    (the only true dependency: H/DC intra prediction + recon feedback).
//  * within an MB, 64 lanes process 4 4x4 blocks at a time, one lane per
//    coefficient; the 4x4 forward/inverse transforms are done with
//    cross-lane __shfl butterflies inside each 16-lane group (no LDS
//    round-trips, wave-synchronous by construction).
//  * k_me does P-frame mode decision (skip / inter MV / intra) for all MBs
//    of scheduled P rows in parallel, one workgroup per MB, before the row
//    kernel runs.
//
// Integer semantics match cpu/h264/transform.h exactly (same quant/dequant
// formulas, same floor shifts) so GPU streams decode bit-identically under
// tests/h264_ref_decoder.py.
#include <hip/hip_runtime.h>

#include "h264_gpu_layout.h"
#include "h264_kernels.h"
#include "h264_rows_common.h"

namespace hipflux {
namespace h264gpu {

// ---------------------------------------------------------------------------
// The row kernel: one workgroup of 128 threads per MB-row slice.
// Wave 0 encodes the row's LUMA, wave 1 its CHROMA — the two pipelines
// share no state (chroma_mode is reported through meta word 1, which is
// unused for intra MBs), so the waves run fully decoupled and the per-MB
// serial chain is the LUMA path only.
//
// Cross-lane data movement uses shuffles (source pixels, DC terms) instead
// of LDS staging; the only LDS state is the left-neighbor recon columns,
// updated once per MB behind a wave-level fence.

// ---- luma row pipeline (one wave) ----------------------------------------
__device__ void luma_row(const uint8_t* __restrict__ srcY, int ypitch, int w,
                         int h, const uint8_t* __restrict__ refY,
                         uint8_t* __restrict__ curY, int mbw, int mby,
                         int qp, bool i_slice,
                         const int16_t* __restrict__ levels_base,
                         int* __restrict__ meta, int lane, int mbx0,
                         int seg_mbw) {
  const int y0 = mby * 16;
  const int r = lane >> 2, cq = (lane & 3) * 4;
  const int g = lane >> 4, c = lane & 15;
  const int zz = c_zz_of_pos[c];
  bool have_left = false;
  // left-neighbor recon column lives in REGISTERS: lane l (<16) holds the
  // reconstructed pixel of row l, column 15 of the previous MB. All state
  // movement is shuffles -> the serial per-MB chain has no LDS round
  // trips and no fences.
  int leftreg = 0;

  // The serial MB chain is latency-bound: issue the NEXT MB's loads
  // (source pixels, zero-mv reference, meta) while computing the current
  // one so global-memory latency overlaps the transform work.
  auto load_psrc = [&](int mbx2) -> uint32_t {
    const uint8_t* sr = srcY + (size_t)min(y0 + r, h - 1) * ypitch;
    int xb = mbx2 * 16 + cq;
    return (uint32_t)sr[min(xb + 0, w - 1)] |
           ((uint32_t)sr[min(xb + 1, w - 1)] << 8) |
           ((uint32_t)sr[min(xb + 2, w - 1)] << 16) |
           ((uint32_t)sr[min(xb + 3, w - 1)] << 24);
  };
  auto load_refp = [&](int mbx2) -> uint32_t {
    const uint8_t* s2 = refY + (size_t)(y0 + r) * ypitch + mbx2 * 16 + cq;
    return (uint32_t)s2[0] | ((uint32_t)s2[1] << 8) |
           ((uint32_t)s2[2] << 16) | ((uint32_t)s2[3] << 24);
  };
  auto load_refcol = [&](int mbx2) -> int {
    return lane < 16
               ? refY[(size_t)(y0 + lane) * ypitch + mbx2 * 16 + 15]
               : 0;
  };
  uint32_t psrc_cur = load_psrc(mbx0);
  uint32_t refp_cur = i_slice ? 0 : load_refp(mbx0);
  int refcol_cur = i_slice ? 0 : load_refcol(mbx0);
  int m0_cur = 0, m1_cur = 0;
  if (!i_slice) {
    m0_cur = meta[((size_t)mby * mbw + mbx0) * kMetaPerMb + 0];
    m1_cur = meta[((size_t)mby * mbw + mbx0) * kMetaPerMb + 1];
  }

  const int mbx_end = mbx0 + seg_mbw;
  for (int mbx = mbx0; mbx < mbx_end; ++mbx) {
    const int x0 = mbx * 16;
    const size_t mb_index = (size_t)mby * mbw + mbx;
    int16_t* L = const_cast<int16_t*>(levels_base) +
                 mb_index * kLevelsPerMb;
    int* M = meta + mb_index * kMetaPerMb;

    uint32_t psrc_nxt = 0, refp_nxt = 0;
    int refcol_nxt = 0, m0_nxt = 0, m1_nxt = 0;
    if (mbx + 1 < mbx_end) {
      psrc_nxt = load_psrc(mbx + 1);
      if (!i_slice) {
        refp_nxt = load_refp(mbx + 1);
        refcol_nxt = load_refcol(mbx + 1);
        m0_nxt = M[kMetaPerMb + 0];
        m1_nxt = M[kMetaPerMb + 1];
      }
    }

    int mode = kIntra, mvx = 0, mvy = 0;
    if (!i_slice) {
      mode = m0_cur & 3;
      mvx = (short)(m1_cur & 0xFFFF);
      mvy = m1_cur >> 16;
    }

    if (mode == kSkip) {
      uint8_t* d = curY + (size_t)(y0 + r) * ypitch + x0 + cq;
      *reinterpret_cast<uint32_t*>(d) = refp_cur;
      leftreg = refcol_cur;
      have_left = true;
    } else if (mode == kInter) {
      // P_L0_16x16 with coded residual: full 16-coeff blocks, no DC
      // Hadamard, inter quant rounding (f = 2^qbits/6). cbp==0 falls out
      // naturally (idct of zeros is zero -> recon == MC pred).
      const int ix = mvx >> 2, iy = mvy >> 2;
      const int fx = mvx & 3, fy = mvy & 3;
      uint32_t psrc = psrc_cur, ppred;
      {
        if ((fx | fy | ix | iy) == 0) {
          ppred = refp_cur;
        } else if ((fx | fy) == 0) {
          const uint8_t* pr =
              refY + (size_t)(y0 + iy + r) * ypitch + x0 + ix + cq;
          ppred = (uint32_t)pr[0] | ((uint32_t)pr[1] << 8) |
                  ((uint32_t)pr[2] << 16) | ((uint32_t)pr[3] << 24);
        } else {
          ppred = 0;
          for (int j = 0; j < 4; ++j)
            ppred |= (uint32_t)luma_interp(refY, ypitch, x0 + ix + cq + j,
                                           y0 + iy + r, fx, fy) << (8 * j);
        }
      }
      auto pix_at = [&](int py, int px) -> int {
        uint32_t v = __shfl(psrc, py * 4 + (px >> 2));
        return (v >> (8 * (px & 3))) & 0xFF;
      };
      auto pred_at = [&](int py, int px) -> int {
        uint32_t v = __shfl(ppred, py * 4 + (px >> 2));
        return (v >> (8 * (px & 3))) & 0xFF;
      };
      int lvl_p[4];
#pragma unroll
      for (int pass = 0; pass < 4; ++pass) {
        int blk = pass * 4 + g;
        int bx = blk & 3, by = blk >> 2;
        int py = by * 4 + (c >> 2), px = bx * 4 + (c & 3);
        int resid = pix_at(py, px) - pred_at(py, px);
        int coefv = fdct4_wave(resid, lane);
        int lvl = quant_coeff(coefv, qp, coeff_cls(c), false);
        lvl = cap12_group(lvl, zz, true, lane);
        lvl_p[pass] = lvl;
        store_lvl_pair(L + kLumaAcOff + blk * 16, c, lvl, lane);
      }
      int anyl = (lvl_p[0] | lvl_p[1] | lvl_p[2] | lvl_p[3]) != 0;
      const int cbp_luma = __ballot(anyl) ? 15 : 0;
      int newleft = leftreg;
#pragma unroll
      for (int pass = 0; pass < 4; ++pass) {
        int blk = pass * 4 + g;
        int bx = blk & 3, by = blk >> 2;
        int d = cbp_luma ? dequant_c(lvl_p[pass], qp, coeff_cls(c)) : 0;
        int rec = idct4_wave(d, lane);
        int py = by * 4 + (c >> 2), px = bx * 4 + (c & 3);
        int pix = clip8(rec + pred_at(py, px));
        int p1 = __shfl(pix, lane + 1), p2 = __shfl(pix, lane + 2),
            p3 = __shfl(pix, lane + 3);
        if ((c & 3) == 0) {
          *reinterpret_cast<uint32_t*>(
              curY + (size_t)(y0 + py) * ypitch + x0 + px) =
              (uint32_t)pix | ((uint32_t)p1 << 8) | ((uint32_t)p2 << 16) |
              ((uint32_t)p3 << 24);
        }
        // collect the MB's right-edge recon into the register column:
        // row l's col-15 pixel lives on lane 51 + 4*(l&3) in pass l>>2
        int cand = __shfl(pix, 51 + 4 * (lane & 3));
        if (lane < 16 && (lane >> 2) == pass) newleft = cand;
      }
      leftreg = newleft;
      have_left = true;
    } else {
    // ---- source pixels come from the prefetched register
    uint32_t psrc = psrc_cur;
    // DC value + mode costs
    int dcval = 128;
    if (have_left) {
      int part = (lane < 16) ? leftreg : 0;
      dcval = (wave_sum_i(part) + 8) >> 4;
    }
    int lv_all = __shfl(leftreg, r);
    int lv = have_left ? lv_all : 0;
    int costH = 0, costDC = 0;
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      int pv = (psrc >> (8 * k)) & 0xFF;
      costDC += abs(pv - dcval);
      costH += abs(pv - lv);
    }
    costDC = wave_sum_i(costDC);
    costH = wave_sum_i(costH);
    const int luma_mode = (have_left && costH < costDC) ? 1 : 2;

    // fetch pixel (py,px) of the MB from the wave's packed registers
    auto pix_at = [&](int py, int px) -> int {
      uint32_t v = __shfl(psrc, py * 4 + (px >> 2));
      return (v >> (8 * (px & 3))) & 0xFF;
    };

    // ---- 4 passes x 4 blocks: fdct + quant (+ DC collection in regs)
    int lvl_p[4];       // this lane's level per pass
    int dcreg = 0;      // lanes 0..15: DC coefficient of block `lane`
#pragma unroll
    for (int pass = 0; pass < 4; ++pass) {
      int blk = pass * 4 + g;
      int bx = blk & 3, by = blk >> 2;
      int py = by * 4 + (c >> 2), px = bx * 4 + (c & 3);
      int lcol = __shfl(leftreg, py);
      int pred = luma_mode == 1 ? lcol : dcval;
      int resid = pix_at(py, px) - pred;
      int coefv = fdct4_wave(resid, lane);
      // collect DC of block (pass*4+gg) into lane (pass*4+gg)
      int dcv = __shfl(coefv, (lane & 3) * 16);
      if (lane < 16 && (lane >> 2) == pass) dcreg = dcv;
      int lvl = (c == 0) ? 0 : quant_coeff(coefv, qp, coeff_cls(c));
      lvl = cap12_group(lvl, zz, c != 0, lane);
      lvl_p[pass] = lvl;
      store_lvl_pair(L + kLumaAcOff + blk * 16, c, lvl, lane);
    }

    // ---- luma DC: hadamard + quant + inverse + dequant, all in-wave
    int had = hadamard4_wave(dcreg, lane, true);
    int qdc = quant_dc_v(had, qp);
    qdc = cap12_group(qdc, zz, lane < 16, lane);
    if (lane < 16) store_lvl_pair(L + kLumaDcOff, lane, qdc, lane);
    int ih = hadamard4_wave(qdc, lane, false);
    int dcrec = dequant_luma_dc_v(ih, qp);  // valid on lanes 0..15

    // cbp_luma: any AC nonzero
    int anyl = (lvl_p[0] | lvl_p[1] | lvl_p[2] | lvl_p[3]) != 0 && c != 0;
    const int cbp_luma = __ballot(anyl) ? 15 : 0;

    // ---- recon passes
    int newleft = leftreg;
#pragma unroll
    for (int pass = 0; pass < 4; ++pass) {
      int blk = pass * 4 + g;
      int bx = blk & 3, by = blk >> 2;
      // shuffle OUTSIDE the divergent branch: ds_bpermute from a lane that
      // is not executing the instruction returns undefined data
      int dcb = __shfl(dcrec, blk);
      int d = 0;
      if (c == 0)
        d = dcb;
      else if (cbp_luma)
        d = dequant_c(lvl_p[pass], qp, coeff_cls(c));
      int rec = idct4_wave(d, lane);
      int py = by * 4 + (c >> 2), px = bx * 4 + (c & 3);
      int lcol = __shfl(leftreg, py);
      int pred = luma_mode == 1 ? lcol : dcval;
      int pix = clip8(rec + pred);
      int p1 = __shfl(pix, lane + 1), p2 = __shfl(pix, lane + 2),
          p3 = __shfl(pix, lane + 3);
      if ((c & 3) == 0) {
        *reinterpret_cast<uint32_t*>(
            curY + (size_t)(y0 + py) * ypitch + x0 + px) =
            (uint32_t)pix | ((uint32_t)p1 << 8) | ((uint32_t)p2 << 16) |
            ((uint32_t)p3 << 24);
      }
      int cand = __shfl(pix, 51 + 4 * (lane & 3));
      if (lane < 16 && (lane >> 2) == pass) newleft = cand;
    }
    leftreg = newleft;

    if (lane == 0)
      M[0] = kIntra | (luma_mode << 2) | (m0_cur & 0x7FFF00);
    have_left = true;
    }

    // rotate the prefetched state
    psrc_cur = psrc_nxt;
    refp_cur = refp_nxt;
    refcol_cur = refcol_nxt;
    m0_cur = m0_nxt;
    m1_cur = m1_nxt;
  }
}

// ---- chroma row pipeline (one wave) ---------------------------------------
__device__ void chroma_row(const uint8_t* __restrict__ srcCb,
                           const uint8_t* __restrict__ srcCr, int cpitch,
                           int w, int h, const uint8_t* __restrict__ refCb,
                           const uint8_t* __restrict__ refCr,
                           uint8_t* __restrict__ curCb,
                           uint8_t* __restrict__ curCr, int mbw, int mby,
                           int qpc, bool i_slice,
                           const int16_t* __restrict__ levels_base,
                           int* __restrict__ meta, int lane, int mbx0,
                           int seg_mbw) {
  const int cy0 = mby * 8;
  const int g = lane >> 4, c = lane & 15;
  const int zz = c_zz_of_pos[c];
  const int cw = (w + 1) / 2, chh = (h + 1) / 2;
  bool have_left = false;
  // left-neighbor recon columns in registers: lane l (<8) holds row l,
  // col 7 of the previous MB for each component (no LDS, no fences).
  int leftcb = 0, leftcr = 0;

  // prefetch helpers (see luma_row): next MB's loads issue early so the
  // serial chain overlaps global latency with compute
  auto cload_src = [&](int mbx2) -> uint32_t {
    if (lane >= 32) return 0;
    int comp = lane >> 4;
    int rr = (lane & 15) >> 1, ccq = (lane & 1) * 4;
    const uint8_t* sp = comp ? srcCr : srcCb;
    int sr = min(cy0 + rr, chh - 1);
    int xb = mbx2 * 8 + ccq;
    return (uint32_t)sp[(size_t)sr * cpitch + min(xb + 0, cw - 1)] |
           ((uint32_t)sp[(size_t)sr * cpitch + min(xb + 1, cw - 1)] << 8) |
           ((uint32_t)sp[(size_t)sr * cpitch + min(xb + 2, cw - 1)] << 16) |
           ((uint32_t)sp[(size_t)sr * cpitch + min(xb + 3, cw - 1)] << 24);
  };
  auto cload_ref = [&](int mbx2) -> uint32_t {
    if (lane >= 32) return 0;
    int comp = lane >> 4;
    int rr = (lane & 15) >> 1, ccq = (lane & 1) * 4;
    const uint8_t* sp = comp ? refCr : refCb;
    const uint8_t* s2 = sp + (size_t)(cy0 + rr) * cpitch + mbx2 * 8 + ccq;
    return (uint32_t)s2[0] | ((uint32_t)s2[1] << 8) |
           ((uint32_t)s2[2] << 16) | ((uint32_t)s2[3] << 24);
  };
  auto cload_col = [&](int mbx2, const uint8_t* plane) -> int {
    return lane < 8
               ? plane[(size_t)(cy0 + lane) * cpitch + mbx2 * 8 + 7]
               : 0;
  };
  uint32_t csrc_cur = cload_src(mbx0);
  uint32_t cref_cur = i_slice ? 0 : cload_ref(mbx0);
  int colcb_cur = i_slice ? 0 : cload_col(mbx0, refCb);
  int colcr_cur = i_slice ? 0 : cload_col(mbx0, refCr);
  int m0_cur = 0, m1_cur = 0;
  if (!i_slice) {
    m0_cur = meta[((size_t)mby * mbw + mbx0) * kMetaPerMb + 0];
    m1_cur = meta[((size_t)mby * mbw + mbx0) * kMetaPerMb + 1];
  }

  const int mbx_end = mbx0 + seg_mbw;
  for (int mbx = mbx0; mbx < mbx_end; ++mbx) {
    const int cx0 = mbx * 8;
    const size_t mb_index = (size_t)mby * mbw + mbx;
    int16_t* L = const_cast<int16_t*>(levels_base) +
                 mb_index * kLevelsPerMb;
    int* M = meta + mb_index * kMetaPerMb;

    uint32_t csrc_nxt = 0, cref_nxt = 0;
    int colcb_nxt = 0, colcr_nxt = 0, m0_nxt = 0, m1_nxt = 0;
    if (mbx + 1 < mbx_end) {
      csrc_nxt = cload_src(mbx + 1);
      if (!i_slice) {
        cref_nxt = cload_ref(mbx + 1);
        colcb_nxt = cload_col(mbx + 1, refCb);
        colcr_nxt = cload_col(mbx + 1, refCr);
        m0_nxt = M[kMetaPerMb + 0];
        m1_nxt = M[kMetaPerMb + 1];
      }
    }

    int mode = kIntra, mvx = 0, mvy = 0;
    if (!i_slice) {
      mode = m0_cur & 3;
      mvx = (short)(m1_cur & 0xFFFF);
      mvy = m1_cur >> 16;
    }

    if (mode == kSkip) {
      if (lane < 32) {
        int comp = lane >> 4;
        int r = (lane & 15) >> 1, cq = (lane & 1) * 4;
        uint8_t* dp = comp ? curCr : curCb;
        uint8_t* d = dp + (size_t)(cy0 + r) * cpitch + cx0 + cq;
        *reinterpret_cast<uint32_t*>(d) = cref_cur;
      }
      leftcb = colcb_cur;
      leftcr = colcr_cur;
      have_left = true;
    } else if (mode == kInter) {
      // chroma inter residual: MC pred (integer, mv/2) + DC Hadamard + AC,
      // inter quant rounding. cbp falls out of the quantized levels.
      const int cix = mvx >> 3, ciy = mvy >> 3;
      const int cdx = mvx & 7, cdy = mvy & 7;
      uint32_t csrc = csrc_cur, cprd = 0;
      if ((cix | ciy | cdx | cdy) == 0) {
        cprd = cref_cur;
      } else if (lane < 32) {
        int comp = lane >> 4;
        int rr = (lane & 15) >> 1, ccq = (lane & 1) * 4;
        const uint8_t* rp = comp ? refCr : refCb;
        for (int j = 0; j < 4; ++j)
          cprd |= (uint32_t)chroma_interp(rp, cpitch, cx0 + cix + ccq + j,
                                          cy0 + ciy + rr, cdx, cdy)
                  << (8 * j);
      }
      auto cpix_at = [&](int comp, int rr, int cc) -> int {
        uint32_t v = __shfl(csrc, (comp << 4) | (rr << 1) | (cc >> 2));
        return (v >> (8 * (cc & 3))) & 0xFF;
      };
      auto cprd_at = [&](int comp, int rr, int cc) -> int {
        uint32_t v = __shfl(cprd, (comp << 4) | (rr << 1) | (cc >> 2));
        return (v >> (8 * (cc & 3))) & 0xFF;
      };
      int lvl_p[2];
      int dcpass[2];
#pragma unroll
      for (int pass = 0; pass < 2; ++pass) {
        int comp = pass;
        int sub = g;
        int scx = (sub & 1) * 4, scy = (sub >> 1) * 4;
        int rr = scy + (c >> 2), cc2 = scx + (c & 3);
        int resid = cpix_at(comp, rr, cc2) - cprd_at(comp, rr, cc2);
        int coefv = fdct4_wave(resid, lane);
        dcpass[pass] = __shfl(coefv, (lane & 3) * 16);
        int lvl = (c == 0) ? 0 : quant_coeff(coefv, qpc, coeff_cls(c),
                                             false);
        lvl = cap12_group(lvl, zz, c != 0, lane);
        lvl_p[pass] = lvl;
        store_lvl_pair(L + kChromaAcOff + (comp * 4 + sub) * 16, c, lvl,
                       lane);
      }
      int qdc_all[8], dcr_all[8];
      bool any_cdc = false;
#pragma unroll
      for (int comp = 0; comp < 2; ++comp) {
        int d0 = __shfl(dcpass[comp], 0);
        int d1 = __shfl(dcpass[comp], 1);
        int d2 = __shfl(dcpass[comp], 2);
        int d3 = __shfl(dcpass[comp], 3);
        int w0 = d0 + d1 + d2 + d3, w1 = d0 - d1 + d2 - d3;
        int w2 = d0 + d1 - d2 - d3, w3 = d0 - d1 - d2 + d3;
        int q0 = quant_dc_v(w0, qpc, false), q1 = quant_dc_v(w1, qpc, false);
        int q2 = quant_dc_v(w2, qpc, false), q3 = quant_dc_v(w3, qpc, false);
        qdc_all[comp * 4 + 0] = q0;
        qdc_all[comp * 4 + 1] = q1;
        qdc_all[comp * 4 + 2] = q2;
        qdc_all[comp * 4 + 3] = q3;
        any_cdc |= (q0 | q1 | q2 | q3) != 0;
      }
      if (lane < 8)
        L[kChromaDcOff + lane] = (int16_t)qdc_all[lane];
      int anyc = (lvl_p[0] | lvl_p[1]) != 0 && c != 0;
      const int cbp_chroma = __ballot(anyc) ? 2 : (any_cdc ? 1 : 0);
      int ncb = leftcb, ncr = leftcr;
#pragma unroll
      for (int comp = 0; comp < 2; ++comp) {
        int dq0 = 0, dq1 = 0, dq2 = 0, dq3 = 0;
        if (cbp_chroma >= 1) {
          int* q = &qdc_all[comp * 4];
          int w0 = q[0] + q[1] + q[2] + q[3], w1 = q[0] - q[1] + q[2] - q[3];
          int w2 = q[0] + q[1] - q[2] - q[3], w3 = q[0] - q[1] - q[2] + q[3];
          dq0 = dequant_chroma_dc_v(w0, qpc);
          dq1 = dequant_chroma_dc_v(w1, qpc);
          dq2 = dequant_chroma_dc_v(w2, qpc);
          dq3 = dequant_chroma_dc_v(w3, qpc);
        }
        dcr_all[comp * 4 + 0] = dq0;
        dcr_all[comp * 4 + 1] = dq1;
        dcr_all[comp * 4 + 2] = dq2;
        dcr_all[comp * 4 + 3] = dq3;
      }
#pragma unroll
      for (int pass = 0; pass < 2; ++pass) {
        int comp = pass;
        int sub = g;
        int scx = (sub & 1) * 4, scy = (sub >> 1) * 4;
        int rr = scy + (c >> 2), cc2 = scx + (c & 3);
        int d = (c == 0) ? dcr_all[comp * 4 + sub]
                         : (cbp_chroma == 2
                                ? dequant_c(lvl_p[pass], qpc, coeff_cls(c))
                                : 0);
        int rec = idct4_wave(d, lane);
        int pix = clip8(rec + cprd_at(comp, rr, cc2));
        uint8_t* dp = comp ? curCr : curCb;
        int p1 = __shfl(pix, lane + 1), p2 = __shfl(pix, lane + 2),
            p3 = __shfl(pix, lane + 3);
        if ((c & 3) == 0) {
          *reinterpret_cast<uint32_t*>(
              dp + (size_t)(cy0 + rr) * cpitch + cx0 + cc2) =
              (uint32_t)pix | ((uint32_t)p1 << 8) | ((uint32_t)p2 << 16) |
              ((uint32_t)p3 << 24);
        }
        // right-edge recon of row l lives on lane 19+4l (rows 0-3) /
        // 35+4l (rows 4-7); collect into the register columns
        int cand = __shfl(pix, lane < 4 ? 19 + 4 * lane : 35 + 4 * lane);
        if (lane < 8) {
          if (pass == 0) ncb = cand;
          else ncr = cand;
        }
      }
      leftcb = ncb;
      leftcr = ncr;
      have_left = true;
    } else {
    // source pixels come from the prefetched register
    uint32_t csrc = csrc_cur;
    auto cpix_at = [&](int comp, int rr, int cc) -> int {
      uint32_t v = __shfl(csrc, (comp << 4) | (rr << 1) | (cc >> 2));
      return (v >> (8 * (cc & 3))) & 0xFF;
    };

    // per-sub-row DC-prediction averages from the register columns
    // (uniform across the wave; default 128 when no left neighbor)
    int cbt = 128, cbb = 128, crt = 128, crb = 128;
    if (have_left) {
      cbt = (__shfl(leftcb, 0) + __shfl(leftcb, 1) + __shfl(leftcb, 2) +
             __shfl(leftcb, 3) + 2) >> 2;
      cbb = (__shfl(leftcb, 4) + __shfl(leftcb, 5) + __shfl(leftcb, 6) +
             __shfl(leftcb, 7) + 2) >> 2;
      crt = (__shfl(leftcr, 0) + __shfl(leftcr, 1) + __shfl(leftcr, 2) +
             __shfl(leftcr, 3) + 2) >> 2;
      crb = (__shfl(leftcr, 4) + __shfl(leftcr, 5) + __shfl(leftcr, 6) +
             __shfl(leftcr, 7) + 2) >> 2;
    }

    // ---- mode decision: DC(0) vs H(1), summed across both components
    int ccH = 0, ccDC = 0;
    {
      int comp = lane >> 5;
      int idx = lane & 31;
      int rr = idx >> 2, ccq = (idx & 3) * 2;
      int lcb = __shfl(leftcb, rr), lcr = __shfl(leftcr, rr);
      int dd = comp ? ((rr & 4) ? crb : crt) : ((rr & 4) ? cbb : cbt);
      int lvv = have_left ? (comp ? lcr : lcb) : 0;
      for (int k = 0; k < 2; ++k) {
        int sv = cpix_at(comp, rr, ccq + k);
        ccDC += abs(sv - dd);
        ccH += abs(sv - lvv);
      }
      ccDC = wave_sum_i(ccDC);
      ccH = wave_sum_i(ccH);
    }
    const int chroma_mode = (have_left && ccH < ccDC) ? 1 : 0;

    // ---- 2 passes x 4 sub-blocks: fdct + quant; DC terms kept in regs
    int lvl_p[2];
    int dcpass[2];          // lane holds shfl-collected DC (see below)
#pragma unroll
    for (int pass = 0; pass < 2; ++pass) {
      int comp = pass;
      int sub = g;
      int scx = (sub & 1) * 4, scy = (sub >> 1) * 4;
      int rr = scy + (c >> 2), cc2 = scx + (c & 3);
      int lH = comp ? __shfl(leftcr, rr) : __shfl(leftcb, rr);
      int pdc = comp ? ((rr & 4) ? crb : crt) : ((rr & 4) ? cbb : cbt);
      int pred = chroma_mode == 1 ? lH : pdc;
      int resid = cpix_at(comp, rr, cc2) - pred;
      int coefv = fdct4_wave(resid, lane);
      // every lane grabs all 4 sub-block DCs of this pass (sub k from
      // lane k*16): lane's slot (lane&3) -> dcpass[pass]
      dcpass[pass] = __shfl(coefv, (lane & 3) * 16);
      int lvl = (c == 0) ? 0 : quant_coeff(coefv, qpc, coeff_cls(c));
      lvl = cap12_group(lvl, zz, c != 0, lane);
      lvl_p[pass] = lvl;
      store_lvl_pair(L + kChromaAcOff + (comp * 4 + sub) * 16, c, lvl, lane);
    }

    // ---- chroma DC 2x2 hadamard/quant/recon, redundantly on all lanes
    // lane's dcpass[p] holds dc of sub-block (lane&3) of comp p; gather the
    // full set via 4 shuffles per comp.
    int qdc_all[8], dcr_all[8];
    bool any_cdc = false;
#pragma unroll
    for (int comp = 0; comp < 2; ++comp) {
      int d0 = __shfl(dcpass[comp], 0);   // lane with slot 0
      int d1 = __shfl(dcpass[comp], 1);
      int d2 = __shfl(dcpass[comp], 2);
      int d3 = __shfl(dcpass[comp], 3);
      int w0 = d0 + d1 + d2 + d3, w1 = d0 - d1 + d2 - d3;
      int w2 = d0 + d1 - d2 - d3, w3 = d0 - d1 - d2 + d3;
      int q0 = quant_dc_v(w0, qpc), q1 = quant_dc_v(w1, qpc);
      int q2 = quant_dc_v(w2, qpc), q3 = quant_dc_v(w3, qpc);
      qdc_all[comp * 4 + 0] = q0;
      qdc_all[comp * 4 + 1] = q1;
      qdc_all[comp * 4 + 2] = q2;
      qdc_all[comp * 4 + 3] = q3;
      any_cdc |= (q0 | q1 | q2 | q3) != 0;
    }
    if (lane < 8)
      L[kChromaDcOff + lane] = (int16_t)qdc_all[lane];
    int anyc = (lvl_p[0] | lvl_p[1]) != 0 && c != 0;
    const int cbp_chroma = __ballot(anyc) ? 2 : (any_cdc ? 1 : 0);
#pragma unroll
    for (int comp = 0; comp < 2; ++comp) {
      int dq0 = 0, dq1 = 0, dq2 = 0, dq3 = 0;
      if (cbp_chroma >= 1) {
        int* q = &qdc_all[comp * 4];
        int w0 = q[0] + q[1] + q[2] + q[3], w1 = q[0] - q[1] + q[2] - q[3];
        int w2 = q[0] + q[1] - q[2] - q[3], w3 = q[0] - q[1] - q[2] + q[3];
        dq0 = dequant_chroma_dc_v(w0, qpc);
        dq1 = dequant_chroma_dc_v(w1, qpc);
        dq2 = dequant_chroma_dc_v(w2, qpc);
        dq3 = dequant_chroma_dc_v(w3, qpc);
      }
      dcr_all[comp * 4 + 0] = dq0;
      dcr_all[comp * 4 + 1] = dq1;
      dcr_all[comp * 4 + 2] = dq2;
      dcr_all[comp * 4 + 3] = dq3;
    }

    // ---- recon
    int ncb2 = leftcb, ncr2 = leftcr;
#pragma unroll
    for (int pass = 0; pass < 2; ++pass) {
      int comp = pass;
      int sub = g;
      int scx = (sub & 1) * 4, scy = (sub >> 1) * 4;
      int rr = scy + (c >> 2), cc2 = scx + (c & 3);
      int d = (c == 0) ? dcr_all[comp * 4 + sub]
                       : (cbp_chroma == 2
                              ? dequant_c(lvl_p[pass], qpc, coeff_cls(c))
                              : 0);
      int rec = idct4_wave(d, lane);
      int lH = comp ? __shfl(leftcr, rr) : __shfl(leftcb, rr);
      int pdc = comp ? ((rr & 4) ? crb : crt) : ((rr & 4) ? cbb : cbt);
      int pred = chroma_mode == 1 ? lH : pdc;
      int pix = clip8(rec + pred);
      uint8_t* dp = comp ? curCr : curCb;
      int p1 = __shfl(pix, lane + 1), p2 = __shfl(pix, lane + 2),
          p3 = __shfl(pix, lane + 3);
      if ((c & 3) == 0) {
        *reinterpret_cast<uint32_t*>(
            dp + (size_t)(cy0 + rr) * cpitch + cx0 + cc2) =
            (uint32_t)pix | ((uint32_t)p1 << 8) | ((uint32_t)p2 << 16) |
            ((uint32_t)p3 << 24);
      }
      int cand = __shfl(pix, lane < 4 ? 19 + 4 * lane : 35 + 4 * lane);
      if (lane < 8) {
        if (pass == 0) ncb2 = cand;
        else ncr2 = cand;
      }
    }
    leftcb = ncb2;
    leftcr = ncr2;

    if (lane == 0) M[1] = chroma_mode;   // m1 is unused for intra MBs
    have_left = true;
    }

    // rotate the prefetched state
    csrc_cur = csrc_nxt;
    cref_cur = cref_nxt;
    colcb_cur = colcb_nxt;
    colcr_cur = colcr_nxt;
    m0_cur = m0_nxt;
    m1_cur = m1_nxt;
  }
}

__global__ void __launch_bounds__(128) k_h264_rows(
    const uint8_t* __restrict__ srcY, const uint8_t* __restrict__ srcCb,
    const uint8_t* __restrict__ srcCr, int ypitch, int cpitch, int w, int h,
    const uint8_t* __restrict__ refY, const uint8_t* __restrict__ refCb,
    const uint8_t* __restrict__ refCr, uint8_t* __restrict__ curY,
    uint8_t* __restrict__ curCb, uint8_t* __restrict__ curCr, int mbw,
    const RowJob* __restrict__ jobs, int16_t* __restrict__ levels,
    int* __restrict__ meta) {
  const RowJob job = jobs[blockIdx.x];
  const int qp = job.qp;
  const bool i_slice = (job.flags & 1) != 0;
  const int mby = job.mb_row;
  const int lane = threadIdx.x & 63;

  if (threadIdx.x < 64) {
    luma_row(srcY, ypitch, w, h, refY, curY, mbw, mby, qp, i_slice, levels,
             meta, lane, job.mbx0, job.seg_mbw);
  } else {
    chroma_row(srcCb, srcCr, cpitch, w, h, refCb, refCr, curCb, curCr, mbw,
               mby, dev_chroma_qp(qp), i_slice, levels, meta, lane,
               job.mbx0, job.seg_mbw);
  }
}

// ---------------------------------------------------------------------------
// P-frame mode decision (diamond search reference path): one workgroup
// (64 lanes) per MB of each P row. Thresholds match the CPU encoder.
__global__ void __launch_bounds__(64) k_h264_me(
    const uint8_t* __restrict__ srcY, int ypitch, int w, int h,
    const uint8_t* __restrict__ refY, int mbw, int seg_max,
    int frame_w_mb16, const RowJob* __restrict__ jobs,
    int* __restrict__ meta) {
  const int job_idx = blockIdx.x / seg_max;
  const RowJob job = jobs[job_idx];
  const int mbx = job.mbx0 + blockIdx.x % seg_max;
  if (blockIdx.x % seg_max >= job.seg_mbw) return;
  if (job.flags & 1) return;  // I rows have no ME
  const int lane = threadIdx.x;
  const int mby = job.mb_row;
  const int x0 = mbx * 16, y0 = mby * 16;
  const size_t mb_index = (size_t)mby * mbw + mbx;

  const int r = lane >> 2, cq = (lane & 3) * 4;
  int sp[4];
  {
    const uint8_t* s = srcY + (size_t)min(y0 + r, h - 1) * ypitch;
    sp[0] = s[min(x0 + cq + 0, w - 1)];
    sp[1] = s[min(x0 + cq + 1, w - 1)];
    sp[2] = s[min(x0 + cq + 2, w - 1)];
    sp[3] = s[min(x0 + cq + 3, w - 1)];
  }
  auto sad_at = [&](int mx, int my) -> int {
    const uint8_t* rp = refY + (size_t)(y0 + my + r) * ypitch + x0 + mx + cq;
    int s = abs(sp[0] - rp[0]) + abs(sp[1] - rp[1]) + abs(sp[2] - rp[2]) +
            abs(sp[3] - rp[3]);
    return wave_sum_i(s);
  };
  // interpolated SAD at a quarter-pel mv (half-pel grid)
  auto sad_q = [&](int qx, int qy) -> int {
    int ixq = qx >> 2, iyq = qy >> 2, fxq = qx & 3, fyq = qy & 3;
    int s = 0;
    for (int j = 0; j < 4; ++j)
      s += abs(sp[j] - luma_interp(refY, ypitch, x0 + ixq + cq + j,
                                   y0 + iyq + r, fxq, fyq));
    return wave_sum_i(s);
  };
  // the full interpolation window must stay inside frame + stripe
  auto window_ok = [&](int qx, int qy) -> bool {
    int ixq = qx >> 2, iyq = qy >> 2;
    // +20: quarter positions may also read the x+1 / y+1 half sample
    return x0 + ixq - 2 >= 0 && x0 + ixq + 20 <= frame_w_mb16 &&
           y0 + iyq - 2 >= job.stripe_y0 && y0 + iyq + 20 <= job.stripe_y1;
  };

  const int qp = job.qp;
  const int skip_thresh = 48 << (qp / 6);
  // residual coding makes inter viable whenever MC is a decent predictor
  const int inter_thresh = 6 * skip_thresh;

  int sad0 = sad_at(0, 0);
  int mode, bqx = 0, bqy = 0;
  if (sad0 <= skip_thresh) {
    mode = kSkip;
  } else {
    int best = sad0;
    int bmx = 0, bmy = 0;
    // seed with the previous frame's MV/hint (meta still holds it)
    {
      int pm0 = meta[mb_index * kMetaPerMb + 0];
      int pm1 = meta[mb_index * kMetaPerMb + 1];
      int pvx, pvy;
      if ((pm0 & 3) == kInter) {
        pvx = ((int)(short)(pm1 & 0xFFFF)) >> 2;
        pvy = (pm1 >> 16) >> 2;
      } else {
        pvx = ((pm0 >> 8) & 127) - 64;
        pvy = ((pm0 >> 15) & 127) - 64;
      }
      pvx = max(-48, min(48, pvx));
      pvy = max(-48, min(48, pvy));
      if ((pvx | pvy) != 0 && window_ok(pvx * 4, pvy * 4)) {
        int s = sad_at(pvx, pvy);
        if (s < best) {
          best = s;
          bmx = pvx;
          bmy = pvy;
        }
      }
    }
    static const int pat[8][2] = {{-1, 0}, {1, 0},  {0, -1}, {0, 1},
                                  {-1, -1}, {1, 1}, {-1, 1}, {1, -1}};
    for (int iter = 0; iter < 16; ++iter) {
      int cx = bmx, cy = bmy;
      bool improved = false;
      for (int pi = 0; pi < 8; ++pi) {
        int mx = cx + pat[pi][0], my = cy + pat[pi][1];
        if (abs(mx) > 16 || abs(my) > 16) continue;
        if (!(mx == 0 && my == 0) && !window_ok(mx * 4, my * 4)) continue;
        int s = sad_at(mx, my);
        if (s < best) {
          best = s;
          bmx = mx;
          bmy = my;
          improved = true;
        }
      }
      if (!improved) break;
    }
    // half- then quarter-pel refinement rings (skipped when the MB is
    // headed to intra regardless)
    bqx = bmx * 4;
    bqy = bmy * 4;
    if (best <= 2 * inter_thresh) {
      for (int step = 2; step >= 1; --step) {
        int cqx = bqx, cqy = bqy;
        for (int pi = 0; pi < 8; ++pi) {
          int qx = cqx + step * pat[pi][0], qy = cqy + step * pat[pi][1];
          if (!window_ok(qx, qy)) continue;
          int s = sad_q(qx, qy);
          if (s < best) {
            best = s;
            bqx = qx;
            bqy = qy;
          }
        }
      }
    }
    if (best <= inter_thresh)
      mode = kInter;   // mv may be (0,0): residuals carry the change
    else
      mode = kIntra;
  }
  if (lane == 0) {
    int hx = max(-48, min(48, bqx >> 2)) + 64;
    int hy = max(-48, min(48, bqy >> 2)) + 64;
    meta[mb_index * kMetaPerMb + 0] = mode | (hx << 8) | (hy << 15);
    meta[mb_index * kMetaPerMb + 1] =
        (bqx & 0xFFFF) | (int)((unsigned)bqy << 16);
  }
}

// ---------------------------------------------------------------------------
// MFMA-int8 full-search motion estimation.
//
// Scores ALL candidates of a ±8 even-step grid (9x9 = 81) by SSD using the
// matrix cores:  SSD_n = S2 - 2*X_n + R2_n  with pixels centered to int8
// (p-128; SSD is shift-invariant). Per 16-candidate group:
//   X_n  : 4x mfma_i32_16x16x64_i8, A rows = broadcast src chunk (64 px),
//          B cols = candidate chunks       -> D[0][n] = sum(a*b_n)
//   R2_n : 4x mfma with A = B              -> diag D[n][n] = sum(b_n^2)
// One wave per MB. The final skip/inter decision uses the winner's SAD so
// thresholds stay identical to the CPU reference encoder.
using i32x4 = __attribute__((__vector_size__(16))) int;

__global__ void __launch_bounds__(64) k_h264_me_mfma(
    const uint8_t* __restrict__ srcY, int ypitch, int w, int h,
    const uint8_t* __restrict__ refY, const uint8_t* __restrict__ srcY2,
    const uint8_t* __restrict__ refY2, int ypitch2, int mbw, int seg_max,
    int frame_w_mb16, const RowJob* __restrict__ jobs,
    int* __restrict__ meta) {
  const int job_idx = blockIdx.x / seg_max;
  const RowJob job = jobs[job_idx];
  const int mbx = job.mbx0 + blockIdx.x % seg_max;
  if (blockIdx.x % seg_max >= job.seg_mbw) return;
  if (job.flags & 1) return;
  const int lane = threadIdx.x;
  const int mby = job.mb_row;
  const int x0 = mbx * 16, y0 = mby * 16;
  const size_t mb_index = (size_t)mby * mbw + mbx;

  // --- src MB into LDS (+ centered copy for fragments)
  __shared__ int8_t s_a[256];
  {
    int r = lane >> 2, cq = (lane & 3) * 4;
    const uint8_t* s = srcY + (size_t)min(y0 + r, h - 1) * ypitch;
    for (int j = 0; j < 4; ++j)
      s_a[r * 16 + cq + j] =
          (int8_t)((int)s[min(x0 + cq + j, w - 1)] - 128);
  }
  __syncthreads();

  // S2 = sum(a^2)
  int s2 = 0;
  {
    int r = lane >> 2, cq = (lane & 3) * 4;
    for (int j = 0; j < 4; ++j) {
      int v = s_a[r * 16 + cq + j];
      s2 += v * v;
    }
    s2 = wave_sum_i(s2);
  }

  // A fragment per k-chunk-group kg = lane>>4: bytes a_q[kg*16 .. +15]
  // (identical for the 16 lanes of a group -> broadcast rows). Chunk q is
  // selected per MFMA call; precompute all 4 chunks' fragments.
  i32x4 afrag[4];
  {
    int kg = lane >> 4;
    for (int q = 0; q < 4; ++q) {
      // chunk q = MB rows 4q..4q+3; k = r*16+c (r<4). bytes kg*16..+15 of
      // the chunk = row kg of the chunk.
      const int8_t* p = &s_a[(4 * q + kg) * 16];
      int words[4];
      for (int t = 0; t < 4; ++t)
        words[t] = (uint8_t)p[4 * t] | ((uint8_t)p[4 * t + 1] << 8) |
                   ((uint8_t)p[4 * t + 2] << 16) |
                   ((uint8_t)p[4 * t + 3] << 24);
      afrag[q] = i32x4{words[0], words[1], words[2], words[3]};
    }
  }

  // candidate grids: 81 candidates (9x9) in 6 groups of up to 16, scored
  // by MFMA SSD around an arbitrary center and step
  const int NC = 81;
  int best_score = INT_MAX, best_mvx = 0, best_mvy = 0;
  const int col = lane & 15, kg = lane >> 4;

  auto score_grid = [&](int cx0, int cy0, int step) {
    for (int g0 = 0; g0 < NC; g0 += 16) {
      int cand = g0 + col;
      int mx = 0, my = 0;
      bool valid = cand < NC;
      if (valid) {
        mx = cx0 + ((cand % 9) - 4) * step;
        my = cy0 + ((cand / 9) - 4) * step;
        if (x0 + mx < 0 || x0 + mx + 16 > frame_w_mb16 ||
            y0 + my < job.stripe_y0 || y0 + my + 16 > job.stripe_y1)
          valid = false;
      }
      i32x4 cross = {0, 0, 0, 0};
      i32x4 norm = {0, 0, 0, 0};
      for (int q = 0; q < 4; ++q) {
        // B fragment: lane holds B[k = kg*16 + j][col] = candidate col's
        // chunk-q byte (row kg of chunk, cols 0..15 -> j)
        int8_t bb[16];
        if (valid) {
          const uint8_t* rp =
              refY + (size_t)(y0 + my + 4 * q + kg) * ypitch + x0 + mx;
          for (int j = 0; j < 16; ++j) bb[j] = (int8_t)((int)rp[j] - 128);
        } else {
          for (int j = 0; j < 16; ++j) bb[j] = 0;
        }
        int words[4];
        for (int t = 0; t < 4; ++t)
          words[t] = (uint8_t)bb[4 * t] | ((uint8_t)bb[4 * t + 1] << 8) |
                     ((uint8_t)bb[4 * t + 2] << 16) |
                     ((uint8_t)bb[4 * t + 3] << 24);
        i32x4 bfrag = i32x4{words[0], words[1], words[2], words[3]};
        cross = __builtin_amdgcn_mfma_i32_16x16x64_i8(afrag[q], bfrag,
                                                      cross, 0, 0, 0);
        norm = __builtin_amdgcn_mfma_i32_16x16x64_i8(bfrag, bfrag, norm,
                                                     0, 0, 0);
      }
      // D mapping (16x16): X_n in row 0 (lane n, acc 0); R2_n on the
      // diagonal (lane ((n>>2)<<4)|n, acc n&3). Reduce on lane 0.
      for (int n_base = 0; n_base < 16; ++n_base) {
        int xl = __shfl(cross[0], n_base);
        int diag_lane = ((n_base >> 2) << 4) | n_base;
        int rl;
        switch (n_base & 3) {
          case 0: rl = __shfl(norm[0], diag_lane); break;
          case 1: rl = __shfl(norm[1], diag_lane); break;
          case 2: rl = __shfl(norm[2], diag_lane); break;
          default: rl = __shfl(norm[3], diag_lane); break;
        }
        int cand2 = g0 + n_base;
        if (cand2 < NC && lane == 0) {
          int mx2 = cx0 + ((cand2 % 9) - 4) * step;
          int my2 = cy0 + ((cand2 / 9) - 4) * step;
          bool v2 = !(x0 + mx2 < 0 || x0 + mx2 + 16 > frame_w_mb16 ||
                      y0 + my2 < job.stripe_y0 ||
                      y0 + my2 + 16 > job.stripe_y1);
          if (v2) {
            int ssd = s2 - 2 * xl + rl;
            // small center bias keeps MVs compact on flat content
            int score = ssd + (abs(mx2) + abs(my2)) * 4;
            if (score < best_score) {
              best_score = score;
              best_mvx = mx2;
              best_mvy = my2;
            }
          }
        }
      }
    }
  };

  // pass 0: previous-frame tracking state (meta still holds last
  // frame's values). An MB whose LAST frame was hopeless (unmatchable
  // at any candidate — noise) skips every search pass except a
  // 1-in-8-frame probe; the sad(0,0) skip/inter test below still runs
  // every frame, so content that settles becomes P_Skip immediately —
  // and going skip clears the hopeless bit, resuming full search.
  const int pm0 = meta[mb_index * kMetaPerMb + 0];
  const int pm1 = meta[mb_index * kMetaPerMb + 1];
  const bool prev_hopeless = (pm0 >> 22) & 1;
  if (!(prev_hopeless && (job.frame_num & 7) != 0)) {
  // pass 1: fine grid at (0,0)
  score_grid(0, 0, 2);
  // pass 2: fine grid at the previous frame's MV/hint for this MB (meta
  // still holds last frame's values here) — tracks sustained motion
  // beyond +-8 even across intra fallbacks
  {
    int pvx = 0, pvy = 0;
    if ((pm0 & 3) == kInter) {
      pvx = ((int)(short)(pm1 & 0xFFFF)) >> 2;
      pvy = (pm1 >> 16) >> 2;
    } else {
      pvx = ((pm0 >> 8) & 127) - 64;    // intra/skip hint (bits 8..21)
      pvy = ((pm0 >> 15) & 127) - 64;
    }
    pvx = max(-48, min(48, pvx));
    pvy = max(-48, min(48, pvy));
    if ((pvx | pvy) != 0 && (abs(pvx) > 2 || abs(pvy) > 2))
      score_grid(pvx, pvy, 2);
  }
  // pass 3: pyramid acquisition when nothing fits yet — first frame of
  // fast new motion. A +-12 step-1 SAD search on the QUARTER-res luma
  // pyramid covers +-48 at full res with correlation intact at that
  // scale; the winner seeds a fine full-res grid.
  if (__shfl(best_score, 0) > 256 * 180) {
    // this MB at quarter res: 4x4 proxy block
    const int qx0 = x0 >> 2, qy0 = y0 >> 2;
    const int qw = frame_w_mb16 >> 2;
    const int qs_y0 = job.stripe_y0 >> 2, qs_y1 = job.stripe_y1 >> 2;
    int px[4];
    {
      int rr = lane & 3;
      const uint8_t* sr = srcY2 + (size_t)min(qy0 + rr, (h >> 2) - 1) *
                                      ypitch2;
      for (int j = 0; j < 4; ++j)
        px[j] = sr[min(qx0 + j, qw - 1)];
    }
    int bs = INT_MAX, bmx2 = 0, bmy2 = 0;
    // 25x25 candidates, distributed over lanes (lane covers cands with
    // index % 16 == lane>>2 pattern via strided loop)
    for (int ci = lane >> 2; ci < 625; ci += 16) {
      int mx = (ci % 25) - 12, my = (ci / 25) - 12;
      if (qx0 + mx < 0 || qx0 + mx + 4 > qw || qy0 + my < qs_y0 ||
          qy0 + my + 4 > qs_y1)
        continue;
      const uint8_t* rp =
          refY2 + (size_t)(qy0 + my + (lane & 3)) * ypitch2 + qx0 + mx;
      int sad = abs(px[0] - rp[0]) + abs(px[1] - rp[1]) +
                abs(px[2] - rp[2]) + abs(px[3] - rp[3]);
      // sum the 4 row-partials within the 4-lane group
      sad += __shfl_xor(sad, 1);
      sad += __shfl_xor(sad, 2);
      if (sad < bs) {
        bs = sad;
        bmx2 = mx;
        bmy2 = my;
      }
    }
    // wave-reduce the best candidate
    for (int d = 4; d < 64; d <<= 1) {
      int os = __shfl_xor(bs, d);
      int ox = __shfl_xor(bmx2, d);
      int oy = __shfl_xor(bmy2, d);
      if (os < bs || (os == bs && (oy < bmy2 || (oy == bmy2 && ox < bmx2)))) {
        bs = os;
        bmx2 = ox;
        bmy2 = oy;
      }
    }
    // fine full-res verification grid at the pyramid seed
    if (bs < INT_MAX) score_grid(bmx2 * 4, bmy2 * 4, 2);
  }
  }  // end non-hopeless search passes

  best_score = __shfl(best_score, 0);
  best_mvx = __shfl(best_mvx, 0);
  best_mvy = __shfl(best_mvy, 0);

  // SAD of (0,0) and of the SSD winner -> same thresholds as CPU encoder
  const int r = lane >> 2, cq = (lane & 3) * 4;
  int sp[4];
  {
    const uint8_t* s = srcY + (size_t)min(y0 + r, h - 1) * ypitch;
    for (int j = 0; j < 4; ++j) sp[j] = s[min(x0 + cq + j, w - 1)];
  }
  auto sad_at = [&](int mx, int my) -> int {
    const uint8_t* rp = refY + (size_t)(y0 + my + r) * ypitch + x0 + mx + cq;
    int s = abs(sp[0] - rp[0]) + abs(sp[1] - rp[1]) + abs(sp[2] - rp[2]) +
            abs(sp[3] - rp[3]);
    return wave_sum_i(s);
  };
  // interpolated SAD at a quarter-pel mv (half-pel grid)
  auto sad_q = [&](int qx, int qy) -> int {
    int ixq = qx >> 2, iyq = qy >> 2, fxq = qx & 3, fyq = qy & 3;
    int s = 0;
    for (int j = 0; j < 4; ++j)
      s += abs(sp[j] - luma_interp(refY, ypitch, x0 + ixq + cq + j,
                                   y0 + iyq + r, fxq, fyq));
    return wave_sum_i(s);
  };
  auto window_ok = [&](int qx, int qy) -> bool {
    int ixq = qx >> 2, iyq = qy >> 2;
    // +20: quarter positions may also read the x+1 / y+1 half sample
    return x0 + ixq - 2 >= 0 && x0 + ixq + 20 <= frame_w_mb16 &&
           y0 + iyq - 2 >= job.stripe_y0 && y0 + iyq + 20 <= job.stripe_y1;
  };

  const int qp = job.qp;
  const int skip_thresh = 48 << (qp / 6);
  const int inter_thresh = 6 * skip_thresh;
  int sad0 = sad_at(0, 0);
  int mode, oqx = 0, oqy = 0;
  int hintx = 0, hinty = 0;
  if (sad0 <= skip_thresh) {
    mode = kSkip;
  } else {
    int best_sad = (best_mvx || best_mvy) ? sad_at(best_mvx, best_mvy)
                                          : sad0;
    // refine the even-grid SSD winner: integer +-1 ring, then half-pel
    // (skipped when the MB is headed to intra regardless)
    int bqx = best_mvx * 4, bqy = best_mvy * 4;
    static const int pat[8][2] = {{-1, 0}, {1, 0},  {0, -1}, {0, 1},
                                  {-1, -1}, {1, 1}, {-1, 1}, {1, -1}};
    if (best_sad <= 2 * inter_thresh) {
      for (int iter = 0; iter < 8; ++iter) {
        int cqx = bqx, cqy = bqy;
        bool improved = false;
        for (int pi = 0; pi < 8; ++pi) {
          int qx = cqx + 4 * pat[pi][0], qy = cqy + 4 * pat[pi][1];
          if (!window_ok(qx, qy)) continue;
          int s = sad_at(qx >> 2, qy >> 2);
          if (s < best_sad) {
            best_sad = s;
            bqx = qx;
            bqy = qy;
            improved = true;
          }
        }
        if (!improved) break;
      }
      for (int step = 2; step >= 1; --step) {
        int cqx2 = bqx, cqy2 = bqy;
        for (int pi = 0; pi < 8; ++pi) {
          int qx = cqx2 + step * pat[pi][0], qy = cqy2 + step * pat[pi][1];
          if (!window_ok(qx, qy)) continue;
          int s = sad_q(qx, qy);
          if (s < best_sad) {
            best_sad = s;
            bqx = qx;
            bqy = qy;
          }
        }
      }
    }
    if (best_sad <= inter_thresh) {
      mode = kInter;   // mv may be (0,0): residuals carry the change
      oqx = bqx;
      oqy = bqy;
    } else {
      mode = kIntra;
    }
    hintx = bqx >> 2;
    hinty = bqy >> 2;
  }
  bool hopeless = mode == kIntra && __shfl(best_score, 0) > 256 * 1200;
  if (hopeless) hintx = hinty = 0;
  if (lane == 0) {
    // best-found integer mv persists as a tracking hint even when the MB
    // goes intra; the luma row wave preserves these bits on its M[0]
    // rewrite
    int hx = max(-48, min(48, hintx)) + 64;
    int hy = max(-48, min(48, hinty)) + 64;
    meta[mb_index * kMetaPerMb + 0] =
        mode | (hx << 8) | (hy << 15) | ((hopeless ? 1 : 0) << 22);
    meta[mb_index * kMetaPerMb + 1] =
        (oqx & 0xFFFF) | (int)((unsigned)oqy << 16);
  }
}

// ---------------------------------------------------------------------------
// host launchers
void launch_h264_me(const uint8_t* srcY, int ypitch, int w, int h,
                    const uint8_t* refY, const uint8_t* srcY2,
                    const uint8_t* refY2, int ypitch2, int mbw, int n_jobs,
                    const RowJob* d_jobs, int* d_meta, hipStream_t stream,
                    bool use_mfma) {
  if (n_jobs == 0) return;
  const int seg_max = mbw < kMaxSegMbw ? mbw : kMaxSegMbw;
  if (use_mfma) {
    hipLaunchKernelGGL(k_h264_me_mfma, dim3(n_jobs * seg_max), dim3(64), 0,
                       stream, srcY, ypitch, w, h, refY, srcY2, refY2,
                       ypitch2, mbw, seg_max, mbw * 16, d_jobs, d_meta);
  } else {
    hipLaunchKernelGGL(k_h264_me, dim3(n_jobs * seg_max), dim3(64), 0,
                       stream, srcY, ypitch, w, h, refY, mbw, seg_max,
                       mbw * 16, d_jobs, d_meta);
  }
}

void launch_h264_rows(const uint8_t* srcY, const uint8_t* srcCb,
                      const uint8_t* srcCr, int ypitch, int cpitch, int w,
                      int h, const uint8_t* refY, const uint8_t* refCb,
                      const uint8_t* refCr, uint8_t* curY, uint8_t* curCb,
                      uint8_t* curCr, int mbw, int n_jobs,
                      const RowJob* d_jobs, int16_t* d_levels, int* d_meta,
                      hipStream_t stream) {
  if (n_jobs == 0) return;
  hipLaunchKernelGGL(k_h264_rows, dim3(n_jobs), dim3(128), 0, stream, srcY,
                     srcCb, srcCr, ypitch, cpitch, w, h, refY, refCb, refCr,
                     curY, curCb, curCr, mbw, d_jobs, d_levels, d_meta);
}


// ---------------------------------------------------------------------------
// 2x2 box downsample (luma pyramid for ME acquisition). One thread per
// output pixel.
__global__ void k_downsample2(const uint8_t* __restrict__ src, int spitch,
                              int sw, int sh, uint8_t* __restrict__ dst,
                              int dpitch) {
  int x = blockIdx.x * blockDim.x + threadIdx.x;
  int y = blockIdx.y * blockDim.y + threadIdx.y;
  int dw = sw >> 1, dh = sh >> 1;
  if (x >= dw || y >= dh) return;
  const uint8_t* r0 = src + (size_t)(2 * y) * spitch + 2 * x;
  const uint8_t* r1 = r0 + spitch;
  dst[(size_t)y * dpitch + x] =
      (uint8_t)((r0[0] + r0[1] + r1[0] + r1[1] + 2) >> 2);
}

void launch_downsample2(const uint8_t* src, int spitch, int sw, int sh,
                        uint8_t* dst, int dpitch, hipStream_t stream) {
  dim3 b(16, 16);
  dim3 g(((sw >> 1) + 15) / 16, ((sh >> 1) + 15) / 16);
  hipLaunchKernelGGL(k_downsample2, g, b, 0, stream, src, spitch, sw, sh,
                     dst, dpitch);
}

}  // namespace h264gpu
}  // namespace hipflux
