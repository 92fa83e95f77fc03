// CDNA4 (gfx950) H.264 row kernel, 4-wave variant.
//
// The 2-wave kernel's per-MB serial chain is ISSUE-RATE bound: a lone
// wave on a CDNA SIMD issues one instruction every 4 cycles, so the MB
// walk's latency is proportional to the per-wave instruction count. This
// variant splits each MB-row slice across FOUR waves:
//   wave 0: luma transform passes 0,1 (blocks 0..7)
//   wave 1: luma transform passes 2,3 (blocks 8..15)
//   wave 2: chroma Cb (all 4 sub-blocks)
//   wave 3: chroma Cr
// Cross-wave state (luma DC coefficients, mode-decision cost halves,
// left-neighbor recon columns) moves through LDS behind s_barriers — two
// per intra MB, one per skip/inter MB. The cbp gates on reconstruction
// are dropped entirely: when cbp is 0 every level is 0 and dequant(0)=0,
// so gated and ungated recon are bit-identical — this removes the only
// other cross-wave dependency.
//
// Bit-exactness contract: identical streams to the 2-wave kernel (and so
// to the CPU reference encoder) — enforced by tests/test_gpu_h264.py.
#include <hip/hip_runtime.h>

#include "h264_gpu_layout.h"
#include "h264_kernels.h"
#include "h264_rows_common.h"

namespace hipflux {
namespace h264gpu {

namespace rows4 {

struct Shared {
  int dc[16];          // luma DC coefficients (raster block order)
  int ccost[2][2];     // [comp][0]=H cost, [1]=DC cost
  // left-neighbor recon columns, double-buffered by MB parity:
  // lcol[buf][0..15] luma, ccol[buf][0..7] Cb, ccol[buf][8..15] Cr
  uint8_t lcol[2][16];
  uint8_t ccol[2][16];
};

// ---- luma: wave w handles NP consecutive passes starting at w*NP ---------
template <int NP>
__device__ void luma_wave(const uint8_t* __restrict__ srcY, int ypitch,
                          int w_, int h, const uint8_t* __restrict__ refY,
                          uint8_t* __restrict__ curY, int mbw, int mby,
                          int qp, bool i_slice,
                          const int16_t* __restrict__ levels_base,
                          int* __restrict__ meta, int lane, int w,
                          int mbx0, int seg_mbw, Shared* sh) {
  const int y0 = mby * 16;
  const int r = lane >> 2, cq = (lane & 3) * 4;
  const int g = lane >> 4, c = lane & 15;
  const int zz = c_zz_of_pos[c];
  bool have_left = false;

  const int mbx_end = mbx0 + seg_mbw;
  for (int mbx = mbx0; mbx < mbx_end; ++mbx) {
    const int x0 = mbx * 16;
    const size_t mb_index = (size_t)mby * mbw + mbx;
    int16_t* L = const_cast<int16_t*>(levels_base) + mb_index * kLevelsPerMb;
    int* M = meta + mb_index * kMetaPerMb;
    const int cur_buf = mbx & 1, nxt_buf = cur_buf ^ 1;
    uint8_t* lcol = sh->lcol[cur_buf];
    uint8_t* lcol_n = sh->lcol[nxt_buf];

    int mode = kIntra, mvx = 0, mvy = 0;
    if (!i_slice) {
      int m0 = M[0];
      mode = m0 & 3;
      int m1 = M[1];
      mvx = (short)(m1 & 0xFFFF);
      mvy = m1 >> 16;
    }

    if (mode == kSkip) {
      // both waves cover all 16 rows: wave w copies rows where r parity
      // is irrelevant — split by halves: wave 0 rows 0..7, wave 1 8..15
      if (lane < 16 * NP) {
        int rr = w * 4 * NP + (lane >> 2);   // wave covers 4*NP rows
        int qq = (lane & 3) * 4;
        const uint8_t* s = refY + (size_t)(y0 + rr) * ypitch + x0 + qq;
        uint8_t* d = curY + (size_t)(y0 + rr) * ypitch + x0 + qq;
        *reinterpret_cast<uint32_t*>(d) = (uint32_t)s[0] |
            ((uint32_t)s[1] << 8) | ((uint32_t)s[2] << 16) |
            ((uint32_t)s[3] << 24);
      }
      if (lane < 4 * NP) {
        int row = w * 4 * NP + lane;
        lcol_n[row] = refY[(size_t)(y0 + row) * ypitch + x0 + 15];
      }
      __syncthreads();
      have_left = true;
      continue;
    }

    if (mode == kInter) {
      const int ix = mvx >> 2, iy = mvy >> 2;
      const int fx = mvx & 3, fy = mvy & 3;
      uint32_t psrc, ppred;
      {
        const uint8_t* sr = srcY + (size_t)min(y0 + r, h - 1) * ypitch;
        psrc = (uint32_t)sr[min(x0 + cq + 0, w_ - 1)] |
               ((uint32_t)sr[min(x0 + cq + 1, w_ - 1)] << 8) |
               ((uint32_t)sr[min(x0 + cq + 2, w_ - 1)] << 16) |
               ((uint32_t)sr[min(x0 + cq + 3, w_ - 1)] << 24);
        if ((fx | fy) == 0) {
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
      int lvl_p[NP];
#pragma unroll
      for (int pi = 0; pi < NP; ++pi) {
        int pass = NP * w + pi;
        int blk = pass * 4 + g;
        int bx = blk & 3, by = blk >> 2;
        int py = by * 4 + (c >> 2), px = bx * 4 + (c & 3);
        int resid = pix_at(py, px) - pred_at(py, px);
        int coefv = fdct4_wave(resid, lane);
        int lvl = quant_coeff(coefv, qp, coeff_cls(c), false);
        lvl = cap12_group(lvl, zz, true, lane);
        lvl_p[pi] = lvl;
        store_lvl_pair(L + kLumaAcOff + blk * 16, c, lvl, lane);
      }
      // recon: no cbp gate needed — zero levels dequantize to zero
#pragma unroll
      for (int pi = 0; pi < NP; ++pi) {
        int pass = NP * w + pi;
        int blk = pass * 4 + g;
        int bx = blk & 3, by = blk >> 2;
        int d = dequant_c(lvl_p[pi], qp, coeff_cls(c));
        int rec = idct4_wave(d, lane);
        int py = by * 4 + (c >> 2), px = bx * 4 + (c & 3);
        int pix = clip8(rec + pred_at(py, px));
        int p1 = __shfl(pix, lane + 1), p2 = __shfl(pix, lane + 2),
            p3 = __shfl(pix, lane + 3);
        if ((c & 3) == 0)
          *reinterpret_cast<uint32_t*>(
              curY + (size_t)(y0 + py) * ypitch + x0 + px) =
              (uint32_t)pix | ((uint32_t)p1 << 8) | ((uint32_t)p2 << 16) |
              ((uint32_t)p3 << 24);
        if (px == 15) lcol_n[py] = (uint8_t)pix;
      }
      __syncthreads();
      have_left = true;
      continue;
    }

    // ---- intra I16x16
    uint32_t psrc;
    {
      const uint8_t* s = srcY + (size_t)min(y0 + r, h - 1) * ypitch;
      psrc = (uint32_t)s[min(x0 + cq + 0, w_ - 1)] |
             ((uint32_t)s[min(x0 + cq + 1, w_ - 1)] << 8) |
             ((uint32_t)s[min(x0 + cq + 2, w_ - 1)] << 16) |
             ((uint32_t)s[min(x0 + cq + 3, w_ - 1)] << 24);
    }
    int dcval = 128;
    if (have_left) {
      int part = (lane < 16) ? lcol[lane] : 0;
      dcval = (wave_sum_i(part) + 8) >> 4;
    }
    int lv = have_left ? lcol[r] : 0;
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

    auto pix_at = [&](int py, int px) -> int {
      uint32_t v = __shfl(psrc, py * 4 + (px >> 2));
      return (v >> (8 * (px & 3))) & 0xFF;
    };

    int lvl_p[NP];
#pragma unroll
    for (int pi = 0; pi < NP; ++pi) {
      int pass = NP * w + pi;
      int blk = pass * 4 + g;
      int bx = blk & 3, by = blk >> 2;
      int py = by * 4 + (c >> 2), px = bx * 4 + (c & 3);
      int pred = luma_mode == 1 ? lcol[py] : dcval;
      int resid = pix_at(py, px) - pred;
      int coefv = fdct4_wave(resid, lane);
      if (c == 0) sh->dc[blk] = coefv;
      int lvl = (c == 0) ? 0 : quant_coeff(coefv, qp, coeff_cls(c));
      lvl = cap12_group(lvl, zz, c != 0, lane);
      lvl_p[pi] = lvl;
      store_lvl_pair(L + kLumaAcOff + blk * 16, c, lvl, lane);
    }
    __syncthreads();

    // ---- luma DC chain: computed once on wave 0; the reconstructed DCs
    // are broadcast back through sh->dc (the raw DCs it held are dead
    // after this). Saves the redundant hadamard/quant/cap chain on the
    // other three luma waves — they wait at the barrier instead.
    if (w == 0) {
      int dcreg = (lane < 16) ? sh->dc[lane] : 0;
      int had = hadamard4_wave(dcreg, lane, true);
      int qdc = quant_dc_v(had, qp);
      qdc = cap12_group(qdc, zz, lane < 16, lane);
      if (lane < 16) store_lvl_pair(L + kLumaDcOff, lane, qdc, lane);
      int ih = hadamard4_wave(qdc, lane, false);
      if (lane < 16) sh->dc[lane] = dequant_luma_dc_v(ih, qp);
    }
    __syncthreads();

    // recon (AC gate dropped: zero levels dequantize to zero)
#pragma unroll
    for (int pi = 0; pi < NP; ++pi) {
      int pass = NP * w + pi;
      int blk = pass * 4 + g;
      int bx = blk & 3, by = blk >> 2;
      int dcb = sh->dc[blk];
      int d = (c == 0) ? dcb : dequant_c(lvl_p[pi], qp, coeff_cls(c));
      int rec = idct4_wave(d, lane);
      int py = by * 4 + (c >> 2), px = bx * 4 + (c & 3);
      int pred = luma_mode == 1 ? lcol[py] : dcval;
      int pix = clip8(rec + pred);
      int p1 = __shfl(pix, lane + 1), p2 = __shfl(pix, lane + 2),
          p3 = __shfl(pix, lane + 3);
      if ((c & 3) == 0)
        *reinterpret_cast<uint32_t*>(
            curY + (size_t)(y0 + py) * ypitch + x0 + px) =
            (uint32_t)pix | ((uint32_t)p1 << 8) | ((uint32_t)p2 << 16) |
            ((uint32_t)p3 << 24);
      if (px == 15) lcol_n[py] = (uint8_t)pix;
    }

    if (w == 0 && lane == 0) {
      // preserve the ME tracking hint (bits 8..21) across the rewrite
      int pm0 = i_slice ? 0 : M[0];
      M[0] = kIntra | (luma_mode << 2) | (pm0 & 0x7FFF00);
    }
    __syncthreads();
    have_left = true;
  }
}

// ---- chroma: one wave per component ---------------------------------------
__device__ void chroma_wave(const uint8_t* __restrict__ srcC,
                            const uint8_t* __restrict__ refC,
                            uint8_t* __restrict__ curC, int cpitch, int w_,
                            int h, int mbw, int mby, int qpc, bool i_slice,
                            const int16_t* __restrict__ levels_base,
                            int* __restrict__ meta, int lane, int comp,
                            int mbx0, int seg_mbw, Shared* sh) {
  const int cy0 = mby * 8;
  const int g = lane >> 4, c = lane & 15;   // sub-block g, coeff c
  const int zz = c_zz_of_pos[c];
  const int cw = (w_ + 1) / 2, chh = (h + 1) / 2;
  bool have_left = false;

  const int mbx_end = mbx0 + seg_mbw;
  for (int mbx = mbx0; mbx < mbx_end; ++mbx) {
    const int cx0 = mbx * 8;
    const size_t mb_index = (size_t)mby * mbw + mbx;
    int16_t* L = const_cast<int16_t*>(levels_base) + mb_index * kLevelsPerMb;
    int* M = meta + mb_index * kMetaPerMb;
    const int cur_buf = mbx & 1, nxt_buf = cur_buf ^ 1;
    uint8_t* ccol = sh->ccol[cur_buf] + comp * 8;
    uint8_t* ccol_n = sh->ccol[nxt_buf] + comp * 8;

    int mode = kIntra, mvx = 0, mvy = 0;
    if (!i_slice) {
      int m0 = M[0];
      mode = m0 & 3;
      int m1 = M[1];
      mvx = (short)(m1 & 0xFFFF);
      mvy = m1 >> 16;
    }

    if (mode == kSkip) {
      if (lane < 16) {
        int rr = lane >> 1, qq = (lane & 1) * 4;
        const uint8_t* s = refC + (size_t)(cy0 + rr) * cpitch + cx0 + qq;
        uint8_t* d = curC + (size_t)(cy0 + rr) * cpitch + cx0 + qq;
        *reinterpret_cast<uint32_t*>(d) =
            (uint32_t)s[0] | ((uint32_t)s[1] << 8) | ((uint32_t)s[2] << 16) |
            ((uint32_t)s[3] << 24);
      }
      if (lane < 8)
        ccol_n[lane] = refC[(size_t)(cy0 + lane) * cpitch + cx0 + 7];
      __syncthreads();
      have_left = true;
      continue;
    }

    // one source/pred pixel per lane: (row = lane>>3, col = lane&7)
    const int prow = lane >> 3, pcol = lane & 7;
    int spx, ppx = 0;
    {
      int sr = min(cy0 + prow, chh - 1);
      spx = srcC[(size_t)sr * cpitch + min(cx0 + pcol, cw - 1)];
    }
    if (mode == kInter) {
      const int cix = mvx >> 3, ciy = mvy >> 3;
      const int cdx = mvx & 7, cdy = mvy & 7;
      ppx = chroma_interp(refC, cpitch, cx0 + cix + pcol, cy0 + ciy + prow,
                          cdx, cdy);
    }
    auto cpix = [&](int rr, int cc) -> int {
      return __shfl(spx, rr * 8 + cc);
    };
    auto cprd = [&](int rr, int cc) -> int {
      return __shfl(ppx, rr * 8 + cc);
    };

    int chroma_mode = 0;
    if (mode == kIntra) {
      // cost halves: this wave covers its component only; the decision
      // sums both components (published through LDS, combined after the
      // barrier below)
      int dtop = 128, dbot = 128;
      if (have_left) {
        dtop = (ccol[0] + ccol[1] + ccol[2] + ccol[3] + 2) >> 2;
        dbot = (ccol[4] + ccol[5] + ccol[6] + ccol[7] + 2) >> 2;
      }
      int dd = (prow & 4) ? dbot : dtop;
      int lvv = have_left ? ccol[prow] : 0;
      int ch = abs(spx - lvv), cdc0 = abs(spx - dd);
      ch = wave_sum_i(ch);
      cdc0 = wave_sum_i(cdc0);
      if (lane == 0) {
        sh->ccost[comp][0] = ch;
        sh->ccost[comp][1] = cdc0;
      }
      __syncthreads();
      int th = sh->ccost[0][0] + sh->ccost[1][0];
      int tdc = sh->ccost[0][1] + sh->ccost[1][1];
      chroma_mode = (have_left && th < tdc) ? 1 : 0;
      __syncthreads();   // pairs with the luma DC-broadcast barrier
    }

    const int scx = (g & 1) * 4, scy = (g >> 1) * 4;
    const int rr = scy + (c >> 2), cc2 = scx + (c & 3);
    int dtop = 128, dbot = 128;
    if (mode == kIntra && have_left) {
      dtop = (ccol[0] + ccol[1] + ccol[2] + ccol[3] + 2) >> 2;
      dbot = (ccol[4] + ccol[5] + ccol[6] + ccol[7] + 2) >> 2;
    }
    int pred;
    if (mode == kInter)
      pred = cprd(rr, cc2);
    else if (chroma_mode == 1)
      pred = ccol[rr];
    else
      pred = (rr & 4) ? dbot : dtop;

    int resid = cpix(rr, cc2) - pred;
    int coefv = fdct4_wave(resid, lane);
    int dcslot = __shfl(coefv, (lane & 3) * 16);   // dc of sub (lane&3)
    int lvl = (c == 0) ? 0
                       : quant_coeff(coefv, qpc, coeff_cls(c),
                                     mode == kIntra);
    lvl = cap12_group(lvl, zz, c != 0, lane);
    store_lvl_pair(L + kChromaAcOff + (comp * 4 + g) * 16, c, lvl, lane);

    // 2x2 DC Hadamard (redundant across lanes)
    int d0 = __shfl(dcslot, 0), d1 = __shfl(dcslot, 1);
    int d2 = __shfl(dcslot, 2), d3 = __shfl(dcslot, 3);
    int w0 = d0 + d1 + d2 + d3, w1 = d0 - d1 + d2 - d3;
    int w2 = d0 + d1 - d2 - d3, w3 = d0 - d1 - d2 + d3;
    bool intra = mode == kIntra;
    int q0 = quant_dc_v(w0, qpc, intra), q1 = quant_dc_v(w1, qpc, intra);
    int q2 = quant_dc_v(w2, qpc, intra), q3 = quant_dc_v(w3, qpc, intra);
    if (lane < 4) {
      int qv = lane == 0 ? q0 : (lane == 1 ? q1 : (lane == 2 ? q2 : q3));
      L[kChromaDcOff + comp * 4 + lane] = (int16_t)qv;
    }
    int iw0 = q0 + q1 + q2 + q3, iw1 = q0 - q1 + q2 - q3;
    int iw2 = q0 + q1 - q2 - q3, iw3 = q0 - q1 - q2 + q3;
    int dcr0 = dequant_chroma_dc_v(iw0, qpc);
    int dcr1 = dequant_chroma_dc_v(iw1, qpc);
    int dcr2 = dequant_chroma_dc_v(iw2, qpc);
    int dcr3 = dequant_chroma_dc_v(iw3, qpc);
    int dcr = g == 0 ? dcr0 : (g == 1 ? dcr1 : (g == 2 ? dcr2 : dcr3));

    // recon (no cbp gates: zero levels/DCs dequantize to zero)
    int d = (c == 0) ? dcr : dequant_c(lvl, qpc, coeff_cls(c));
    int rec = idct4_wave(d, lane);
    int pix = clip8(rec + pred);
    int p1 = __shfl(pix, lane + 1), p2 = __shfl(pix, lane + 2),
        p3 = __shfl(pix, lane + 3);
    if ((c & 3) == 0)
      *reinterpret_cast<uint32_t*>(
          curC + (size_t)(cy0 + rr) * cpitch + cx0 + cc2) =
          (uint32_t)pix | ((uint32_t)p1 << 8) | ((uint32_t)p2 << 16) |
          ((uint32_t)p3 << 24);
    if (cc2 == 7) ccol_n[rr] = (uint8_t)pix;

    if (mode == kIntra && comp == 0 && lane == 0) M[1] = chroma_mode;
    __syncthreads();
    have_left = true;
  }
}

}  // namespace rows4

__global__ void __launch_bounds__(384) k_h264_rows4(
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
  const int wave = threadIdx.x >> 6;

  __shared__ rows4::Shared sh;

  if (wave < 4) {
    rows4::luma_wave<1>(srcY, ypitch, w, h, refY, curY, mbw, mby, qp,
                        i_slice, levels, meta, lane, wave, job.mbx0,
                        job.seg_mbw, &sh);
  } else if (wave == 4) {
    rows4::chroma_wave(srcCb, refCb, curCb, cpitch, w, h, mbw, mby,
                       dev_chroma_qp(qp), i_slice, levels, meta, lane, 0,
                       job.mbx0, job.seg_mbw, &sh);
  } else {
    rows4::chroma_wave(srcCr, refCr, curCr, cpitch, w, h, mbw, mby,
                       dev_chroma_qp(qp), i_slice, levels, meta, lane, 1,
                       job.mbx0, job.seg_mbw, &sh);
  }
}

void launch_h264_rows4(const uint8_t* srcY, const uint8_t* srcCb,
                       const uint8_t* srcCr, int ypitch, int cpitch, int w,
                       int h, const uint8_t* refY, const uint8_t* refCb,
                       const uint8_t* refCr, uint8_t* curY, uint8_t* curCb,
                       uint8_t* curCr, int mbw, int n_jobs,
                       const RowJob* d_jobs, int16_t* d_levels, int* d_meta,
                       hipStream_t stream) {
  if (n_jobs == 0) return;
  hipLaunchKernelGGL(k_h264_rows4, dim3(n_jobs), dim3(384), 0, stream, srcY,
                     srcCb, srcCr, ypitch, cpitch, w, h, refY, refCb, refCr,
                     curY, curCb, curCr, mbw, d_jobs, d_levels, d_meta);
}

}  // namespace h264gpu
}  // namespace hipflux
