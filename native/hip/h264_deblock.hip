// H.264 in-loop deblocking kernel (gfx950). One workgroup (64 lanes) per
// slice segment: the segment's 16-px luma row tile (and 8-px chroma
// tiles) stage through LDS, edges filter per §8.7 in macroblock order
// (vertical then horizontal per MB — MB m+1's left edge reads MB m's
// horizontally-filtered columns), and the tile writes back. Scalar
// reference: native/cpu/h264/deblock.h (byte-identical by tests).
#include <hip/hip_runtime.h>

#include "h264_gpu_layout.h"
#include "h264_kernels.h"

namespace hipflux {
namespace h264gpu {

__constant__ uint8_t cDbAlpha[52] = {
    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
    4, 4, 5, 6, 7, 8, 9, 10, 12, 13, 15, 17, 20, 22, 25, 28,
    32, 36, 40, 45, 50, 56, 63, 71, 80, 90, 101, 113, 127, 144,
    162, 182, 203, 226, 255, 255};
__constant__ uint8_t cDbBeta[52] = {
    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
    2, 2, 2, 3, 3, 3, 3, 4, 4, 4, 6, 6, 7, 7, 8, 8,
    9, 9, 10, 10, 11, 11, 12, 12, 13, 13, 14, 14, 15, 15,
    16, 16, 17, 17, 18, 18};
__constant__ uint8_t cDbTc0[52][3] = {
    {0,0,0},{0,0,0},{0,0,0},{0,0,0},{0,0,0},{0,0,0},{0,0,0},{0,0,0},
    {0,0,0},{0,0,0},{0,0,0},{0,0,0},{0,0,0},{0,0,0},{0,0,0},{0,0,0},
    {0,0,0},{0,0,1},{0,0,1},{0,0,1},{0,0,1},{0,1,1},{0,1,1},{1,1,1},
    {1,1,1},{1,1,1},{1,1,1},{1,1,2},{1,1,2},{1,1,2},{1,1,2},{1,2,3},
    {1,2,3},{2,2,3},{2,2,4},{2,3,4},{2,3,4},{3,3,5},{3,4,6},{3,4,6},
    {4,5,7},{4,5,8},{5,6,9},{6,7,10},{6,8,11},{7,9,12},{8,10,13},
    {9,12,15},{10,13,17},{11,16,20},{13,18,23},{14,20,25}};

__constant__ int cDbChromaQp[52] = {
    0, 1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11, 12, 13, 14, 15, 16, 17, 18,
    19, 20, 21, 22, 23, 24, 25, 26, 27, 28, 29, 29, 30, 31, 32, 32, 33,
    34, 34, 35, 35, 36, 36, 37, 37, 37, 38, 38, 38, 39, 39, 39, 39};

__device__ __forceinline__ int db_clip3d(int lo, int hi, int v) {
  return v < lo ? lo : v > hi ? hi : v;
}
__device__ __forceinline__ int db_clip8d(int v) {
  return v < 0 ? 0 : v > 255 ? 255 : v;
}

struct DbInfo {
  int intra;
  int mvx, mvy;
  unsigned nz;
};

__device__ __forceinline__ int db_bs_d(const DbInfo& mp, int pblk,
                                       const DbInfo& mq, int qblk,
                                       bool mb_edge) {
  if (mp.intra || mq.intra) return mb_edge ? 4 : 3;
  if (((mp.nz >> pblk) & 1) || ((mq.nz >> qblk) & 1)) return 2;
  if (abs(mp.mvx - mq.mvx) >= 4 || abs(mp.mvy - mq.mvy) >= 4) return 1;
  return 0;
}

// Filter one luma line in a byte array with stride `st` between samples
// across the edge; `e` points at q0 (p0 = e - st).
__device__ void db_luma_line(uint8_t* e, int st, int bs, int alpha,
                             int beta, int tc0) {
  const int p0 = e[-st], p1 = e[-2 * st], p2 = e[-3 * st], p3 = e[-4 * st];
  const int q0 = e[0], q1 = e[st], q2 = e[2 * st], q3 = e[3 * st];
  if (abs(p0 - q0) >= alpha || abs(p1 - p0) >= beta ||
      abs(q1 - q0) >= beta)
    return;
  const bool ap = abs(p2 - p0) < beta;
  const bool aq = abs(q2 - q0) < beta;
  if (bs < 4) {
    const int tc = tc0 + (ap ? 1 : 0) + (aq ? 1 : 0);
    const int delta =
        db_clip3d(-tc, tc, ((q0 - p0) * 4 + (p1 - q1) + 4) >> 3);
    e[-st] = (uint8_t)db_clip8d(p0 + delta);
    e[0] = (uint8_t)db_clip8d(q0 - delta);
    if (ap)
      e[-2 * st] = (uint8_t)(p1 + db_clip3d(-tc0, tc0,
                       (p2 + ((p0 + q0 + 1) >> 1) - 2 * p1) >> 1));
    if (aq)
      e[st] = (uint8_t)(q1 + db_clip3d(-tc0, tc0,
                  (q2 + ((p0 + q0 + 1) >> 1) - 2 * q1) >> 1));
  } else {
    const bool small = abs(p0 - q0) < ((alpha >> 2) + 2);
    if (ap && small) {
      e[-st] = (uint8_t)((p2 + 2 * p1 + 2 * p0 + 2 * q0 + q1 + 4) >> 3);
      e[-2 * st] = (uint8_t)((p2 + p1 + p0 + q0 + 2) >> 2);
      e[-3 * st] = (uint8_t)((2 * p3 + 3 * p2 + p1 + p0 + q0 + 4) >> 3);
    } else {
      e[-st] = (uint8_t)((2 * p1 + p0 + q1 + 2) >> 2);
    }
    if (aq && small) {
      e[0] = (uint8_t)((q2 + 2 * q1 + 2 * q0 + 2 * p0 + p1 + 4) >> 3);
      e[st] = (uint8_t)((q2 + q1 + q0 + p0 + 2) >> 2);
      e[2 * st] = (uint8_t)((2 * q3 + 3 * q2 + q1 + q0 + p0 + 4) >> 3);
    } else {
      e[0] = (uint8_t)((2 * q1 + q0 + p1 + 2) >> 2);
    }
  }
}

__device__ void db_chroma_line(uint8_t* e, int st, int bs, int alpha,
                               int beta, int tc0) {
  const int p0 = e[-st], p1 = e[-2 * st];
  const int q0 = e[0], q1 = e[st];
  if (abs(p0 - q0) >= alpha || abs(p1 - p0) >= beta ||
      abs(q1 - q0) >= beta)
    return;
  if (bs < 4) {
    const int tc = tc0 + 1;
    const int delta =
        db_clip3d(-tc, tc, ((q0 - p0) * 4 + (p1 - q1) + 4) >> 3);
    e[-st] = (uint8_t)db_clip8d(p0 + delta);
    e[0] = (uint8_t)db_clip8d(q0 - delta);
  } else {
    e[-st] = (uint8_t)((2 * p1 + p0 + q1 + 2) >> 2);
    e[0] = (uint8_t)((2 * q1 + q0 + p1 + 2) >> 2);
  }
}

// LDS tile: luma 16 rows x (kMaxSegMbw*16), chroma 8 x (kMaxSegMbw*8) x2.
struct DbShared {
  uint8_t y[16][kMaxSegMbw * 16];
  uint8_t cb[8][kMaxSegMbw * 8];
  uint8_t cr[8][kMaxSegMbw * 8];
  DbInfo info[kMaxSegMbw];
};

__global__ __launch_bounds__(64) void k_h264_deblock(
    uint8_t* __restrict__ curY, uint8_t* __restrict__ curCb,
    uint8_t* __restrict__ curCr, int ypitch, int cpitch, int mbw,
    const RowJob* __restrict__ jobs, const int16_t* __restrict__ levels,
    const int* __restrict__ meta) {
  __shared__ DbShared sh;
  const RowJob job = jobs[blockIdx.x];
  if (job.flags & 1) {
    // IDR rows: all-intra, every edge is bS 3/4 — still filter
  }
  const int lane = threadIdx.x;
  const int segw = job.seg_mbw;
  const int y0 = job.mb_row * 16, cy0 = job.mb_row * 8;
  const int px0 = job.mbx0 * 16, cpx0 = job.mbx0 * 8;
  const int wpx = segw * 16, cwpx = segw * 8;

  // ---- load tile (coalesced rows) + per-MB info
  for (int r = 0; r < 16; ++r)
    for (int x = lane; x < wpx; x += 64)
      sh.y[r][x] = curY[(size_t)(y0 + r) * ypitch + px0 + x];
  for (int r = 0; r < 8; ++r)
    for (int x = lane; x < cwpx; x += 64) {
      sh.cb[r][x] = curCb[(size_t)(cy0 + r) * cpitch + cpx0 + x];
      sh.cr[r][x] = curCr[(size_t)(cy0 + r) * cpitch + cpx0 + x];
    }
  // info: one lane per MB; nz from the levels buffer (inter AC blocks)
  for (int m = lane; m < segw; m += 64) {
    const size_t mb = (size_t)job.mb_row * mbw + job.mbx0 + m;
    const int m0 = meta[mb * kMetaPerMb + 0];
    const int m1 = meta[mb * kMetaPerMb + 1];
    const int mode = m0 & 3;
    DbInfo di;
    di.intra = mode == kIntra || (job.flags & 1);
    di.mvx = mode == kInter ? (int)(short)(m1 & 0xFFFF) : 0;
    di.mvy = mode == kInter ? (m1 >> 16) : 0;
    di.nz = 0;
    if (mode == kInter) {
      const int16_t* lv = levels + mb * kLevelsPerMb + kLumaAcOff;
      for (int b = 0; b < 16; ++b) {
        int any = 0;
        for (int i = 0; i < 16; ++i) any |= lv[b * 16 + i] != 0;
        di.nz |= (unsigned)any << b;
      }
    }
    sh.info[m] = di;
  }
  __syncthreads();

  const int qp = db_clip3d(0, 51, job.qp);
  const int qpc = cDbChromaQp[qp];
  const int alpha = cDbAlpha[qp], beta = cDbBeta[qp];
  const int alpha_c = cDbAlpha[qpc], beta_c = cDbBeta[qpc];

  // ---- per-MB edge filtering (spec order: MB raster; V then H).
  // Chroma edges never read luma pixels and their bS comes from the MB
  // info only, so the chroma chain (V0, V1, H) runs on lanes 16..31
  // UNDER the first three luma vertical steps: 7 barrier steps per MB
  // instead of 10 (measured: deblock is ~20% of the GPU frame).
  for (int m = 0; m < segw; ++m) {
    const int x0 = m * 16, cx0 = m * 8;
    // steps 0..3: luma vertical edge e on lanes 0..15; chroma work
    // (V e=0, V e=1, H) on lanes 16..31 during steps 0..2
    for (int e = 0; e < 4; ++e) {
      if (lane < 16) {
        if (!(e == 0 && m == 0)) {          // slice/segment boundary
          const int pblk = (lane >> 2) * 4 + (e == 0 ? 3 : e - 1);
          const int qblk = (lane >> 2) * 4 + e;
          const DbInfo& mp = e == 0 ? sh.info[m - 1] : sh.info[m];
          const int bs = db_bs_d(mp, pblk, sh.info[m], qblk, e == 0);
          if (bs) {
            const int tc0 = bs < 4 ? cDbTc0[qp][bs - 1] : 0;
            db_luma_line(&sh.y[lane][x0 + e * 4], 1, bs, alpha, beta,
                         tc0);
          }
        }
      } else if (lane < 32 && e < 2) {      // chroma vertical edge e
        if (!(e == 0 && m == 0)) {
          const int pl = (lane - 16) >> 3, yy = (lane - 16) & 7;
          const int lrow = (yy * 2) >> 2;
          const int pblk = lrow * 4 + (e == 0 ? 3 : 1);
          const int qblk = lrow * 4 + (e == 0 ? 0 : 2);
          const DbInfo& mp = e == 0 ? sh.info[m - 1] : sh.info[m];
          const int bs = db_bs_d(mp, pblk, sh.info[m], qblk, e == 0);
          if (bs) {
            const int tc0 = bs < 4 ? cDbTc0[qpc][bs - 1] : 0;
            uint8_t* base = pl ? &sh.cr[yy][cx0 + e * 4]
                               : &sh.cb[yy][cx0 + e * 4];
            db_chroma_line(base, 1, bs, alpha_c, beta_c, tc0);
          }
        }
      } else if (lane < 32 && e == 2) {     // chroma horizontal edge
        const int pl = (lane - 16) >> 3, xx = (lane - 16) & 7;
        const int lcol = (xx * 2) >> 2;
        const int pblk = 1 * 4 + lcol;
        const int qblk = 2 * 4 + lcol;
        const int bs = db_bs_d(sh.info[m], pblk, sh.info[m], qblk, false);
        if (bs) {
          const int tc0 = bs < 4 ? cDbTc0[qpc][bs - 1] : 0;
          uint8_t* base = pl ? &sh.cr[4][cx0 + xx] : &sh.cb[4][cx0 + xx];
          db_chroma_line(base, kMaxSegMbw * 8, bs, alpha_c, beta_c, tc0);
        }
      }
      __syncthreads();
    }
    // steps 4..6: horizontal luma edges (y = 4, 8, 12), lanes = columns
    for (int e = 1; e < 4; ++e) {
      if (lane < 16) {
        const int pblk = (e - 1) * 4 + (lane >> 2);
        const int qblk = e * 4 + (lane >> 2);
        const int bs = db_bs_d(sh.info[m], pblk, sh.info[m], qblk, false);
        if (bs) {
          const int tc0 = bs < 4 ? cDbTc0[qp][bs - 1] : 0;
          db_luma_line(&sh.y[e * 4][x0 + lane], kMaxSegMbw * 16, bs,
                       alpha, beta, tc0);
        }
      }
      __syncthreads();
    }
  }

  // ---- write back
  for (int r = 0; r < 16; ++r)
    for (int x = lane; x < wpx; x += 64)
      curY[(size_t)(y0 + r) * ypitch + px0 + x] = sh.y[r][x];
  for (int r = 0; r < 8; ++r)
    for (int x = lane; x < cwpx; x += 64) {
      curCb[(size_t)(cy0 + r) * cpitch + cpx0 + x] = sh.cb[r][x];
      curCr[(size_t)(cy0 + r) * cpitch + cpx0 + x] = sh.cr[r][x];
    }
}

void launch_h264_deblock(uint8_t* d_curY, uint8_t* d_curCb,
                         uint8_t* d_curCr, int ypitch, int cpitch, int mbw,
                         int n_jobs, const RowJob* d_jobs,
                         const int16_t* d_levels, const int* d_meta,
                         hipStream_t stream) {
  if (n_jobs == 0) return;
  hipLaunchKernelGGL(k_h264_deblock, dim3(n_jobs), dim3(64), 0, stream,
                     d_curY, d_curCb, d_curCr, ypitch, cpitch, mbw, d_jobs,
                     d_levels, d_meta);
}

}  // namespace h264gpu
}  // namespace hipflux
