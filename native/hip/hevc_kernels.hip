// HEVC gfx950 kernels: CTU row wavefront (intra decision + transform +
// quant + recon) and per-slice CABAC entropy.
//
// Parallel structure (designed for CDNA4, not ported from anywhere):
//  * k_hevc_rows: one 256-thread workgroup (4 waves) per slice-segment
//    job. CTUs march left->right (the only dependency is the left
//    neighbor's reconstructed right column, kept in LDS). All 256 luma
//    pixels of a CTU are processed by one thread each; the 16x16
//    transforms run as two 16-MAC-per-thread stages through LDS; chroma
//    8x8 planes run on threads 0..63 / 64..127 concurrently.
//  * k_hevc_cabac: CABAC is bin-serial per slice, so one LANE encodes one
//    slice segment; parallelism comes from the many independent segments
//    (4K: 135 rows x slices/row). The bin semantics transliterate
//    native/cpu/hevc/entropy.h exactly — byte equality with the CPU
//    encoder is asserted in tests/test_gpu_hevc.py.
//
// Reference-sample note: with one slice per CTU row, only LEFT neighbors
// exist. The spec reference array (§8.4.4.2.2 substitution + §8.4.4.2.3
// smoothing) then collapses to closed forms of the 16 left samples:
//   corner/top/top-right = left[0] replicated, below-left = left[15];
//   [1 2 1] filtering leaves the replicated arms unchanged and smooths
//   only the real column (edges replicated). These forms are used
//   directly; equality with the generic CPU path is covered by the
//   byte-exactness tests.
#include <hip/hip_runtime.h>

#include "hevc_gpu_layout.h"
#include "hevc_kernels.h"

namespace hipflux {
namespace hevcgpu {

// ---- constant tables (mirror native/cpu/hevc/tables.h) -------------------
__constant__ int8_t cT16[16][16] = {
    {64, 64, 64, 64, 64, 64, 64, 64, 64, 64, 64, 64, 64, 64, 64, 64},
    {90, 87, 80, 70, 57, 43, 25, 9, -9, -25, -43, -57, -70, -80, -87, -90},
    {89, 75, 50, 18, -18, -50, -75, -89, -89, -75, -50, -18, 18, 50, 75, 89},
    {87, 57, 9, -43, -80, -90, -70, -25, 25, 70, 90, 80, 43, -9, -57, -87},
    {83, 36, -36, -83, -83, -36, 36, 83, 83, 36, -36, -83, -83, -36, 36, 83},
    {80, 9, -70, -87, -25, 57, 90, 43, -43, -90, -57, 25, 87, 70, -9, -80},
    {75, -18, -89, -50, 50, 89, 18, -75, -75, 18, 89, 50, -50, -89, -18, 75},
    {70, -43, -87, 9, 90, 25, -80, -57, 57, 80, -25, -90, -9, 87, 43, -70},
    {64, -64, -64, 64, 64, -64, -64, 64, 64, -64, -64, 64, 64, -64, -64, 64},
    {57, -80, -25, 90, -9, -87, 43, 70, -70, -43, 87, 9, -90, 25, 80, -57},
    {50, -89, 18, 75, -75, -18, 89, -50, -50, 89, -18, -75, 75, 18, -89, 50},
    {43, -90, 57, 25, -87, 70, 9, -80, 80, -9, -70, 87, -25, -57, 90, -43},
    {36, -83, 83, -36, -36, 83, -83, 36, 36, -83, 83, -36, -36, 83, -83, 36},
    {25, -70, 90, -80, 43, 9, -57, 87, -87, 57, -9, -43, 80, -90, 70, -25},
    {18, -50, 75, -89, 89, -75, 50, -18, -18, 50, -75, 89, -89, 75, -50, 18},
    {9, -25, 43, -57, 70, -80, 87, -90, 90, -87, 80, -70, 57, -43, 25, -9}};

__constant__ int8_t cT8[8][8] = {
    {64, 64, 64, 64, 64, 64, 64, 64},
    {89, 75, 50, 18, -18, -50, -75, -89},
    {83, 36, -36, -83, -83, -36, 36, 83},
    {75, -18, -89, -50, 50, 89, 18, -75},
    {64, -64, -64, 64, 64, -64, -64, 64},
    {50, -89, 18, 75, -75, -18, 89, -50},
    {36, -83, 83, -36, -36, 83, -83, 36},
    {18, -50, 75, -89, 89, -75, 50, -18}};

__constant__ int cQuantScale[6] = {26214, 23302, 20560, 18396, 16384, 14564};
__constant__ int cDequantScale[6] = {40, 45, 51, 57, 64, 72};

__device__ __forceinline__ int clip16d(int v) {
  return v < -32768 ? -32768 : v > 32767 ? 32767 : v;
}
__device__ __forceinline__ int clip8d(int v) {
  return v < 0 ? 0 : v > 255 ? 255 : v;
}

// ---- rows kernel ---------------------------------------------------------

// LDS working set for one workgroup.
struct RowsShared {
  int lcolY[16];     // left CTU's recon right column (luma)
  int lcolCb[8];
  int lcolCr[8];
  int lfY[17];       // [1 2 1]-filtered left column + below-left (planar)
  short res[256];    // residual / reused as stage buffers
  int tmp[256];      // transform intermediate
  short coef[256];   // quantized-dequantized coefficients
  int pred[256];     // chosen prediction (luma), then per-plane chroma
  int sad[4];
  int best;          // chosen mode index 0..3
  int cbf;           // bit0 luma, bit1 cb, bit2 cr
  int predC[2][64];
  short resC[2][64];
  int tmpC[2][64];
  short coefC[2][64];
};

__device__ const int kModeOrder[4] = {1, 10, 26, 0};  // DC, H, V, Planar

__global__ __launch_bounds__(256) void k_hevc_rows(
    const uint8_t* __restrict__ srcY, const uint8_t* __restrict__ srcCb,
    const uint8_t* __restrict__ srcCr, int ypitch, int cpitch, int w, int h,
    uint8_t* __restrict__ curY, uint8_t* __restrict__ curCb,
    uint8_t* __restrict__ curCr, int ctbw, const HevcJob* __restrict__ jobs,
    int16_t* __restrict__ levels, int* __restrict__ meta) {
  __shared__ RowsShared sh;
  const HevcJob job = jobs[blockIdx.x];
  const int tid = threadIdx.x;
  const int tx = tid & 15, ty = tid >> 4;
  const int qp = job.qp, qpc = job.qpc;
  const int y0 = job.ctu_row * 16;
  const int cy0 = job.ctu_row * 8;
  // visible chroma dims (CSC writes visible pixels only; clamped reads
  // reproduce the CPU encoder's edge replication)
  const int cw = (w + 1) >> 1, chh = (h + 1) >> 1;

  for (int ci = 0; ci < job.seg_w; ++ci) {
    const int cx = job.ctu_x0 + ci;
    const int x0 = cx * 16;
    const bool has_left = ci > 0;

    // ---- load source pixel + build refs (closed forms; see header note)
    const int sx_c = min(x0 + tx, w - 1), sy_c = min(y0 + ty, h - 1);
    const int s = srcY[(size_t)sy_c * ypitch + sx_c];
    if (tid < 16) {
      int l = has_left ? sh.lcolY[tid] : 128;
      // filtered column for planar: lf[i] = (l[i-1] + 2 l[i] + l[i+1]
      // + 2) >> 2 with l[-1] = l[0] (corner replica), l[16] = l[15]
      int lm = has_left ? sh.lcolY[tid == 0 ? 0 : tid - 1] : 128;
      int lp = has_left ? sh.lcolY[tid == 15 ? 15 : tid + 1] : 128;
      sh.lfY[tid] = (lm + 2 * l + lp + 2) >> 2;
      if (tid == 15) sh.lfY[16] = has_left ? sh.lcolY[15] : 128;
    }
    if (tid < 4) sh.sad[tid] = 0;
    __syncthreads();

    const int l0 = has_left ? sh.lcolY[0] : 128;
    const int lty = has_left ? sh.lcolY[ty] : 128;
    const int l15 = has_left ? sh.lcolY[15] : 128;

    // ---- 4 candidate predictions + SAD (order: DC, H, V, Planar)
    int dcsum = 0;
    if (tid == 0) {
      dcsum = 16 * l0 + 16;
      for (int i = 0; i < 16; ++i)
        dcsum += has_left ? sh.lcolY[i] : 128;
      sh.best = dcsum >> 5;  // stash DC value in best temporarily
    }
    __syncthreads();
    const int dc = sh.best;
    int pm[4];
    // DC with luma edge filtering (§8.4.4.2.5, nT<32)
    if (tx == 0 && ty == 0)
      pm[0] = (lty + 2 * dc + l0 + 2) >> 2;
    else if (ty == 0)
      pm[0] = (l0 + 3 * dc + 2) >> 2;
    else if (tx == 0)
      pm[0] = (lty + 3 * dc + 2) >> 2;
    else
      pm[0] = dc;
    // Horizontal: pred = left[y]; first-row edge adjust is identity here
    // (top == corner == l0)
    pm[1] = lty;
    // Vertical: top is l0 everywhere; column-0 edge adjust
    pm[2] = (tx == 0) ? clip8d(l0 + ((lty - l0) >> 1)) : l0;
    // Planar on filtered refs. The [1 2 1] filter runs over the LINEAR
    // reference array, where the replicated top/corner/below-left arms
    // are flat — their filtered values stay the RAW l0 / l15; only the
    // real left column smooths.
    {
      int lf = has_left ? sh.lfY[ty] : 128;
      pm[3] = ((15 - tx) * lf + (tx + 1) * l0 + (15 - ty) * l0 +
               (ty + 1) * l15 + 16) >> 5;
    }
    // wave-reduce SADs, then pick first strict-min in order DC,H,V,Planar
    for (int m = 0; m < 4; ++m) {
      int d = s - pm[m];
      d = d < 0 ? -d : d;
      for (int off = 32; off; off >>= 1) d += __shfl_down(d, off, 64);
      if ((tid & 63) == 0) atomicAdd(&sh.sad[m], d);
    }
    __syncthreads();
    if (tid == 0) {
      int best = 0;
      long bs = (long)sh.sad[0];
      for (int m = 1; m < 4; ++m)
        if ((long)sh.sad[m] < bs) {
          bs = sh.sad[m];
          best = m;
        }
      sh.best = best;
    }
    __syncthreads();
    const int best = sh.best;
    const int mode = kModeOrder[best];
    const int p = pm[best];
    sh.pred[tid] = p;
    sh.res[ty * 16 + tx] = (short)(s - p);
    __syncthreads();

    // ---- forward T16: stage1 over columns, stage2 over rows
    {
      int acc = 0;
      // thread (u=ty, x=tx): sum_y T[u][y] * res[y][x]
      for (int y = 0; y < 16; ++y) acc += cT16[ty][y] * sh.res[y * 16 + tx];
      sh.tmp[ty * 16 + tx] = (acc + 4) >> 3;
    }
    __syncthreads();
    {
      int acc = 0;
      // thread (u=ty, v=tx): sum_x tmp[u][x] * T[v][x]
      for (int x = 0; x < 16; ++x) acc += sh.tmp[ty * 16 + x] * cT16[tx][x];
      int coef = clip16d((acc + 512) >> 10);
      // quant
      const int qbits = 14 + qp / 6 + 3;
      const int add = 171 << (qbits - 9);
      int a = coef < 0 ? -coef : coef;
      int lv = (a * cQuantScale[qp % 6] + add) >> qbits;
      if (lv > 32767) lv = 32767;
      if (coef < 0) lv = -lv;
      sh.coef[ty * 16 + tx] = (short)lv;
      if (tid == 0) sh.cbf = 0;
    }
    __syncthreads();
    {
      int nz = sh.coef[tid] != 0;
      if (__any(nz) && (tid & 63) == 0) atomicOr(&sh.cbf, 1);
    }
    __syncthreads();
    const bool cbf_y = sh.cbf & 1;

    // levels out
    const size_t ctu_base =
        (size_t)(job.ctu_row * ctbw + cx) * kHevcLevelsPerCtu;
    levels[ctu_base + tid] = sh.coef[tid];

    // ---- dequant + inverse + recon (only if cbf; else recon = pred)
    int recon;
    if (cbf_y) {
      {
        const int bd_shift = 7;
        const long long scale =
            ((long long)cDequantScale[qp % 6] << (qp / 6)) * 16;
        long long d = ((long long)sh.coef[ty * 16 + tx] * scale +
                       (1 << (bd_shift - 1))) >> bd_shift;
        sh.tmp[ty * 16 + tx] = clip16d((int)d);
      }
      __syncthreads();
      // reuse res as stage buffer: stage1 (y=ty, v=tx)
      {
        int acc = 0;
        for (int u = 0; u < 16; ++u)
          acc += cT16[u][ty] * sh.tmp[u * 16 + tx];
        sh.res[ty * 16 + tx] = (short)clip16d((acc + 64) >> 7);
      }
      __syncthreads();
      {
        int acc = 0;
        for (int v = 0; v < 16; ++v)
          acc += sh.res[ty * 16 + v] * cT16[v][tx];
        recon = clip8d(sh.pred[tid] + clip16d((acc + 2048) >> 12));
      }
    } else {
      recon = p;
    }
    __syncthreads();
    curY[(size_t)(y0 + ty) * ypitch + x0 + tx] = (uint8_t)recon;
    if (tx == 15) sh.lcolY[ty] = recon;

    // ---- chroma: plane pl = 0 (Cb, threads 0..63) / 1 (Cr, 64..127)
    const int cpl = tid >> 6;          // 0..3 (waves)
    if (cpl < 2) {
      const int ctid = tid & 63;
      const int ctx8 = ctid & 7, cty8 = ctid >> 3;
      const uint8_t* srcC = cpl ? srcCr : srcCb;
      int* lcol = cpl ? sh.lcolCr : sh.lcolCb;
      const int cx0 = cx * 8;
      const int cs = srcC[(size_t)min(cy0 + cty8, chh - 1) * cpitch +
                          min(cx0 + ctx8, cw - 1)];
      const int cl0 = has_left ? lcol[0] : 128;
      const int clty = has_left ? lcol[cty8] : 128;
      const int cl7 = has_left ? lcol[7] : 128;
      int cp;
      if (mode == 1) {  // DC: no chroma edge filter
        int sum = 8 * cl0 + 8;
        for (int i = 0; i < 8; ++i) sum += has_left ? lcol[i] : 128;
        cp = sum >> 4;
      } else if (mode == 10) {
        cp = clty;
      } else if (mode == 26) {
        cp = cl0;
      } else {  // planar (unfiltered refs for chroma)
        cp = ((7 - ctx8) * clty + (ctx8 + 1) * cl0 + (7 - cty8) * cl0 +
              (cty8 + 1) * cl7 + 8) >> 4;
      }
      sh.predC[cpl][ctid] = cp;
      sh.resC[cpl][cty8 * 8 + ctx8] = (short)(cs - cp);
    }
    __syncthreads();
    if (cpl < 2) {
      const int ctid = tid & 63;
      const int u8 = ctid >> 3, x8 = ctid & 7;
      int acc = 0;
      for (int y = 0; y < 8; ++y)
        acc += cT8[u8][y] * sh.resC[cpl][y * 8 + x8];
      sh.tmpC[cpl][u8 * 8 + x8] = (acc + 2) >> 2;
    }
    __syncthreads();
    if (cpl < 2) {
      const int ctid = tid & 63;
      const int u8 = ctid >> 3, v8 = ctid & 7;
      int acc = 0;
      for (int x = 0; x < 8; ++x)
        acc += sh.tmpC[cpl][u8 * 8 + x] * cT8[v8][x];
      int coef = clip16d((acc + 256) >> 9);
      const int qbits = 14 + qpc / 6 + 4;
      const int add = 171 << (qbits - 9);
      int a = coef < 0 ? -coef : coef;
      int lv = (a * cQuantScale[qpc % 6] + add) >> qbits;
      if (lv > 32767) lv = 32767;
      if (coef < 0) lv = -lv;
      sh.coefC[cpl][u8 * 8 + v8] = (short)lv;
    }
    __syncthreads();
    if (cpl < 2) {
      const int ctid = tid & 63;
      int nz = sh.coefC[cpl][ctid] != 0;
      if (__any(nz) && ctid == 0) atomicOr(&sh.cbf, 2 << cpl);
      levels[ctu_base + 256 + cpl * 64 + ctid] = sh.coefC[cpl][ctid];
    }
    __syncthreads();
    // chroma dequant + inverse: barriers at top level (uniform), work
    // guarded per-thread
    {
      const int ctid = tid & 63;
      const int y8 = ctid >> 3, x8 = ctid & 7;
      const int cbf_c = cpl < 2 ? ((sh.cbf >> (1 + cpl)) & 1) : 0;
      if (cbf_c) {
        const int bd_shift = 6;
        const long long scale =
            ((long long)cDequantScale[qpc % 6] << (qpc / 6)) * 16;
        long long d = ((long long)sh.coefC[cpl][y8 * 8 + x8] * scale +
                       (1 << (bd_shift - 1))) >> bd_shift;
        sh.tmpC[cpl][y8 * 8 + x8] = clip16d((int)d);
      }
      __syncthreads();
      if (cbf_c) {
        int acc = 0;
        for (int u = 0; u < 8; ++u)
          acc += cT8[u][y8] * sh.tmpC[cpl][u * 8 + x8];
        sh.resC[cpl][y8 * 8 + x8] = (short)clip16d((acc + 64) >> 7);
      }
      __syncthreads();
      if (cpl < 2) {
        int creconv;
        if (cbf_c) {
          int acc = 0;
          for (int v = 0; v < 8; ++v)
            acc += sh.resC[cpl][y8 * 8 + v] * cT8[v][x8];
          creconv = clip8d(sh.predC[cpl][ctid] +
                           clip16d((acc + 2048) >> 12));
        } else {
          creconv = sh.predC[cpl][ctid];
        }
        uint8_t* curC = cpl ? curCr : curCb;
        curC[(size_t)(cy0 + y8) * cpitch + cx * 8 + x8] = (uint8_t)creconv;
        int* lcol = cpl ? sh.lcolCr : sh.lcolCb;
        if (x8 == 7) lcol[y8] = creconv;
      }
    }
    __syncthreads();
    if (tid == 0) {
      meta[(size_t)(job.ctu_row * ctbw + cx) * kHevcMetaPerCtu + 0] = mode;
      meta[(size_t)(job.ctu_row * ctbw + cx) * kHevcMetaPerCtu + 1] = sh.cbf;
    }
    __syncthreads();
  }
}

void launch_hevc_rows(const uint8_t* d_srcY, const uint8_t* d_srcCb,
                      const uint8_t* d_srcCr, int ypitch, int cpitch, int w,
                      int h, uint8_t* d_curY, uint8_t* d_curCb,
                      uint8_t* d_curCr, int ctbw, int n_jobs,
                      const HevcJob* d_jobs, int16_t* d_levels, int* d_meta,
                      hipStream_t stream) {
  if (n_jobs == 0) return;
  hipLaunchKernelGGL(k_hevc_rows, dim3(n_jobs), dim3(256), 0, stream,
                     d_srcY, d_srcCb, d_srcCr, ypitch, cpitch, w, h, d_curY,
                     d_curCb, d_curCr, ctbw, d_jobs, d_levels, d_meta);
}

// ---- CABAC kernel --------------------------------------------------------
// Scalar transliteration of native/cpu/hevc/{cabac.h,entropy.h}; one lane
// per slice-segment job.

__constant__ uint8_t cRangeTabLps[64][4] = {
    {128, 176, 208, 240}, {128, 167, 197, 227}, {128, 158, 187, 216},
    {123, 150, 178, 205}, {116, 142, 169, 195}, {111, 135, 160, 185},
    {105, 128, 152, 175}, {100, 122, 144, 166}, {95, 116, 137, 158},
    {90, 110, 130, 150},  {85, 104, 123, 142},  {81, 99, 117, 135},
    {77, 94, 111, 128},   {73, 89, 105, 122},   {69, 85, 100, 116},
    {66, 80, 95, 110},    {62, 76, 90, 104},    {59, 72, 86, 99},
    {56, 69, 81, 94},     {53, 65, 77, 89},     {51, 62, 73, 85},
    {48, 59, 69, 80},     {46, 56, 66, 76},     {43, 53, 63, 72},
    {41, 50, 59, 69},     {39, 48, 56, 65},     {37, 45, 54, 62},
    {35, 43, 51, 59},     {33, 41, 48, 56},     {32, 39, 46, 53},
    {30, 37, 43, 50},     {29, 35, 41, 48},     {27, 33, 39, 45},
    {26, 31, 37, 43},     {24, 30, 35, 41},     {23, 28, 33, 39},
    {22, 27, 32, 37},     {21, 26, 30, 35},     {20, 24, 29, 33},
    {19, 23, 27, 31},     {18, 22, 26, 30},     {17, 21, 25, 28},
    {16, 20, 23, 27},     {15, 19, 22, 25},     {14, 18, 21, 24},
    {14, 17, 20, 23},     {13, 16, 19, 22},     {12, 15, 18, 21},
    {12, 14, 17, 20},     {11, 14, 16, 19},     {11, 13, 15, 18},
    {10, 12, 15, 17},     {10, 12, 14, 16},     {9, 11, 13, 15},
    {9, 11, 12, 14},      {8, 10, 12, 14},      {8, 9, 11, 13},
    {7, 9, 11, 12},       {7, 9, 10, 12},       {7, 8, 10, 11},
    {6, 8, 9, 11},        {6, 7, 9, 10},        {6, 7, 8, 9},
    {2, 2, 2, 2}};

__constant__ uint8_t cTransIdxLps[64] = {
    0,  0,  1,  2,  2,  4,  4,  5,  6,  7,  8,  9,  9,  11, 11, 12,
    13, 13, 15, 15, 16, 16, 18, 18, 19, 19, 21, 21, 22, 22, 23, 24,
    24, 25, 26, 26, 27, 27, 28, 29, 29, 30, 30, 30, 31, 32, 32, 33,
    33, 33, 34, 34, 35, 35, 35, 36, 36, 36, 37, 37, 37, 38, 38, 63};

__constant__ uint8_t cRenorm[32] = {6, 5, 4, 4, 3, 3, 3, 3, 2, 2, 2,
                                    2, 2, 2, 2, 2, 1, 1, 1, 1, 1, 1,
                                    1, 1, 1, 1, 1, 1, 1, 1, 1, 1};

// context bank offsets (must match cpu/hevc/tables.h CtxOffset)
#define HG_CTX_SPLIT 0
#define HG_CTX_PREVINTRA 3
#define HG_CTX_CHROMA 4
#define HG_CTX_CBFY 5
#define HG_CTX_CBFC 7
#define HG_CTX_LASTX 11
#define HG_CTX_LASTY 29
#define HG_CTX_CSBF 47
#define HG_CTX_SIG 51
#define HG_CTX_GT1 93
#define HG_CTX_GT2 117
#define HG_NUM_CTX 123

__constant__ uint8_t cInitVals[HG_NUM_CTX] = {
    // split_cu (3)
    139, 141, 157,
    // prev_intra (1), chroma_mode (1)
    184, 63,
    // cbf_luma (2)
    111, 141,
    // cbf_chroma (4)
    94, 138, 182, 154,
    // last_x (18)
    110, 110, 124, 125, 140, 153, 125, 127, 140, 109, 111, 143, 127, 111,
    79, 108, 123, 63,
    // last_y (18)
    110, 110, 124, 125, 140, 153, 125, 127, 140, 109, 111, 143, 127, 111,
    79, 108, 123, 63,
    // csbf (4)
    91, 171, 134, 141,
    // sig (42)
    111, 111, 125, 110, 110, 94, 124, 108, 124, 107, 125, 141, 179, 153,
    125, 107, 125, 141, 179, 153, 125, 107, 125, 141, 179, 153, 125, 140,
    139, 182, 182, 152, 136, 152, 136, 153, 136, 139, 111, 136, 139, 111,
    // gt1 (24)
    140, 92, 137, 138, 140, 152, 138, 139, 153, 74, 149, 92, 139, 107,
    122, 152, 140, 179, 166, 182, 140, 227, 122, 197,
    // gt2 (6)
    138, 153, 136, 167, 152, 152};

// diagonal scan (x | y<<4) for 4x4; 2x2 sub-block scan likewise
__constant__ uint8_t cScan4[16] = {0x00, 0x10, 0x01, 0x20, 0x11, 0x02,
                                   0x30, 0x21, 0x12, 0x03, 0x31, 0x22,
                                   0x13, 0x32, 0x23, 0x33};
__constant__ uint8_t cScan2[4] = {0x00, 0x10, 0x01, 0x11};

struct DevCabac {
  uint32_t low, range, buffered;
  int bits_left, num_buffered, count;
  uint8_t* out;
  const uint32_t* ptab;  // packed per-(state,q) entry, see k_hevc_cabac

  __device__ void init(uint8_t* o) {
    low = 0;
    range = 510;
    bits_left = 23;
    buffered = 0xFF;
    num_buffered = 0;
    count = 0;
    out = o;
  }
  __device__ void put(uint8_t b) { out[count++] = b; }
  __device__ void test_write() {
    if (bits_left >= 12) return;
    uint32_t lead = low >> (24 - bits_left);
    bits_left += 8;
    low &= 0xFFFFFFFFu >> bits_left;
    if (lead == 0xFF) {
      ++num_buffered;
    } else if (num_buffered > 0) {
      uint32_t carry = lead >> 8;
      put((uint8_t)(buffered + carry));
      uint8_t fill = (uint8_t)((0xFF + carry) & 0xFF);
      while (num_buffered > 1) {
        put(fill);
        --num_buffered;
      }
      buffered = lead & 0xFF;
    } else {
      num_buffered = 1;
      buffered = lead & 0xFF;
    }
  }
  // packed entry: lps | transIdxLps<<8 | transIdxMps<<14 | renorm<<20
  __device__ void bin(uint32_t* st, int b) {
    const int state = (int)(*st >> 1);
    int mps = (int)(*st & 1);
    const uint32_t e = ptab[(state << 2) | ((range >> 6) & 3)];
    const uint32_t lps = e & 0xFF;
    range -= lps;
    if (b != mps) {
      const int n = (e >> 20) & 7;
      low = (low + range) << n;
      range = lps << n;
      if (state == 0) mps ^= 1;
      *st = ((((e >> 8) & 63)) << 1) | (uint32_t)mps;
      bits_left -= n;
      test_write();
    } else {
      *st = (((e >> 14) & 63) << 1) | (uint32_t)mps;
      if (range >= 256) return;
      low <<= 1;
      range <<= 1;
      --bits_left;
      test_write();
    }
  }
  __device__ void bypass(int b) {
    low <<= 1;
    if (b) low += range;
    --bits_left;
    test_write();
  }
  // n bypass bins at once: low' = low*2^n + range*value (associativity of
  // the per-bit recurrence low = 2*low + b*range). Chunked so bits_left
  // stays positive between flushes; byte-identical to bit-at-a-time.
  __device__ void bypass_bins(uint32_t v, int n) {
    while (n > 0) {
      const int take = n < 10 ? n : 10;
      n -= take;
      low = (low << take) + range * ((v >> n) & ((1u << take) - 1));
      bits_left -= take;
      test_write();
    }
  }
  // a run of `ones` 1-bins followed by one 0-bin (unary prefix)
  __device__ void bypass_unary(int ones) {
    while (ones > 0) {
      const int take = ones < 10 ? ones : 10;
      ones -= take;
      bypass_bins((1u << take) - 1, take);
    }
    bypass(0);
  }
  __device__ void terminate(int b) {
    range -= 2;
    if (b) {
      low = (low + range) << 7;
      range = 2 << 7;
      bits_left -= 7;
    } else if (range >= 256) {
      return;
    } else {
      low <<= 1;
      range <<= 1;
      --bits_left;
    }
    test_write();
  }
  // returns tail bits packed (nbits<<16 | bits)
  __device__ int finish() {
    if ((low >> (32 - bits_left)) != 0) {
      put((uint8_t)(buffered + 1));
      while (num_buffered > 1) {
        put(0x00);
        --num_buffered;
      }
      low -= 1u << (32 - bits_left);
    } else {
      if (num_buffered > 0) put((uint8_t)buffered);
      while (num_buffered > 1) {
        put(0xFF);
        --num_buffered;
      }
    }
    int nbits = 24 - bits_left;
    uint32_t val = low >> 8;
    while (nbits >= 8) {
      put((uint8_t)(val >> (nbits - 8)));
      nbits -= 8;
    }
    return (nbits << 16) | (val & ((1u << nbits) - 1));
  }
};

// Transliteration of entropy.h code_residual<N>, restructured for the
// GPU: contexts live in LDS (lane-strided words, no scratch), the
// sub-block significance map is a register bitmask with constant
// neighbor tables, and the in-sub-block scans are fully unrolled so
// every level access has a static offset (the 16 loads of a sub-block
// issue together and pipeline instead of serializing).

// scan-index of the right / below neighbor sub-block (-1 = none)
__constant__ int8_t cSbRight4[16] = {2, 4, 5, 7, 8, 9, 10, 11,
                                     12, -1, 13, 14, -1, 15, -1, -1};
__constant__ int8_t cSbBelow4[16] = {1, 3, 4, 6, 7, 8, -1, 10,
                                     11, 12, -1, 13, 14, -1, 15, -1};
__constant__ int8_t cSbRight2[4] = {2, 3, -1, -1};
__constant__ int8_t cSbBelow2[4] = {1, -1, 3, -1};
// sig_coeff_flag base context by prevCsbf pattern and scan position
// (§9.3.4.2.5; xp/yp are the static in-sub-block coordinates)
__constant__ uint8_t cSigPat[4][16] = {
    // prev 0: (xp+yp==0)?2:(xp+yp<3)?1:0
    {2, 1, 1, 1, 1, 1, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0},
    // prev 1: yp==0?2:yp==1?1:0
    {2, 1, 2, 0, 1, 2, 0, 0, 1, 2, 0, 0, 1, 0, 0, 0},
    // prev 2: xp==0?2:xp==1?1:0
    {2, 2, 1, 2, 1, 0, 2, 1, 0, 0, 1, 0, 0, 0, 0, 0},
    // prev 3
    {2, 2, 2, 2, 2, 2, 2, 2, 2, 2, 2, 2, 2, 2, 2, 2}};

// per-block LDS context bank (one slice segment per block; only lane 0
// runs the bin-serial coder, so jobs never serialize each other through
// wave lockstep — that costs ~64x with one job per LANE)
#define CTX_LDS(i) ctx_lds[(i)]

template <int N>
__device__ void dev_code_residual(DevCabac& cab, uint32_t* ctx_lds, int lane,
                                  const int16_t* __restrict__ level,
                                  int cidx) {
  const int log2n = N == 8 ? 3 : 4;
  constexpr int kSbW = N / 4;
  constexpr int n_sb = kSbW * kSbW;
  const int8_t* sb_right = (N == 8) ? cSbRight2 : cSbRight4;
  const int8_t* sb_below = (N == 8) ? cSbBelow2 : cSbBelow4;
  const uint8_t* sb_scan = (N == 8) ? cScan2 : cScan4;

  // ---- pass 1: sub-block significance map + last position.
  // Each sub-block's 16 levels load with STATIC offsets (unrolled scan),
  // so they live in registers — no per-lane scratch arrays.
  uint32_t csbf_mask = 0;
  int last_sb = -1, last_pos = -1;
  for (int i = 0; i < n_sb; ++i) {
    const int sx = (sb_scan[i] & 15) * 4, sy = (sb_scan[i] >> 4) * 4;
    const int16_t* base = level + sy * N + sx;
    uint32_t nzm = 0;
#pragma unroll
    for (int n = 0; n < 16; ++n) {
      int16_t v = base[(cScan4[n] >> 4) * N + (cScan4[n] & 15)];
      nzm |= (v != 0 ? 1u : 0u) << n;
    }
    if (nzm) {
      csbf_mask |= 1u << i;
      last_sb = i;
      last_pos = 31 - __builtin_clz(nzm);
    }
  }
  const int last_x = (sb_scan[last_sb] & 15) * 4 + (cScan4[last_pos] & 15);
  const int last_y = (sb_scan[last_sb] >> 4) * 4 + (cScan4[last_pos] >> 4);

  int ctx_off, ctx_shift;
  if (cidx == 0) {
    ctx_off = 3 * (log2n - 2) + ((log2n - 1) >> 2);
    ctx_shift = (log2n + 1) >> 2;
  } else {
    ctx_off = 15;
    ctx_shift = log2n - 2;
  }
  const int g_max = (log2n << 1) - 1;
  auto group_idx = [](int v) {
    int g = 0;
    while (true) {
      int base = g < 4 ? g : (2 + (g & 1)) << ((g >> 1) - 1);
      int nx = (g + 1) < 4 ? (g + 1)
                           : (2 + ((g + 1) & 1)) << (((g + 1) >> 1) - 1);
      if (v >= base && v < nx) return g;
      ++g;
    }
  };
  int px = group_idx(last_x), py = group_idx(last_y);
  for (int b = 0; b < (px < g_max ? px : g_max); ++b)
    cab.bin(&CTX_LDS(HG_CTX_LASTX + ctx_off + (b >> ctx_shift)), 1);
  if (px < g_max)
    cab.bin(&CTX_LDS(HG_CTX_LASTX + ctx_off + (px >> ctx_shift)), 0);
  for (int b = 0; b < (py < g_max ? py : g_max); ++b)
    cab.bin(&CTX_LDS(HG_CTX_LASTY + ctx_off + (b >> ctx_shift)), 1);
  if (py < g_max)
    cab.bin(&CTX_LDS(HG_CTX_LASTY + ctx_off + (py >> ctx_shift)), 0);
  if (px > 3) {
    int nbits = (px >> 1) - 1;
    cab.bypass_bins(last_x - ((2 + (px & 1)) << nbits), nbits);
  }
  if (py > 3) {
    int nbits = (py >> 1) - 1;
    cab.bypass_bins(last_y - ((2 + (py & 1)) << nbits), nbits);
  }

  int prev_g1_zero = -1;
  for (int i = last_sb; i >= 0; --i) {
    const bool sb_nz = (csbf_mask >> i) & 1;
    bool coded_explicit = false;
    // prevCsbf pattern from the constant neighbor tables
    const int ri = sb_right[i], bi = sb_below[i];
    const int right = ri >= 0 ? ((csbf_mask >> ri) & 1) : 0;
    const int below = bi >= 0 ? ((csbf_mask >> bi) & 1) : 0;
    if (i < last_sb && i > 0) {
      int c = (right + below > 0 ? 1 : 0) + (cidx ? 2 : 0);
      cab.bin(&CTX_LDS(HG_CTX_CSBF + c), sb_nz ? 1 : 0);
      coded_explicit = true;
    }
    if (!sb_nz && i != last_sb && i != 0) continue;
    bool infer_dc = coded_explicit;

    // reload this sub-block's levels (static offsets -> registers)
    int lvn[16];
    {
      const int sx = (sb_scan[i] & 15) * 4, sy = (sb_scan[i] >> 4) * 4;
      const int16_t* base = level + sy * N + sx;
#pragma unroll
      for (int n = 0; n < 16; ++n)
        lvn[n] = base[(cScan4[n] >> 4) * N + (cScan4[n] & 15)];
    }

    const int prev = right + (below << 1);
    // context adders shared by every coefficient of this sub-block
    int sig_add;
    if (cidx == 0)
      sig_add = (i > 0 ? 3 : 0) + ((log2n == 3) ? 9 : 21);
    else
      sig_add = 27 + ((log2n == 3) ? 9 : 12);
    const int sig_luma_off = cidx == 0 ? 0 : 0;  // folded into sig_add
    (void)sig_luma_off;

    // significance flags: unrolled reverse scan, mask accumulates
    uint32_t sig_mask = 0;
    const int start = (i == last_sb) ? last_pos - 1 : 15;
    if (i == last_sb) sig_mask |= 1u << last_pos;
#pragma unroll
    for (int n = 15; n >= 0; --n) {
      if (n > start) continue;
      const int sig = lvn[n] != 0;
      if (n > 0 || !infer_dc) {
        int sctx;
        if (i == 0 && n == 0) {
          sctx = cidx == 0 ? 0 : 27;
        } else {
          sctx = cSigPat[prev][n] + sig_add;
        }
        cab.bin(&CTX_LDS(HG_CTX_SIG + sctx), sig);
        if (sig) infer_dc = false;
      }
      if (sig) sig_mask |= 1u << n;
    }
    if (!sig_mask) continue;

    // greater1 / greater2 (reverse scan over significant positions)
    int ctx_set = (i == 0 || cidx > 0) ? 0 : 2;
    if (i != last_sb && prev_g1_zero == 1) ctx_set += 1;
    int g1_ctx = 1, first_g1 = -1, k = 0, abs_first_g1 = 0;
#pragma unroll
    for (int n = 15; n >= 0; --n) {
      if (!((sig_mask >> n) & 1)) continue;
      if (k < 8) {
        const int a = lvn[n] < 0 ? -lvn[n] : lvn[n];
        const int g1 = a > 1;
        const int gc = g1_ctx < 3 ? g1_ctx : 3;
        cab.bin(&CTX_LDS(HG_CTX_GT1 + ctx_set * 4 + gc + (cidx ? 16 : 0)),
                g1);
        if (g1) {
          g1_ctx = 0;
          if (first_g1 < 0) { first_g1 = n; abs_first_g1 = a; }
        } else if (g1_ctx > 0 && g1_ctx < 3) {
          ++g1_ctx;
        }
      }
      ++k;
    }
    prev_g1_zero = (g1_ctx == 0) ? 1 : 0;
    if (first_g1 >= 0) {
      const int a = abs_first_g1;
      cab.bin(&CTX_LDS(HG_CTX_GT2 + ctx_set + (cidx ? 4 : 0)), a > 2);
    }
    // signs (bypass)
#pragma unroll
    for (int n = 15; n >= 0; --n)
      if ((sig_mask >> n) & 1) cab.bypass(lvn[n] < 0 ? 1 : 0);
    // remaining levels
    int rice = 0;
    k = 0;
#pragma unroll
    for (int n = 15; n >= 0; --n) {
      if (!((sig_mask >> n) & 1)) continue;
      const int a = lvn[n] < 0 ? -lvn[n] : lvn[n];
      const int base = (k < 8) ? ((n == first_g1) ? 3 : 2) : 1;
      if (a >= base) {
        uint32_t rem = a - base;
        if (rem < (3u << rice)) {
          cab.bypass_unary(rem >> rice);
          if (rice) cab.bypass_bins(rem & ((1 << rice) - 1), rice);
        } else {
          int len = rice;
          uint32_t v = rem - (3u << rice);
          while (v >= (1u << len)) {
            v -= 1u << len;
            ++len;
          }
          cab.bypass_unary(3 + len - rice);
          cab.bypass_bins(v, len);
        }
        if (a > (3 << rice) && rice < 4) ++rice;
      }
      ++k;
    }
  }
}

__device__ void dev_code_ctu(DevCabac& cab, uint32_t* ctx_lds, int lane,
                             int mode, int left_mode, int cbf_mask,
                             const int16_t* __restrict__ lv) {
  cab.bin(&CTX_LDS(HG_CTX_SPLIT), 0);
  int cand_a = left_mode >= 0 ? left_mode : 1;
  int cand_b = 1;
  int list[3];
  if (cand_a == cand_b) {
    if (cand_a < 2) {
      list[0] = 0;
      list[1] = 1;
      list[2] = 26;
    } else {
      list[0] = cand_a;
      list[1] = 2 + ((cand_a + 29) % 32);
      list[2] = 2 + ((cand_a - 2 + 1) % 32);
    }
  } else {
    list[0] = cand_a;
    list[1] = cand_b;
    list[2] = (cand_a != 0 && cand_b != 0) ? 0
              : (cand_a != 1 && cand_b != 1) ? 1
                                             : 26;
  }
  int mpm_idx = -1;
  for (int i = 0; i < 3; ++i)
    if (list[i] == mode) {
      mpm_idx = i;
      break;
    }
  cab.bin(&CTX_LDS(HG_CTX_PREVINTRA), mpm_idx >= 0 ? 1 : 0);
  if (mpm_idx >= 0) {
    cab.bypass(mpm_idx > 0 ? 1 : 0);
    if (mpm_idx > 0) cab.bypass(mpm_idx - 1);
  } else {
    int a = list[0], b = list[1], c = list[2], t;
    if (a > b) { t = a; a = b; b = t; }
    if (b > c) { t = b; b = c; c = t; }
    if (a > b) { t = a; a = b; b = t; }
    int rem = mode;
    if (mode > c) --rem;
    if (mode > b) --rem;
    if (mode > a) --rem;
    cab.bypass_bins(rem, 5);
  }
  cab.bin(&CTX_LDS(HG_CTX_CHROMA), 0);
  cab.bin(&CTX_LDS(HG_CTX_CBFC), (cbf_mask >> 1) & 1);
  cab.bin(&CTX_LDS(HG_CTX_CBFC), (cbf_mask >> 2) & 1);
  cab.bin(&CTX_LDS(HG_CTX_CBFY + 1), cbf_mask & 1);
  if (cbf_mask & 1) dev_code_residual<16>(cab, ctx_lds, lane, lv, 0);
  if (cbf_mask & 2) dev_code_residual<8>(cab, ctx_lds, lane, lv + 256, 1);
  if (cbf_mask & 4) dev_code_residual<8>(cab, ctx_lds, lane, lv + 320, 2);
}

__global__ __launch_bounds__(64) void k_hevc_cabac(
    const int16_t* __restrict__ levels, const int* __restrict__ meta,
    int ctbw, int n_jobs, const HevcJob* __restrict__ jobs,
    uint8_t* __restrict__ out, int out_stride, int* __restrict__ counts) {
  __shared__ uint32_t ctx_lds[HG_NUM_CTX];
  __shared__ uint32_t ptab[256];   // packed (state, qRangeIdx) entries
  const int lane = threadIdx.x;
  const int j = blockIdx.x;
  if (j >= n_jobs) return;

  const HevcJob job = jobs[j];
  const int qp = job.qp < 0 ? 0 : job.qp > 51 ? 51 : job.qp;
  for (int i = lane; i < 256; i += 64) {
    const int state = i >> 2, q = i & 3;
    const uint32_t lps = cRangeTabLps[state][q];
    const uint32_t nlps = cTransIdxLps[state];
    const uint32_t nmps = state < 62 ? state + 1 : state;
    ptab[i] = lps | (nlps << 8) | (nmps << 14)
              | ((uint32_t)cRenorm[lps >> 3] << 20);
  }
  for (int i = lane; i < HG_NUM_CTX; i += 64) {
    int iv = cInitVals[i];
    int slope = (iv >> 4) * 5 - 45;
    int off = ((iv & 15) << 3) - 16;
    int pre = ((slope * qp) >> 4) + off;
    pre = pre < 1 ? 1 : pre > 126 ? 126 : pre;
    ctx_lds[i] = pre <= 63 ? (uint32_t)((63 - pre) << 1)
                           : (uint32_t)(((pre - 64) << 1) | 1);
  }
  __syncthreads();
  if (lane != 0) return;

  DevCabac cab;
  cab.init(out + (size_t)j * out_stride);
  cab.ptab = ptab;
  int left_mode = -1;
  for (int ci = 0; ci < job.seg_w; ++ci) {
    const int cx = job.ctu_x0 + ci;
    const size_t mbase = (size_t)(job.ctu_row * ctbw + cx);
    const int mode = meta[mbase * kHevcMetaPerCtu + 0];
    const int cbf = meta[mbase * kHevcMetaPerCtu + 1];
    dev_code_ctu(cab, ctx_lds, lane, mode, left_mode, cbf,
                 levels + mbase * kHevcLevelsPerCtu);
    left_mode = mode;
    cab.terminate(ci == job.seg_w - 1 ? 1 : 0);
  }
  int tail = cab.finish();
  counts[j * 3 + 0] = cab.count;
  counts[j * 3 + 1] = tail & 0xFFFF;
  counts[j * 3 + 2] = tail >> 16;
}

void launch_hevc_cabac(const int16_t* d_levels, const int* d_meta, int ctbw,
                       int n_jobs, const HevcJob* d_jobs, uint8_t* d_out,
                       int out_stride_bytes, int* d_counts,
                       hipStream_t stream) {
  if (n_jobs == 0) return;
  hipLaunchKernelGGL(k_hevc_cabac, dim3(n_jobs), dim3(64), 0,
                     stream, d_levels, d_meta, ctbw, n_jobs, d_jobs, d_out,
                     out_stride_bytes, d_counts);
}

}  // namespace hevcgpu
}  // namespace hipflux
