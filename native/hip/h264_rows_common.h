// Shared device helpers for the H.264 row kernels (both TUs).
// __constant__ tables are `static` so each TU gets its own copy (no RDC).
#pragma once
#include <hip/hip_runtime.h>

#include "h264_gpu_layout.h"

namespace hipflux {
namespace h264gpu {

static __constant__ int c_quant_mf[6][3] = {
    {13107, 5243, 8066}, {11916, 4660, 7490}, {10082, 4194, 6554},
    {9362, 3647, 5825},  {8192, 3355, 5243},  {7282, 2893, 4559}};
static __constant__ int c_dequant_v[6][3] = {{10, 16, 13}, {11, 18, 14},
                                      {13, 20, 16}, {14, 23, 18},
                                      {16, 25, 20}, {18, 29, 23}};
// zigzag index of each raster position (inverse of the scan)
static __constant__ int c_zz_of_pos[16] = {0, 1, 5, 6, 2, 4, 7, 12,
                                    3, 8, 11, 13, 9, 10, 14, 15};
static __constant__ int c_chroma_qp[22] = {29, 30, 31, 32, 32, 33, 34, 34,
                                    35, 35, 36, 36, 37, 37, 37, 38,
                                    38, 38, 39, 39, 39, 39};

__device__ inline int dev_chroma_qp(int qp) {
  return qp < 30 ? qp : c_chroma_qp[qp - 30];
}

__device__ inline int coeff_cls(int pos) {
  int i = pos >> 2, j = pos & 3;
  bool ei = (i & 1) == 0, ej = (j & 1) == 0;
  return (ei && ej) ? 0 : ((!ei && !ej) ? 1 : 2);
}

__device__ inline int quant_coeff(int w, int qp, int cls,
                                  bool intra = true) {
  int qbits = 15 + qp / 6;
  int f = (1 << qbits) / (intra ? 3 : 6);
  int az = abs(w);
  int level = (az * c_quant_mf[qp % 6][cls] + f) >> qbits;
  level = min(level, 2063);
  return w < 0 ? -level : level;
}

__device__ inline int quant_dc_v(int w, int qp, bool intra = true) {
  int qbits = 15 + qp / 6;
  int f = (1 << qbits) / (intra ? 3 : 6);
  int az = abs(w);
  int level = (az * c_quant_mf[qp % 6][0] + 2 * f) >> (qbits + 1);
  level = min(level, 2063);
  return w < 0 ? -level : level;
}

__device__ inline int dequant_c(int level, int qp, int cls) {
  // multiply, not shift: level may be negative (<< on negatives is UB)
  return level * c_dequant_v[qp % 6][cls] * (1 << (qp / 6));
}

__device__ inline int dequant_luma_dc_v(int c, int qp) {
  int v = c_dequant_v[qp % 6][0];
  if (qp >= 12) return c * v * (1 << (qp / 6 - 2));
  return (c * v + (1 << (1 - qp / 6))) >> (2 - qp / 6);
}

__device__ inline int dequant_chroma_dc_v(int c, int qp) {
  int v = c_dequant_v[qp % 6][0];
  if (qp >= 6) return c * v * (1 << (qp / 6 - 1));
  return (c * v) >> 1;
}

__device__ inline uint8_t clip8(int v) { return (uint8_t)max(0, min(255, v)); }

// ---- H.264 half-sample interpolation (8.4.2.2.1 subset) -------------------
// MVs live on the half-pel grid (quarter-pel units, mv % 2 == 0).
__device__ inline int tap6i(int a, int b, int c, int d, int e, int f) {
  return a - 5 * b + 20 * c + 20 * d - 5 * e + f;
}

__device__ inline int hsum6g(const uint8_t* row, int x) {
  return tap6i(row[x - 2], row[x - 1], row[x], row[x + 1], row[x + 2],
               row[x + 3]);
}

// half samples at integer base (x,y)
__device__ inline int half_b(const uint8_t* p, int pitch, int x, int y) {
  return clip8((hsum6g(p + (size_t)y * pitch, x) + 16) >> 5);
}

__device__ inline int half_h(const uint8_t* p, int pitch, int x, int y) {
  const uint8_t* c = p + (size_t)(y - 2) * pitch + x;
  int v = tap6i(c[0], c[pitch], c[2 * pitch], c[3 * pitch], c[4 * pitch],
                c[5 * pitch]);
  return clip8((v + 16) >> 5);
}

__device__ inline int half_j(const uint8_t* p, int pitch, int x, int y) {
  int v = tap6i(hsum6g(p + (size_t)(y - 2) * pitch, x),
                hsum6g(p + (size_t)(y - 1) * pitch, x),
                hsum6g(p + (size_t)y * pitch, x),
                hsum6g(p + (size_t)(y + 1) * pitch, x),
                hsum6g(p + (size_t)(y + 2) * pitch, x),
                hsum6g(p + (size_t)(y + 3) * pitch, x));
  return clip8((v + 512) >> 10);
}

// predicted luma sample at integer (x,y), frac (fx,fy) in 0..3: 6-tap
// half samples (8.4.2.2.1) + rounded-average quarters (Table 8-12)
__device__ inline int luma_interp(const uint8_t* p, int pitch, int x, int y,
                                  int fx, int fy) {
  if ((fx | fy) == 0) return p[(size_t)y * pitch + x];
  if (fy == 0) {
    int b = half_b(p, pitch, x, y);
    if (fx == 2) return b;
    int g = p[(size_t)y * pitch + x + (fx == 1 ? 0 : 1)];
    return (g + b + 1) >> 1;
  }
  if (fx == 0) {
    int hh = half_h(p, pitch, x, y);
    if (fy == 2) return hh;
    int g = p[(size_t)(y + (fy == 1 ? 0 : 1)) * pitch + x];
    return (g + hh + 1) >> 1;
  }
  if (fx == 2 && fy == 2) return half_j(p, pitch, x, y);
  if (fx == 2) {                  // f / q: avg of b (above/below) and j
    int j = half_j(p, pitch, x, y);
    int b = half_b(p, pitch, x, y + (fy == 1 ? 0 : 1));
    return (b + j + 1) >> 1;
  }
  if (fy == 2) {                  // i / k: avg of h (left/right) and j
    int j = half_j(p, pitch, x, y);
    int hh = half_h(p, pitch, x + (fx == 1 ? 0 : 1), y);
    return (hh + j + 1) >> 1;
  }
  int b = half_b(p, pitch, x, y + (fy == 1 ? 0 : 1));
  int hh = half_h(p, pitch, x + (fx == 1 ? 0 : 1), y);
  return (b + hh + 1) >> 1;
}

// predicted chroma sample: bilinear with eighth-pel weights
__device__ inline int chroma_interp(const uint8_t* p, int pitch, int x,
                                    int y, int dx, int dy) {
  const uint8_t* r0 = p + (size_t)y * pitch + x;
  int a = r0[0];
  int b = dx ? r0[1] : a;
  int c = dy ? r0[pitch] : a;
  int d = dy ? (dx ? r0[pitch + 1] : c) : b;
  return ((8 - dx) * (8 - dy) * a + dx * (8 - dy) * b + (8 - dx) * dy * c +
          dx * dy * d + 32) >> 6;
}

// ---------------------------------------------------------------------------
// 16-lane-group 4x4 transforms via shuffles. lane c in [0,16): r=c>>2, x=c&3.
// base = (lane & ~15) is the group's first lane in the wave.

// branchless forward butterfly: out(u) for inputs s0..s3
//   u0: (s0+s3)+(s1+s2)   u1: 2(s0-s3)+(s1-s2)
//   u2: (s0+s3)-(s1+s2)   u3: (s0-s3)-2(s1-s2)
// selected via cndmask-style arithmetic (no lane divergence).
__device__ inline int fdct_bfly(int u, int s0, int s1, int s2, int s3) {
  bool odd = u & 1;            // u1/u3 use differences, u0/u2 sums
  bool hi = u & 2;             // u2/u3 subtract the second term
  int a = odd ? (s0 - s3) : (s0 + s3);
  int b = odd ? (s1 - s2) : (s1 + s2);
  // weights: u0: a+b; u1: 2a+b; u2: a-b; u3: a-2b
  int wa = (u == 1) ? 2 : 1;
  int wb = (u == 3) ? 2 : 1;
  return hi ? (wa * a - wb * b) : (wa * a + wb * b);
}

__device__ inline int fdct4_wave(int v, int lane) {
  int base = lane & ~15;
  int r = (lane >> 2) & 3, u = lane & 3;
  int s0 = __shfl(v, base + r * 4 + 0);
  int s1 = __shfl(v, base + r * 4 + 1);
  int s2 = __shfl(v, base + r * 4 + 2);
  int s3 = __shfl(v, base + r * 4 + 3);
  int t = fdct_bfly(u, s0, s1, s2, s3);
  int c0 = __shfl(t, base + 0 * 4 + u);
  int c1 = __shfl(t, base + 1 * 4 + u);
  int c2 = __shfl(t, base + 2 * 4 + u);
  int c3 = __shfl(t, base + 3 * 4 + u);
  return fdct_bfly(r, c0, c1, c2, c3);
}

// inverse core transform; input dequantized coeff per lane; result includes
// (x+32)>>6
// branchless inverse butterfly: x0:e0+e3 x1:e1+e2 x2:e1-e2 x3:e0-e3
__device__ inline int idct_bfly(int x, int d0, int d1, int d2, int d3) {
  int e0 = d0 + d2, e1 = d0 - d2;
  int e2 = (d1 >> 1) - d3, e3 = d1 + (d3 >> 1);
  bool mid = (x == 1) || (x == 2);   // e1/e2 pair
  int a = mid ? e1 : e0;
  int b = mid ? e2 : e3;
  return (x & 2) ? (a - b) : (a + b);
}

__device__ inline int idct4_wave(int d, int lane) {
  int base = lane & ~15;
  int r = (lane >> 2) & 3, x = lane & 3;
  int d0 = __shfl(d, base + r * 4 + 0);
  int d1 = __shfl(d, base + r * 4 + 1);
  int d2 = __shfl(d, base + r * 4 + 2);
  int d3 = __shfl(d, base + r * 4 + 3);
  int t = idct_bfly(x, d0, d1, d2, d3);
  int c0 = __shfl(t, base + 0 * 4 + x);
  int c1 = __shfl(t, base + 1 * 4 + x);
  int c2 = __shfl(t, base + 2 * 4 + x);
  int c3 = __shfl(t, base + 3 * 4 + x);
  return (idct_bfly(r, c0, c1, c2, c3) + 32) >> 6;
}

// branchless Hadamard butterfly: u0:a+b u1:d03+d12 u2:a-b u3:d03-d12
__device__ inline int had_bfly(int u, int s0, int s1, int s2, int s3) {
  bool odd = u & 1;
  int a = odd ? (s0 - s3) : (s0 + s3);
  int b = odd ? (s1 - s2) : (s1 + s2);
  return (u & 2) ? (a - b) : (a + b);
}

// 4x4 Hadamard (fwd includes >>1) on lanes 0..15 of the wave
__device__ inline int hadamard4_wave(int v, int lane, bool fwd) {
  int r = (lane >> 2) & 3, u = lane & 3;
  int base = lane & ~15;
  int s0 = __shfl(v, base + r * 4 + 0);
  int s1 = __shfl(v, base + r * 4 + 1);
  int s2 = __shfl(v, base + r * 4 + 2);
  int s3 = __shfl(v, base + r * 4 + 3);
  int t = had_bfly(u, s0, s1, s2, s3);
  int c0 = __shfl(t, base + 0 * 4 + u);
  int c1 = __shfl(t, base + 1 * 4 + u);
  int c2 = __shfl(t, base + 2 * 4 + u);
  int c3 = __shfl(t, base + 3 * 4 + u);
  int o = had_bfly(r, c0, c1, c2, c3);
  return fwd ? (o >> 1) : o;
}

// store a 16-lane group's int16 levels as paired u32 writes (even lanes)
__device__ inline void store_lvl_pair(int16_t* base, int c, int lvl,
                                      int lane) {
  int nxt = __shfl(lvl, lane + 1);
  if ((c & 1) == 0)
    *reinterpret_cast<uint32_t*>(base + c) =
        (uint16_t)lvl | ((uint32_t)(uint16_t)nxt << 16);
}

// cap nonzero count at 12 within a 16-lane group (zero highest zigzag)
// active: whether this lane's coefficient participates (e.g. AC excludes 0)
__device__ inline int cap12_group(int level, int zz, bool active, int lane) {
  for (;;) {
    unsigned long long m = __ballot(active && level != 0);
    int cnt = __popcll((m >> (lane & ~15)) & 0xFFFFULL);
    if (cnt <= 12) break;
    int key = (active && level != 0) ? zz : -1;
    int mx = key;
    for (int d = 1; d < 16; d <<= 1) mx = max(mx, __shfl_xor(mx, d));
    if (key == mx) level = 0;
  }
  return level;
}

__device__ inline int wave_sum_i(int v) {
  for (int d = 1; d < 64; d <<= 1) v += __shfl_xor(v, d);
  return v;
}

__device__ inline int wave_min_i(int v) {
  for (int d = 1; d < 64; d <<= 1) v = min(v, __shfl_xor(v, d));
  return v;
}


}  // namespace h264gpu
}  // namespace hipflux
