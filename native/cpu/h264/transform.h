// H.264 4x4 integer transform, Hadamard DC transforms, quant/dequant.
// Shared by the CPU reference encoder and (as scalar reference) the HIP
// kernels — both must produce identical integer results.
#pragma once

#include <cstdint>
#include <cstdlib>

namespace hipflux {
namespace h264 {

// Quantization multipliers MF (per QP%6) for coefficient classes
// a=(0,0),(0,2),(2,0),(2,2); b=(1,1),(1,3),(3,1),(3,3); c=others.
inline constexpr int kQuantMF[6][3] = {
    {13107, 5243, 8066}, {11916, 4660, 7490}, {10082, 4194, 6554},
    {9362, 3647, 5825},  {8192, 3355, 5243},  {7282, 2893, 4559}};

// Dequant scale V (per QP%6), same classes.
inline constexpr int kDequantV[6][3] = {{10, 16, 13}, {11, 18, 14},
                                        {13, 20, 16}, {14, 23, 18},
                                        {16, 25, 20}, {18, 29, 23}};

inline int coeff_class(int i, int j) {  // 0=a 1=b 2=c
  bool ei = (i & 1) == 0, ej = (j & 1) == 0;
  return ei && ej ? 0 : (!ei && !ej ? 1 : 2);
}

// zigzag scan for 4x4 (index -> raster position)
inline constexpr int kZigzag4[16] = {0, 1, 4, 8, 5, 2, 3, 6,
                                     9, 12, 13, 10, 7, 11, 14, 15};

// chroma QP from luma QP (chroma_qp_index_offset = 0), Table 8-15
inline int chroma_qp(int qp) {
  static const int tab[22] = {29, 30, 31, 32, 32, 33, 34, 34, 35, 35, 36,
                              36, 37, 37, 37, 38, 38, 38, 39, 39, 39, 39};
  if (qp < 30) return qp;
  return tab[qp - 30];
}

// Forward 4x4 core transform (residual in, coefficients out; raster order).
inline void fdct4x4(const int* in, int* out) {
  int tmp[16];
  for (int i = 0; i < 4; ++i) {
    const int* r = in + 4 * i;
    int s03 = r[0] + r[3], d03 = r[0] - r[3];
    int s12 = r[1] + r[2], d12 = r[1] - r[2];
    tmp[4 * i + 0] = s03 + s12;
    tmp[4 * i + 1] = 2 * d03 + d12;
    tmp[4 * i + 2] = s03 - s12;
    tmp[4 * i + 3] = d03 - 2 * d12;
  }
  for (int j = 0; j < 4; ++j) {
    int a = tmp[j], b = tmp[4 + j], c = tmp[8 + j], d = tmp[12 + j];
    int s03 = a + d, d03 = a - d, s12 = b + c, d12 = b - c;
    out[j] = s03 + s12;
    out[4 + j] = 2 * d03 + d12;
    out[8 + j] = s03 - s12;
    out[12 + j] = d03 - 2 * d12;
  }
}

// Inverse 4x4 core transform. Input: dequantized coefficients; output adds
// (x+32)>>6 rounding. Caller adds prediction and clips.
inline void idct4x4(const int* in, int* out) {
  int tmp[16];
  for (int i = 0; i < 4; ++i) {
    const int* r = in + 4 * i;
    int e0 = r[0] + r[2], e1 = r[0] - r[2];
    int e2 = (r[1] >> 1) - r[3], e3 = r[1] + (r[3] >> 1);
    tmp[4 * i + 0] = e0 + e3;
    tmp[4 * i + 1] = e1 + e2;
    tmp[4 * i + 2] = e1 - e2;
    tmp[4 * i + 3] = e0 - e3;
  }
  for (int j = 0; j < 4; ++j) {
    int a = tmp[j], b = tmp[4 + j], c = tmp[8 + j], d = tmp[12 + j];
    int e0 = a + c, e1 = a - c;
    int e2 = (b >> 1) - d, e3 = b + (d >> 1);
    out[j] = (e0 + e3 + 32) >> 6;
    out[4 + j] = (e1 + e2 + 32) >> 6;
    out[8 + j] = (e1 - e2 + 32) >> 6;
    out[12 + j] = (e0 - e3 + 32) >> 6;
  }
}

// 4x4 Hadamard for I16x16 luma DC (forward includes the /2).
inline void hadamard4x4_fwd(const int* in, int* out) {
  int tmp[16];
  for (int i = 0; i < 4; ++i) {
    const int* r = in + 4 * i;
    int s03 = r[0] + r[3], d03 = r[0] - r[3];
    int s12 = r[1] + r[2], d12 = r[1] - r[2];
    tmp[4 * i + 0] = s03 + s12;
    tmp[4 * i + 1] = d03 + d12;
    tmp[4 * i + 2] = s03 - s12;
    tmp[4 * i + 3] = d03 - d12;
  }
  for (int j = 0; j < 4; ++j) {
    int a = tmp[j], b = tmp[4 + j], c = tmp[8 + j], d = tmp[12 + j];
    int s03 = a + d, d03 = a - d, s12 = b + c, d12 = b - c;
    out[j] = (s03 + s12) >> 1;
    out[4 + j] = (d03 + d12) >> 1;
    out[8 + j] = (s03 - s12) >> 1;
    out[12 + j] = (d03 - d12) >> 1;
  }
}

// Inverse 4x4 Hadamard (no scaling).
inline void hadamard4x4_inv(const int* in, int* out) {
  int tmp[16];
  for (int i = 0; i < 4; ++i) {
    const int* r = in + 4 * i;
    int s03 = r[0] + r[3], d03 = r[0] - r[3];
    int s12 = r[1] + r[2], d12 = r[1] - r[2];
    tmp[4 * i + 0] = s03 + s12;
    tmp[4 * i + 1] = d03 + d12;
    tmp[4 * i + 2] = s03 - s12;
    tmp[4 * i + 3] = d03 - d12;
  }
  for (int j = 0; j < 4; ++j) {
    int a = tmp[j], b = tmp[4 + j], c = tmp[8 + j], d = tmp[12 + j];
    int s03 = a + d, d03 = a - d, s12 = b + c, d12 = b - c;
    out[j] = s03 + s12;
    out[4 + j] = d03 + d12;
    out[8 + j] = s03 - s12;
    out[12 + j] = d03 - d12;
  }
}

// Max |level| we emit: keeps CAVLC level codes within the level_prefix<=15
// escape (the prefix>=16 extension is not Baseline-safe). Only reachable
// near QP 0 on extreme residuals.
constexpr int kMaxLevel = 2063;

// Quantize one AC/4x4 coefficient. intra: f = 2^qbits/3, inter: /6.
inline int quant_coeff(int w, int qp, int cls, bool intra) {
  int qbits = 15 + qp / 6;
  int mf = kQuantMF[qp % 6][cls];
  int f = (1 << qbits) / (intra ? 3 : 6);
  int az = std::abs(w);
  int level = (az * mf + f) >> qbits;
  if (level > kMaxLevel) level = kMaxLevel;
  return w < 0 ? -level : level;
}

inline int dequant_coeff(int level, int qp, int cls) {
  // multiply, not shift: level may be negative (<< on negatives is UB)
  return level * kDequantV[qp % 6][cls] * (1 << (qp / 6));
}

// DC (Hadamard-domain) quant: double shift, doubled rounding.
inline int quant_dc(int w, int qp, bool intra) {
  int qbits = 15 + qp / 6;
  int mf = kQuantMF[qp % 6][0];
  int f = (1 << qbits) / (intra ? 3 : 6);
  int az = std::abs(w);
  int level = (az * mf + 2 * f) >> (qbits + 1);
  if (level > kMaxLevel) level = kMaxLevel;
  return w < 0 ? -level : level;
}

// Luma DC dequant (after inverse Hadamard).
inline int dequant_luma_dc(int c, int qp) {
  int v = kDequantV[qp % 6][0];
  if (qp >= 12) return c * v * (1 << (qp / 6 - 2));
  return (c * v + (1 << (1 - qp / 6))) >> (2 - qp / 6);
}

// Chroma DC dequant (after inverse 2x2 Hadamard).
inline int dequant_chroma_dc(int c, int qp) {
  int v = kDequantV[qp % 6][0];
  if (qp >= 6) return c * v * (1 << (qp / 6 - 1));
  return (c * v) >> 1;
}

}  // namespace h264
}  // namespace hipflux
