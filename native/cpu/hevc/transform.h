// HEVC core transform + quantization (§8.6) for TB16 (luma) and TB8
// (chroma), bit depth 8. Forward path mirrors the standard two-stage
// integer DCT with HM shift conventions; inverse matches the spec decode
// process exactly (it is re-implemented independently in
// tests/hevc_ref_decoder.py).
#pragma once

#include <cstdint>

#include "tables.h"

namespace hipflux {
namespace hevc {

inline int16_t clip16(int v) {
  return static_cast<int16_t>(v < -32768 ? -32768 : v > 32767 ? 32767 : v);
}

// Forward transform: C = T * R * T^t with stage shifts
// s1 = log2N + BitDepth - 9, s2 = log2N + 6.
template <int N>
inline void fwd_transform(const int16_t* res, int res_stride, int16_t* coef) {
  const int8_t(*T)[N] =
      reinterpret_cast<const int8_t(*)[N]>(N == 8 ? &kT8[0][0] : &kT16[0][0]);
  const int log2n = N == 8 ? 3 : 4;
  const int s1 = log2n + 8 - 9, s2 = log2n + 6;
  int32_t tmp[N][N];
  for (int u = 0; u < N; ++u)
    for (int x = 0; x < N; ++x) {
      int32_t acc = 0;
      for (int y = 0; y < N; ++y) acc += T[u][y] * res[y * res_stride + x];
      tmp[u][x] = (acc + (1 << (s1 - 1))) >> s1;
    }
  for (int u = 0; u < N; ++u)
    for (int v = 0; v < N; ++v) {
      int32_t acc = 0;
      for (int x = 0; x < N; ++x) acc += tmp[u][x] * T[v][x];
      coef[u * N + v] = clip16((acc + (1 << (s2 - 1))) >> s2);
    }
}

// Inverse transform per §8.6.4.2: stage1 shift 7 (clip to 16 bits),
// stage2 shift 20 - BitDepth = 12.
template <int N>
inline void inv_transform(const int16_t* coef, int16_t* res, int res_stride) {
  const int8_t(*T)[N] =
      reinterpret_cast<const int8_t(*)[N]>(N == 8 ? &kT8[0][0] : &kT16[0][0]);
  int32_t tmp[N][N];
  for (int y = 0; y < N; ++y)
    for (int v = 0; v < N; ++v) {
      int32_t acc = 0;
      for (int u = 0; u < N; ++u) acc += T[u][y] * coef[u * N + v];
      tmp[y][v] = clip16((acc + 64) >> 7);
    }
  for (int y = 0; y < N; ++y)
    for (int x = 0; x < N; ++x) {
      int32_t acc = 0;
      for (int v = 0; v < N; ++v) acc += tmp[y][v] * T[v][x];
      res[y * res_stride + x] = clip16((acc + 2048) >> 12);
    }
}

// Quantization (HM rounding, intra offset 171/512).
template <int N>
inline int quantize(const int16_t* coef, int qp, int16_t* level) {
  const int log2n = N == 8 ? 3 : 4;
  const int qbits = 14 + qp / 6 + (15 - 8 - log2n);
  const int scale = kQuantScale[qp % 6];
  const int add = 171 << (qbits - 9);
  int nz = 0;
  for (int i = 0; i < N * N; ++i) {
    int c = coef[i];
    int a = c < 0 ? -c : c;
    int l = (a * scale + add) >> qbits;
    if (l > 32767) l = 32767;
    level[i] = static_cast<int16_t>(c < 0 ? -l : l);
    nz += l != 0;
  }
  return nz;
}

// Dequantization per §8.6.3 (flat scaling list m = 16).
template <int N>
inline void dequantize(const int16_t* level, int qp, int16_t* coef) {
  const int log2n = N == 8 ? 3 : 4;
  const int bd_shift = 8 + log2n - 5;
  const int scale = (kDequantScale[qp % 6] << (qp / 6)) * 16;
  for (int i = 0; i < N * N; ++i) {
    int64_t d = (static_cast<int64_t>(level[i]) * scale +
                 (1 << (bd_shift - 1))) >> bd_shift;
    coef[i] = clip16(static_cast<int>(d));
  }
}

}  // namespace hevc
}  // namespace hipflux
