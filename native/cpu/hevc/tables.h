// HEVC (ITU-T H.265) constant tables — transcribed from the spec for the
// subset this encoder emits (all-intra, CTU16, TB16 luma / TB8 chroma,
// CABAC). Single source of truth: the GPU entropy stage includes this
// header too. Table numbers cite the 2013-04 H.265 text.
//
// The reference project carries no HEVC encoder (pixelflux is
// H.264/JPEG only; HEVC is a BASELINE.json config-3 target for this
// framework) — nothing here derives from /root/reference.
#pragma once

#include <cstdint>

namespace hipflux {
namespace hevc {

// ---- CABAC arithmetic core (Table 9-46 / 9-47 / 9-48; identical family
// to H.264's engine) -------------------------------------------------------
inline const uint8_t kRangeTabLps[64][4] = {
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

inline const uint8_t kTransIdxLps[64] = {
    0,  0,  1,  2,  2,  4,  4,  5,  6,  7,  8,  9,  9,  11, 11, 12,
    13, 13, 15, 15, 16, 16, 18, 18, 19, 19, 21, 21, 22, 22, 23, 24,
    24, 25, 26, 26, 27, 27, 28, 29, 29, 30, 30, 30, 31, 32, 32, 33,
    33, 33, 34, 34, 35, 35, 35, 36, 36, 36, 37, 37, 37, 38, 38, 63};

inline const uint8_t kTransIdxMps[64] = {
    1,  2,  3,  4,  5,  6,  7,  8,  9,  10, 11, 12, 13, 14, 15, 16,
    17, 18, 19, 20, 21, 22, 23, 24, 25, 26, 27, 28, 29, 30, 31, 32,
    33, 34, 35, 36, 37, 38, 39, 40, 41, 42, 43, 44, 45, 46, 47, 48,
    49, 50, 51, 52, 53, 54, 55, 56, 57, 58, 59, 60, 61, 62, 62, 63};

// HM renormalization lookup: number of shift bits for an LPS range,
// indexed by lps >> 3 (lps in [2, 240]).
inline const uint8_t kRenormTable[32] = {
    6, 5, 4, 4, 3, 3, 3, 3, 2, 2, 2, 2, 2, 2, 2, 2,
    1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1};

// ---- Context initValues, initType 0 (I slices) — Tables 9-5..9-32 -------
// Layout: one flat array per syntax element, indexed by ctxInc.
inline const uint8_t kInitSplitCuFlag[3] = {139, 141, 157};
inline const uint8_t kInitPrevIntraLumaPredFlag = 184;
inline const uint8_t kInitIntraChromaPredMode = 63;
inline const uint8_t kInitCbfLuma[2] = {111, 141};
inline const uint8_t kInitCbfChroma[4] = {94, 138, 182, 154};
inline const uint8_t kInitLastSigXPrefix[18] = {
    110, 110, 124, 125, 140, 153, 125, 127, 140,
    109, 111, 143, 127, 111, 79,  108, 123, 63};
inline const uint8_t kInitLastSigYPrefix[18] = {
    110, 110, 124, 125, 140, 153, 125, 127, 140,
    109, 111, 143, 127, 111, 79,  108, 123, 63};
inline const uint8_t kInitCodedSubBlockFlag[4] = {91, 171, 134, 141};
inline const uint8_t kInitSigCoeffFlag[42] = {
    111, 111, 125, 110, 110, 94,  124, 108, 124, 107, 125, 141, 179, 153,
    125, 107, 125, 141, 179, 153, 125, 107, 125, 141, 179, 153, 125, 140,
    139, 182, 182, 152, 136, 152, 136, 153, 136, 139, 111, 136, 139, 111};
inline const uint8_t kInitGreater1Flag[24] = {
    140, 92,  137, 138, 140, 152, 138, 139, 153, 74,  149, 92,
    139, 107, 122, 152, 140, 179, 166, 182, 140, 227, 122, 197};
inline const uint8_t kInitGreater2Flag[6] = {138, 153, 136, 167, 152, 152};

// Context-bank offsets into one flat context array.
enum CtxOffset {
  kCtxSplitCu = 0,                 // 3
  kCtxPrevIntraLuma = 3,           // 1
  kCtxIntraChroma = 4,             // 1
  kCtxCbfLuma = 5,                 // 2
  kCtxCbfChroma = 7,               // 4
  kCtxLastSigX = 11,               // 18
  kCtxLastSigY = 29,               // 18
  kCtxCodedSubBlock = 47,          // 4
  kCtxSigCoeff = 51,               // 42
  kCtxGreater1 = 93,               // 24
  kCtxGreater2 = 117,              // 6
  kNumContexts = 123,
};

// ---- Scan (§6.5.3 up-right diagonal) -------------------------------------
// Positions in coding order for a 4x4 block: kDiagScan4[i] = (x, y).
// Used both for coefficients within a sub-block and sub-blocks within a TB.
struct ScanPos { uint8_t x, y; };
inline const ScanPos kDiagScan4[16] = {
    {0, 0}, {0, 1}, {1, 0}, {0, 2}, {1, 1}, {2, 0}, {0, 3}, {1, 2},
    {2, 1}, {3, 0}, {1, 3}, {2, 2}, {3, 1}, {2, 3}, {3, 2}, {3, 3}};
// Sub-block diagonal scan for a 2x2 grid of sub-blocks (8x8 TB).
inline const ScanPos kDiagScan2[4] = {{0, 0}, {0, 1}, {1, 0}, {1, 1}};

// ---- Quantization (§8.6.3 / HM QUANT_SHIFT=14) --------------------------
inline const int kQuantScale[6] = {26214, 23302, 20560, 18396, 16384, 14564};
inline const int kDequantScale[6] = {40, 45, 51, 57, 64, 72};

// Chroma QP mapping (Table 8-10, 4:2:0).
inline int chroma_qp(int qp_y) {
  static const int map[14] = {29, 30, 31, 32, 33, 33, 34,
                              34, 35, 35, 36, 36, 37, 37};
  if (qp_y < 30) return qp_y;
  if (qp_y > 43) return qp_y - 6;
  return map[qp_y - 30];
}

// ---- Transform matrices (§8.6.4.2; integer DCT-II approximations) --------
// T8 full matrix; T16 full matrix. Row k of the inverse transform uses
// column k of these (matrices are listed in forward orientation:
// coeff = T * residual * T^t with stage shifts).
inline const int8_t kT8[8][8] = {
    {64, 64, 64, 64, 64, 64, 64, 64},
    {89, 75, 50, 18, -18, -50, -75, -89},
    {83, 36, -36, -83, -83, -36, 36, 83},
    {75, -18, -89, -50, 50, 89, 18, -75},
    {64, -64, -64, 64, 64, -64, -64, 64},
    {50, -89, 18, 75, -75, -18, 89, -50},
    {36, -83, 83, -36, -36, 83, -83, 36},
    {18, -50, 75, -89, 89, -75, 50, -18}};

inline const int8_t kT16[16][16] = {
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

}  // namespace hevc
}  // namespace hipflux
