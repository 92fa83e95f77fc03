// Shared GPU<->CPU layout for the HIP H.264 pipeline.
// The GPU kernels write per-MB quantized levels + mode metadata; the CPU
// entropy stage (cpu/h264/gpu_entropy.cpp) packs CAVLC slices from them.
#pragma once

#include <cstdint>

namespace hipflux {
namespace h264gpu {

// per-MB int16 levels block (raster order inside each 4x4):
//   [0..15]    luma DC (raster over the 4x4 grid of block DCs)
//   [16..271]  luma AC: 16 blocks x 16 (position 0 unused)
//   [272..279] chroma DC: cb c00,c01,c10,c11 then cr
//   [280..407] chroma AC: cb0..3, cr0..3 x 16 (position 0 unused)
constexpr int kLevelsPerMb = 408;
constexpr int kLumaDcOff = 0;
constexpr int kLumaAcOff = 16;
constexpr int kChromaDcOff = 272;
constexpr int kChromaAcOff = 280;

// per-MB meta (2 x int32):
//   m0: mode(2) | luma_mode<<2 | chroma_mode<<5  (mode: 0 skip, 1 inter,
//       2 intra)
//   m1: mvx (int16, quarter-pel) | mvy<<16
constexpr int kMetaPerMb = 2;

enum MbMode : int { kSkip = 0, kInter = 1, kIntra = 2 };

// One job = one SLICE = a horizontal SEGMENT of an MB row. Narrow frames
// use one segment per row; wide frames split rows into multiple slices
// (H.264 allows a slice to start at any MB) so the serial left-neighbor
// chain — the row kernels' latency bound — halves at 4K and quarters at
// 8K. Segment boundaries reset intra/MVP/skip context exactly like row
// starts do.
struct RowJob {
  int mb_row;      // absolute MB row in the frame
  int qp;          // luma QP for this slice
  int flags;       // bit0: I slice (IDR)
  int stripe_y0;   // stripe pixel bounds (for ME clamping)
  int stripe_y1;
  // slice-header fields for the GPU entropy kernel
  int first_mb;    // first_mb_in_slice (stripe-relative)
  int frame_num;
  int idr_pic_id;
  int mbx0;        // first MB column of this segment
  int seg_mbw;     // segment width in MBs
};

// rows wider than this are split into ceil(mbw / kMaxSegMbw) slices
constexpr int kMaxSegMbw = 128;

// GPU entropy staging layout: per row, items =
//   [0] slice header, [1 + mb*28 + slot] per-MB items, [last] trailing
//   skip run.  Per-MB slots: 0 header, 1 lumaDC, 2..17 lumaAC (Z-order),
//   18..19 chromaDC, 20..27 chromaAC.
constexpr int kSlotsPerMb = 28;
constexpr int kStageWordsPerItem = 16;   // 64 B staging per item

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
__host__ __device__
#endif
inline int items_per_row(int mbw) { return 2 + mbw * kSlotsPerMb; }

}  // namespace h264gpu
}  // namespace hipflux
