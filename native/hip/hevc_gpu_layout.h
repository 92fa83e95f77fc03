// Shared GPU<->host layout for the HEVC pipeline (mirrors the role of
// h264_gpu_layout.h). Levels/meta are frame-wide, CTU-granular:
//   levels[(ctu_row * ctbw + ctu_x) * kHevcLevelsPerCtu]
//     +0    .. 255 : luma 16x16 quantized levels, raster [y*16+x]
//     +256  .. 319 : Cb 8x8 levels
//     +320  .. 383 : Cr 8x8 levels
//   meta[(ctu_row * ctbw + ctu_x) * kHevcMetaPerCtu]
//     +0 : chosen intra mode (0 planar / 1 dc / 10 hor / 26 ver)
//     +1 : cbf mask (bit0 luma, bit1 cb, bit2 cr)
#pragma once

#include <cstdint>

namespace hipflux {
namespace hevcgpu {

constexpr int kHevcLevelsPerCtu = 384;
constexpr int kHevcMetaPerCtu = 2;

// One slice-segment job (one CABAC stream; one row-kernel workgroup).
struct HevcJob {
  int ctu_row;      // absolute CTU row in the frame
  int ctu_x0;       // first CTU x of this segment
  int seg_w;        // CTUs in this segment
  int qp;           // luma QP
  int qpc;          // chroma QP (host-computed Table 8-10 mapping)
  int first_slice;  // 1 = first slice segment of its stripe picture
  int slice_addr;   // CTU address within the stripe picture
  int addr_bits;    // bits of slice_segment_address in this stripe
  int last_in_pic;  // 1 = this segment codes end_of_slice = 1 at its end
};

}  // namespace hevcgpu
}  // namespace hipflux
