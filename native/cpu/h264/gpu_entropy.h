// CAVLC entropy packing from GPU-produced level/meta buffers
// (layout: native/hip/h264_gpu_layout.h). One call packs one stripe's
// frame: SPS/PPS (on IDR) + one slice NAL per MB row.
#pragma once

#include <cstdint>
#include <vector>

namespace hipflux {
namespace h264 {

struct GpuStripeParams {
  const int16_t* levels;   // frame-wide levels buffer (host copy)
  const int* meta;         // frame-wide meta buffer (host copy)
  int mbw;                 // frame MBs per row
  int mb_row0;             // first absolute MB row of this stripe
  int n_mb_rows;           // MB rows in this stripe
  int width;               // stripe logical width (pixels)
  int height;              // stripe logical height (pixels)
  int qp;
  bool idr;
  bool deblock = false;   // slice headers signal idc=2
  uint32_t frame_num;
  uint32_t idr_pic_id;
};

void encode_stripe_from_gpu(const GpuStripeParams& p,
                            std::vector<uint8_t>& out);

// One MB-row slice NAL (row is stripe-relative). Used for row-parallel
// entropy; concatenating SPS/PPS (idr) + all rows reproduces
// encode_stripe_from_gpu's output exactly.
void encode_seg_nal_from_gpu(const GpuStripeParams& p, int row, int mbx0,
                             int seg_mbw, bool long_startcode,
                             std::vector<uint8_t>& out);

void encode_row_nal_from_gpu(const GpuStripeParams& p, int row,
                             std::vector<uint8_t>& out);

// Wrap a GPU-packed slice RBSP (MSB-first u32 words, `bits` long) into an
// Annex-B NAL: stop bit, emulation prevention, start code + header.
void assemble_gpu_row_nal(const uint32_t* words, int bits, bool idr,
                          bool long_startcode, std::vector<uint8_t>& out);

}  // namespace h264
}  // namespace hipflux
