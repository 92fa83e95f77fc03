// HEVC bitstream assembly from GPU-produced buffers
// (layout: native/hip/hevc_gpu_layout.h).
#pragma once

#include <cstdint>
#include <vector>

namespace hipflux {
namespace hevc {

// Wrap one slice segment's finished CABAC payload (bytes + tail bits from
// the GPU kernel's finish()) into an IDR NAL with slice header.
void assemble_hevc_slice_nal(const uint8_t* cabac_bytes, int n_bytes,
                             int tail_bits, int tail_nbits, bool first_slice,
                             int slice_addr, int addr_bits, int qp,
                             std::vector<uint8_t>& out);

// CPU-entropy fallback: encode one slice segment's CABAC from GPU
// levels/meta (frame-wide layout) and wrap it. Byte-identical to the GPU
// CABAC kernel; used to isolate rows-kernel vs entropy-kernel bugs.
void encode_hevc_job_nal(const int16_t* levels, const int* meta, int ctbw,
                         int ctu_row, int ctu_x0, int seg_w, int qp,
                         bool first_slice, int slice_addr, int addr_bits,
                         std::vector<uint8_t>& out);

// Per-stripe parameter sets (VPS+SPS+PPS for a w x h stripe picture).
void write_hevc_stripe_headers(int coded_w, int coded_h, int vis_w,
                               int vis_h, std::vector<uint8_t>& out);

}  // namespace hevc
}  // namespace hipflux
