#pragma once
#include <hip/hip_runtime.h>

#include <cstdint>

namespace hipflux {

void upload_dct_tables(hipStream_t stream);

void launch_bgrx_to_planes(const void* bgrx, int width, int height,
                           int stride_px, uint8_t* yp, uint8_t* cbp,
                           uint8_t* crp, int ypitch, int cpitch,
                           bool fullcolor, hipStream_t stream);

// DCT+quant one plane into MCU-scan-order int16 blocks (see .hip for the
// output indexing contract shared with the CPU entropy coder).
void launch_dct_quant(const uint8_t* plane, int pw, int ph, int pitch,
                      const float* rq, int16_t* out, int plane_kind,
                      bool fullcolor, int mcux, int mcu_rows_per_stripe,
                      int stripe_mcu_count, hipStream_t stream);

}  // namespace hipflux

namespace hipflux {
namespace jpeggpu {
// GPU JPEG Huffman entropy (jpeg_entropy.hip): one workgroup per MCU
// row, restart-interval framing. Tables: 12+256+12+256 packed
// code|(size<<16) words (dc luma, ac luma, dc chroma, ac chroma).
struct JRow {
  int coeff_off;
};
constexpr int kJStageWords = 56;
void launch_jpeg_entropy(const int16_t* d_coeff, const JRow* d_jobs,
                         int n_rows, int mcux, int per_mcu,
                         const uint32_t* d_tabs, uint32_t* d_stage,
                         int* d_nbits, uint32_t* d_out,
                         int out_stride_words, int* d_out_bits,
                         hipStream_t stream);
}  // namespace jpeggpu
}  // namespace hipflux
