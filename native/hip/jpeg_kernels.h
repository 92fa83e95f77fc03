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
