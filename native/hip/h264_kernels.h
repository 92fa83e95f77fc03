#pragma once
#include <hip/hip_runtime.h>

#include <cstdint>

#include "h264_gpu_layout.h"

namespace hipflux {
namespace h264gpu {

void launch_downsample2(const uint8_t* src, int spitch, int sw, int sh,
                        uint8_t* dst, int dpitch, hipStream_t stream);

void launch_h264_me(const uint8_t* srcY, int ypitch, int w, int h,
                    const uint8_t* refY, const uint8_t* srcY2,
                    const uint8_t* refY2, int ypitch2, int mbw, int n_jobs,
                    const RowJob* d_jobs, int* d_meta, hipStream_t stream,
                    bool use_mfma = true);

void launch_h264_rows4(const uint8_t* srcY, const uint8_t* srcCb,
                       const uint8_t* srcCr, int ypitch, int cpitch, int w,
                       int h, const uint8_t* refY, const uint8_t* refCb,
                       const uint8_t* refCr, uint8_t* curY, uint8_t* curCb,
                       uint8_t* curCr, int mbw, int n_jobs,
                       const struct RowJob* d_jobs, int16_t* d_levels,
                       int* d_meta, hipStream_t stream);

void launch_h264_rows(const uint8_t* srcY, const uint8_t* srcCb,
                      const uint8_t* srcCr, int ypitch, int cpitch, int w,
                      int h, const uint8_t* refY, const uint8_t* refCb,
                      const uint8_t* refCr, uint8_t* curY, uint8_t* curCb,
                      uint8_t* curCr, int mbw, int n_jobs,
                      const RowJob* d_jobs, int16_t* d_levels, int* d_meta,
                      hipStream_t stream);

void launch_h264_deblock(uint8_t* d_curY, uint8_t* d_curCb,
                         uint8_t* d_curCr, int ypitch, int cpitch, int mbw,
                         int n_jobs, const RowJob* d_jobs,
                         const int16_t* d_levels, const int* d_meta,
                         hipStream_t stream);

void launch_h264_cavlc(const int16_t* d_levels, const int* d_meta, int mbw,
                       int n_jobs, const RowJob* d_jobs, uint32_t* d_stage,
                       int* d_nbits, uint32_t* d_out, int out_stride_words,
                       int* d_out_bits, hipStream_t stream);

}  // namespace h264gpu
}  // namespace hipflux
