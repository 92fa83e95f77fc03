// HEVC gfx950 kernel launchers (host side). Kernels in hevc_kernels.hip.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

#include "hevc_gpu_layout.h"

namespace hipflux {
namespace hevcgpu {

// Per-job CTU wavefront: intra decision + transform/quant + recon.
// One workgroup (256 threads) per job, marching the segment's CTUs
// left to right. Writes recon planes, levels and meta.
void launch_hevc_rows(const uint8_t* d_srcY, const uint8_t* d_srcCb,
                      const uint8_t* d_srcCr, int ypitch, int cpitch, int w,
                      int h, uint8_t* d_curY, uint8_t* d_curCb,
                      uint8_t* d_curCr, int ctbw, int n_jobs,
                      const HevcJob* d_jobs, int16_t* d_levels, int* d_meta,
                      hipStream_t stream);

// Per-job CABAC: one lane per job encodes the slice segment's bins from
// levels/meta. Output bytes at d_out[job * out_stride_bytes]; d_counts
// packs {n_bytes, tail_bits, tail_nbits} as 3 ints per job.
void launch_hevc_cabac(const int16_t* d_levels, const int* d_meta, int ctbw,
                       int n_jobs, const HevcJob* d_jobs, uint8_t* d_out,
                       int out_stride_bytes, int* d_counts,
                       hipStream_t stream);

}  // namespace hevcgpu
}  // namespace hipflux
