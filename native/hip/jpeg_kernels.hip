// CDNA4 (gfx950) kernels for the striped-JPEG pipeline:
//   k_bgrx_to_planes_420 / _444 — BGRX -> Y + (subsampled) Cb/Cr planes
//   k_dct_quant               — 8x8 forward DCT + quantize, one WAVE per block
//
// Design notes (cdna_hip_programming.md):
//  * wavefront = 64 lanes -> one lane per 8x8 coefficient, a natural fit:
//    the whole block transform lives in one wave, LDS-staged, no divergence.
//  * CSC kernels are memory-bound: uchar4/uint4 vectorized loads, grid-stride,
//    grid capped so the scheduler has room (Guideline 11).
//  * Output is written in stripe-major MCU scan order so the CPU entropy
//    coder for stripe s reads one contiguous range (no gather).
#include <hip/hip_runtime.h>

#include <algorithm>

#include "jpeg_kernels.h"

using std::min;

namespace hipflux {

// BT.601 full-range (JFIF) coefficients — match cpu/jpeg_enc.cpp exactly.
// fp contract OFF: the host reference compiles without FMA (x86-64
// baseline), so fused mul-adds here would produce rare ±1 rounding
// differences on noise content and break the HEVC/H.264 byte-equality
// contract between the CPU and GPU pipelines.
#pragma clang fp contract(off)
__device__ inline float dev_y(float r, float g, float b) {
  return 0.299f * r + 0.587f * g + 0.114f * b;
}
__device__ inline float dev_cb(float r, float g, float b) {
  return -0.168736f * r - 0.331264f * g + 0.5f * b + 128.f;
}
__device__ inline float dev_cr(float r, float g, float b) {
  return 0.5f * r - 0.418688f * g - 0.081312f * b + 128.f;
}

// ---------------------------------------------------------------------------
// BGRX -> planar Y + 2x2-subsampled Cb/Cr. Grid-stride over 2x2 quads.
// Edge replication for odd sizes is handled by clamping reads.
__global__ void k_bgrx_to_planes_420(const uchar4* __restrict__ bgrx,
                                     int width, int height, int stride_px,
                                     uint8_t* __restrict__ yp,
                                     uint8_t* __restrict__ cbp,
                                     uint8_t* __restrict__ crp,
                                     int ypitch, int cpitch) {
  int cw = (width + 1) >> 1, ch = (height + 1) >> 1;
  int total = cw * ch;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * blockDim.x) {
    int qx = (i % cw) * 2, qy = (i / cw) * 2;
    float cbs = 0.f, crs = 0.f;
#pragma unroll
    for (int dy = 0; dy < 2; ++dy) {
#pragma unroll
      for (int dx = 0; dx < 2; ++dx) {
        int x = min(qx + dx, width - 1), y = min(qy + dy, height - 1);
        uchar4 p = bgrx[y * stride_px + x];
        float r = p.z, g = p.y, b = p.x;
        float yv = dev_y(r, g, b);
        cbs += dev_cb(r, g, b);
        crs += dev_cr(r, g, b);
        if (qx + dx < width && qy + dy < height)
          yp[(qy + dy) * ypitch + qx + dx] =
              (uint8_t)__float2int_rn(fminf(fmaxf(yv, 0.f), 255.f));
      }
    }
    cbp[(qy >> 1) * cpitch + (qx >> 1)] =
        (uint8_t)__float2int_rn(fminf(fmaxf(cbs * 0.25f, 0.f), 255.f));
    crp[(qy >> 1) * cpitch + (qx >> 1)] =
        (uint8_t)__float2int_rn(fminf(fmaxf(crs * 0.25f, 0.f), 255.f));
  }
}

__global__ void k_bgrx_to_planes_444(const uchar4* __restrict__ bgrx,
                                     int width, int height, int stride_px,
                                     uint8_t* __restrict__ yp,
                                     uint8_t* __restrict__ cbp,
                                     uint8_t* __restrict__ crp, int pitch) {
  int total = width * height;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * blockDim.x) {
    int x = i % width, y = i / width;
    uchar4 p = bgrx[y * stride_px + x];
    float r = p.z, g = p.y, b = p.x;
    yp[y * pitch + x] =
        (uint8_t)__float2int_rn(fminf(fmaxf(dev_y(r, g, b), 0.f), 255.f));
    cbp[y * pitch + x] =
        (uint8_t)__float2int_rn(fminf(fmaxf(dev_cb(r, g, b), 0.f), 255.f));
    crp[y * pitch + x] =
        (uint8_t)__float2int_rn(fminf(fmaxf(dev_cr(r, g, b), 0.f), 255.f));
  }
}

// ---------------------------------------------------------------------------
// 8x8 forward DCT + quantize. One wave per block; lane = one coefficient.
// in: 8-bit plane; out: int16 blocks at MCU-scan-order offsets.
//
// Output indexing (4:2:0): for plane_kind 0 (Y), block (bx,by) belongs to
// MCU (bx>>1, by>>1), sub-block (by&1)*2+(bx&1) at MCU offset sub*64.
// For Cb (1) / Cr (2): MCU (bx,by), offset (3+plane_kind)*64.
// MCUs are laid out stripe-major: stripe = my / mcu_rows_per_stripe;
// within a stripe, MCU scan order (my_local * mcux + mx) * 384.
// For 4:4:4 the per-MCU block count is 3 and sub offsets are plane_kind.
__constant__ float c_dctmat[64];  // m[u*8+x] = c(u)/2 cos((2x+1)u pi/16)

__global__ void __launch_bounds__(256)
    k_dct_quant(const uint8_t* __restrict__ plane, int pw, int ph, int pitch,
                const float* __restrict__ rq,  // reciprocal quant, natural
                int16_t* __restrict__ out, int plane_kind, bool fullcolor,
                int mcux, int mcu_rows_per_stripe, int stripe_mcu_count) {
  int wave = threadIdx.x >> 6;          // 4 waves per workgroup
  int lane = threadIdx.x & 63;
  int bx_count = (pw + 7) >> 3, by_count = (ph + 7) >> 3;
  int nblocks = bx_count * by_count;
  int block = blockIdx.x * 4 + wave;
  if (block >= nblocks) return;
  int bx = block % bx_count, by = block / bx_count;

  // load pixel (lane = y*8+x), clamped edge replication, center at 0
  int px = bx * 8 + (lane & 7), py = by * 8 + (lane >> 3);
  px = min(px, pw - 1);
  py = min(py, ph - 1);
  float v = (float)plane[py * pitch + px] - 128.f;

  // Wave-synchronous 2D DCT via cross-lane shuffles (no LDS, no barriers):
  // row pass: lane (y,u) = sum_x pixel[y][x] * m[u][x]
  //   pixel[y][x] lives in lane (lane & ~7) + x.
  int u = lane & 7;
  float t = 0.f;
#pragma unroll
  for (int x = 0; x < 8; ++x)
    t += __shfl(v, (lane & ~7) + x) * c_dctmat[u * 8 + x];
  // col pass: lane (v,u) = sum_y tmp[y][u] * m[v][y];  tmp[y][u] in lane y*8+u
  int vv = lane >> 3;
  float s = 0.f;
#pragma unroll
  for (int y = 0; y < 8; ++y)
    s += __shfl(t, y * 8 + u) * c_dctmat[vv * 8 + y];
  int q = __float2int_rn(s * rq[lane]);

  // MCU-scan-order output offset
  int mx, my, sub, per_mcu;
  if (!fullcolor) {
    per_mcu = 6;
    if (plane_kind == 0) {
      mx = bx >> 1;
      my = by >> 1;
      sub = (by & 1) * 2 + (bx & 1);
    } else {
      mx = bx;
      my = by;
      sub = 3 + plane_kind;
    }
  } else {
    per_mcu = 3;
    mx = bx;
    my = by;
    sub = plane_kind;
  }
  int stripe = my / mcu_rows_per_stripe;
  int my_local = my - stripe * mcu_rows_per_stripe;
  size_t base =
      ((size_t)stripe * stripe_mcu_count + (size_t)my_local * mcux + mx) *
          per_mcu * 64 +
      sub * 64;
  out[base + lane] = (int16_t)q;
}

// ---------------------------------------------------------------------------
// host-side wrappers
void upload_dct_tables(hipStream_t stream) {
  float m[64];
  for (int u = 0; u < 8; ++u) {
    double cu = (u == 0) ? 0.70710678118654752440 : 1.0;
    for (int x = 0; x < 8; ++x)
      m[u * 8 + x] = (float)(0.5 * cu * cos((2 * x + 1) * u * M_PI / 16.0));
  }
  (void)hipMemcpyToSymbolAsync(HIP_SYMBOL(c_dctmat), m, sizeof(m), 0,
                               hipMemcpyHostToDevice, stream);
}

void launch_bgrx_to_planes(const void* bgrx, int width, int height,
                           int stride_px, uint8_t* yp, uint8_t* cbp,
                           uint8_t* crp, int ypitch, int cpitch,
                           bool fullcolor, hipStream_t stream) {
  int threads = 256;
  if (!fullcolor) {
    int total = ((width + 1) / 2) * ((height + 1) / 2);
    int blocks = min((total + threads - 1) / threads, 2048);
    hipLaunchKernelGGL(k_bgrx_to_planes_420, dim3(blocks), dim3(threads), 0,
                       stream, (const uchar4*)bgrx, width, height, stride_px,
                       yp, cbp, crp, ypitch, cpitch);
  } else {
    int total = width * height;
    int blocks = min((total + threads - 1) / threads, 2048);
    hipLaunchKernelGGL(k_bgrx_to_planes_444, dim3(blocks), dim3(threads), 0,
                       stream, (const uchar4*)bgrx, width, height, stride_px,
                       yp, cbp, crp, ypitch);
  }
}

void launch_dct_quant(const uint8_t* plane, int pw, int ph, int pitch,
                      const float* rq, int16_t* out, int plane_kind,
                      bool fullcolor, int mcux, int mcu_rows_per_stripe,
                      int stripe_mcu_count, hipStream_t stream) {
  int bx = (pw + 7) / 8, by = (ph + 7) / 8;
  int nblocks = bx * by;
  int wgs = (nblocks + 3) / 4;
  hipLaunchKernelGGL(k_dct_quant, dim3(wgs), dim3(256), 0, stream, plane, pw,
                     ph, pitch, rq, out, plane_kind, fullcolor, mcux,
                     mcu_rows_per_stripe, stripe_mcu_count);
}

}  // namespace hipflux
