// GPU JPEG Huffman entropy (gfx950) — the CAVLC machinery applied to JFIF.
//
// One workgroup per MCU ROW. Restart markers between rows (DRI = mcux)
// make rows independent restart intervals, so every row packs in
// parallel; within a row each BLOCK's bitstring is encodable in parallel
// because the DC predictor is derivable straight from the coefficient
// buffer (previous block's DC of the same component). Phases:
//   A. each lane Huffman-encodes whole blocks into 224-byte staging slots
//   B. segment prefix-scan of bit lengths
//   C. carry-accumulator scatter, atomics only at item-boundary words
// The host adds byte padding (1-fill), 0xFF00 stuffing, RSTn markers and
// the JFIF headers (cpu/jpeg_enc.cpp writes the same framing for the
// byte-equality test).
#include <hip/hip_runtime.h>

#include "jpeg_kernels.h"

namespace hipflux {
namespace jpeggpu {

// zigzag scan for 8x8 (index -> natural position), JPEG standard order
__device__ constexpr int kZig8[64] = {
    0,  1,  8,  16, 9,  2,  3,  10, 17, 24, 32, 25, 18, 11, 4,  5,
    12, 19, 26, 33, 40, 48, 41, 34, 27, 20, 13, 6,  7,  14, 21, 28,
    35, 42, 49, 56, 57, 50, 43, 36, 29, 22, 15, 23, 30, 37, 44, 51,
    58, 59, 52, 45, 38, 31, 39, 46, 53, 60, 61, 54, 47, 55, 62, 63};

struct DevBW {
  uint64_t acc = 0;
  int na = 0;
  uint32_t* out;
  int word = 0;
  __device__ void u(uint32_t v, int n) {
    if (n == 0) return;
    acc = (acc << n) | (uint64_t)(v & ((n == 32 ? 0u : (1u << n)) - 1u));
    na += n;
    while (na >= 32) {
      out[word++] = (uint32_t)(acc >> (na - 32));
      na -= 32;
    }
  }
  __device__ int flush() {
    int bits = word * 32 + na;
    if (na) out[word++] = (uint32_t)(acc << (32 - na));
    return bits;
  }
};

__device__ inline int bitlen(int v) { return 32 - __clz(v); }

// encode one block: tables packed code | (size << 16)
__device__ void enc_block(DevBW& bw, const int16_t* blk, int dc_pred,
                          const uint32_t* dct, const uint32_t* act) {
  union U {
    int4 v;
    short s[8];
  } u[8];
#pragma unroll
  for (int i = 0; i < 8; ++i)
    u[i].v = reinterpret_cast<const int4*>(blk)[i];

  int diff = u[0].s[0] - dc_pred;
  int mag = diff < 0 ? -diff : diff;
  int nb = bitlen(mag);
  uint32_t t = dct[nb];
  bw.u(t & 0xFFFF, t >> 16);
  if (nb) bw.u(diff < 0 ? diff + ((1 << nb) - 1) : diff, nb);

  int run = 0;
#pragma unroll
  for (int i = 1; i < 64; ++i) {
    constexpr int kZ[64] = {
        0,  1,  8,  16, 9,  2,  3,  10, 17, 24, 32, 25, 18, 11, 4,  5,
        12, 19, 26, 33, 40, 48, 41, 34, 27, 20, 13, 6,  7,  14, 21, 28,
        35, 42, 49, 56, 57, 50, 43, 36, 29, 22, 15, 23, 30, 37, 44, 51,
        58, 59, 52, 45, 38, 31, 39, 46, 53, 60, 61, 54, 47, 55, 62, 63};
    int pos = kZ[i];
    int v = u[pos >> 3].s[pos & 7];
    if (v == 0) {
      ++run;
      continue;
    }
    while (run >= 16) {
      uint32_t z = act[0xF0];
      bw.u(z & 0xFFFF, z >> 16);
      run -= 16;
    }
    int m = v < 0 ? -v : v;
    int nb2 = bitlen(m);
    uint32_t a = act[(run << 4) | nb2];
    bw.u(a & 0xFFFF, a >> 16);
    bw.u(v < 0 ? v + ((1 << nb2) - 1) : v, nb2);
    run = 0;
  }
  if (run) {
    uint32_t e = act[0x00];
    bw.u(e & 0xFFFF, e >> 16);
  }
}

__global__ void __launch_bounds__(256) k_jpeg_entropy_rows(
    const int16_t* __restrict__ coeff, const JRow* __restrict__ jobs,
    int mcux, int per_mcu, const uint32_t* __restrict__ tabs,
    uint32_t* __restrict__ stage, int* __restrict__ nbits,
    uint32_t* __restrict__ out, int out_stride_words,
    int* __restrict__ out_bits) {
  const int tid = threadIdx.x;
  const JRow job = jobs[blockIdx.x];
  const int16_t* row = coeff + job.coeff_off;
  const int nitems = mcux * per_mcu;
  const uint32_t* dcl = tabs;
  const uint32_t* acl = tabs + 12;
  const uint32_t* dcc = tabs + 12 + 256;
  const uint32_t* acc = tabs + 12 + 256 + 12;
  uint32_t* row_stage = stage + (size_t)blockIdx.x * nitems * kJStageWords;
  int* row_nbits = nbits + (size_t)blockIdx.x * nitems;
  uint32_t* row_out = out + (size_t)blockIdx.x * out_stride_words;

  // ---- phase A: each item = one block
  for (int item = tid; item < nitems; item += 256) {
    int m = item / per_mcu, slot = item % per_mcu;
    const int16_t* blk = row + ((size_t)m * per_mcu + slot) * 64;
    // DC predictor: previous block of the same component in scan order
    int pred = 0;
    if (per_mcu == 6) {                     // 4:2:0 (Y0..Y3, Cb, Cr)
      if (slot > 0 && slot < 4)
        pred = row[((size_t)m * per_mcu + slot - 1) * 64];
      else if (slot == 0 && m > 0)
        pred = row[((size_t)(m - 1) * per_mcu + 3) * 64];
      else if (slot >= 4 && m > 0)
        pred = row[((size_t)(m - 1) * per_mcu + slot) * 64];
    } else {                                // 4:4:4 (Y, Cb, Cr)
      if (m > 0) pred = row[((size_t)(m - 1) * per_mcu + slot) * 64];
    }
    const bool luma = per_mcu == 6 ? slot < 4 : slot == 0;
    DevBW bw;
    bw.out = row_stage + (size_t)item * kJStageWords;
    enc_block(bw, blk, pred, luma ? dcl : dcc, luma ? acl : acc);
    row_nbits[item] = bw.flush();
  }
  __syncthreads();

  // ---- phase B: exclusive prefix sum (4 waves)
  const int seg_lo = (int)((long)nitems * tid / 256);
  const int seg_hi = (int)((long)nitems * (tid + 1) / 256);
  int seg_sum = 0;
  for (int i = seg_lo; i < seg_hi; ++i) seg_sum += row_nbits[i];
  __shared__ int s_total;
  {
    const int lane = tid & 63, wid = tid >> 6;
    int inc = seg_sum;
    for (int d = 1; d < 64; d <<= 1) {
      int other = __shfl_up(inc, d);
      if (lane >= d) inc += other;
    }
    __shared__ int s_wsum[4];
    if (lane == 63) s_wsum[wid] = inc;
    __syncthreads();
    int wbase = 0;
    for (int w = 0; w < wid; ++w) wbase += s_wsum[w];
    int seg_off = wbase + inc - seg_sum;
    if (tid == 255) s_total = wbase + inc;
    int off = seg_off;
    for (int i = seg_lo; i < seg_hi; ++i) {
      int t = row_nbits[i];
      row_nbits[i] = off;
      off += t;
    }
    __syncthreads();
  }
  int total_bits = s_total;
  if (tid == 0) out_bits[blockIdx.x] = total_bits;

  int total_words = (total_bits + 31) / 32 + 1;
  for (int wdx = tid; wdx < total_words; wdx += 256) row_out[wdx] = 0;
  __syncthreads();

  // ---- phase C: scatter
  for (int item = tid; item < nitems; item += 256) {
    int off = row_nbits[item];
    int next_off = item + 1 < nitems ? row_nbits[item + 1] : total_bits;
    int bits = next_off - off;
    if (bits <= 0) continue;
    const uint32_t* src = row_stage + (size_t)item * kJStageWords;
    int nwords = (bits + 31) / 32;
    int shift = off & 31;
    int w0 = off >> 5;
    int last_dst = (off + bits - 1) >> 5;
    uint32_t carry = 0;
    for (int k = 0; k < nwords; ++k) {
      uint32_t w = src[k];
      if (k == nwords - 1 && (bits & 31))
        w &= ~((1u << (32 - (bits & 31))) - 1u);
      uint32_t val = shift ? (carry | (w >> shift)) : w;
      int d = w0 + k;
      if (d == w0 || d == last_dst)
        atomicOr(&row_out[d], val);
      else
        row_out[d] = val;
      carry = shift ? (w << (32 - shift)) : 0;
    }
    if (shift && (w0 + nwords) <= last_dst)
      atomicOr(&row_out[w0 + nwords], carry);
  }
}

void launch_jpeg_entropy(const int16_t* d_coeff, const JRow* d_jobs,
                         int n_rows, int mcux, int per_mcu,
                         const uint32_t* d_tabs, uint32_t* d_stage,
                         int* d_nbits, uint32_t* d_out,
                         int out_stride_words, int* d_out_bits,
                         hipStream_t stream) {
  if (n_rows == 0) return;
  hipLaunchKernelGGL(k_jpeg_entropy_rows, dim3(n_rows), dim3(256), 0,
                     stream, d_coeff, d_jobs, mcux, per_mcu, d_tabs,
                     d_stage, d_nbits, d_out, out_stride_words, d_out_bits);
}

}  // namespace jpeggpu
}  // namespace hipflux
