// GPU CAVLC entropy stage (gfx950).
//
// One wave per MB-row slice. Three phases inside the kernel:
//  A. each lane encodes independent bitstring ITEMS (slice header, MB
//     headers, residual blocks) into 64-byte staging slots — legal because
//     every CAVLC context (nC neighbor totals, skip runs, MV predictors,
//     CBP) is derivable in parallel from the levels/meta buffers without
//     decoding bits;
//  B. a wave prefix-sum over item bit-lengths yields every item's global
//     bit offset;
//  C. lanes scatter their items into the row's output bitstream with
//     atomicOr word writes (bit order: MSB-first within u32 words).
// The CPU keeps only RBSP stop-bit, emulation prevention and NAL wrap.
//
// Bit-exactness contract: identical output to the CPU packer
// (cpu/h264/gpu_entropy.cpp) for the same levels/meta — enforced by
// tests/test_gpu_h264.py::test_gpu_cavlc_matches_cpu.
#include <hip/hip_runtime.h>

#include "cavlc_tables_gpu.h"
#include "h264_gpu_layout.h"
#include "h264_kernels.h"

namespace hipflux {
namespace h264gpu {

__constant__ int c_zig4[16] = {0, 1, 4, 8, 5, 2, 3, 6,
                               9, 12, 13, 10, 7, 11, 14, 15};
// compile-time copy of the zigzag so unrolled loops index registers,
// not memory (a __constant__ lookup defeats the register promotion)
__device__ constexpr int kZig4[16] = {0, 1, 4, 8, 5, 2, 3, 6,
                                      9, 12, 13, 10, 7, 11, 14, 15};

// one 4x4 block's levels via two int4 loads; zigzag applied with
// compile-time indices so everything stays in registers. `first` = 1
// drops position 0 (AC-only blocks).
template <int FIRST>
__device__ __forceinline__ void load_zz16(const int16_t* L, int off,
                                          int* zz) {
  union U {
    int4 v;
    short s[8];
  } a, b;
  a.v = *reinterpret_cast<const int4*>(L + off);
  b.v = *reinterpret_cast<const int4*>(L + off + 8);
#pragma unroll
  for (int i = FIRST; i < 16; ++i) {
    constexpr int kZ[16] = {0, 1, 4, 8, 5, 2, 3, 6,
                            9, 12, 13, 10, 7, 11, 14, 15};
    int pos = kZ[i];
    zz[i - FIRST] = pos < 8 ? a.s[pos] : b.s[pos - 8];
  }
}

// Z-order block -> raster index within the MB
__constant__ int c_zorder_raster[16] = {0, 1, 4, 5, 2, 3, 6, 7,
                                        8, 9, 12, 13, 10, 11, 14, 15};

struct DevBW {
  uint64_t acc = 0;
  int na = 0;
  uint32_t* out;
  int word = 0;
  __device__ void u(uint32_t v, int n) {
    if (n == 0) return;
    acc = (acc << n) |
          (uint64_t)(v & (uint32_t)((n == 32 ? 0ull : (1ull << n)) - 1ull));
    na += n;
    while (na >= 32) {
      out[word++] = (uint32_t)(acc >> (na - 32));
      na -= 32;
    }
  }
  __device__ void ue(uint32_t v) {
    uint32_t cw = v + 1;
    int len = 31 - __clz(cw);
    u(0, len);
    u(cw, len + 1);
  }
  __device__ void se(int v) {
    ue(v > 0 ? 2 * (uint32_t)v - 1 : 2 * (uint32_t)(-v));
  }
  __device__ int flush() {
    int bits = word * 32 + na;
    if (na) out[word++] = (uint32_t)(acc << (32 - na));
    return bits;
  }
};

// ---- residual block encode (mirrors cpu/h264/cavlc.h bit-exactly) ---------
// Streaming over the zigzag values with STATIC trip counts (template N):
// no runtime-indexed arrays -> everything stays in registers (a
// coeffs[]/pos[] formulation spills to scratch and was 2.5x slower).
template <int N>
__device__ __forceinline__ void dev_cavlc_residual(DevBW& bw,
                                                   const int* zz, int nC) {
  // pass 1: tc, trailing ones, last nonzero position
  int tc = 0, t1 = 0, last_pos = -1;
  bool blocked = false;
#pragma unroll
  for (int i = N - 1; i >= 0; --i) {
    int v = zz[i];
    if (v) {
      ++tc;
      if (last_pos < 0) last_pos = i;
      if (!blocked) {
        if ((v == 1 || v == -1) && t1 < 3)
          ++t1;
        else
          blocked = true;
      }
    }
  }
  unsigned packed;
  if (nC == -1)
    packed = c_ctcdc[tc * 4 + t1];
  else if (nC < 2)
    packed = c_ct0[tc * 4 + t1];
  else if (nC < 4)
    packed = c_ct1[tc * 4 + t1];
  else if (nC < 8)
    packed = c_ct2[tc * 4 + t1];
  else
    packed = VLCPACK(6, tc == 0 ? 3 : (((tc - 1) << 2) | t1));
  bw.u(packed & 0xFFFF, packed >> 16);
  if (tc == 0) return;

  // pass 2: trailing-one signs (highest frequency first)
  {
    int seen = 0;
#pragma unroll
    for (int i = N - 1; i >= 0; --i) {
      int v = zz[i];
      if (v) {
        if (seen < t1) bw.u(v > 0 ? 0 : 1, 1);
        ++seen;
      }
    }
  }

  // pass 3: remaining levels (highest frequency first)
  {
    int suffix_len = (tc > 10 && t1 < 3) ? 1 : 0;
    int seen = 0;
#pragma unroll
    for (int i = N - 1; i >= 0; --i) {
      int v = zz[i];
      if (!v) continue;
      ++seen;
      if (seen <= t1) continue;
      const int true_level = v;
      int level = v;
      if (seen == t1 + 1 && t1 < 3) level += level > 0 ? -1 : 1;
      int code = level > 0 ? 2 * level - 2 : -2 * level - 1;
      if (suffix_len == 0) {
        if (code < 14) {
          bw.u(1, code + 1);
        } else if (code < 30) {
          bw.u(1, 15);
          bw.u(code - 14, 4);
        } else {
          bw.u(1, 16);
          bw.u(code - 30, 12);
        }
      } else {
        int prefix = code >> suffix_len;
        if (prefix < 15) {
          bw.u(1, prefix + 1);
          bw.u(code & ((1 << suffix_len) - 1), suffix_len);
        } else {
          bw.u(1, 16);
          bw.u(code - (15 << suffix_len), 12);
        }
      }
      if (suffix_len == 0) suffix_len = 1;
      if (abs(true_level) > (3 << (suffix_len - 1)) && suffix_len < 6)
        ++suffix_len;
    }
  }

  // pass 4: total_zeros
  int total_zeros = last_pos + 1 - tc;
  if (tc < N) {
    unsigned p = (nC == -1) ? c_tzcdc[(tc - 1) * 4 + total_zeros]
                            : c_tz[(tc - 1) * 16 + total_zeros];
    bw.u(p & 0xFFFF, p >> 16);
  }

  // pass 5: run_before (gap below each nonzero, highest first; the lowest
  // nonzero carries no run; stop once all zeros are accounted for)
  {
    int zeros_left = total_zeros;
    int prev_pos = -1;
#pragma unroll
    for (int i = N - 1; i >= 0; --i) {
      if (!zz[i]) continue;
      if (prev_pos >= 0 && zeros_left > 0) {
        int run = prev_pos - i - 1;
        int zl = zeros_left < 7 ? zeros_left : 7;
        unsigned p = c_rb[(zl - 1) * 15 + run];
        bw.u(p & 0xFFFF, p >> 16);
        zeros_left -= run;
      }
      prev_pos = i;
    }
  }
}

// ---- per-MB derived state ---------------------------------------------------
// totals of one 16-coeff block via two int4 (8 x int16) loads; `first`
// is 0 (full block) or 1 (AC-only: position 0 excluded)
__device__ inline int blk_total16v(const int16_t* L, int off, int first) {
  union U { int4 v; short s[8]; };
  U a, b;
  a.v = *reinterpret_cast<const int4*>(L + off);
  b.v = *reinterpret_cast<const int4*>(L + off + 8);
  int t = 0;
#pragma unroll
  for (int i = 0; i < 8; ++i) t += (a.s[i] != 0) + (b.s[i] != 0);
  if (first) t -= a.s[0] != 0;
  return t;
}

// coded_block_pattern me(v) codeNum, Inter column of Table 9-4 — only
// the six values our encoder emits (see cpu/h264/encoder.cpp).
__device__ inline int dev_inter_cbp_codenum(int cbp) {
  switch (cbp) {
    case 0: return 0;
    case 16: return 1;
    case 32: return 6;
    case 15: return 11;
    case 47: return 12;
    default: return 19;  // 31
  }
}

// Per-MB info, precomputed once per row into LDS:
//   flags: mode(2) | luma_mode<<2 | chroma_mode<<5 | cbp_luma1<<8 |
//          cbp_chroma<<9
//   totals: 16 luma-AC + 8 chroma-AC per-block nonzero counts
struct MbInfo {
  short flags;
  short mvx, mvy;
  uint8_t ltot[16];
  uint8_t ctot[8];
};

#define MB_MODE(f) ((f) & 3)
#define MB_LMODE(f) (((f) >> 2) & 7)
#define MB_CMODE(f) (((f) >> 5) & 7)
#define MB_CBPL(f) (((f) >> 8) & 1)
#define MB_CBPC(f) (((f) >> 9) & 3)

__device__ void precompute_mb(const int16_t* levels, const int* meta,
                              size_t mb_index, bool i_slice, MbInfo* out) {
  int m0 = meta[mb_index * kMetaPerMb + 0];
  int mode = i_slice ? kIntra : (m0 & 3);
  int m1 = meta[mb_index * kMetaPerMb + 1];
  out->mvx = (short)(m1 & 0xFFFF);
  out->mvy = (short)(m1 >> 16);
  int cbp_luma1 = 0, cbp_chroma = 0;
  if (mode == kIntra || mode == kInter) {
    // intra: luma totals over the 15 AC coeffs; inter: full 16-coeff
    // blocks (no DC Hadamard for inter residuals)
    const int first = (mode == kIntra) ? 1 : 0;
    const int16_t* L = levels + mb_index * kLevelsPerMb;
    int any_ac = 0;
    for (int b = 0; b < 16; ++b) {
      int t = blk_total16v(L, kLumaAcOff + b * 16, first);
      out->ltot[b] = (uint8_t)t;
      any_ac |= t;
    }
    int any_cac = 0, any_cdc = 0;
    for (int b = 0; b < 8; ++b) {
      int t = blk_total16v(L, kChromaAcOff + b * 16, 1);
      out->ctot[b] = (uint8_t)t;
      any_cac |= t;
    }
    for (int i = 0; i < 8; ++i) any_cdc |= L[kChromaDcOff + i] != 0;
    cbp_luma1 = any_ac ? 1 : 0;
    cbp_chroma = any_cac ? 2 : (any_cdc ? 1 : 0);
  } else {
    for (int b = 0; b < 16; ++b) out->ltot[b] = 0;
    for (int b = 0; b < 8; ++b) out->ctot[b] = 0;
  }
  // chroma mode rides in meta word 1 for intra MBs (see k_h264_rows)
  int cmode = (mode == kIntra) ? (m1 & 7) : 0;
  out->flags = (short)(mode | (((m0 >> 2) & 7) << 2) | (cmode << 5) |
                       (cbp_luma1 << 8) | (cbp_chroma << 9));
}

__device__ inline int lds_luma_nc(const MbInfo* info, int mbx, int bx,
                                  int by) {
  if (bx > 0) {
    if (!MB_CBPL(info[mbx].flags)) return 0;
    return info[mbx].ltot[by * 4 + (bx - 1)];
  }
  if (mbx == 0) return 0;
  const MbInfo& l = info[mbx - 1];
  if (!MB_CBPL(l.flags)) return 0;  // skip/uncoded neighbors -> 0 totals
  return l.ltot[by * 4 + 3];
}

__device__ inline int lds_chroma_nc(const MbInfo* info, int mbx, int comp,
                                    int cx, int cy) {
  if (cx > 0) {
    if (MB_CBPC(info[mbx].flags) != 2) return 0;
    return info[mbx].ctot[comp * 4 + cy * 2];
  }
  if (mbx == 0) return 0;
  const MbInfo& l = info[mbx - 1];
  if (MB_CBPC(l.flags) != 2) return 0;
  return l.ctot[comp * 4 + cy * 2 + 1];
}

// ---- the kernel -------------------------------------------------------------
__global__ void __launch_bounds__(1024) k_h264_cavlc_rows(
    const int16_t* __restrict__ levels, const int* __restrict__ meta,
    int mbw, const RowJob* __restrict__ jobs,
    uint32_t* __restrict__ stage,       // [row][item][kStageWordsPerItem]
    int* __restrict__ nbits,            // [row][item]
    uint32_t* __restrict__ out,         // [row][out_stride_words]
    int out_stride_words,
    int* __restrict__ out_bits) {       // [row]
  const RowJob job = jobs[blockIdx.x];
  const int tid = threadIdx.x;
  const int NT = blockDim.x;   // 512 (many blocks: throughput regime)
                               // or 1024 (few blocks: latency regime)
  const bool i_slice = (job.flags & 1) != 0;
  const int seg_mbw = job.seg_mbw;
  const size_t row_base = (size_t)job.mb_row * mbw + job.mbx0;
  const int nitems = items_per_row(seg_mbw);
  uint32_t* row_stage =
      stage + (size_t)blockIdx.x * nitems * kStageWordsPerItem;
  int* row_nbits = nbits + (size_t)blockIdx.x * nitems;
  uint32_t* row_out = out + (size_t)blockIdx.x * out_stride_words;

  // ---- precompute per-MB info into LDS (totals, cbp, modes, mvs)
  __shared__ MbInfo s_mb[kMaxSegMbw];
  __shared__ short s_skiprun[kMaxSegMbw];
  __shared__ int s_trailing;
  for (int mb = tid; mb < seg_mbw; mb += NT)
    precompute_mb(levels, meta, row_base + mb, i_slice, &s_mb[mb]);
  __syncthreads();
  if (tid == 0) {
    int run = 0;
    for (int mb = 0; mb < seg_mbw; ++mb) {
      if (!i_slice && MB_MODE(s_mb[mb].flags) == kSkip) {
        s_skiprun[mb] = -1;  // skipped MB: no items
        ++run;
      } else {
        s_skiprun[mb] = (short)run;
        run = 0;
      }
    }
    s_trailing = run;
  }
  __syncthreads();

  // ---- phase A: encode items
  for (int item = tid; item < nitems; item += NT) {
    DevBW bw;
    bw.out = row_stage + (size_t)item * kStageWordsPerItem;
    int bits = 0;
    if (item == 0) {
      // slice header
      bw.ue(job.first_mb);
      bw.ue(i_slice ? 7 : 5);
      bw.ue(0);
      bw.u(job.frame_num & 0xFFFF, 16);
      if (i_slice) bw.ue(job.idr_pic_id);
      if (!i_slice) {
        bw.u(0, 1);
        bw.u(0, 1);
      }
      if (i_slice) {
        bw.u(0, 1);
        bw.u(0, 1);
      } else {
        bw.u(0, 1);
      }
      bw.se(job.qp - 26);
      bw.ue((job.flags & 2) ? 2 : 1);  // disable_deblocking_filter_idc
      bits = bw.flush();
    } else if (item == nitems - 1) {
      if (!i_slice && s_trailing > 0) bw.ue(s_trailing);
      bits = bw.flush();
    } else {
      int mb = (item - 1) / kSlotsPerMb;
      int slot = (item - 1) % kSlotsPerMb;
      const MbInfo& m = s_mb[mb];
      const int mode = MB_MODE(m.flags);
      const int cbp_chroma = MB_CBPC(m.flags);
      const int16_t* L = levels + (row_base + mb) * kLevelsPerMb;
      if (s_skiprun[mb] < 0) {
        bits = 0;  // skipped MB
      } else if (slot == 0) {
        if (!i_slice) bw.ue(s_skiprun[mb]);
        if (mode == kInter) {
          bw.ue(0);  // P_L0_16x16
          int mvpx = 0, mvpy = 0;
          if (mb > 0 && MB_MODE(s_mb[mb - 1].flags) != kIntra) {
            mvpx = s_mb[mb - 1].mvx;   // skip or inter both carry MVs
            mvpy = s_mb[mb - 1].mvy;
          }
          bw.se(m.mvx - mvpx);
          bw.se(m.mvy - mvpy);
          int cbp = (cbp_chroma << 4) | (MB_CBPL(m.flags) ? 15 : 0);
          bw.ue(dev_inter_cbp_codenum(cbp));
          if (cbp) bw.se(0);  // mb_qp_delta
        } else {
          int i16 = 1 + MB_LMODE(m.flags) + 4 * cbp_chroma +
                    12 * MB_CBPL(m.flags);
          bw.ue(i_slice ? i16 : 5 + i16);
          bw.ue(MB_CMODE(m.flags));
          bw.se(0);  // mb_qp_delta
        }
        bits = bw.flush();
      } else if (mode == kSkip) {
        bits = 0;
      } else if (slot == 1) {
        // luma DC (intra only; inter has no DC Hadamard)
        if (mode == kIntra) {
          int zz[16];
          load_zz16<0>(L, kLumaDcOff, zz);
          int nC = lds_luma_nc(s_mb, mb, 0, 0);
          dev_cavlc_residual<16>(bw, zz, nC);
          bits = bw.flush();
        }
      } else if (slot < 18) {
        if (MB_CBPL(m.flags)) {
          int blk = slot - 2;                      // Z-order index
          int r = c_zorder_raster[blk];
          int bx = r & 3, by = r >> 2;
          int nC = lds_luma_nc(s_mb, mb, bx, by);
          if (mode == kIntra) {
            int zz[15];
            load_zz16<1>(L, kLumaAcOff + r * 16, zz);
            dev_cavlc_residual<15>(bw, zz, nC);
          } else {
            int zz[16];
            load_zz16<0>(L, kLumaAcOff + r * 16, zz);
            dev_cavlc_residual<16>(bw, zz, nC);
          }
          bits = bw.flush();
        }
      } else if (slot < 20) {
        if (cbp_chroma > 0) {
          int comp = slot - 18;
          int zz[4];
          for (int i = 0; i < 4; ++i) zz[i] = L[kChromaDcOff + comp * 4 + i];
          dev_cavlc_residual<4>(bw, zz, -1);
          bits = bw.flush();
        }
      } else {
        if (cbp_chroma == 2) {
          int b = slot - 20;                       // cb0..3 then cr0..3
          int comp = b >> 2, sub = b & 3;
          int zz[15];
          load_zz16<1>(L, kChromaAcOff + (comp * 4 + sub) * 16, zz);
          int nC = lds_chroma_nc(s_mb, mb, comp, sub & 1, sub >> 1);
          dev_cavlc_residual<15>(bw, zz, nC);
          bits = bw.flush();
        }
      }
    }
    row_nbits[item] = bits;
  }
  __syncthreads();

  // ---- phase B: exclusive prefix sum via per-thread segments
  const int seg_lo = (int)((long)nitems * tid / NT);
  const int seg_hi = (int)((long)nitems * (tid + 1) / NT);
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
    __shared__ int s_wsum[16];
    if (lane == 63) s_wsum[wid] = inc;
    __syncthreads();
    int wbase = 0;
    for (int w = 0; w < wid; ++w) wbase += s_wsum[w];
    int seg_off = wbase + inc - seg_sum;
    if (tid == NT - 1) s_total = wbase + inc;
    // rewrite nbits -> offsets for this thread's segment
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

  // ---- zero the needed output words, then scatter
  int total_words = (total_bits + 31) / 32 + 1;
  for (int wdx = tid; wdx < total_words; wdx += NT) row_out[wdx] = 0;
  __syncthreads();

  for (int item = tid; item < nitems; item += NT) {
    int off = row_nbits[item];
    int next_off = item + 1 < nitems
                       ? row_nbits[item + 1]
                       : total_bits;
    int bits = next_off - off;
    if (bits <= 0) continue;
    // Build each destination word fully (carry accumulator across source
    // words), then: plain store for interior words, atomicOr only at the
    // first/last destination word (the only ones a neighbor item shares).
    const uint32_t* src = row_stage + (size_t)item * kStageWordsPerItem;
    int nwords = (bits + 31) / 32;
    int shift = off & 31;
    int w0 = off >> 5;
    int last_dst = (off + bits - 1) >> 5;
    uint32_t carry = 0;
    for (int k = 0; k < nwords; ++k) {
      uint32_t w = src[k];
      if (k == nwords - 1 && (bits & 31))
        w &= ~((1u << (32 - (bits & 31))) - 1u);  // mask tail garbage
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

void launch_h264_cavlc(const int16_t* d_levels, const int* d_meta, int mbw,
                       int n_jobs, const RowJob* d_jobs, uint32_t* d_stage,
                       int* d_nbits, uint32_t* d_out, int out_stride_words,
                       int* d_out_bits, hipStream_t stream) {
  if (n_jobs == 0) return;
  // few blocks (one 1024-thread workgroup per CU fits up to the 256-CU
  // count, per-row latency-bound): widest blocks shorten each row's
  // serial item chain. Many blocks (throughput-bound, e.g. 4K/8K):
  // 512-thread blocks co-reside 3-per-CU and win on occupancy
  // (measured: 1024 threads = +16% @1080p but -13% @8K).
  const int nt = n_jobs <= 256 ? 1024 : 512;
  hipLaunchKernelGGL(k_h264_cavlc_rows, dim3(n_jobs), dim3(nt), 0, stream,
                     d_levels, d_meta, mbw, d_jobs, d_stage, d_nbits, d_out,
                     out_stride_words, d_out_bits);
}

}  // namespace h264gpu
}  // namespace hipflux
