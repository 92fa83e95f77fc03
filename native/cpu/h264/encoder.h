// H.264 stripe encoder — CPU reference implementation.
//
// Design (chosen for GPU-parallelism, not translated from any reference
// encoder):
//  * Constrained Baseline profile, CAVLC, progressive, 4:2:0.
//  * ONE SLICE PER MACROBLOCK ROW: rows share no intra/MV/nC state, so rows
//    encode (and on the HIP path, reconstruct) fully in parallel; only
//    left-neighbor dependencies remain inside a row.
//  * I slices: I16x16 only (H / DC prediction; V needs the top row, which is
//    in another slice). P slices: P_Skip, P_L0_16x16 with zero residual and
//    even-integer MVs (chroma MC stays integer), I16x16 fallback.
//  * Deblocking disabled per slice header (disable_deblocking_filter_idc=1)
//    and matching encoder reconstruction.
//  * Infinite GOP: IDR only on demand (keyframe_interval handled upstream).
//
// Each stripe of the striped encoder is one instance (its own bitstream,
// frame_num, recon state) — the stripe-parallel seam described in
// SURVEY.md §5.7.
#pragma once

#include <cstdint>
#include <memory>
#include <vector>

namespace hipflux {
namespace h264 {

struct EncodeStats {
  int frame_qp = 0;
  bool is_idr = false;
  int mb_intra = 0, mb_inter = 0, mb_skip = 0;
  size_t bytes = 0;
};

class StripeEncoder {
 public:
  // fullcolor: Hi444 (profile 244) with separate_colour_plane_flag=1 --
  // each plane is coded as monochrome through the same luma machinery
  // (the reference's pixelflux `h264_fullcolor` / I444 streaming mode);
  // encode_frame then takes FULL-resolution cb/cr planes with `cpitch`.
  StripeEncoder(int width, int height, bool deblock = true,
                bool fullcolor = false);
  ~StripeEncoder();

  // Encode one frame from planar YUV420 (or I444 when fullcolor; pitch
  // in bytes). qp in [0,51]. force_idr resets the stream (SPS/PPS +
  // IDR). Appends Annex-B to `out`.
  void encode_frame(const uint8_t* y, int ypitch, const uint8_t* cb,
                    const uint8_t* cr, int cpitch, int qp, bool force_idr,
                    std::vector<uint8_t>& out, EncodeStats* stats = nullptr);

  bool fullcolor() const { return fullcolor_; }

  // Access to the reconstructed reference (for tests and drift checks).
  const uint8_t* recon_y() const;
  const uint8_t* recon_cb() const;
  const uint8_t* recon_cr() const;
  int recon_ypitch() const;
  int recon_cpitch() const;

  int width() const { return width_; }
  int height() const { return height_; }

 private:
  struct Impl;
  std::unique_ptr<Impl> impl_, impl_cb_, impl_cr_;
  int width_, height_;
  bool fullcolor_ = false;
};

// BGRX -> planar YUV420 (BT.601 full range), edge-replicated to MB-aligned
// planes. Shared scalar reference for the HIP CSC kernel.
void bgrx_to_yuv420(const uint8_t* bgrx, int stride, int width, int height,
                    uint8_t* y, int ypitch, uint8_t* cb, uint8_t* cr,
                    int cpitch);

// BGRX -> planar I444 (BT.601 full range), for the Hi444 fullcolor mode.
void bgrx_to_yuv444(const uint8_t* bgrx, int stride, int width, int height,
                    uint8_t* y, uint8_t* cb, uint8_t* cr, int pitch);

}  // namespace h264
}  // namespace hipflux
