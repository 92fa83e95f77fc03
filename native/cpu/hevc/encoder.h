// HEVC stripe encoder — CPU reference implementation (BASELINE config 3).
//
// Design (GPU-parallel-first, mirroring the H.264 engine's proven shape —
// see native/cpu/h264/encoder.h; no reference-project counterpart exists,
// pixelflux has no HEVC):
//  * Main profile, all-intra (every frame IDR), CABAC, 4:2:0, bit depth 8.
//  * CTU = CU = PU = 16x16, one 16x16 luma TB + two 8x8 chroma TBs, no
//    residual quadtree. This keeps the flat per-16px-row geometry the
//    stripe wire format and the HIP row-wavefront kernels are built
//    around, and needs only the well-conditioned T16/T8 transforms.
//  * ONE SLICE SEGMENT PER CTU ROW (optionally split into segments):
//    rows share no prediction or CABAC state, so rows encode and
//    reconstruct fully in parallel; only left-neighbor dependencies
//    remain inside a row.
//  * Intra modes Planar/DC/H/V with spec MPM signaling; deblocking and
//    SAO disabled (signaled off in PPS/SPS).
//  * Each stripe is an independent bitstream (own VPS/SPS/PPS + IDR per
//    emission) — the stripe-parallel seam of SURVEY.md §5.7.
#pragma once

#include <cstdlib>

#include <cstdint>
#include <memory>
#include <vector>

namespace hipflux {
namespace hevc {

struct EncodeStats {
  int frame_qp = 0;
  int ctu_count = 0;
  size_t bytes = 0;
};

// Slice segments per CTU row, shared by the CPU and HIP pipelines so their
// bitstreams stay byte-identical. More segments shorten the serial
// left-neighbor chain (GPU wavefront) and multiply CABAC parallelism, at
// ~0.1-0.5% bitrate cost per extra segment boundary.
inline int default_slices_per_row(int width) {
  // HIPFLUX_HEVC_SPR overrides for tuning runs (CABAC jobs are serial
  // per slice segment, so more segments = shorter GPU critical path at
  // ~10 bytes/slice header cost)
  if (const char* e = std::getenv("HIPFLUX_HEVC_SPR")) {
    int v = std::atoi(e);
    if (v >= 1 && v <= 256) return v;
  }
  // measured on MI355X (profiles/NOTES.md): the CABAC critical path
  // is serial per segment, fps plateaus at ~32 segments/row at 4K
  // for ~2% per-frame bitrate cost
  // ~120 px per segment at 4K+ (32 segments at 3840, 64 at 7680)
  if (width >= 3840) return width / 120;
  return width >= 2560 ? 8 : width >= 1280 ? 2 : 1;
}

class StripeEncoder {
 public:
  // width/height: visible stripe dims (coded dims round up to 16 with a
  // conformance-window crop). slices_per_row splits each CTU row into
  // independent slice segments for entropy parallelism.
  StripeEncoder(int width, int height, int slices_per_row = 1);
  ~StripeEncoder();

  // Encode one frame from planar YUV420 (pitch in bytes, planes at least
  // coded-size with any padding content; qp in [0,51]). Appends Annex-B
  // (VPS+SPS+PPS+IDR slices) to `out`.
  void encode_frame(const uint8_t* y, int ypitch, const uint8_t* cb,
                    const uint8_t* cr, int cpitch, int qp,
                    std::vector<uint8_t>& out, EncodeStats* stats = nullptr);

  const uint8_t* recon_y() const;
  const uint8_t* recon_cb() const;
  const uint8_t* recon_cr() const;
  int recon_ypitch() const;
  int recon_cpitch() const;

  int width() const { return width_; }
  int height() const { return height_; }

 private:
  struct Impl;
  std::unique_ptr<Impl> impl_;
  int width_, height_;
};

}  // namespace hevc
}  // namespace hipflux
