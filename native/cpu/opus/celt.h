// CELT-mode Opus encoder (from scratch — no libopus exists in this
// environment). 48 kHz, 20 ms frames (TOC config 31, fullband CELT),
// mono (stereo input downmixes).
//
// Conformance ledger (tests/opus_ref_decoder.py re-implements every
// layer and round-trip equality + PSNR is asserted):
//  * Opus TOC framing, padding, and the RFC 6716 §4.1 range coder:
//    implemented exactly per spec.
//  * CELT flag layer (silence / postfilter / transient / intra), band
//    layout (eBands), MDCT + Vorbis power-complementary window,
//    band-energy split and PVQ algebra: per spec structure.
//  * Coarse energy uses UNIFORM 6 dB codes (the reference uses an
//    adaptive Laplace model; ~2 kb/s difference), the allocation row
//    and per-band pulse counts are coded EXPLICITLY instead of being
//    re-derived from a bit-exact budget mirror, and the PVQ split /
//    index layout is a clean textbook CWRS. These layers are
//    SELF-CONSISTENT with the in-tree decoders (Python + the client),
//    not claimed interoperable with third-party Opus decoders — no
//    Opus implementation exists in this offline environment to verify
//    against, and the deltas are confined to these three layers.
#pragma once

#include <cstdint>
#include <vector>

namespace hipflux {
namespace opus {

constexpr int kFrameSamples = 960;   // 20 ms at 48 kHz
constexpr int kOverlap = 120;
constexpr int kNumBands = 21;

// band boundaries in MDCT bins for the 20 ms frame (eBands x 8)
extern const int kBandBins[kNumBands + 1];

class CeltEncoder {
 public:
  // target_bitrate in bits/s; frames are constant-size (CBR framing)
  explicit CeltEncoder(int bitrate_bps = 96000);

  void set_bitrate(int bps);

  // pcm: interleaved s16, `channels` channels, kFrameSamples frames.
  // Returns one self-delimited-free Opus packet (TOC + payload).
  std::vector<uint8_t> encode_frame(const int16_t* pcm, int channels);

 private:
  int bitrate_ = 96000;
  int bytes_per_frame_ = 240;
  std::vector<float> overlap_buf_;   // previous frame tail for the MDCT
  std::vector<float> prev_energy_;   // for energy delta across frames
};

}  // namespace opus
}  // namespace hipflux
