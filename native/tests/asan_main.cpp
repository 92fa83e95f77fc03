// Sanitizer harness for the host-side codec paths (ASan + UBSan).
// the reference has no native sanitizer coverage (pure
// Python + external Rust); the HIP engine's host code gets real ASan
// coverage here. Exercises the H.264 stripe encoder (all mode paths,
// odd sizes, QP extremes), the JPEG encoder, the box downscaler and the
// NAL assembly helpers with pseudo-random inputs; any heap error or UB
// aborts with a nonzero exit.
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <vector>

#include "../cpu/h264/encoder.h"
#include "../cpu/h264/gpu_entropy.h"
#include "../cpu/jpeg_enc.h"
#include "../cpu/opus/celt.h"
#include "../cpu/scale.h"

using namespace hipflux;

static uint64_t rng_state = 0x9E3779B97F4A7C15ull;
static uint32_t rnd() {
  rng_state ^= rng_state << 13;
  rng_state ^= rng_state >> 7;
  rng_state ^= rng_state << 17;
  return static_cast<uint32_t>(rng_state);
}

int main() {
  // H.264: odd sizes, QP extremes, IDR + P chains with motion
  const int sizes[][2] = {{64, 48}, {100, 52}, {176, 144}, {33, 17}};
  for (auto& wh : sizes) {
    int w = wh[0], h = wh[1];
    h264::StripeEncoder enc(w, h);
    std::vector<uint8_t> frame(static_cast<size_t>(w) * h * 4);
    for (int f = 0; f < 6; ++f) {
      for (auto& b : frame) b = static_cast<uint8_t>(rnd());
      std::vector<uint8_t> y(static_cast<size_t>(w) * h),
          cb(static_cast<size_t>((w + 1) / 2) * ((h + 1) / 2)),
          cr(cb.size());
      h264::bgrx_to_yuv420(frame.data(), w * 4, w, h, y.data(), w,
                           cb.data(), cr.data(), (w + 1) / 2);
      std::vector<uint8_t> out;
      int qp = f == 0 ? 0 : (f == 1 ? 51 : 10 + (rnd() % 35));
      enc.encode_frame(y.data(), w, cb.data(), cr.data(), (w + 1) / 2, qp,
                       f == 0, out, nullptr);
      if (out.empty()) {
        std::fprintf(stderr, "empty stream at %dx%d f%d\n", w, h, f);
        return 1;
      }
    }
  }
  // JPEG both chroma modes
  for (int fullcolor = 0; fullcolor < 2; ++fullcolor) {
    int w = 120, h = 66;
    std::vector<uint8_t> frame(static_cast<size_t>(w) * h * 4);
    for (auto& b : frame) b = static_cast<uint8_t>(rnd());
    std::vector<uint8_t> out;
    jpeg_encode_bgrx(frame.data(), w * 4, w, h, 5 + (rnd() % 90),
                     fullcolor, out);
    if (out.size() < 100) return 2;
  }
  // box downscale all divisors
  for (int div = 1; div <= 4; ++div) {
    int w = 97, h = 61;
    std::vector<uint8_t> src(static_cast<size_t>(w) * h * 4);
    for (auto& b : src) b = static_cast<uint8_t>(rnd());
    std::vector<uint8_t> out;
    int ow, oh, ostride;
    box_downscale_bgrx(src.data(), w * 4, w, h, div, out, ow, oh, ostride);
    if (ow != w / div || oh != h / div) return 3;
  }
  // fractional bilinear downscale across awkward ratios
  for (float sc : {0.26f, 0.5f, 0.667f, 0.99f}) {
    int w = 103, h = 57;
    std::vector<uint8_t> src(static_cast<size_t>(w) * h * 4);
    for (auto& b : src) b = static_cast<uint8_t>(rnd());
    std::vector<uint8_t> out;
    int ow, oh, ostride;
    bilinear_downscale_bgrx(src.data(), w * 4, w, h, sc, out, ow, oh,
                            ostride);
    if (ow <= 0 || oh <= 0) return 4;
  }
  // Hi444 fullcolor encode (mono planes x3, IDR + P with residual
  // fallback) under the sanitizers
  {
    int w = 80, h = 48;
    h264::StripeEncoder enc(w, h, true, true);
    std::vector<uint8_t> y(static_cast<size_t>(w) * h),
        cb(y.size()), cr(y.size());
    for (int f = 0; f < 3; ++f) {
      for (auto& v : y) v = static_cast<uint8_t>(rnd());
      for (auto& v : cb) v = static_cast<uint8_t>(rnd());
      for (auto& v : cr) v = static_cast<uint8_t>(rnd());
      std::vector<uint8_t> out;
      enc.encode_frame(y.data(), w, cb.data(), cr.data(), w, 24, f == 0,
                       out);
      if (out.empty()) return 5;
    }
  }
  // Opus encode (range coder + MDCT + PVQ) under the sanitizers
  {
    opus::CeltEncoder enc(128000);
    std::vector<int16_t> pcm(960 * 2);
    for (int f = 0; f < 4; ++f) {
      for (auto& v : pcm) v = static_cast<int16_t>(rnd());
      auto pkt = enc.encode_frame(pcm.data(), 2);
      if (pkt.empty()) return 6;
    }
  }
  // NAL assembly: zero-heavy bit patterns hit the emulation-prevention path
  for (int t = 0; t < 50; ++t) {
    int nbits = 1 + (rnd() % 2000);
    std::vector<uint32_t> words((nbits + 31) / 32 + 2, 0);
    for (auto& wv : words)
      wv = (t % 3 == 0) ? 0 : (t % 3 == 1 ? rnd() : rnd() & 0x03030303);
    std::vector<uint8_t> out;
    h264::assemble_gpu_row_nal(words.data(), nbits, t & 1, t & 2, out);
  }
  std::puts("sanitizer harness ok");
  return 0;
}
