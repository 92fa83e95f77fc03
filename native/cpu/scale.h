// Exact integer box downscale for BGRX capture frames.
// out(x,y) = round(mean of the div x div source block), per channel.
// Integer semantics are identical everywhere (engine CPU path), so tests
// can compare against an independent numpy implementation bit-exactly.
#pragma once

#include <algorithm>
#include <cstdint>
#include <vector>

namespace hipflux {

inline void box_downscale_bgrx(const uint8_t* src, int sstride, int w,
                               int h, int div, std::vector<uint8_t>& out,
                               int& ow, int& oh, int& ostride) {
  ow = w / div;
  oh = h / div;
  ostride = ow * 4;
  out.resize(static_cast<size_t>(ostride) * oh);
  const int n = div * div, half = n / 2;
  for (int y = 0; y < oh; ++y) {
    uint8_t* d = out.data() + static_cast<size_t>(y) * ostride;
    for (int x = 0; x < ow; ++x) {
      int acc[4] = {0, 0, 0, 0};
      for (int dy = 0; dy < div; ++dy) {
        const uint8_t* s = src + static_cast<size_t>(y * div + dy) * sstride +
                           static_cast<size_t>(x) * div * 4;
        for (int dx = 0; dx < div; ++dx) {
          acc[0] += s[dx * 4 + 0];
          acc[1] += s[dx * 4 + 1];
          acc[2] += s[dx * 4 + 2];
          acc[3] += s[dx * 4 + 3];
        }
      }
      d[x * 4 + 0] = static_cast<uint8_t>((acc[0] + half) / n);
      d[x * 4 + 1] = static_cast<uint8_t>((acc[1] + half) / n);
      d[x * 4 + 2] = static_cast<uint8_t>((acc[2] + half) / n);
      d[x * 4 + 3] = static_cast<uint8_t>((acc[3] + half) / n);
    }
  }
}

// Fractional bilinear downscale (0 < scale < 1), fixed-point 16.16
// sample positions with 8-bit interpolation weights. The reference's
// `scale` knob (Wayland logical->pixel ratios like 1.5x) needs
// non-integer ratios the box filter can't express. Integer math only,
// so the engine path and the numpy mirror in tests agree bit-exactly.
inline void bilinear_downscale_bgrx(const uint8_t* src, int sstride, int w,
                                    int h, float scale,
                                    std::vector<uint8_t>& out, int& ow,
                                    int& oh, int& ostride) {
  ow = std::max(1, static_cast<int>(w * scale + 0.5f));
  oh = std::max(1, static_cast<int>(h * scale + 0.5f));
  ostride = ow * 4;
  out.resize(static_cast<size_t>(ostride) * oh);
  // source position of output pixel center i: (i + 0.5) / scale - 0.5,
  // in 16.16 fixed point computed from the integer output size
  const int64_t xstep = (static_cast<int64_t>(w) << 16) / ow;
  const int64_t ystep = (static_cast<int64_t>(h) << 16) / oh;
  std::vector<int> xi(ow);
  std::vector<int> xw(ow);
  for (int x = 0; x < ow; ++x) {
    int64_t sx = ((2 * static_cast<int64_t>(x) + 1) * xstep - (1 << 16)) / 2;
    if (sx < 0) sx = 0;
    int ix = static_cast<int>(sx >> 16);
    if (ix > w - 2) ix = w - 2;
    int fr = static_cast<int>((sx >> 8) & 0xFF);
    if (w == 1) { ix = 0; fr = 0; }
    xi[x] = ix;
    xw[x] = fr;
  }
  for (int y = 0; y < oh; ++y) {
    int64_t sy = ((2 * static_cast<int64_t>(y) + 1) * ystep - (1 << 16)) / 2;
    if (sy < 0) sy = 0;
    int iy = static_cast<int>(sy >> 16);
    if (iy > h - 2) iy = h - 2;
    int fy = static_cast<int>((sy >> 8) & 0xFF);
    if (h == 1) { iy = 0; fy = 0; }
    const uint8_t* r0 = src + static_cast<size_t>(iy) * sstride;
    const uint8_t* r1 = r0 + (h > 1 ? sstride : 0);
    uint8_t* d = out.data() + static_cast<size_t>(y) * ostride;
    for (int x = 0; x < ow; ++x) {
      const uint8_t* a = r0 + static_cast<size_t>(xi[x]) * 4;
      const uint8_t* b = a + (w > 1 ? 4 : 0);
      const uint8_t* c = r1 + static_cast<size_t>(xi[x]) * 4;
      const uint8_t* e = c + (w > 1 ? 4 : 0);
      const int fx = xw[x];
      for (int ch = 0; ch < 4; ++ch) {
        int top = (a[ch] << 8) + (b[ch] - a[ch]) * fx;       // 8.8
        int bot = (c[ch] << 8) + (e[ch] - c[ch]) * fx;
        int v = (top << 8) + (bot - top) * fy;               // 8.16
        d[x * 4 + ch] = static_cast<uint8_t>((v + (1 << 15)) >> 16);
      }
    }
  }
}

}  // namespace hipflux
