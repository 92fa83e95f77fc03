// Exact integer box downscale for BGRX capture frames.
// out(x,y) = round(mean of the div x div source block), per channel.
// Integer semantics are identical everywhere (engine CPU path), so tests
// can compare against an independent numpy implementation bit-exactly.
#pragma once

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

}  // namespace hipflux
