#include <algorithm>
#include <chrono>
#include <cstring>
#include <random>

#include "capture.h"

namespace hipflux {

double now_ms() {
  using clock = std::chrono::steady_clock;
  return std::chrono::duration<double, std::milli>(
             clock::now().time_since_epoch())
      .count();
}

namespace {

// xorshift64* — fast deterministic pixel noise
inline uint64_t xs64(uint64_t& s) {
  s ^= s >> 12;
  s ^= s << 25;
  s ^= s >> 27;
  return s * 0x2545F4914F6CDD1DULL;
}

class SyntheticSource : public FrameSource {
 public:
  SyntheticSource(int w, int h, std::string pattern, uint64_t seed)
      : w_(w), h_(h), pattern_(std::move(pattern)), seed_(seed) {
    buf_.resize(static_cast<size_t>(w_) * h_ * 4);
    uint64_t s = seed_;
    auto* p64 = reinterpret_cast<uint64_t*>(buf_.data());
    for (size_t i = 0; i < buf_.size() / 8; ++i) p64[i] = xs64(s);
    mask_x();
  }

  bool acquire(RawFrame& out) override {
    ++frame_;  // also advances the synthetic cursor on static patterns
    if (pattern_ == "noise") {
      uint64_t s = seed_ + frame_;
      auto* p64 = reinterpret_cast<uint64_t*>(buf_.data());
      for (size_t i = 0; i < buf_.size() / 8; ++i) p64[i] = xs64(s);
      mask_x();
    } else if (pattern_ == "desktop") {
      // moving 256x192 "window" bouncing over the static background
      int box_w = std::min(256, w_), box_h = std::min(192, h_);
      int span_x = std::max(1, w_ - box_w), span_y = std::max(1, h_ - box_h);
      int px = static_cast<int>((frame_ * 7) % (2 * span_x));
      int py = static_cast<int>((frame_ * 3) % (2 * span_y));
      if (px >= span_x) px = 2 * span_x - px - 1;
      if (py >= span_y) py = 2 * span_y - py - 1;
      // erase previous window area back to background, then draw
      if (last_px_ >= 0) fill_rect(last_px_, last_py_, box_w, box_h, true);
      fill_rect(px, py, box_w, box_h, false);
      last_px_ = px;
      last_py_ = py;
    }  // "static": nothing changes after the first frame
    out.data = buf_.data();
    out.width = w_;
    out.height = h_;
    out.stride = w_ * 4;
    out.ts_ms = now_ms();
    return true;
  }

  int width() const override { return w_; }
  int height() const override { return h_; }

  bool cursor(CursorImage& out) override {
    // synthetic cursor: 12x16 arrow bouncing over the frame
    out.width = 12;
    out.height = 16;
    out.hot_x = 0;
    out.hot_y = 0;
    out.serial = 1;
    int sx = std::max(1, w_ - 12), sy = std::max(1, h_ - 16);
    int px = static_cast<int>((frame_ * 5) % (2 * sx));
    int py = static_cast<int>((frame_ * 4) % (2 * sy));
    if (px >= sx) px = 2 * sx - px - 1;
    if (py >= sy) py = 2 * sy - py - 1;
    out.x = px;
    out.y = py;
    if (out.argb.empty()) {
      out.argb.assign(12 * 16, 0);
      for (int y = 0; y < 16; ++y)
        for (int x = 0; x < 12; ++x)
          if (x <= y && y < 14 - x / 3)
            out.argb[y * 12 + x] =
                (x == 0 || x == y || y == 13 - x / 3) ? 0xFF000000
                                                      : 0xFFFFFFFF;
    }
    return true;
  }

 private:
  void mask_x() {
    // keep the X byte deterministic (matches real BGRX captures)
    for (size_t i = 3; i < buf_.size(); i += 4) buf_[i] = 0xFF;
  }
  void fill_rect(int x, int y, int w, int h, bool background) {
    for (int yy = y; yy < y + h && yy < h_; ++yy) {
      uint8_t* row = buf_.data() + (static_cast<size_t>(yy) * w_ + x) * 4;
      if (background) {
        uint64_t s = seed_ ^ (static_cast<uint64_t>(yy) << 20);
        // regenerate the original background row segment deterministically
        // (same generator order as the constructor per-row would differ;
        // a fixed per-row seed gives a stable background)
        for (int xx = 0; xx < w && x + xx < w_; ++xx) {
          uint64_t v = xs64(s) + static_cast<uint64_t>(x + xx) * 0x9E3779B97F4A7C15ULL;
          row[xx * 4 + 0] = static_cast<uint8_t>(v);
          row[xx * 4 + 1] = static_cast<uint8_t>(v >> 8);
          row[xx * 4 + 2] = static_cast<uint8_t>(v >> 16);
          row[xx * 4 + 3] = 0xFF;
        }
      } else {
        uint8_t shade = static_cast<uint8_t>(64 + ((yy - y) * 191) / std::max(1, h));
        for (int xx = 0; xx < w && x + xx < w_; ++xx) {
          row[xx * 4 + 0] = shade;
          row[xx * 4 + 1] = static_cast<uint8_t>(255 - shade);
          row[xx * 4 + 2] = static_cast<uint8_t>(frame_ & 0xFF);
          row[xx * 4 + 3] = 0xFF;
        }
      }
    }
  }

  int w_, h_;
  std::string pattern_;
  uint64_t seed_;
  uint64_t frame_ = 0;
  int last_px_ = -1, last_py_ = -1;
  std::vector<uint8_t> buf_;
};

}  // namespace

std::unique_ptr<FrameSource> make_synthetic_source(int width, int height,
                                                   const std::string& pattern,
                                                   uint64_t seed) {
  return std::make_unique<SyntheticSource>(width, height, pattern, seed);
}

}  // namespace hipflux
