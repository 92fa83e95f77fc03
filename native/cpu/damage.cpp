#include "damage.h"

#include <algorithm>
#include <cstring>

namespace hipflux {

void DamageTracker::reset(int width, int height, int block) {
  w_ = width;
  h_ = height;
  block_ = block;
  bx_ = (width + block - 1) / block;
  by_ = (height + block - 1) / block;
  prev_.assign(static_cast<size_t>(width) * height * 4, 0);
  age_.assign(static_cast<size_t>(bx_) * by_, 0);
  have_prev_ = false;
  still_frames_ = 0;
}

void DamageTracker::update(const uint8_t* cur, int stride, int threshold,
                           int duration) {
  bool any_change = false;
  if (!have_prev_) {
    any_change = true;
    for (auto& a : age_) a = static_cast<uint16_t>(std::max(1, duration));
  } else {
    for (int by = 0; by < by_; ++by) {
      int y0 = by * block_, y1 = std::min(y0 + block_, h_);
      for (int bx = 0; bx < bx_; ++bx) {
        int x0 = bx * block_, x1 = std::min(x0 + block_, w_);
        bool dirty = false;
        for (int y = y0; y < y1 && !dirty; ++y) {
          const uint8_t* c = cur + static_cast<size_t>(y) * stride + x0 * 4;
          const uint8_t* p = prev_.data() + (static_cast<size_t>(y) * w_ + x0) * 4;
          int n = (x1 - x0) * 4;
          if (threshold <= 0) {
            dirty = std::memcmp(c, p, n) != 0;
          } else {
            for (int i = 0; i < n; ++i) {
              int d = static_cast<int>(c[i]) - static_cast<int>(p[i]);
              if (d > threshold || d < -threshold) { dirty = true; break; }
            }
          }
        }
        auto& a = age_[static_cast<size_t>(by) * bx_ + bx];
        if (dirty) {
          a = static_cast<uint16_t>(std::max(1, duration));
          any_change = true;
        } else if (a > 0) {
          --a;
        }
      }
    }
  }
  // keep a copy of the current frame for the next diff
  for (int y = 0; y < h_; ++y)
    std::memcpy(prev_.data() + static_cast<size_t>(y) * w_ * 4,
                cur + static_cast<size_t>(y) * stride,
                static_cast<size_t>(w_) * 4);
  have_prev_ = true;
  still_frames_ = any_change ? 0 : still_frames_ + 1;
}

bool DamageTracker::stripe_damaged(int y0, int y1) const {
  int b0 = y0 / block_, b1 = std::min((y1 + block_ - 1) / block_, by_);
  for (int by = b0; by < b1; ++by)
    for (int bx = 0; bx < bx_; ++bx)
      if (age_[static_cast<size_t>(by) * bx_ + bx] > 0) return true;
  return false;
}

bool DamageTracker::any_damaged() const {
  for (auto a : age_)
    if (a > 0) return true;
  return false;
}

}  // namespace hipflux
