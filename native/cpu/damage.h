// Block damage detection + paint-over state (CPU reference path).
// Mirrors the reference engine's damage-gated encode behavior
// (damage_block_threshold / damage_block_duration knobs;
// "damage-gated encoders are application-limited", webrtc_mode.py:2186).
#pragma once

#include <cstdint>
#include <vector>

namespace hipflux {

class DamageTracker {
 public:
  void reset(int width, int height, int block = 16);

  // Compare cur vs the previously submitted frame; update per-block damage
  // ages. First frame after reset marks everything damaged.
  // threshold: max |channel diff| that still counts as unchanged.
  // duration: frames a block stays damaged after its last change.
  void update(const uint8_t* cur, int stride, int threshold, int duration);

  // True if any block intersecting rows [y0, y1) is currently damaged.
  bool stripe_damaged(int y0, int y1) const;
  bool any_damaged() const;
  int consecutive_still_frames() const { return still_frames_; }

  int blocks_x() const { return bx_; }
  int blocks_y() const { return by_; }
  const std::vector<uint16_t>& ages() const { return age_; }

 private:
  int w_ = 0, h_ = 0, block_ = 16, bx_ = 0, by_ = 0;
  std::vector<uint8_t> prev_;
  std::vector<uint16_t> age_;
  bool have_prev_ = false;
  int still_frames_ = 0;
};

}  // namespace hipflux
