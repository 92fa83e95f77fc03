// Framebuffer sources: synthetic pattern generator + X11/XShm capture.
#pragma once

#include <memory>
#include <string>

#include "include/hipflux/common.h"

namespace hipflux {

// Cursor snapshot (ARGB premultiplied-ish, as X11 XFixes delivers).
struct CursorImage {
  int width = 0, height = 0;
  int hot_x = 0, hot_y = 0;
  int x = 0, y = 0;              // current position
  uint64_t serial = 0;           // changes when the shape changes
  std::vector<uint32_t> argb;
};

class FrameSource {
 public:
  virtual ~FrameSource() = default;
  // Acquire the next framebuffer. The returned pointers stay valid until the
  // next acquire() or destruction. Returns false on unrecoverable error.
  virtual bool acquire(RawFrame& out) = 0;
  virtual int width() const = 0;
  virtual int height() const = 0;
  // Fill the current cursor state; false if the source has no cursor.
  virtual bool cursor(CursorImage& out) { (void)out; return false; }
};

// Synthetic BGRX generator for benches and GPU boxes without an X server
// (the BASELINE measurement surface: "synthetic framebuffer with random
// pixel content"). Patterns:
//   "noise"   — every pixel re-randomized every frame (worst case: 100%
//               damage, defeats any gating — used by bench.py)
//   "desktop" — static random background + moving window + blinking cursor
//               (exercises damage gating / paint-over)
//   "static"  — one random frame, never changes (exercises paint-over)
std::unique_ptr<FrameSource> make_synthetic_source(int width, int height,
                                                   const std::string& pattern,
                                                   uint64_t seed = 0x5eed);

// X11 XShm capture of (x, y, w, h) from `display`; nullptr if the display
// cannot be opened. w/h of 0 = full root window.
std::unique_ptr<FrameSource> make_x11_source(const std::string& display,
                                             int x, int y, int w, int h);

// Seqlock shared-memory frame source ("shm:<path>" capture backend):
// the Wayland-compositor capture seam and a generic producer bridge.
std::unique_ptr<FrameSource> make_shm_source(const std::string& path);

}  // namespace hipflux
