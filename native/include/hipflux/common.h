// hipflux — MI355X-native capture+encode engine for selkies_amd.
// Common types shared by capture sources, CPU encoders and HIP pipeline.
#pragma once

#include <cstdint>
#include <cstring>
#include <functional>
#include <string>
#include <vector>

namespace hipflux {

// Output frame/stripe type tags (wire protocol; 0x06 is
// this framework's HEVC extension — the reference has no HEVC encoder).
enum class StripeType : uint8_t { kJpeg = 0x03, kH264 = 0x04, kHevc = 0x06 };

enum class OutputMode : int { kJpeg = 0, kH264 = 1, kHevc = 2 };

// Mirrors the pixelflux CaptureSettings contract consumed by the reference
// control plane (reference display_utils.py:2250-2354, selkies.py:5364-5426;
// field inventory in SURVEY.md §2.3). Defaults follow the reference.
struct CaptureSettings {
  int capture_width = 1920;
  int capture_height = 1080;
  int capture_x = 0;
  int capture_y = 0;
  double target_fps = 60.0;
  bool capture_cursor = false;

  int output_mode = 1;              // 0 = JPEG, 1 = H.264, 2 = HEVC
  float capture_scale = 1.0f;       // fractional bilinear downscale
                                    // (0.25..1.0); takes precedence over
                                    // capture_scale_div when != 1
  int capture_scale_div = 1;        // integer box-downscale (1..4): encode
                                    // at capture/div (e.g. 4K capture -> 2
                                    // -> 1080p stream)
  bool video_fullframe = false;     // disable damage gating
  bool use_cpu = false;             // force CPU encode path
  int gpu_id = 0;                   // HIP device ordinal (encode_node_index)

  int video_bitrate_kbps = 16000;
  int video_crf = 25;
  bool video_cbr_mode = false;
  int video_min_qp = 2;
  int video_max_qp = 48;
  double vbv_multiplier = 1.5;
  double keyframe_interval_s = 0.0; // 0 = infinite GOP, IDR on demand
  bool video_streaming_mode = false;
  bool video_deblock = true;       // in-loop deblocking (idc=2 within-slice)
  int pipeline_depth = 1;           // 2 = one encode frame in flight
                                    // (throughput mode; emission lags one
                                    // frame — recording/transcode use)
  bool video_fullcolor = false;     // 4:4:4
  bool use_paint_over_quality = true;
  int paint_over_trigger_frames = 15;
  int video_paintover_crf = 18;
  int video_paintover_burst_frames = 5;
  int damage_block_threshold = 15;
  int damage_block_duration = 30;
  int jpeg_quality = 80;
  int jpeg_paintover_quality = 95;
  int stripe_height = 64;           // must be a multiple of 16
  bool omit_stripe_headers = false;

  std::string watermark_path;
  int watermark_location = 0;

  std::string capture_backend = "auto";  // auto|x11|synthetic
  std::string display;                   // X DISPLAY for x11 backend
  std::string recording_path;            // raw ES tap ('' = off)
  bool debug_logging = false;
};

// One encoded stripe handed to the frame callback (native thread!).
struct EncodedStripe {
  StripeType type;
  const uint8_t* data = nullptr;  // payload WITHOUT wire header
  size_t size = 0;
  uint32_t frame_id = 0;
  int y = 0;
  int width = 0;
  int height = 0;
  bool is_keyframe = false;
  double capture_ts_ms = 0.0;     // monotonic clock at framebuffer acquire
  double encode_done_ms = 0.0;    // monotonic clock when stripe bytes ready
};

using StripeCallback = std::function<void(const EncodedStripe&)>;

// A raw BGRX framebuffer (4 bytes/pixel, little-endian B,G,R,X).
struct RawFrame {
  const uint8_t* data = nullptr;
  int width = 0;
  int height = 0;
  int stride = 0;        // bytes per row
  double ts_ms = 0.0;
};

double now_ms();

}  // namespace hipflux
