#include "engine.h"

#include "cpu/scale.h"

#include <algorithm>
#include <chrono>
#include <cstdio>
#include <cstring>
#include <map>
#include <string>

namespace hipflux {

ScreenCapture::~ScreenCapture() { stop_capture(); }

void ScreenCapture::start_capture(StripeCallback cb,
                                  const CaptureSettings& settings) {
  stop_capture();
  settings_ = settings;
  cb_ = std::move(cb);
  fps_.store(settings.target_fps);
  bitrate_kbps_.store(settings.video_bitrate_kbps);
  crf_.store(settings.video_crf);
  jpeg_quality_.store(settings.jpeg_quality);
  vbv_mult_.store(settings.vbv_multiplier);
  stop_.store(false);
  running_.store(true);
  thread_ = std::thread([this] { run(); });
}

void ScreenCapture::stop_capture() {
  stop_.store(true);
  if (thread_.joinable()) thread_.join();
  running_.store(false);
}

void ScreenCapture::update_capture_region(int x, int y, int w, int h) {
  std::lock_guard<std::mutex> lk(region_mutex_);
  region_[0] = x; region_[1] = y; region_[2] = w; region_[3] = h;
  region_changed_.store(true);
}

namespace {

inline uint8_t clip8u(long v) {
  return static_cast<uint8_t>(v < 0 ? 0 : (v > 255 ? 255 : v));
}

// Watermark: raw ".bgra" file = u32 width, u32 height, then BGRA pixels.
// (The Python layer converts PNG -> .bgra via PIL; keeps libpng out of the
// native build.) Locations: 1=TL 2=TR 3=BL 4=BR 5=center 6=animated.
struct Watermark {
  int w = 0, h = 0;
  std::vector<uint8_t> bgra;
  bool load(const std::string& path) {
    FILE* f = std::fopen(path.c_str(), "rb");
    if (!f) return false;
    uint32_t wh[2];
    if (std::fread(wh, 4, 2, f) != 2) {
      std::fclose(f);
      return false;
    }
    w = static_cast<int>(wh[0]);
    h = static_cast<int>(wh[1]);
    if (w <= 0 || h <= 0 || w > 8192 || h > 8192) {
      std::fclose(f);
      return false;
    }
    bgra.resize(static_cast<size_t>(w) * h * 4);
    size_t got = std::fread(bgra.data(), 1, bgra.size(), f);
    std::fclose(f);
    return got == bgra.size();
  }

  void composite(uint8_t* frame, int fw, int fh, int stride, int location,
                 uint64_t frame_idx) const {
    if (w == 0) return;
    int x0 = 0, y0 = 0, margin = 16;
    switch (location) {
      case 1: x0 = margin; y0 = margin; break;
      case 2: x0 = fw - w - margin; y0 = margin; break;
      case 3: x0 = margin; y0 = fh - h - margin; break;
      case 4: x0 = fw - w - margin; y0 = fh - h - margin; break;
      case 5: x0 = (fw - w) / 2; y0 = (fh - h) / 2; break;
      case 6: {  // bounce animation
        int sx = std::max(1, fw - w), sy = std::max(1, fh - h);
        x0 = static_cast<int>((frame_idx * 3) % (2 * sx));
        y0 = static_cast<int>((frame_idx * 2) % (2 * sy));
        if (x0 >= sx) x0 = 2 * sx - x0 - 1;
        if (y0 >= sy) y0 = 2 * sy - y0 - 1;
        break;
      }
      default: return;
    }
    for (int y = 0; y < h; ++y) {
      int fy = y0 + y;
      if (fy < 0 || fy >= fh) continue;
      uint8_t* dst = frame + static_cast<size_t>(fy) * stride;
      const uint8_t* src = bgra.data() + static_cast<size_t>(y) * w * 4;
      for (int x = 0; x < w; ++x) {
        int fx = x0 + x;
        if (fx < 0 || fx >= fw) continue;
        int a = src[x * 4 + 3];
        if (a == 0) continue;
        uint8_t* d = dst + fx * 4;
        for (int c = 0; c < 3; ++c)
          d[c] = static_cast<uint8_t>(
              (src[x * 4 + c] * a + d[c] * (255 - a) + 127) / 255);
      }
    }
  }
};

}  // namespace

void pack_wire_stripe(const EncodedStripe& s, std::vector<uint8_t>& out) {
  if (s.type == StripeType::kH264 || s.type == StripeType::kHevc) {
    // [tag, keyflag, frame_id:u16be, y:u16be, w:u16be, h:u16be] (10 bytes)
    out.push_back(static_cast<uint8_t>(s.type));
    out.push_back(s.is_keyframe ? 1 : 0);
    out.push_back(static_cast<uint8_t>((s.frame_id >> 8) & 0xFF));
    out.push_back(static_cast<uint8_t>(s.frame_id & 0xFF));
    out.push_back(static_cast<uint8_t>((s.y >> 8) & 0xFF));
    out.push_back(static_cast<uint8_t>(s.y & 0xFF));
    out.push_back(static_cast<uint8_t>((s.width >> 8) & 0xFF));
    out.push_back(static_cast<uint8_t>(s.width & 0xFF));
    out.push_back(static_cast<uint8_t>((s.height >> 8) & 0xFF));
    out.push_back(static_cast<uint8_t>(s.height & 0xFF));
  } else {
    // [0x03, flags, frame_id:u16be, y:u16be] (6 bytes)
    out.push_back(0x03);
    out.push_back(s.is_keyframe ? 1 : 0);
    out.push_back(static_cast<uint8_t>((s.frame_id >> 8) & 0xFF));
    out.push_back(static_cast<uint8_t>(s.frame_id & 0xFF));
    out.push_back(static_cast<uint8_t>((s.y >> 8) & 0xFF));
    out.push_back(static_cast<uint8_t>(s.y & 0xFF));
  }
  out.insert(out.end(), s.data, s.data + s.size);
}

void ScreenCapture::run() {
  using clock = std::chrono::steady_clock;

  // --- frame source -------------------------------------------------------
  std::unique_ptr<FrameSource> src;
  std::string backend = settings_.capture_backend;
  if (backend == "auto")
    backend = settings_.display.empty() ? "synthetic" : "x11";
  if (backend == "x11") {
    src = make_x11_source(settings_.display, settings_.capture_x,
                          settings_.capture_y, settings_.capture_width,
                          settings_.capture_height);
    if (!src) {
      std::fprintf(stderr,
                   "hipflux: cannot open X display '%s', falling back to "
                   "synthetic source\n",
                   settings_.display.c_str());
    }
  }
  if (!src && settings_.capture_backend.rfind("shm:", 0) == 0) {
    // shared-memory producer (the Wayland compositor capture seam)
    src = make_shm_source(settings_.capture_backend.substr(4));
    if (!src)
      std::fprintf(stderr,
                   "hipflux: shm capture source unavailable, falling back "
                   "to synthetic\n");
  }
  if (!src) {
    // synthetic pattern override: "synthetic:<pattern>" in capture_backend
    std::string pattern = "desktop";
    auto pos = settings_.capture_backend.find(':');
    if (pos != std::string::npos)
      pattern = settings_.capture_backend.substr(pos + 1);
    src = make_synthetic_source(settings_.capture_width,
                                settings_.capture_height, pattern);
  }

  // --- pipeline ------------------------------------------------------------
  std::unique_ptr<EncodePipeline> pipeline;
  if (!settings_.use_cpu && settings_.gpu_id >= 0)
    pipeline = make_hip_pipeline(settings_);
  if (!pipeline) {
    pipeline = settings_.output_mode == 0   ? make_cpu_jpeg_pipeline(settings_)
               : settings_.output_mode == 2 ? make_cpu_hevc_pipeline(settings_)
                                            : make_cpu_h264_pipeline(settings_);
  }
  if (!pipeline) {
    std::fprintf(stderr,
                 "hipflux: requested pipeline unavailable, using cpu-jpeg\n");
    pipeline = make_cpu_jpeg_pipeline(settings_);
  }
  pipeline_name_ = pipeline->name();
  pipeline->set_pipeline_depth(settings_.pipeline_depth);

  const int scale_div =
      std::min(4, std::max(1, settings_.capture_scale_div));
  const float fscale =
      std::min(1.0f, std::max(0.25f, settings_.capture_scale));
  const bool frac_scale = fscale < 0.9999f;
  DamageTracker damage;
  if (frac_scale)
    damage.reset(std::max(1, int(src->width() * fscale + 0.5f)),
                 std::max(1, int(src->height() * fscale + 0.5f)));
  else
    damage.reset(src->width() / scale_div, src->height() / scale_div);
  std::vector<uint8_t> scale_buf;

  Watermark watermark;
  std::vector<uint8_t> wm_scratch;
  if (!settings_.watermark_path.empty() && settings_.watermark_location > 0) {
    if (!watermark.load(settings_.watermark_path))
      std::fprintf(stderr, "hipflux: cannot load watermark %s\n",
                   settings_.watermark_path.c_str());
  }
  uint64_t wm_frame = 0;
  uint64_t last_cursor_serial = ~0ULL;

  // CBR rate control (video_cbr_mode): VBV-fullness PI controller mapping
  // produced bytes to a per-frame QP within [min_qp, max_qp]. CRF mode
  // uses the fixed crf value. (Reference knob semantics: SURVEY.md §2.3
  // video_cbr_mode / vbv fields / min-max QP clamp.)
  double rc_qp = static_cast<double>(crf_.load());
  double vbv_fullness = 0.0;

  const int stripe_h = std::max(16, settings_.stripe_height & ~15);
  uint32_t frame_id = 0;
  int paintover_pending = 0;   // remaining paint-over burst frames
  bool paintover_done = false; // paint-over completed for current still period
  double keyframe_due_ms = 0;  // next forced keyframe time (0 = disabled)
  if (settings_.keyframe_interval_s > 0)
    keyframe_due_ms = now_ms() + settings_.keyframe_interval_s * 1000.0;

  std::vector<uint8_t> wire;
  // raw-ES recording tap (reference recording_socket semantics,
  // SURVEY.md §2.3): one elementary-stream file per stripe row.
  std::map<int, FILE*> rec_files;  // y0 -> file
  auto next_tick = clock::now();

  while (!stop_.load()) {
    double fps = std::max(1.0, fps_.load());
    next_tick += std::chrono::microseconds(static_cast<int64_t>(1e6 / fps));

    RawFrame frame;
    if (!src->acquire(frame)) break;
    frames_captured_.fetch_add(1);

    CursorImage cursor;
    bool have_cursor = false;
    if (settings_.capture_cursor || cursor_cb_)
      have_cursor = src->cursor(cursor);

    if (watermark.w > 0 || (settings_.capture_cursor && have_cursor)) {
      // composite on a scratch copy (the source buffer may persist)
      size_t bytes = static_cast<size_t>(frame.stride) * frame.height;
      wm_scratch.resize(bytes);
      std::memcpy(wm_scratch.data(), frame.data, bytes);
      if (watermark.w > 0)
        watermark.composite(wm_scratch.data(), frame.width, frame.height,
                            frame.stride, settings_.watermark_location,
                            wm_frame++);
      if (settings_.capture_cursor && have_cursor) {
        for (int cy = 0; cy < cursor.height; ++cy) {
          int fy = cursor.y + cy;
          if (fy < 0 || fy >= frame.height) continue;
          uint8_t* dst = wm_scratch.data() +
                         static_cast<size_t>(fy) * frame.stride;
          for (int cx = 0; cx < cursor.width; ++cx) {
            int fx = cursor.x + cx;
            if (fx < 0 || fx >= frame.width) continue;
            uint32_t p = cursor.argb[static_cast<size_t>(cy) * cursor.width +
                                     cx];
            int a = p >> 24;
            if (a == 0) continue;
            uint8_t* d = dst + fx * 4;
            // XFixes pixels are premultiplied ARGB
            d[0] = clip8u(((p & 0xFF) * 255 + d[0] * (255 - a)) / 255);
            d[1] = clip8u((((p >> 8) & 0xFF) * 255 + d[1] * (255 - a)) / 255);
            d[2] = clip8u((((p >> 16) & 0xFF) * 255 + d[2] * (255 - a)) / 255);
          }
        }
      }
      frame.data = wm_scratch.data();
    }
    if (cursor_cb_ && have_cursor && cursor.serial != last_cursor_serial) {
      last_cursor_serial = cursor.serial;
      cursor_cb_(cursor.width, cursor.height, cursor.hot_x, cursor.hot_y,
                 cursor.argb.data(), cursor.argb.size());
    }

    // capture downscale: fractional bilinear (capture_scale) or exact
    // integer box (capture_scale_div); cursor/watermark were composited
    // at native size above
    if (frac_scale) {
      int ow, oh, ostride;
      bilinear_downscale_bgrx(frame.data, frame.stride, frame.width,
                              frame.height, fscale, scale_buf, ow, oh,
                              ostride);
      frame.data = scale_buf.data();
      frame.width = ow;
      frame.height = oh;
      frame.stride = ostride;
    } else if (scale_div > 1) {
      int ow, oh, ostride;
      box_downscale_bgrx(frame.data, frame.stride, frame.width,
                         frame.height, scale_div, scale_buf, ow, oh,
                         ostride);
      frame.data = scale_buf.data();
      frame.width = ow;
      frame.height = oh;
      frame.stride = ostride;
    }

    // damage update (skipped in fullframe mode to save the diff cost)
    bool force_all = settings_.video_fullframe;
    if (!force_all)
      damage.update(frame.data, frame.stride, settings_.damage_block_threshold,
                    settings_.damage_block_duration);

    bool idr = idr_requested_.exchange(false);
    double tnow = now_ms();
    if (keyframe_due_ms > 0 && tnow >= keyframe_due_ms) {
      idr = true;
      keyframe_due_ms = tnow + settings_.keyframe_interval_s * 1000.0;
    }

    // paint-over state machine
    bool paintover_frame = false;
    if (!force_all && settings_.use_paint_over_quality) {
      if (damage.consecutive_still_frames() == 0) {
        paintover_done = false;
        paintover_pending = 0;
      } else if (!paintover_done &&
                 damage.consecutive_still_frames() >=
                     settings_.paint_over_trigger_frames) {
        paintover_pending = settings_.output_mode == 0
                                ? 1
                                : settings_.video_paintover_burst_frames;
        paintover_done = true;
      }
      if (paintover_pending > 0) {
        paintover_frame = true;
        --paintover_pending;
      }
    }

    // build stripe jobs
    FrameContext ctx;
    ctx.frame_id = frame_id;
    ctx.idr = idr;
    ctx.paintover = paintover_frame;
    ctx.jpeg_quality = paintover_frame ? settings_.jpeg_paintover_quality
                                       : jpeg_quality_.load();
    if (settings_.video_cbr_mode && settings_.output_mode >= 1) {
      ctx.crf = static_cast<int>(rc_qp + 0.5);
      if (paintover_frame)
        ctx.crf = std::min(ctx.crf, settings_.video_paintover_crf);
    } else {
      ctx.crf = paintover_frame ? settings_.video_paintover_crf : crf_.load();
    }
    ctx.crf = std::clamp(ctx.crf, settings_.video_min_qp,
                         settings_.video_max_qp);
    ctx.bitrate_kbps = bitrate_kbps_.load();
    bool any = false;
    for (int y = 0; y < frame.height; y += stripe_h) {
      StripeJob job;
      job.y0 = y;
      job.y1 = std::min(y + stripe_h, frame.height);
      job.encode = force_all || idr || paintover_frame ||
                   damage.stripe_damaged(job.y0, job.y1);
      job.paintover = paintover_frame;
      any |= job.encode;
      ctx.stripes.push_back(job);
    }

    auto record_stripe = [&](const EncodedStripe& s) {
      if (settings_.recording_path.empty()) return;
      FILE*& f = rec_files[s.y];
      if (!f) {
        std::string path = settings_.recording_path + ".s" +
                           std::to_string(s.y) +
                           (s.type == StripeType::kH264   ? ".h264"
                            : s.type == StripeType::kHevc ? ".h265"
                                                          : ".mjpeg");
        f = std::fopen(path.c_str(), "wb");
      }
      if (f) std::fwrite(s.data, 1, s.size, f);
    };

    size_t frame_bytes = 0;
    auto handle_stripe = [&](EncodedStripe& s) {
      frame_bytes += s.size;
      record_stripe(s);
      // pipelined emission stamps its own capture time (the stripe may
      // belong to the PREVIOUS frame); only backfill when unset
      if (s.capture_ts_ms == 0) s.capture_ts_ms = frame.ts_ms;
      s.encode_done_ms = now_ms();
      if (settings_.omit_stripe_headers) {
        if (cb_) cb_(s);
      } else {
        wire.clear();
        pack_wire_stripe(s, wire);
        EncodedStripe ws = s;
        ws.data = wire.data();
        ws.size = wire.size();
        if (cb_) cb_(ws);
      }
      stripes_emitted_.fetch_add(1);
    };
    if (!any) {
      // no damage this frame: drain any pipelined frame promptly so
      // depth-2 emission latency is bounded by the frame interval
      pipeline->flush(handle_stripe);
    }
    if (any) {
      double t0 = now_ms();
      pipeline->encode_frame(frame, ctx, handle_stripe);
      last_encode_ms_.store(now_ms() - t0);
      frames_encoded_.fetch_add(1);
      ++frame_id;

      if (settings_.video_cbr_mode && settings_.output_mode >= 1) {
        double target_bpf =
            bitrate_kbps_.load() * 1000.0 / 8.0 / std::max(1.0, fps);
        // VBV buffer: vbv_multiplier x ~quarter-second of stream
        double vbv_size = target_bpf * std::max(0.2, vbv_mult_.load()) *
                          std::max(2.0, fps / 4.0);
        vbv_fullness += static_cast<double>(frame_bytes) - target_bpf;
        vbv_fullness = std::clamp(vbv_fullness, 0.0, vbv_size);
        double err = frame_bytes / std::max(1.0, target_bpf);
        // proportional on instantaneous overshoot + integral via fullness
        rc_qp += 0.6 * std::log2(std::max(0.05, err)) +
                 2.0 * (vbv_fullness / vbv_size - 0.4);
        rc_qp = std::clamp(rc_qp,
                           static_cast<double>(settings_.video_min_qp),
                           static_cast<double>(settings_.video_max_qp));
      }
    }

    // pacing (skip sleeping if we're behind)
    auto now = clock::now();
    if (next_tick > now) {
      std::this_thread::sleep_until(next_tick);
    } else {
      next_tick = now;
    }
  }
  for (auto& kv : rec_files)
    if (kv.second) std::fclose(kv.second);
  running_.store(false);
}

}  // namespace hipflux
