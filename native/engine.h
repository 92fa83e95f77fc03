// hipflux engine: capture thread + damage gating + encode pipeline dispatch.
//
// Implements the pixelflux ScreenCapture contract consumed by the reference
// control plane (start_capture(cb, settings)/stop_capture/
// is_capturing, request_idr_frame collapse semantics per selkies.py:120-123,
// live tunables update_framerate/update_video_bitrate/update_tunables).
#pragma once

#include <atomic>
#include <memory>
#include <mutex>
#include <thread>
#include <vector>

#include "capture.h"
#include "cpu/damage.h"
#include "include/hipflux/common.h"

namespace hipflux {

// One stripe's work descriptor for a frame.
struct StripeJob {
  int y0 = 0;
  int y1 = 0;
  bool encode = false;      // damaged (or forced) this frame
  bool paintover = false;   // encode at paint-over quality
};

struct FrameContext {
  uint32_t frame_id = 0;
  bool idr = false;             // force intra / full refresh
  bool paintover = false;       // this frame is a paint-over pass
  int jpeg_quality = 80;
  int crf = 25;
  int bitrate_kbps = 16000;
  std::vector<StripeJob> stripes;
};

// An encode pipeline turns a raw frame + job list into encoded stripes.
// Implementations: CpuJpegPipeline, CpuH264Pipeline, HipPipeline (GPU).
class EncodePipeline {
 public:
  virtual ~EncodePipeline() = default;
  using Emit = std::function<void(EncodedStripe&)>;
  virtual void encode_frame(const RawFrame& frame, const FrameContext& ctx,
                            const Emit& emit) = 0;
  virtual const char* name() const = 0;

  // Throughput mode (reference has no equivalent; x264-style frame
  // pipelining for the MI355X pipeline): with depth 2 an implementation
  // may defer emission by one frame — encode_frame(N) submits N's GPU
  // work and emits frame N-1's stripes, overlapping host bitstream
  // assembly with GPU compute of the next frame. flush() emits any
  // deferred frame. Depth 1 (default) is strictly synchronous; the
  // streaming engine uses depth 1 so interactive latency is unchanged.
  virtual void set_pipeline_depth(int) {}
  virtual void flush(const Emit&) {}

  // Device pointer to the luma reconstruction/reference plane (GPU
  // pipelines only). Tile-parallel encode reads boundary rows from here
  // for the RCCL exchange (rccl_comm.h). Returns false on CPU paths.
  virtual bool recon_dev(void** y, int* ypitch, int* height) {
    (void)y; (void)ypitch; (void)height;
    return false;
  }

  // Test-only introspection: reconstruction planes + (GPU) level/meta
  // buffers of the last encoded frame. Returns false when unsupported.
  struct DebugDump {
    int w = 0, h = 0, ypitch = 0, cpitch = 0;
    std::vector<uint8_t> y, cb, cr;
    std::vector<int16_t> levels;
    std::vector<int> meta;
  };
  virtual bool debug_dump(DebugDump&) { return false; }
};

std::unique_ptr<EncodePipeline> make_cpu_jpeg_pipeline(const CaptureSettings&);
std::unique_ptr<EncodePipeline> make_cpu_h264_pipeline(const CaptureSettings&);
std::unique_ptr<EncodePipeline> make_cpu_hevc_pipeline(const CaptureSettings&);
// GPU (HIP/gfx950) pipeline; returns nullptr if no usable HIP device.
std::unique_ptr<EncodePipeline> make_hip_pipeline(const CaptureSettings&);

class ScreenCapture {
 public:
  ScreenCapture() = default;
  ~ScreenCapture();

  // Spawns the native capture+encode thread. The callback runs ON THAT
  // THREAD — the Python side must trampoline to its event loop
  // (call_soon_threadsafe), same contract as the reference engine.
  void start_capture(StripeCallback cb, const CaptureSettings& settings);
  void stop_capture();
  // Drop the stored callback (must not be running). Needed by the Python
  // binding so a py::function is destroyed while the GIL is held.
  void clear_callback() { cb_ = nullptr; }
  bool is_capturing() const { return running_.load(); }

  // Collapses concurrent requests into one per-frame flag.
  void request_idr_frame() { idr_requested_.store(true); }

  // Cursor shape/position callback (native thread): fired when the cursor
  // shape changes (reference set_cursor_callback contract).
  using CursorCallback =
      std::function<void(int w, int h, int hot_x, int hot_y,
                         const uint32_t* argb, size_t count)>;
  void set_cursor_callback(CursorCallback cb) { cursor_cb_ = std::move(cb); }
  void clear_cursor_callback() { cursor_cb_ = nullptr; }

  // Live tunables (no restart).
  void update_framerate(double fps) { fps_.store(fps); }
  void update_video_bitrate(int kbps) { bitrate_kbps_.store(kbps); }
  void update_crf(int crf) { crf_.store(crf); }
  void update_jpeg_quality(int q) { jpeg_quality_.store(q); }
  void update_vbv_multiplier(double m) { vbv_mult_.store(m); }
  void update_capture_region(int x, int y, int w, int h);

  // Stats
  uint64_t frames_captured() const { return frames_captured_.load(); }
  uint64_t frames_encoded() const { return frames_encoded_.load(); }
  uint64_t stripes_emitted() const { return stripes_emitted_.load(); }
  double last_encode_ms() const { return last_encode_ms_.load(); }
  const char* pipeline_name() const { return pipeline_name_; }

 private:
  void run();

  CaptureSettings settings_;
  StripeCallback cb_;
  CursorCallback cursor_cb_;
  std::thread thread_;
  std::atomic<bool> running_{false};
  std::atomic<bool> stop_{false};
  std::atomic<bool> idr_requested_{false};
  std::atomic<double> fps_{60.0};
  std::atomic<int> bitrate_kbps_{16000};
  std::atomic<int> crf_{25};
  std::atomic<int> jpeg_quality_{80};
  std::atomic<double> vbv_mult_{1.5};
  std::atomic<uint64_t> frames_captured_{0};
  std::atomic<uint64_t> frames_encoded_{0};
  std::atomic<uint64_t> stripes_emitted_{0};
  std::atomic<double> last_encode_ms_{0.0};
  std::mutex region_mutex_;
  int region_[4] = {0, 0, 0, 0};
  std::atomic<bool> region_changed_{false};
  const char* pipeline_name_ = "none";
};

// Pack the wire header in front of an encoded stripe payload
// (SURVEY.md §3.2 frame formats). Appends header+payload to out.
void pack_wire_stripe(const EncodedStripe& s, std::vector<uint8_t>& out);

}  // namespace hipflux
