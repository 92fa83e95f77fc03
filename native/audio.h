// pcmflux-equivalent audio engine.
//
// Implements the AudioCapture API contract the reference control plane
// consumes (AudioCaptureSettings fields, AudioCapture.
// start_capture(settings, cb), frames carrying a [0x01, n_red] wire
// header, update_audio_bitrate, AudioPlayback for mic downlink).
//
// This image has no PulseAudio/ALSA and no libopus, so the capture source
// is synthetic (or a raw-PCM file tap) and the payload codec is s16le PCM
// with optional RED-style redundancy (previous frames appended — same
// loss-resilience semantics the reference's RED Opus gives).
// The codec stage is pluggable so an Opus backend can drop in when the
// library exists.
#pragma once

#include <atomic>
#include <cstdint>
#include <functional>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

namespace hipflux {

struct AudioCaptureSettings {
  std::string device_name = "synthetic";  // synthetic | silence | file:<path>
  int sample_rate = 48000;
  int channels = 2;                        // 1..6
  std::string codec = "opus";              // "opus" (CELT-class) | "pcm"
  int opus_bitrate = 128000;
  int frame_duration_ms = 20;
  int red_distance = 0;                    // redundant previous frames
  bool omit_audio_header = false;
  bool debug_logging = false;
};

// One encoded audio frame handed to the callback (native thread).
struct AudioFrame {
  const uint8_t* data;   // wire payload ([0x01, n_red] header unless omitted)
  size_t size;
  double pts_ms;         // monotonically increasing presentation time
};

class AudioCapture {
 public:
  using Callback = std::function<void(const AudioFrame&)>;

  ~AudioCapture() { stop_capture(); }

  void start_capture(const AudioCaptureSettings& s, Callback cb);
  void stop_capture();
  bool is_capturing() const { return running_.load(); }
  void update_audio_bitrate(int bps) { bitrate_.store(bps); }
  void clear_callback() { cb_ = nullptr; }

 private:
  void run();
  AudioCaptureSettings settings_;
  Callback cb_;
  std::thread thread_;
  std::atomic<bool> running_{false};
  std::atomic<bool> stop_{false};
  std::atomic<int> bitrate_{128000};
};

// Mic downlink sink: client PCM frames written into a ring buffer that a
// local consumer (virtual mic) can drain. (Reference: AudioPlayback /
// provision_virtual_microphone, selkies.py:356.)
struct AudioPlaybackSettings {
  int sample_rate = 48000;
  int channels = 1;
  size_t max_buffer_bytes = 1 << 20;
};

class AudioPlayback {
 public:
  explicit AudioPlayback(const AudioPlaybackSettings& s) : settings_(s) {}
  // append s16le PCM; drops oldest when over budget. Returns stored bytes.
  size_t write(const uint8_t* data, size_t n);
  size_t read(uint8_t* out, size_t n);      // consumer side
  size_t buffered() const { return buf_.size(); }

 private:
  AudioPlaybackSettings settings_;
  std::vector<uint8_t> buf_;
  std::mutex mu_;
};

}  // namespace hipflux
