#include "audio.h"

#include "cpu/opus/celt.h"

#include <chrono>
#include <cmath>
#include <cstdio>
#include <cstring>
#include <mutex>

namespace hipflux {

void AudioCapture::start_capture(const AudioCaptureSettings& s, Callback cb) {
  stop_capture();
  settings_ = s;
  cb_ = std::move(cb);
  bitrate_.store(s.opus_bitrate);
  stop_.store(false);
  running_.store(true);
  thread_ = std::thread([this] { run(); });
}

void AudioCapture::stop_capture() {
  stop_.store(true);
  if (thread_.joinable()) thread_.join();
  running_.store(false);
}

void AudioCapture::run() {
  using clock = std::chrono::steady_clock;
  const int rate = settings_.sample_rate;
  const int ch = settings_.channels;
  const int frame_samples = rate * settings_.frame_duration_ms / 1000;
  const size_t frame_bytes = static_cast<size_t>(frame_samples) * ch * 2;
  const int red = std::min(5, std::max(0, settings_.red_distance));

  FILE* src_file = nullptr;
  if (settings_.device_name.rfind("file:", 0) == 0)
    src_file = std::fopen(settings_.device_name.c_str() + 5, "rb");

  std::vector<int16_t> pcm(static_cast<size_t>(frame_samples) * ch);
  const bool use_opus = settings_.codec != "pcm" &&
                        rate == 48000 && frame_samples == 960;
  opus::CeltEncoder opus_enc(bitrate_.load());
  int last_bitrate = bitrate_.load();
  // history for RED redundancy
  std::vector<std::vector<uint8_t>> history;
  std::vector<uint8_t> wire;

  double phase = 0.0;
  uint64_t frame_idx = 0;
  auto next_tick = clock::now();
  const auto tick = std::chrono::microseconds(
      settings_.frame_duration_ms * 1000);

  while (!stop_.load()) {
    // --- produce one frame of source PCM
    if (src_file) {
      size_t got = std::fread(pcm.data(), 1, frame_bytes, src_file);
      if (got < frame_bytes) {
        std::memset(reinterpret_cast<uint8_t*>(pcm.data()) + got, 0,
                    frame_bytes - got);
        std::rewind(src_file);
      }
    } else if (settings_.device_name == "silence") {
      std::memset(pcm.data(), 0, frame_bytes);
    } else {
      // synthetic: gentle stereo test tone (amplitude kept low)
      const double f0 = 440.0, f1 = 554.37;
      for (int i = 0; i < frame_samples; ++i) {
        double t = phase + static_cast<double>(i) / rate;
        int16_t l = static_cast<int16_t>(3000 * std::sin(2 * M_PI * f0 * t));
        int16_t r = static_cast<int16_t>(3000 * std::sin(2 * M_PI * f1 * t));
        for (int c = 0; c < ch; ++c)
          pcm[static_cast<size_t>(i) * ch + c] = (c % 2 == 0) ? l : r;
      }
      phase += static_cast<double>(frame_samples) / rate;
    }

    // --- encode: CELT-class Opus framing by default (native/cpu/opus),
    // raw PCM passthrough when codec == "pcm"
    std::vector<uint8_t> payload;
    if (use_opus) {
      int want = bitrate_.load();
      if (want != last_bitrate) {
        opus_enc.set_bitrate(want);      // live update_audio_bitrate
        last_bitrate = want;
      }
      payload = opus_enc.encode_frame(pcm.data(), ch);
    } else {
      payload.resize(frame_bytes);
      std::memcpy(payload.data(), pcm.data(), frame_bytes);
    }

    // --- wire frame: [0x01, n_red][u16 len]*red oldest..newest + payloads
    wire.clear();
    int n_red = std::min<int>(red, history.size());
    if (!settings_.omit_audio_header) {
      wire.push_back(0x01);
      wire.push_back(static_cast<uint8_t>(n_red));
    }
    for (int k = n_red; k > 0; --k) {
      const auto& h = history[history.size() - k];
      wire.push_back(static_cast<uint8_t>(h.size() >> 8));
      wire.push_back(static_cast<uint8_t>(h.size() & 0xFF));
      wire.insert(wire.end(), h.begin(), h.end());
    }
    wire.insert(wire.end(), payload.begin(), payload.end());

    if (cb_) {
      AudioFrame f;
      f.data = wire.data();
      f.size = wire.size();
      f.pts_ms = static_cast<double>(frame_idx) * settings_.frame_duration_ms;
      cb_(f);
    }
    ++frame_idx;
    if (red > 0) {
      history.push_back(std::move(payload));
      if (history.size() > 8) history.erase(history.begin());
    }

    next_tick += tick;
    auto now = clock::now();
    if (next_tick > now)
      std::this_thread::sleep_until(next_tick);
    else
      next_tick = now;
  }
  if (src_file) std::fclose(src_file);
  running_.store(false);
}

size_t AudioPlayback::write(const uint8_t* data, size_t n) {
  std::lock_guard<std::mutex> lk(mu_);
  buf_.insert(buf_.end(), data, data + n);
  if (buf_.size() > settings_.max_buffer_bytes)
    buf_.erase(buf_.begin(),
               buf_.begin() + (buf_.size() - settings_.max_buffer_bytes));
  return buf_.size();
}

size_t AudioPlayback::read(uint8_t* out, size_t n) {
  std::lock_guard<std::mutex> lk(mu_);
  size_t take = std::min(n, buf_.size());
  std::memcpy(out, buf_.data(), take);
  buf_.erase(buf_.begin(), buf_.begin() + take);
  return take;
}

}  // namespace hipflux
