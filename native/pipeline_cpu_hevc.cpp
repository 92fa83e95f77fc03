// CPU striped HEVC pipeline: one independent hevc::StripeEncoder (own
// bitstream, recon state) per stripe, encoded in parallel on the pool.
// All-intra (every stripe emission is an IDR), so every stripe is a
// keyframe and reference-chain repair never applies. Bit-exact reference
// for the HIP HEVC pipeline.
#include <algorithm>

#include "cpu/h264/encoder.h"  // bgrx_to_yuv420 (shared CSC reference)
#include "cpu/hevc/encoder.h"
#include "engine.h"
#include "thread_pool.h"

namespace hipflux {
namespace {

class CpuHevcPipeline : public EncodePipeline {
 public:
  explicit CpuHevcPipeline(const CaptureSettings& s)
      : settings_(s),
        pool_(std::max(2u, std::thread::hardware_concurrency() / 2)) {}

  void encode_frame(const RawFrame& frame, const FrameContext& ctx,
                    const Emit& emit) override {
    const int stripe_h = std::max(16, settings_.stripe_height & ~15);
    size_t n_stripes = (frame.height + stripe_h - 1) / stripe_h;
    if (encoders_.size() != n_stripes || frame.width != w_) {
      encoders_.clear();
      w_ = frame.width;
      for (size_t i = 0; i < n_stripes; ++i) {
        int y0 = static_cast<int>(i) * stripe_h;
        int hgt = std::min(stripe_h, frame.height - y0);
        encoders_.push_back(std::make_unique<hevc::StripeEncoder>(
            frame.width, hgt, hevc::default_slices_per_row(frame.width)));
      }
      ypitch_ = (frame.width + 15) & ~15;
      cpitch_ = ypitch_ / 2;
      yuv_.resize(n_stripes);
      for (auto& v : yuv_)
        v.resize(static_cast<size_t>(ypitch_) * stripe_h * 3 / 2);
    }

    struct Out {
      std::vector<uint8_t> bytes;
      int y0 = 0, h = 0;
      bool encode = false;
    };
    std::vector<Out> outs(ctx.stripes.size());
    for (size_t i = 0; i < ctx.stripes.size(); ++i) {
      const auto& job = ctx.stripes[i];
      outs[i].y0 = job.y0;
      outs[i].h = job.y1 - job.y0;
      outs[i].encode = job.encode;
      if (!job.encode) continue;
      pool_.submit([&, i] {
        const auto& j = ctx.stripes[i];
        int hgt = j.y1 - j.y0;
        uint8_t* y = yuv_[i].data();
        uint8_t* cb = y + static_cast<size_t>(ypitch_) * stripe_h;
        uint8_t* cr = cb + static_cast<size_t>(cpitch_) * (stripe_h / 2);
        h264::bgrx_to_yuv420(
            frame.data + static_cast<size_t>(j.y0) * frame.stride,
            frame.stride, frame.width, hgt, y, ypitch_, cb, cr, cpitch_);
        encoders_[i]->encode_frame(y, ypitch_, cb, cr, cpitch_, ctx.crf,
                                   outs[i].bytes);
      });
    }
    pool_.wait_all();
    for (auto& o : outs) {
      if (!o.encode || o.bytes.empty()) continue;
      EncodedStripe s;
      s.type = StripeType::kHevc;
      s.data = o.bytes.data();
      s.size = o.bytes.size();
      s.frame_id = ctx.frame_id;
      s.y = o.y0;
      s.width = w_;
      s.height = o.h;
      s.is_keyframe = true;   // all-intra: every emission is an IDR
      emit(s);
    }
  }

  const char* name() const override { return "cpu-hevc"; }

  bool debug_dump(DebugDump& d) override {
    if (encoders_.empty()) return false;
    auto& e = *encoders_[0];
    d.w = e.width();
    d.h = e.height();
    d.ypitch = e.recon_ypitch();
    d.cpitch = e.recon_cpitch();
    int yh = (e.height() + 15) & ~15;
    d.y.assign(e.recon_y(), e.recon_y() + static_cast<size_t>(d.ypitch) * yh);
    d.cb.assign(e.recon_cb(),
                e.recon_cb() + static_cast<size_t>(d.cpitch) * yh / 2);
    d.cr.assign(e.recon_cr(),
                e.recon_cr() + static_cast<size_t>(d.cpitch) * yh / 2);
    return true;
  }

 private:
  CaptureSettings settings_;
  ThreadPool pool_;
  std::vector<std::unique_ptr<hevc::StripeEncoder>> encoders_;
  std::vector<std::vector<uint8_t>> yuv_;
  int w_ = 0, ypitch_ = 0, cpitch_ = 0;
};

}  // namespace

std::unique_ptr<EncodePipeline> make_cpu_hevc_pipeline(
    const CaptureSettings& s) {
  return std::make_unique<CpuHevcPipeline>(s);
}

}  // namespace hipflux
