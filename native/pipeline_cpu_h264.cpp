// CPU striped H.264 pipeline: one independent StripeEncoder (own bitstream,
// frame_num, recon state) per stripe, encoded in parallel on the pool —
// the stripe-parallel seam of SURVEY.md §5.7. Also the bit-exact reference
// for the HIP H.264 pipeline.
#include <algorithm>

#include "cpu/h264/encoder.h"
#include "engine.h"
#include "thread_pool.h"

namespace hipflux {
namespace {

class CpuH264Pipeline : public EncodePipeline {
 public:
  explicit CpuH264Pipeline(const CaptureSettings& s)
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
        encoders_.push_back(
            std::make_unique<h264::StripeEncoder>(
                frame.width, hgt, settings_.video_deblock,
                settings_.video_fullcolor));
      }
      // per-stripe scratch YUV planes (I444 when fullcolor)
      ypitch_ = (frame.width + 15) & ~15;
      fullcolor_ = settings_.video_fullcolor;
      cpitch_ = fullcolor_ ? ypitch_ : ypitch_ / 2;
      yuv_.resize(n_stripes);
      for (auto& v : yuv_)
        v.resize(static_cast<size_t>(ypitch_) * stripe_h *
                 (fullcolor_ ? 6 : 3) / 2);
    }

    struct Out {
      std::vector<uint8_t> bytes;
      h264::EncodeStats st;
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
        uint8_t* cr =
            cb + static_cast<size_t>(cpitch_) *
                     (fullcolor_ ? stripe_h : stripe_h / 2);
        if (fullcolor_)
          h264::bgrx_to_yuv444(
              frame.data + static_cast<size_t>(j.y0) * frame.stride,
              frame.stride, frame.width, hgt, y, cb, cr, ypitch_);
        else
          h264::bgrx_to_yuv420(
              frame.data + static_cast<size_t>(j.y0) * frame.stride,
              frame.stride, frame.width, hgt, y, ypitch_, cb, cr, cpitch_);
        encoders_[i]->encode_frame(y, ypitch_, cb, cr, cpitch_, ctx.crf,
                                   ctx.idr, outs[i].bytes, &outs[i].st);
      });
    }
    pool_.wait_all();
    for (auto& o : outs) {
      if (!o.encode || o.bytes.empty()) continue;
      EncodedStripe s;
      s.type = StripeType::kH264;
      s.data = o.bytes.data();
      s.size = o.bytes.size();
      s.frame_id = ctx.frame_id;
      s.y = o.y0;
      s.width = w_;
      s.height = o.h;
      s.is_keyframe = o.st.is_idr;
      emit(s);
    }
  }

  const char* name() const override { return "cpu-h264"; }

 private:
  CaptureSettings settings_;
  ThreadPool pool_;
  std::vector<std::unique_ptr<h264::StripeEncoder>> encoders_;
  std::vector<std::vector<uint8_t>> yuv_;
  int w_ = 0, ypitch_ = 0, cpitch_ = 0;
  bool fullcolor_ = false;
};

}  // namespace

std::unique_ptr<EncodePipeline> make_cpu_h264_pipeline(
    const CaptureSettings& s) {
  return std::make_unique<CpuH264Pipeline>(s);
}

}  // namespace hipflux
