// CPU striped-MJPEG pipeline: stripe-parallel baseline JPEG over a thread
// pool. This is the software-fallback analogue of the reference's "jpeg"
// encoder and the bit-exact reference for the HIP path.
#include <algorithm>
#include <cstdio>

#include "cpu/jpeg_enc.h"
#include "engine.h"
#include "thread_pool.h"

namespace hipflux {
namespace {

class CpuJpegPipeline : public EncodePipeline {
 public:
  explicit CpuJpegPipeline(const CaptureSettings& s)
      : settings_(s),
        pool_(std::max(2u, std::thread::hardware_concurrency() / 2)) {}

  void encode_frame(const RawFrame& frame, const FrameContext& ctx,
                    const Emit& emit) override {
    struct Out {
      std::vector<uint8_t> bytes;
      int y0 = 0, y1 = 0;
      bool encode = false;
    };
    std::vector<Out> outs(ctx.stripes.size());
    for (size_t i = 0; i < ctx.stripes.size(); ++i) {
      const auto& job = ctx.stripes[i];
      outs[i].y0 = job.y0;
      outs[i].y1 = job.y1;
      outs[i].encode = job.encode;
      if (!job.encode) continue;
      pool_.submit([&, i] {
        const auto& j = ctx.stripes[i];
        const uint8_t* base =
            frame.data + static_cast<size_t>(j.y0) * frame.stride;
        jpeg_encode_bgrx(base, frame.stride, frame.width, j.y1 - j.y0,
                         ctx.jpeg_quality, settings_.video_fullcolor,
                         outs[i].bytes);
      });
    }
    pool_.wait_all();
    for (auto& o : outs) {
      if (!o.encode) continue;
      EncodedStripe s;
      s.type = StripeType::kJpeg;
      s.data = o.bytes.data();
      s.size = o.bytes.size();
      s.frame_id = ctx.frame_id;
      s.y = o.y0;
      s.width = frame.width;
      s.height = o.y1 - o.y0;
      s.is_keyframe = true;  // every JPEG stripe is independently decodable
      emit(s);
    }
  }

  const char* name() const override { return "cpu-jpeg"; }

 private:
  CaptureSettings settings_;
  ThreadPool pool_;
};

}  // namespace

std::unique_ptr<EncodePipeline> make_cpu_jpeg_pipeline(
    const CaptureSettings& s) {
  return std::make_unique<CpuJpegPipeline>(s);
}

}  // namespace hipflux
