// HIP (gfx950) encode pipeline host side.
// JPEG path: upload BGRX -> CSC kernel -> wave-per-block DCT/quant kernel ->
// readback MCU-ordered coefficients -> stripe-parallel CPU Huffman pack.
// The H.264 path extends this file as its kernels land.
#include <hip/hip_runtime.h>

#include <cstdio>
#include <map>
#include <stdexcept>

#include "cpu/jpeg_enc.h"
#include "engine.h"
#include "hip/jpeg_kernels.h"
#include "thread_pool.h"

namespace hipflux {
namespace {

#define HIP_CHECK(expr)                                                    \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess) {                                                \
      throw std::runtime_error(std::string("HIP error: ") +                \
                               hipGetErrorString(_e) + " at " #expr);      \
    }                                                                      \
  } while (0)

class HipJpegPipeline : public EncodePipeline {
 public:
  explicit HipJpegPipeline(const CaptureSettings& s)
      : settings_(s),
        pool_(std::max(2u, std::thread::hardware_concurrency() / 2)) {
    HIP_CHECK(hipSetDevice(std::max(0, s.gpu_id)));
    HIP_CHECK(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking));
    upload_dct_tables(stream_);
    HIP_CHECK(hipMalloc(&d_rqy_, 64 * sizeof(float)));
    HIP_CHECK(hipMalloc(&d_rqc_, 64 * sizeof(float)));
    alloc_for(s.capture_width, s.capture_height);
  }

  ~HipJpegPipeline() override {
    (void)hipStreamSynchronize(stream_);
    for (auto& kv : registered_) (void)hipHostUnregister(kv.first);
    if (d_frame_) (void)hipFree(d_frame_);
    if (d_y_) (void)hipFree(d_y_);
    if (d_cb_) (void)hipFree(d_cb_);
    if (d_cr_) (void)hipFree(d_cr_);
    if (d_coeff_) (void)hipFree(d_coeff_);
    if (d_rqy_) (void)hipFree(d_rqy_);
    if (d_rqc_) (void)hipFree(d_rqc_);
    if (h_coeff_) (void)hipHostFree(h_coeff_);
  }

  void encode_frame(const RawFrame& frame, const FrameContext& ctx,
                    const Emit& emit) override {
    const bool fullcolor = settings_.video_fullcolor;
    if (frame.width != w_ || frame.height != h_)
      alloc_for(frame.width, frame.height);
    ensure_quality(ctx.jpeg_quality);

    // upload (register the capture buffer once; zero-copy DMA afterwards)
    const uint8_t* src = frame.data;
    size_t frame_bytes = static_cast<size_t>(frame.stride) * frame.height;
    if (!registered_.count(const_cast<uint8_t*>(src))) {
      hipError_t e = hipHostRegister(const_cast<uint8_t*>(src), frame_bytes,
                                     hipHostRegisterDefault);
      registered_[const_cast<uint8_t*>(src)] = (e == hipSuccess);
      if (e != hipSuccess) (void)hipGetLastError();  // clear; fall back
    }
    HIP_CHECK(hipMemcpyAsync(d_frame_, src, frame_bytes,
                             hipMemcpyHostToDevice, stream_));

    const int stride_px = frame.stride / 4;
    launch_bgrx_to_planes(d_frame_, w_, h_, stride_px, d_y_, d_cb_, d_cr_,
                          ypitch_, cpitch_, fullcolor, stream_);

    const int stripe_h = std::max(16, settings_.stripe_height & ~15);
    const int mcu = fullcolor ? 8 : 16;
    const int mcux = (w_ + mcu - 1) / mcu;
    const int mcuy = (h_ + mcu - 1) / mcu;
    const int rows_per_stripe = stripe_h / mcu;
    const int stripe_mcu_count = rows_per_stripe * mcux;
    const int per_mcu_real = fullcolor ? 3 : 6;

    if (!fullcolor) {
      launch_dct_quant(d_y_, w_, h_, ypitch_, d_rqy_, d_coeff_, 0, false,
                       mcux, rows_per_stripe, stripe_mcu_count, stream_);
      launch_dct_quant(d_cb_, cw_, ch_, cpitch_, d_rqc_, d_coeff_, 1, false,
                       mcux, rows_per_stripe, stripe_mcu_count, stream_);
      launch_dct_quant(d_cr_, cw_, ch_, cpitch_, d_rqc_, d_coeff_, 2, false,
                       mcux, rows_per_stripe, stripe_mcu_count, stream_);
    } else {
      launch_dct_quant(d_y_, w_, h_, ypitch_, d_rqy_, d_coeff_, 0, true,
                       mcux, rows_per_stripe, stripe_mcu_count, stream_);
      launch_dct_quant(d_cb_, w_, h_, ypitch_, d_rqc_, d_coeff_, 1, true,
                       mcux, rows_per_stripe, stripe_mcu_count, stream_);
      launch_dct_quant(d_cr_, w_, h_, ypitch_, d_rqc_, d_coeff_, 2, true,
                       mcux, rows_per_stripe, stripe_mcu_count, stream_);
    }

    size_t coeff_count =
        static_cast<size_t>(mcux) * mcuy * per_mcu_real * 64;
    HIP_CHECK(hipMemcpyAsync(h_coeff_, d_coeff_, coeff_count * sizeof(int16_t),
                             hipMemcpyDeviceToHost, stream_));
    HIP_CHECK(hipStreamSynchronize(stream_));

    // stripe-parallel entropy packing on the CPU pool
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
        int stripe_idx = j.y0 / stripe_h;
        int rows = std::min(rows_per_stripe, mcuy - stripe_idx * rows_per_stripe);
        const int16_t* blocks =
            h_coeff_ + static_cast<size_t>(stripe_idx) * stripe_mcu_count *
                           per_mcu_real * 64;
        jpeg_entropy_from_blocks(blocks, mcux, rows, w_, j.y1 - j.y0,
                                 ctx.jpeg_quality, fullcolor, outs[i].bytes);
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
      s.width = w_;
      s.height = o.h;
      s.is_keyframe = true;
      emit(s);
    }
  }

  const char* name() const override { return "hip-jpeg"; }

 private:
  void alloc_for(int w, int h) {
    HIP_CHECK(hipStreamSynchronize(stream_));
    if (d_frame_) (void)hipFree(d_frame_);
    if (d_y_) (void)hipFree(d_y_);
    if (d_cb_) (void)hipFree(d_cb_);
    if (d_cr_) (void)hipFree(d_cr_);
    if (d_coeff_) (void)hipFree(d_coeff_);
    if (h_coeff_) (void)hipHostFree(h_coeff_);
    w_ = w;
    h_ = h;
    cw_ = (w + 1) / 2;
    ch_ = (h + 1) / 2;
    ypitch_ = (w + 255) & ~255;
    cpitch_ = (cw_ + 255) & ~255;
    HIP_CHECK(hipMalloc(&d_frame_, static_cast<size_t>(w) * h * 4));
    HIP_CHECK(hipMalloc(&d_y_, static_cast<size_t>(ypitch_) * h));
    // chroma buffers sized for 4:4:4 (the larger case)
    HIP_CHECK(hipMalloc(&d_cb_, static_cast<size_t>(ypitch_) * h));
    HIP_CHECK(hipMalloc(&d_cr_, static_cast<size_t>(ypitch_) * h));
    // coefficients: 6 blocks/MCU covers both 420 (6) and 444 (3)
    size_t mcux = (w + 7) / 8, mcuy = (h + 7) / 8;  // worst case 444
    size_t coeff_bytes = mcux * mcuy * 6 * 64 * sizeof(int16_t);
    HIP_CHECK(hipMalloc(&d_coeff_, coeff_bytes));
    HIP_CHECK(hipHostMalloc(&h_coeff_, coeff_bytes, hipHostMallocDefault));
  }

  void ensure_quality(int q) {
    if (q == cur_quality_) return;
    uint8_t qy[64], qc[64];
    jpeg_quality_tables(q, qy, qc);
    float rqy[64], rqc[64];
    for (int i = 0; i < 64; ++i) {
      rqy[i] = 1.0f / qy[i];
      rqc[i] = 1.0f / qc[i];
    }
    HIP_CHECK(hipMemcpyAsync(d_rqy_, rqy, sizeof(rqy), hipMemcpyHostToDevice,
                             stream_));
    HIP_CHECK(hipMemcpyAsync(d_rqc_, rqc, sizeof(rqc), hipMemcpyHostToDevice,
                             stream_));
    HIP_CHECK(hipStreamSynchronize(stream_));
    cur_quality_ = q;
  }

  CaptureSettings settings_;
  ThreadPool pool_;
  hipStream_t stream_{};
  uint8_t* d_frame_ = nullptr;
  uint8_t* d_y_ = nullptr;
  uint8_t* d_cb_ = nullptr;
  uint8_t* d_cr_ = nullptr;
  int16_t* d_coeff_ = nullptr;
  int16_t* h_coeff_ = nullptr;
  float* d_rqy_ = nullptr;
  float* d_rqc_ = nullptr;
  int w_ = 0, h_ = 0, cw_ = 0, ch_ = 0, ypitch_ = 0, cpitch_ = 0;
  int cur_quality_ = -1;
  std::map<void*, bool> registered_;
};

}  // namespace

std::unique_ptr<EncodePipeline> make_hip_pipeline(const CaptureSettings& s) {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess || n == 0) return nullptr;
  if (s.gpu_id >= n) return nullptr;
  try {
    if (s.output_mode == 0) return std::make_unique<HipJpegPipeline>(s);
    // H.264 HIP pipeline lands next; JPEG covers output_mode 0 only.
    return nullptr;
  } catch (const std::exception& e) {
    std::fprintf(stderr, "hipflux: HIP pipeline init failed: %s\n", e.what());
    return nullptr;
  }
}

}  // namespace hipflux
