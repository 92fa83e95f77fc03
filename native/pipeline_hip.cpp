// HIP (gfx950) encode pipeline host side.
// JPEG path: upload BGRX -> CSC kernel -> wave-per-block DCT/quant kernel ->
// readback MCU-ordered coefficients -> stripe-parallel CPU Huffman pack.
// The H.264 path extends this file as its kernels land.
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <chrono>
#include <map>
#include <stdexcept>

#include "cpu/h264/gpu_entropy.h"
#include "cpu/h264/headers.h"
#include "cpu/jpeg_enc.h"
#include "engine.h"
#include "hip/h264_kernels.h"
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
        pool_(std::min(16u, std::max(
                                2u,
                                std::thread::hardware_concurrency() / 2))) {
    HIP_CHECK(hipSetDevice(std::max(0, s.gpu_id)));
    HIP_CHECK(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking));
    HIP_CHECK(hipStreamCreateWithFlags(&prod_stream_, hipStreamNonBlocking));
    HIP_CHECK(hipStreamCreateWithFlags(&fix_stream_, hipStreamNonBlocking));
    for (int i = 0; i < 2; ++i) {
      HIP_CHECK(hipEventCreateWithFlags(&jev_dct_[i],
                                        hipEventDisableTiming));
      HIP_CHECK(hipEventCreateWithFlags(&jev_done_[i],
                                        hipEventDisableTiming));
    }
    upload_dct_tables(stream_);
    HIP_CHECK(hipMalloc(&d_rqy_, 64 * sizeof(float)));
    HIP_CHECK(hipMalloc(&d_rqc_, 64 * sizeof(float)));
    cpu_jpeg_entropy_ = std::getenv("HIPFLUX_CPU_JPEG_ENTROPY") != nullptr;
    {
      uint32_t tabs[12 + 256 + 12 + 256];
      jpeg_export_huff(tabs, tabs + 12, tabs + 268, tabs + 280);
      HIP_CHECK(hipMalloc(&d_jtabs_, sizeof(tabs)));
      HIP_CHECK(hipMemcpy(d_jtabs_, tabs, sizeof(tabs),
                          hipMemcpyHostToDevice));
    }
    alloc_for(s.capture_width, s.capture_height);
  }

  ~HipJpegPipeline() override {
    (void)hipStreamSynchronize(stream_);
    (void)hipStreamSynchronize(prod_stream_);
    (void)hipStreamSynchronize(fix_stream_);
    (void)hipStreamDestroy(prod_stream_);
    (void)hipStreamDestroy(fix_stream_);
    for (int i = 0; i < 2; ++i) {
      (void)hipEventDestroy(jev_dct_[i]);
      (void)hipEventDestroy(jev_done_[i]);
    }
    for (auto& kv : registered_) (void)hipHostUnregister(kv.first);
    if (d_jtabs_) (void)hipFree(d_jtabs_);
    if (d_jstage_) (void)hipFree(d_jstage_);
    if (d_jnbits_) (void)hipFree(d_jnbits_);
    for (int i = 0; i < 2; ++i) {
      if (d_jrows2_[i]) (void)hipFree(d_jrows2_[i]);
      if (d_jout2_[i]) (void)hipFree(d_jout2_[i]);
      if (d_joutbits2_[i]) (void)hipFree(d_joutbits2_[i]);
      if (h_jout2_[i]) (void)hipHostFree(h_jout2_[i]);
      if (h_joutbits2_[i]) (void)hipHostFree(h_joutbits2_[i]);
      if (h_jrows2_[i]) (void)hipHostFree(h_jrows2_[i]);
      if (d_frame2_[i]) (void)hipFree(d_frame2_[i]);
      if (d_coeff2_[i]) (void)hipFree(d_coeff2_[i]);
    }
    if (d_y_) (void)hipFree(d_y_);
    if (d_cb_) (void)hipFree(d_cb_);
    if (d_cr_) (void)hipFree(d_cr_);
    if (d_rqy_) (void)hipFree(d_rqy_);
    if (d_rqc_) (void)hipFree(d_rqc_);
    if (h_coeff_) (void)hipHostFree(h_coeff_);
  }

  struct JOut {
    std::vector<uint8_t> bytes;
    int y0 = 0, h = 0;
    int row0 = 0, rows = 0;     // MCU-row range (GPU entropy path)
    bool encode = false;
  };
  struct JPending {
    bool active = false;
    int par = 0;
    int n_rows = 0;
    int copy_words = 0;
    int quality = 80;
    bool fullcolor = false;
    uint32_t frame_id = 0;
    int mcux = 0;
    std::vector<JOut> outs;
  };

  JPending submit_frame(const RawFrame& frame, const FrameContext& ctx) {
    const bool fullcolor = settings_.video_fullcolor;
    if (frame.width != w_ || frame.height != h_)
      alloc_for(frame.width, frame.height);
    ensure_quality(ctx.jpeg_quality);
    const int par = jparity_;

    // upload (register the capture buffer once; zero-copy DMA afterwards)
    const uint8_t* src = frame.data;
    size_t frame_bytes = static_cast<size_t>(frame.stride) * frame.height;
    if (!registered_.count(const_cast<uint8_t*>(src))) {
      hipError_t e = hipHostRegister(const_cast<uint8_t*>(src), frame_bytes,
                                     hipHostRegisterDefault);
      registered_[const_cast<uint8_t*>(src)] = (e == hipSuccess);
      if (e != hipSuccess) (void)hipGetLastError();  // clear; fall back
    }
    // produce chain (upload, CSC, DCT) on its own stream so it overlaps
    // the previous frame's entropy kernel + readback on stream_
    HIP_CHECK(hipMemcpyAsync(d_frame2_[par], src, frame_bytes,
                             hipMemcpyHostToDevice, prod_stream_));

    const int stride_px = frame.stride / 4;
    launch_bgrx_to_planes(d_frame2_[par], w_, h_, stride_px, d_y_, d_cb_,
                          d_cr_, ypitch_, cpitch_, fullcolor, prod_stream_);

    const int stripe_h = std::max(16, settings_.stripe_height & ~15);
    const int mcu = fullcolor ? 8 : 16;
    const int mcux = (w_ + mcu - 1) / mcu;
    const int mcuy = (h_ + mcu - 1) / mcu;
    const int rows_per_stripe = stripe_h / mcu;
    const int stripe_mcu_count = rows_per_stripe * mcux;
    const int per_mcu_real = fullcolor ? 3 : 6;

    int16_t* d_coeff = d_coeff2_[par];
    if (!fullcolor) {
      launch_dct_quant(d_y_, w_, h_, ypitch_, d_rqy_, d_coeff, 0, false,
                       mcux, rows_per_stripe, stripe_mcu_count,
                       prod_stream_);
      launch_dct_quant(d_cb_, cw_, ch_, cpitch_, d_rqc_, d_coeff, 1, false,
                       mcux, rows_per_stripe, stripe_mcu_count,
                       prod_stream_);
      launch_dct_quant(d_cr_, cw_, ch_, cpitch_, d_rqc_, d_coeff, 2, false,
                       mcux, rows_per_stripe, stripe_mcu_count,
                       prod_stream_);
    } else {
      launch_dct_quant(d_y_, w_, h_, ypitch_, d_rqy_, d_coeff, 0, true,
                       mcux, rows_per_stripe, stripe_mcu_count,
                       prod_stream_);
      launch_dct_quant(d_cb_, w_, h_, ypitch_, d_rqc_, d_coeff, 1, true,
                       mcux, rows_per_stripe, stripe_mcu_count,
                       prod_stream_);
      launch_dct_quant(d_cr_, w_, h_, ypitch_, d_rqc_, d_coeff, 2, true,
                       mcux, rows_per_stripe, stripe_mcu_count,
                       prod_stream_);
    }
    HIP_CHECK(hipEventRecord(jev_dct_[par], prod_stream_));

    JPending pd;
    pd.par = par;
    pd.quality = ctx.jpeg_quality;
    pd.fullcolor = fullcolor;
    pd.frame_id = ctx.frame_id;
    pd.mcux = mcux;
    pd.outs.resize(ctx.stripes.size());
    for (size_t i = 0; i < ctx.stripes.size(); ++i) {
      const auto& job = ctx.stripes[i];
      pd.outs[i].y0 = job.y0;
      pd.outs[i].h = job.y1 - job.y0;
      pd.outs[i].encode = job.encode;
    }

    if (cpu_jpeg_entropy_) {
      // test/debug path: stays synchronous (depth-1 semantics)
      size_t coeff_count =
          static_cast<size_t>(mcux) * mcuy * per_mcu_real * 64;
      HIP_CHECK(hipMemcpyAsync(h_coeff_, d_coeff,
                               coeff_count * sizeof(int16_t),
                               hipMemcpyDeviceToHost, prod_stream_));
      HIP_CHECK(hipStreamSynchronize(prod_stream_));
      for (size_t i = 0; i < pd.outs.size(); ++i) {
        if (!pd.outs[i].encode) continue;
        pool_.submit([&, i] {
          const JOut& o = pd.outs[i];
          int stripe_idx = o.y0 / stripe_h;
          int rows = std::min(rows_per_stripe,
                              mcuy - stripe_idx * rows_per_stripe);
          const int16_t* blocks =
              h_coeff_ + static_cast<size_t>(stripe_idx) *
                             stripe_mcu_count * per_mcu_real * 64;
          jpeg_entropy_from_blocks(blocks, mcux, rows, w_, o.h,
                                   pd.quality, fullcolor,
                                   pd.outs[i].bytes, true);
        });
      }
      pool_.wait_all();
      pd.active = true;
      pd.n_rows = 0;              // bytes already assembled
      jparity_ ^= 1;
      return pd;
    }

    // GPU entropy on stream_ behind the DCT event; one kernel row per
    // MCU row of every scheduled stripe
    auto* jrows = static_cast<jpeggpu::JRow*>(h_jrows2_[par]);
    int n_rows = 0;
    for (size_t i = 0; i < pd.outs.size(); ++i) {
      if (!pd.outs[i].encode) continue;
      int stripe_idx = pd.outs[i].y0 / stripe_h;
      int rows = std::min(rows_per_stripe,
                          mcuy - stripe_idx * rows_per_stripe);
      pd.outs[i].row0 = n_rows;
      pd.outs[i].rows = rows;
      size_t base = static_cast<size_t>(stripe_idx) * stripe_mcu_count *
                    per_mcu_real * 64;
      for (int r = 0; r < rows; ++r)
        jrows[n_rows++].coeff_off = static_cast<int>(
            base + static_cast<size_t>(r) * mcux * per_mcu_real * 64);
    }
    pd.n_rows = n_rows;
    pd.active = true;
    jparity_ ^= 1;
    if (!n_rows) return pd;

    HIP_CHECK(hipStreamWaitEvent(stream_, jev_dct_[par], 0));
    HIP_CHECK(hipMemcpyAsync(d_jrows2_[par], h_jrows2_[par],
                             sizeof(jpeggpu::JRow) * n_rows,
                             hipMemcpyHostToDevice, stream_));
    jpeggpu::launch_jpeg_entropy(
        d_coeff, static_cast<jpeggpu::JRow*>(d_jrows2_[par]), n_rows, mcux,
        per_mcu_real, d_jtabs_, static_cast<uint32_t*>(d_jstage_),
        static_cast<int*>(d_jnbits_), static_cast<uint32_t*>(d_jout2_[par]),
        jout_stride_, static_cast<int*>(d_joutbits2_[par]), stream_);
    // adaptive compacted readback (2x last frame's max row)
    pd.copy_words = std::min(jout_stride_, jout_copy_words_);
    if (pd.copy_words >= jout_stride_) {
      HIP_CHECK(hipMemcpyAsync(h_jout2_[par], d_jout2_[par],
                               (size_t)n_rows * jout_stride_ * 4,
                               hipMemcpyDeviceToHost, stream_));
    } else {
      HIP_CHECK(hipMemcpy2DAsync(
          h_jout2_[par], (size_t)jout_stride_ * 4, d_jout2_[par],
          (size_t)jout_stride_ * 4, (size_t)pd.copy_words * 4,
          n_rows, hipMemcpyDeviceToHost, stream_));
    }
    HIP_CHECK(hipMemcpyAsync(h_joutbits2_[par], d_joutbits2_[par],
                             4 * n_rows, hipMemcpyDeviceToHost, stream_));
    HIP_CHECK(hipEventRecord(jev_done_[par], stream_));
    return pd;
  }

  void collect_pending(JPending& pd, const Emit& emit) {
    if (!pd.active) return;
    pd.active = false;
    const int par = pd.par;
    if (!cpu_jpeg_entropy_ && pd.n_rows) {
      HIP_CHECK(hipEventSynchronize(jev_done_[par]));
      auto* outbits = static_cast<int*>(h_joutbits2_[par]);
      int max_words = 0;
      bool fix = false;
      for (int r = 0; r < pd.n_rows; ++r) {
        int wds = (outbits[r] + 31) / 32 + 1;
        max_words = std::max(max_words, wds);
        if (wds > pd.copy_words) {
          HIP_CHECK(hipMemcpyAsync(
              static_cast<uint32_t*>(h_jout2_[par]) +
                  (size_t)r * jout_stride_,
              static_cast<uint32_t*>(d_jout2_[par]) +
                  (size_t)r * jout_stride_,
              (size_t)wds * 4, hipMemcpyDeviceToHost, fix_stream_));
          fix = true;
        }
      }
      if (fix) HIP_CHECK(hipStreamSynchronize(fix_stream_));
      jout_copy_words_ = std::min(jout_stride_, max_words * 2 + 64);
      for (size_t i = 0; i < pd.outs.size(); ++i) {
        if (!pd.outs[i].encode) continue;
        pool_.submit([&, i, par, outbits] {
          JOut& o = pd.outs[i];
          jpeg_write_headers(o.bytes, w_, o.h, pd.quality,
                             pd.fullcolor, pd.mcux);
          int rst = 0;
          for (int r = 0; r < o.rows; ++r) {
            if (r > 0) {
              o.bytes.push_back(0xFF);
              o.bytes.push_back(
                  static_cast<uint8_t>(0xD0 + (rst++ & 7)));
            }
            const uint32_t* words =
                static_cast<uint32_t*>(h_jout2_[par]) +
                (size_t)(o.row0 + r) * jout_stride_;
            jpeg_append_row_bits(words, outbits[o.row0 + r], o.bytes);
          }
          o.bytes.push_back(0xFF);
          o.bytes.push_back(0xD9);
        });
      }
      pool_.wait_all();
    }
    for (auto& o : pd.outs) {
      if (!o.encode) continue;
      EncodedStripe s;
      s.type = StripeType::kJpeg;
      s.data = o.bytes.data();
      s.size = o.bytes.size();
      s.frame_id = pd.frame_id;
      s.y = o.y0;
      s.width = w_;
      s.height = o.h;
      s.is_keyframe = true;
      emit(s);
    }
  }

  void encode_frame(const RawFrame& frame, const FrameContext& ctx,
                    const Emit& emit) override {
    JPending cur = submit_frame(frame, ctx);
    if (jdepth_ < 2 || cpu_jpeg_entropy_) {
      collect_pending(cur, emit);
      collect_pending(jprev_, emit);   // drain any leftover from a mode flip
      return;
    }
    collect_pending(jprev_, emit);
    jprev_ = std::move(cur);
    // the caller may reuse its frame buffer once the H2D completes; the
    // DCT event is recorded after CSC+DCT which transitively covers it
    HIP_CHECK(hipEventSynchronize(jev_dct_[jprev_.par]));
  }

  void flush(const Emit& emit) override { collect_pending(jprev_, emit); }

  void set_pipeline_depth(int d) override {
    jdepth_ = std::max(1, std::min(2, d));
  }

  const char* name() const override { return "hip-jpeg"; }

 private:
  void alloc_for(int w, int h) {
    HIP_CHECK(hipStreamSynchronize(stream_));
    HIP_CHECK(hipStreamSynchronize(prod_stream_));
    jprev_ = JPending{};
    jparity_ = 0;
    for (int i = 0; i < 2; ++i) {
      if (d_frame2_[i]) (void)hipFree(d_frame2_[i]);
      if (d_coeff2_[i]) (void)hipFree(d_coeff2_[i]);
      d_frame2_[i] = nullptr;
      d_coeff2_[i] = nullptr;
    }
    if (d_y_) (void)hipFree(d_y_);
    if (d_cb_) (void)hipFree(d_cb_);
    if (d_cr_) (void)hipFree(d_cr_);
    if (h_coeff_) (void)hipHostFree(h_coeff_);
    w_ = w;
    h_ = h;
    cw_ = (w + 1) / 2;
    ch_ = (h + 1) / 2;
    ypitch_ = (w + 255) & ~255;
    cpitch_ = (cw_ + 255) & ~255;
    for (int i = 0; i < 2; ++i)
      HIP_CHECK(hipMalloc(&d_frame2_[i], static_cast<size_t>(w) * h * 4));
    HIP_CHECK(hipMalloc(&d_y_, static_cast<size_t>(ypitch_) * h));
    // chroma buffers sized for 4:4:4 (the larger case)
    HIP_CHECK(hipMalloc(&d_cb_, static_cast<size_t>(ypitch_) * h));
    HIP_CHECK(hipMalloc(&d_cr_, static_cast<size_t>(ypitch_) * h));
    // coefficients: 6 blocks/MCU covers both 420 (6) and 444 (3)
    size_t mcux = (w + 7) / 8, mcuy = (h + 7) / 8;  // worst case 444
    size_t coeff_bytes = mcux * mcuy * 6 * 64 * sizeof(int16_t);
    for (int i = 0; i < 2; ++i)
      HIP_CHECK(hipMalloc(&d_coeff2_[i], coeff_bytes));
    HIP_CHECK(hipHostMalloc(&h_coeff_, coeff_bytes, hipHostMallocDefault));
    // GPU entropy buffers: one slot per MCU row (444 row count is the max)
    if (d_jstage_) (void)hipFree(d_jstage_);
    if (d_jnbits_) (void)hipFree(d_jnbits_);
    for (int i = 0; i < 2; ++i) {
      if (d_jrows2_[i]) (void)hipFree(d_jrows2_[i]);
      if (d_jout2_[i]) (void)hipFree(d_jout2_[i]);
      if (d_joutbits2_[i]) (void)hipFree(d_joutbits2_[i]);
      if (h_jout2_[i]) (void)hipHostFree(h_jout2_[i]);
      if (h_joutbits2_[i]) (void)hipHostFree(h_joutbits2_[i]);
      if (h_jrows2_[i]) (void)hipHostFree(h_jrows2_[i]);
      d_jrows2_[i] = d_jout2_[i] = d_joutbits2_[i] = nullptr;
      h_jout2_[i] = h_joutbits2_[i] = h_jrows2_[i] = nullptr;
    }
    size_t max_rows = mcuy;
    size_t max_items = mcux * 3 > ((w + 15) / 16) * 6
                           ? mcux * 3
                           : ((size_t)(w + 15) / 16) * 6;
    jout_stride_ = (int)(max_items * jpeggpu::kJStageWords);
    HIP_CHECK(hipMalloc(&d_jstage_, max_rows * max_items *
                                        jpeggpu::kJStageWords * 4));
    HIP_CHECK(hipMalloc(&d_jnbits_, max_rows * max_items * 4));
    for (int i = 0; i < 2; ++i) {
      HIP_CHECK(hipMalloc(&d_jrows2_[i],
                          sizeof(jpeggpu::JRow) * max_rows));
      HIP_CHECK(hipMalloc(&d_jout2_[i],
                          max_rows * (size_t)jout_stride_ * 4));
      HIP_CHECK(hipMalloc(&d_joutbits2_[i], max_rows * 4));
      HIP_CHECK(hipHostMalloc(&h_jout2_[i],
                              max_rows * (size_t)jout_stride_ * 4,
                              hipHostMallocDefault));
      HIP_CHECK(hipHostMalloc(&h_joutbits2_[i], max_rows * 4,
                              hipHostMallocDefault));
      HIP_CHECK(hipHostMalloc(&h_jrows2_[i],
                              sizeof(jpeggpu::JRow) * max_rows,
                              hipHostMallocDefault));
    }
    jout_copy_words_ = 1 << 30;
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
    // quality changes are rare: drain BOTH streams so an in-flight
    // frame never sees half-updated tables, then upload on the producer
    HIP_CHECK(hipStreamSynchronize(stream_));
    HIP_CHECK(hipStreamSynchronize(prod_stream_));
    HIP_CHECK(hipMemcpyAsync(d_rqy_, rqy, sizeof(rqy), hipMemcpyHostToDevice,
                             prod_stream_));
    HIP_CHECK(hipMemcpyAsync(d_rqc_, rqc, sizeof(rqc), hipMemcpyHostToDevice,
                             prod_stream_));
    HIP_CHECK(hipStreamSynchronize(prod_stream_));
    cur_quality_ = q;
  }

  CaptureSettings settings_;
  ThreadPool pool_;
  hipStream_t stream_{}, prod_stream_{}, fix_stream_{};
  hipEvent_t jev_dct_[2] = {}, jev_done_[2] = {};
  int jparity_ = 0;
  int jdepth_ = 1;
  JPending jprev_;
  uint8_t* d_frame2_[2] = {};
  uint8_t* d_y_ = nullptr;
  uint8_t* d_cb_ = nullptr;
  uint8_t* d_cr_ = nullptr;
  int16_t* d_coeff2_[2] = {};
  uint32_t* d_jtabs_ = nullptr;
  void* d_jrows2_[2] = {};
  void* d_jstage_ = nullptr;
  void* d_jnbits_ = nullptr;
  void* d_jout2_[2] = {};
  void* d_joutbits2_[2] = {};
  void* h_jout2_[2] = {};
  void* h_joutbits2_[2] = {};
  void* h_jrows2_[2] = {};
  int jout_stride_ = 0;
  int jout_copy_words_ = 1 << 30;
  bool cpu_jpeg_entropy_ = false;
  int16_t* h_coeff_ = nullptr;
  float* d_rqy_ = nullptr;
  float* d_rqc_ = nullptr;
  int w_ = 0, h_ = 0, cw_ = 0, ch_ = 0, ypitch_ = 0, cpitch_ = 0;
  int cur_quality_ = -1;
  std::map<void*, bool> registered_;
};

// ---------------------------------------------------------------------------
// HIP H.264 pipeline: CSC + per-MB-row intra/inter kernels on the GPU,
// stripe-parallel CAVLC packing on the CPU. Each stripe is an independent
// bitstream with its own frame_num / recon region (SURVEY.md §5.7 seam).
class HipH264Pipeline : public EncodePipeline {
 public:
  explicit HipH264Pipeline(const CaptureSettings& s)
      : settings_(s),
        pool_(std::min(16u,
                       std::max(2u, std::thread::hardware_concurrency() / 2))) {
    // pool capped at 16: the per-frame batches are small, and waking a
    // hardware_concurrency/2-sized pool (128 threads on a 256-core box)
    // costs more in cv wakeups than the work itself
    HIP_CHECK(hipSetDevice(std::max(0, s.gpu_id)));
    HIP_CHECK(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking));
    HIP_CHECK(hipStreamCreateWithFlags(&up_stream_, hipStreamNonBlocking));
    HIP_CHECK(hipStreamCreateWithFlags(&db_stream_, hipStreamNonBlocking));
    HIP_CHECK(hipStreamCreateWithFlags(&rows_stream_, hipStreamNonBlocking));
    for (int i = 0; i < 2; ++i) {
      HIP_CHECK(hipEventCreateWithFlags(&ev_rows_[i],
                                        hipEventDisableTiming));
      HIP_CHECK(hipEventCreateWithFlags(&ev_db_[i], hipEventDisableTiming));
    }
    stripe_h_ = std::max(16, s.stripe_height & ~15);
    batch_events_.resize(4);
    for (auto& e : batch_events_)
      HIP_CHECK(hipEventCreateWithFlags(&e, hipEventDisableTiming));
    for (int i = 0; i < 2; ++i) {
      HIP_CHECK(hipEventCreateWithFlags(&ev_done_[i],
                                        hipEventDisableTiming));
      HIP_CHECK(hipEventCreateWithFlags(&ev_h2d_[i],
                                        hipEventDisableTiming));
    }
    alloc_for(s.capture_width, s.capture_height);
  }

  ~HipH264Pipeline() override {
    (void)hipStreamSynchronize(stream_);
    (void)hipStreamSynchronize(up_stream_);
    (void)hipStreamDestroy(up_stream_);
    (void)hipStreamSynchronize(db_stream_);
    (void)hipStreamDestroy(db_stream_);
    (void)hipStreamSynchronize(rows_stream_);
    (void)hipStreamDestroy(rows_stream_);
    for (int i = 0; i < 2; ++i) {
      (void)hipEventDestroy(ev_rows_[i]);
      (void)hipEventDestroy(ev_db_[i]);
    }
    for (auto& e : batch_events_) (void)hipEventDestroy(e);
    for (int i = 0; i < 2; ++i) {
      (void)hipEventDestroy(ev_done_[i]);
      (void)hipEventDestroy(ev_h2d_[i]);
    }
    for (auto& kv : registered_) (void)hipHostUnregister(kv.first);
    for (void* p : device_ptrs_)
      if (p) (void)hipFree(p);
    if (h_levels_) (void)hipHostFree(h_levels_);
    if (h_meta_) (void)hipHostFree(h_meta_);
    if (h_stage_) (void)hipHostFree(h_stage_);
    for (int i = 0; i < 2; ++i) {
      if (h_jobs_[i]) (void)hipHostFree(h_jobs_[i]);
      if (h_entout_[i]) (void)hipHostFree(h_entout_[i]);
      if (h_outbits_[i]) (void)hipHostFree(h_outbits_[i]);
    }
  }

  // ---- frame pipeline -----------------------------------------------------
  // submit_frame enqueues one frame's full GPU sequence (upload -> CSC ->
  // pyramid -> ME -> rows -> CAVLC -> compacted D2H -> ref refresh) with
  // NO sync; collect_pending waits on the frame's completion event and
  // does the host-side bitstream assembly + emission. Depth 1 runs them
  // back to back (today's latency behavior, bit-identical); depth 2 keeps
  // one frame in flight so host assembly of frame N overlaps the GPU
  // compute of frame N+1 (x264-style frame pipelining; throughput mode).
  struct Out {
    std::vector<uint8_t> header;            // SPS/PPS on IDR
    std::vector<std::vector<uint8_t>> rows;
    std::vector<uint8_t> bytes;
    h264::GpuStripeParams p{};
    int y0 = 0, h = 0;
    bool idr = false;
  };
  struct Batch {
    int job0, jobn;        // job index range
    int row0, rown;        // absolute MB row range [row0, rown)
  };
  struct Pending {
    bool active = false;
    int par = 0;
    int n_jobs = 0;
    int copy_words = 0;    // the D2H cap this frame was submitted with
    double ts_ms = 0;      // capture timestamp of this frame
    uint32_t frame_id = 0;
    std::vector<Out> outs;
    std::vector<std::pair<int, int>> job_map;
    std::vector<Batch> batches;
  };

  Pending submit_frame(const RawFrame& frame, const FrameContext& ctx) {
    if (frame.width != w_ || frame.height != h_)
      alloc_for(frame.width, frame.height);
    const int par = parity_;
    const int qp = std::min(51, std::max(0, ctx.crf));

    // upload + CSC. Prefer zero-copy DMA from the (registered) capture
    // buffer; if registration failed, stage through a pinned buffer — an
    // unpinned async copy silently degrades to a slow sync path.
    const uint8_t* src = frame.data;
    size_t frame_bytes = static_cast<size_t>(frame.stride) * frame.height;
    auto reg_it = registered_.find(const_cast<uint8_t*>(src));
    if (reg_it == registered_.end()) {
      hipError_t e = hipHostRegister(const_cast<uint8_t*>(src), frame_bytes,
                                     hipHostRegisterDefault);
      if (e != hipSuccess) (void)hipGetLastError();
      reg_it = registered_.emplace(const_cast<uint8_t*>(src),
                                   e == hipSuccess).first;
    }
    if (!reg_it->second) {
      if (h_stage_bytes_ < frame_bytes) {
        if (h_stage_) (void)hipHostFree(h_stage_);
        HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&h_stage_),
                                frame_bytes, hipHostMallocDefault));
        h_stage_bytes_ = frame_bytes;
      }
      std::memcpy(h_stage_, src, frame_bytes);
      src = h_stage_;
    }
    // the ~8 MB BGRX upload rides its own stream so the SDMA engine
    // overlaps it with the previous frame's kernels (measured ~120 us
    // serialized on the compute stream); the compute stream's CSC waits
    // on the copy via event
    HIP_CHECK(hipMemcpyAsync(d_frame_[par], src, frame_bytes,
                             hipMemcpyHostToDevice, up_stream_));
    // the caller may reuse its frame buffer once this event completes
    HIP_CHECK(hipEventRecord(ev_h2d_[par], up_stream_));
    // the whole produce chain (CSC -> ME -> rows) rides rows_stream_ so
    // it overlaps the PREVIOUS frame's entropy kernel + D2H on stream_
    HIP_CHECK(hipStreamWaitEvent(rows_stream_, ev_h2d_[par], 0));
    launch_bgrx_to_planes(d_frame_[par], w_, h_, frame.stride / 4, d_srcY_,
                          d_srcCb_, d_srcCr_, ypitch_, cpitch_, false,
                          rows_stream_);

    // build row jobs for scheduled stripes (stripe stream state — IDR,
    // frame_num, idr_pic_id — resolves here so the GPU entropy kernel gets
    // final slice-header fields)
    struct SJob {
      int idx;       // stripe index
      int y0, y1;    // pixel bounds (logical)
      bool idr;
      uint32_t frame_num, idr_pic_id;
    };
    std::vector<SJob> sjobs;
    int n_jobs = 0;
    for (size_t i = 0; i < ctx.stripes.size(); ++i) {
      const auto& st = ctx.stripes[i];
      if (!st.encode) continue;
      auto& state = stripes_[st.y0 / stripe_h_];
      bool idr = ctx.idr || state.need_idr;
      if (idr) {
        state.frame_num = 0;
        ++state.idr_pic_id;
        state.need_idr = false;
      }
      SJob sj{static_cast<int>(st.y0 / stripe_h_), st.y0, st.y1, idr,
              state.frame_num, state.idr_pic_id};
      ++state.frame_num;
      sjobs.push_back(sj);
      int row0 = st.y0 / 16;
      int rows = (std::min(st.y1, mbh_ * 16) - st.y0 + 15) / 16;
      // stripe's padded pixel bounds for ME clamping
      int sy0 = st.y0, sy1 = (row0 + rows) * 16;
      for (int r = 0; r < rows; ++r) {
        int x = 0;
        for (int sg = 0; sg < segs_; ++sg) {
          int segw = std::min(seg_w0_, mbw_ - x);
          h_jobs_[par][n_jobs].mb_row = row0 + r;
          h_jobs_[par][n_jobs].qp = qp;
          h_jobs_[par][n_jobs].flags =
              (sj.idr ? 1 : 0) | (settings_.video_deblock ? 2 : 0);
          h_jobs_[par][n_jobs].stripe_y0 = sy0;
          h_jobs_[par][n_jobs].stripe_y1 = sy1;
          h_jobs_[par][n_jobs].first_mb = r * mbw_ + x;
          h_jobs_[par][n_jobs].frame_num = static_cast<int>(sj.frame_num);
          h_jobs_[par][n_jobs].idr_pic_id = static_cast<int>(sj.idr_pic_id);
          h_jobs_[par][n_jobs].mbx0 = x;
          h_jobs_[par][n_jobs].seg_mbw = segw;
          x += segw;
          ++n_jobs;
        }
      }
    }
    if (n_jobs == 0) return Pending{};
    parity_ ^= 1;

    HIP_CHECK(hipMemcpyAsync(d_jobs_[par], h_jobs_[par],
                             sizeof(h264gpu::RowJob) * n_jobs,
                             hipMemcpyHostToDevice, rows_stream_));
    // meta carries cross-frame tracking state (prev MV hints, hopeless
    // bits); seed this parity's buffer from the previous one so the
    // semantics are identical to the single-buffer design even for
    // damage-skipped stripes
    HIP_CHECK(hipMemcpyAsync(d_meta_[par], d_meta_[par ^ 1], meta_bytes_,
                             hipMemcpyDeviceToDevice, rows_stream_));
    // luma pyramid (quarter res) for ME acquisition
    if (!ctx.idr) {
      h264gpu::launch_downsample2(d_srcY_, ypitch_, ypitch_, mbh_ * 16,
                                  d_mip1_, ypitch_ / 2, rows_stream_);
      h264gpu::launch_downsample2(d_mip1_, ypitch_ / 2, ypitch_ / 2,
                                  mbh_ * 8, d_srcY2_, ypitch_ / 4,
                                  rows_stream_);
      h264gpu::launch_downsample2(d_refY_, ypitch_, ypitch_, mbh_ * 16,
                                  d_mip1_, ypitch_ / 2, rows_stream_);
      h264gpu::launch_downsample2(d_mip1_, ypitch_ / 2, ypitch_ / 2,
                                  mbh_ * 8, d_refY2_, ypitch_ / 4,
                                  rows_stream_);
    }
    h264gpu::launch_h264_me(d_srcY_, ypitch_, w_, h_, d_refY_, d_srcY2_,
                            d_refY2_, ypitch_ / 4, mbw_, n_jobs,
                            d_jobs_[par], d_meta_[par], rows_stream_);

    // Single batch: the row kernel's cost is per-row LATENCY (all rows run
    // concurrently), so splitting into sequential batches multiplies GPU
    // time (measured: 4 batches regressed 278->155 fps). Entropy overlap
    // is cross-FRAME instead (depth-2 pipelining).
    const int kBatches = 1;
    std::vector<Batch> batches;
    for (int b = 0; b < kBatches; ++b) {
      int j0 = n_jobs * b / kBatches, j1 = n_jobs * (b + 1) / kBatches;
      if (j0 == j1) continue;
      batches.push_back({j0, j1, h_jobs_[par][j0].mb_row,
                         h_jobs_[par][j1 - 1].mb_row + 1});
    }
    const size_t lvl_row =
        static_cast<size_t>(mbw_) * h264gpu::kLevelsPerMb * sizeof(int16_t);
    const size_t meta_row =
        static_cast<size_t>(mbw_) * h264gpu::kMetaPerMb * sizeof(int);
    for (size_t b = 0; b < batches.size(); ++b) {
      const Batch& bt = batches[b];
      // 4-wave row kernel by default (issue-rate-bound chain split across
      // waves); HIPFLUX_ROWS_V1=1 falls back to the 2-wave variant
      if (rows_v1_)
        h264gpu::launch_h264_rows(
            d_srcY_, d_srcCb_, d_srcCr_, ypitch_, cpitch_, w_, h_, d_refY_,
            d_refCb_, d_refCr_, d_curY_, d_curCb_, d_curCr_, mbw_,
            bt.jobn - bt.job0, d_jobs_[par] + bt.job0, d_levels_[par],
            d_meta_[par], rows_stream_);
      else
        h264gpu::launch_h264_rows4(
            d_srcY_, d_srcCb_, d_srcCr_, ypitch_, cpitch_, w_, h_, d_refY_,
            d_refCb_, d_refCr_, d_curY_, d_curCb_, d_curCr_, mbw_,
            bt.jobn - bt.job0, d_jobs_[par] + bt.job0, d_levels_[par],
            d_meta_[par], rows_stream_);
      if (cpu_entropy_) {
        HIP_CHECK(hipMemcpyAsync(
            reinterpret_cast<uint8_t*>(h_levels_) + bt.row0 * lvl_row,
            reinterpret_cast<uint8_t*>(d_levels_[par]) + bt.row0 * lvl_row,
            (bt.rown - bt.row0) * lvl_row, hipMemcpyDeviceToHost,
            rows_stream_));
        HIP_CHECK(hipMemcpyAsync(
            reinterpret_cast<uint8_t*>(h_meta_) + bt.row0 * meta_row,
            reinterpret_cast<uint8_t*>(d_meta_[par]) + bt.row0 * meta_row,
            (bt.rown - bt.row0) * meta_row, hipMemcpyDeviceToHost,
            rows_stream_));
      }
      HIP_CHECK(hipEventRecord(batch_events_[b], rows_stream_));
    }
    HIP_CHECK(hipEventRecord(ev_rows_[par], rows_stream_));
    // in-loop deblock of the current recon (within-slice edges only;
    // idc=2 is signaled in the slice headers) before it becomes the
    // reference frame. Deblock (reads recon+levels) and CAVLC (reads
    // levels) are independent, so deblock runs on its own stream
    // concurrently with entropy + bitstream D2H; the cur->ref copies
    // below join both streams before the recon becomes the reference.
    const bool forked_db = settings_.video_deblock;
    if (settings_.video_deblock) {
      HIP_CHECK(hipStreamWaitEvent(db_stream_, ev_rows_[par], 0));
      h264gpu::launch_h264_deblock(d_curY_, d_curCb_, d_curCr_, ypitch_,
                                   cpitch_, mbw_, n_jobs, d_jobs_[par],
                                   d_levels_[par], d_meta_[par],
                                   db_stream_);
      HIP_CHECK(hipEventRecord(ev_db_[par], db_stream_));
    }
    const int copy_words = ent_copy_words_;
    if (!cpu_entropy_) {
      // entropy rides stream_ so the NEXT frame's produce chain on
      // rows_stream_ overlaps it (levels/meta are parity-buffered)
      HIP_CHECK(hipStreamWaitEvent(stream_, ev_rows_[par], 0));
      h264gpu::launch_h264_cavlc(d_levels_[par], d_meta_[par], mbw_,
                                 n_jobs, d_jobs_[par], d_stage_, d_nbits_,
                                 d_entout_[par], ent_stride_words_,
                                 d_outbits_[par], stream_);
      // compaction: copy only ~the used prefix of each row's bitstream
      // (adaptive cap = 2x last frame's max row, full stride first frame;
      // rows that overflow the cap are re-copied exactly in collect)
      if (copy_words >= ent_stride_words_) {
        HIP_CHECK(hipMemcpyAsync(h_entout_[par], d_entout_[par],
                                 (size_t)n_jobs * ent_stride_words_ * 4,
                                 hipMemcpyDeviceToHost, stream_));
      } else {
        HIP_CHECK(hipMemcpy2DAsync(
            h_entout_[par], (size_t)ent_stride_words_ * 4, d_entout_[par],
            (size_t)ent_stride_words_ * 4, (size_t)copy_words * 4,
            n_jobs, hipMemcpyDeviceToHost, stream_));
      }
      HIP_CHECK(hipMemcpyAsync(h_outbits_[par], d_outbits_[par],
                               sizeof(int) * n_jobs, hipMemcpyDeviceToHost,
                               stream_));
    }
    // refresh ref from cur for encoded stripes (merge contiguous spans
    // so the all-stripes case is 3 copies, not 3 x n_stripes). They ride
    // rows_stream_ (gated on deblock) so the NEXT frame's rows see the
    // filtered reference without waiting on this frame's entropy.
    if (forked_db)
      HIP_CHECK(hipStreamWaitEvent(rows_stream_, ev_db_[par], 0));
    {
      int span_y0 = -1, span_y1 = -1;
      auto flush_span = [&] {
        if (span_y0 < 0) return;
        int rows16 = (std::min(span_y1, mbh_ * 16) - span_y0 + 15) / 16;
        copy_region(d_refY_, d_curY_, ypitch_, span_y0, rows16 * 16);
        copy_region(d_refCb_, d_curCb_, cpitch_, span_y0 / 2, rows16 * 8);
        copy_region(d_refCr_, d_curCr_, cpitch_, span_y0 / 2, rows16 * 8);
        span_y0 = span_y1 = -1;
      };
      for (const auto& sj : sjobs) {
        if (span_y1 == sj.y0) {
          span_y1 = sj.y1;
        } else {
          flush_span();
          span_y0 = sj.y0;
          span_y1 = sj.y1;
        }
      }
      flush_span();
    }
    HIP_CHECK(hipEventRecord(ev_done_[par],
                             cpu_entropy_ ? rows_stream_ : stream_));

    // host-side output prep (headers, job -> stripe/row mapping)
    Pending pd;
    pd.active = true;
    pd.par = par;
    pd.n_jobs = n_jobs;
    pd.copy_words = copy_words;
    pd.ts_ms = frame.ts_ms;
    pd.frame_id = ctx.frame_id;
    pd.batches = std::move(batches);
    pd.outs.resize(sjobs.size());
    pd.job_map.resize(n_jobs);
    {
      int j = 0;
      for (size_t i = 0; i < sjobs.size(); ++i) {
        const auto& sj = sjobs[i];
        Out& o = pd.outs[i];
        o.y0 = sj.y0;
        o.h = std::min(sj.y1, h_) - sj.y0;
        o.idr = sj.idr;
        h264::GpuStripeParams& p = o.p;
        p.levels = h_levels_;
        p.meta = h_meta_;
        p.mbw = mbw_;
        p.mb_row0 = sj.y0 / 16;
        p.n_mb_rows = (std::min(sj.y1, mbh_ * 16) - sj.y0 + 15) / 16;
        p.width = w_;
        p.height = o.h;
        p.qp = qp;
        p.idr = sj.idr;
        p.deblock = settings_.video_deblock;
        p.frame_num = sj.frame_num;
        p.idr_pic_id = sj.idr_pic_id;
        o.rows.resize(p.n_mb_rows * segs_);
        if (p.idr) {
          h264::write_sps_nal(o.header, (p.width + 15) / 16,
                              p.n_mb_rows, p.width, p.height);
          h264::write_pps_nal(o.header);
        }
        for (int r = 0; r < p.n_mb_rows * segs_; ++r)
          pd.job_map[j++] = {int(i), r};
      }
    }
    return pd;
  }

  void collect_pending(Pending& pd, const Emit& emit) {
    if (!pd.active) return;
    pd.active = false;
    const bool timing = std::getenv("HIPFLUX_TIMES") != nullptr;
    auto tick = [] {
      return std::chrono::duration<double, std::micro>(
                 std::chrono::steady_clock::now().time_since_epoch())
          .count();
    };
    double c0 = timing ? tick() : 0, c1 = 0, c2 = 0;
    const int par = pd.par;
    if (cpu_entropy_) {
      for (size_t b = 0; b < pd.batches.size(); ++b) {
        HIP_CHECK(hipEventSynchronize(batch_events_[b]));
        for (int j = pd.batches[b].job0; j < pd.batches[b].jobn; ++j) {
          auto [si, sl] = pd.job_map[j];
          auto* dst = &pd.outs[si].rows[sl];
          const h264::GpuStripeParams* pp = &pd.outs[si].p;
          int r = sl / segs_;
          int mbx0 = h_jobs_[par][j].mbx0, segw = h_jobs_[par][j].seg_mbw;
          bool long_sc = sl == 0;
          pool_.submit([pp, r, mbx0, segw, long_sc, dst] {
            h264::encode_seg_nal_from_gpu(*pp, r, mbx0, segw, long_sc,
                                          *dst);
          });
        }
      }
      pool_.wait_all();
      HIP_CHECK(hipStreamSynchronize(stream_));
    } else {
      HIP_CHECK(hipEventSynchronize(ev_done_[par]));
      if (timing) c1 = tick();
      int max_words = 0;
      for (int j = 0; j < pd.n_jobs; ++j) {
        int wds = (h_outbits_[par][j] + 31) / 32 + 1;
        max_words = std::max(max_words, wds);
        if (wds > pd.copy_words) {
          // rare: this row outgrew the adaptive cap; fetch it exactly
          // (d_entout_[par] is parity-preserved until this frame's
          // collect, even with the next frame already submitted)
          HIP_CHECK(hipMemcpy(
              h_entout_[par] + (size_t)j * ent_stride_words_,
              d_entout_[par] + (size_t)j * ent_stride_words_,
              (size_t)wds * 4, hipMemcpyDeviceToHost));
        }
      }
      ent_copy_words_ = std::min(ent_stride_words_, max_words * 2 + 64);
      // one strided task per worker (not one per NAL: 136 tiny submits
      // per frame serialized on the pool mutex — measured 370 us/frame
      // of wall time for ~40 us of actual assembly work)
      const int nw = std::min(pool_.size(), pd.n_jobs);
      for (int t = 0; t < nw; ++t) {
        pool_.submit([this, &pd, par, t, nw] {
          for (int j = t; j < pd.n_jobs; j += nw) {
            auto [si, sl] = pd.job_map[j];
            h264::assemble_gpu_row_nal(
                h_entout_[par] + (size_t)j * ent_stride_words_,
                h_outbits_[par][j], pd.outs[si].idr, sl == 0,
                pd.outs[si].rows[sl]);
          }
        });
      }
      pool_.wait_all();
    }
    if (timing) c2 = tick();
    // per-stripe concat on the pool (serial it cost ~90 us/frame)
    if (pd.outs.size() > 1) {
      for (auto& o : pd.outs)
        pool_.submit([&o] {
          o.bytes = std::move(o.header);
          size_t total = o.bytes.size();
          for (auto& r : o.rows) total += r.size();
          o.bytes.reserve(total);
          for (auto& r : o.rows)
            o.bytes.insert(o.bytes.end(), r.begin(), r.end());
        });
      pool_.wait_all();
    } else {
      for (auto& o : pd.outs) {
        o.bytes = std::move(o.header);
        for (auto& r : o.rows)
          o.bytes.insert(o.bytes.end(), r.begin(), r.end());
      }
    }
    for (auto& o : pd.outs) {
      if (o.bytes.empty()) continue;
      EncodedStripe s;
      s.type = StripeType::kH264;
      s.data = o.bytes.data();
      s.size = o.bytes.size();
      s.frame_id = pd.frame_id;
      s.capture_ts_ms = pd.ts_ms;
      s.y = o.y0;
      s.width = w_;
      s.height = o.h;
      s.is_keyframe = o.idr;
      emit(s);
    }
    if (timing && c1 > 0) {
      tm_sync_ += c1 - c0;
      tm_asm_ += c2 - c1;
      tm_concat_ += tick() - c2;
    }
  }

  void encode_frame(const RawFrame& frame, const FrameContext& ctx,
                    const Emit& emit) override {
    const bool timing = std::getenv("HIPFLUX_TIMES") != nullptr;
    auto tick = [] {
      return std::chrono::duration<double, std::micro>(
                 std::chrono::steady_clock::now().time_since_epoch())
          .count();
    };
    double t0 = timing ? tick() : 0;
    Pending cur = submit_frame(frame, ctx);
    double t1 = timing ? tick() : 0;
    if (!cur.active) {
      // nothing scheduled this frame; keep the pipe draining
      collect_pending(prev_, emit);
      return;
    }
    if (depth_ < 2 || cpu_entropy_) {
      collect_pending(cur, emit);
      return;
    }
    // depth 2: emit LAST frame's stripes (host assembly overlaps this
    // frame's GPU work), keep this one in flight
    collect_pending(prev_, emit);
    double t2 = timing ? tick() : 0;
    prev_ = std::move(cur);
    // the caller may reuse its frame buffer after we return
    HIP_CHECK(hipEventSynchronize(ev_h2d_[prev_.par]));
    if (timing) {
      double t3 = tick();
      tm_submit_ += t1 - t0;
      tm_collect_ += t2 - t1;
      tm_h2d_ += t3 - t2;
      if (++tm_n_ == 256) {
        std::fprintf(stderr,
                     "[times us/frame] submit=%.0f collect=%.0f (sync=%.0f "
                     "asm=%.0f concat=%.0f) h2dwait=%.0f\n",
                     tm_submit_ / tm_n_, tm_collect_ / tm_n_,
                     tm_sync_ / tm_n_, tm_asm_ / tm_n_, tm_concat_ / tm_n_,
                     tm_h2d_ / tm_n_);
        tm_submit_ = tm_collect_ = tm_h2d_ = tm_sync_ = tm_asm_ =
            tm_concat_ = 0;
        tm_n_ = 0;
      }
    }
  }

  void flush(const Emit& emit) override { collect_pending(prev_, emit); }

  void set_pipeline_depth(int d) override {
    depth_ = std::max(1, std::min(2, d));
  }

  const char* name() const override { return "hip-h264"; }

 private:
  struct StripeState {
    uint32_t frame_num = 0;
    uint32_t idr_pic_id = 0;
    bool need_idr = true;
  };

  void copy_region(uint8_t* dst, const uint8_t* src, int pitch, int y0,
                   int rows) {
    HIP_CHECK(hipMemcpyAsync(dst + static_cast<size_t>(y0) * pitch,
                             src + static_cast<size_t>(y0) * pitch,
                             static_cast<size_t>(rows) * pitch,
                             hipMemcpyDeviceToDevice, rows_stream_));
  }

 public:
  bool recon_dev(void** y, int* ypitch, int* height) override {
    *y = d_refY_;
    *ypitch = ypitch_;
    *height = mbh_ * 16;
    return d_refY_ != nullptr;
  }

  bool debug_dump(DebugDump& d) override {
    HIP_CHECK(hipStreamSynchronize(stream_));
    d.w = w_;
    d.h = h_;
    d.ypitch = ypitch_;
    d.cpitch = cpitch_;
    size_t ysz = static_cast<size_t>(ypitch_) * mbh_ * 16;
    size_t csz = static_cast<size_t>(cpitch_) * mbh_ * 8;
    d.y.resize(ysz);
    d.cb.resize(csz);
    d.cr.resize(csz);
    // after encode_frame the recon has been copied cur -> ref
    HIP_CHECK(hipMemcpy(d.y.data(), d_refY_, ysz, hipMemcpyDeviceToHost));
    HIP_CHECK(hipMemcpy(d.cb.data(), d_refCb_, csz, hipMemcpyDeviceToHost));
    HIP_CHECK(hipMemcpy(d.cr.data(), d_refCr_, csz, hipMemcpyDeviceToHost));
    size_t nlv = static_cast<size_t>(mbw_) * mbh_ * h264gpu::kLevelsPerMb;
    size_t nmt = static_cast<size_t>(mbw_) * mbh_ * h264gpu::kMetaPerMb;
    d.levels.resize(nlv);
    d.meta.resize(nmt);
    const int last_par = parity_ ^ 1;
    HIP_CHECK(hipMemcpy(d.levels.data(), d_levels_[last_par],
                        nlv * sizeof(int16_t),
                        hipMemcpyDeviceToHost));
    HIP_CHECK(hipMemcpy(d.meta.data(), d_meta_[last_par],
                        nmt * sizeof(int),
                        hipMemcpyDeviceToHost));
    return true;
  }

 private:
  void alloc_for(int w, int h) {
    // resolution change: any pipelined frame still in flight references
    // the old buffers — drop it (streams restart with a fresh IDR anyway)
    prev_.active = false;
    HIP_CHECK(hipStreamSynchronize(stream_));
    for (void* p : device_ptrs_)
      if (p) (void)hipFree(p);
    device_ptrs_.clear();
    if (h_levels_) (void)hipHostFree(h_levels_);
    if (h_meta_) (void)hipHostFree(h_meta_);
    w_ = w;
    h_ = h;
    mbw_ = (w + 15) / 16;
    mbh_ = (h + 15) / 16;
    ypitch_ = mbw_ * 16;
    cpitch_ = mbw_ * 8;
    size_t ysz = static_cast<size_t>(ypitch_) * mbh_ * 16;
    size_t csz = static_cast<size_t>(cpitch_) * mbh_ * 8;
    auto dalloc = [&](size_t bytes) {
      void* p = nullptr;
      HIP_CHECK(hipMalloc(&p, bytes));
      device_ptrs_.push_back(p);
      return static_cast<uint8_t*>(p);
    };
    d_frame_[0] = dalloc(static_cast<size_t>(w) * h * 4);
    d_frame_[1] = dalloc(static_cast<size_t>(w) * h * 4);
    d_srcY_ = dalloc(ysz);
    d_srcCb_ = dalloc(csz);
    d_srcCr_ = dalloc(csz);
    d_refY_ = dalloc(ysz);
    d_refCb_ = dalloc(csz);
    d_refCr_ = dalloc(csz);
    d_curY_ = dalloc(ysz);
    d_curCb_ = dalloc(csz);
    d_curCr_ = dalloc(csz);
    size_t level_bytes = static_cast<size_t>(mbw_) * mbh_ *
                         h264gpu::kLevelsPerMb * sizeof(int16_t);
    size_t meta_bytes =
        static_cast<size_t>(mbw_) * mbh_ * h264gpu::kMetaPerMb * sizeof(int);
    meta_bytes_ = meta_bytes;
    for (int i = 0; i < 2; ++i) {
      d_levels_[i] = reinterpret_cast<int16_t*>(dalloc(level_bytes));
      d_meta_[i] = reinterpret_cast<int*>(dalloc(meta_bytes));
      // meta seeds from parity^1 every frame; zero both so the first
      // seed copy is well-defined
      HIP_CHECK(hipMemset(d_meta_[i], 0, meta_bytes));
    }
    d_mip1_ = dalloc(static_cast<size_t>(ypitch_ / 2) * mbh_ * 8);
    d_srcY2_ = dalloc(static_cast<size_t>(ypitch_ / 4) * mbh_ * 4);
    d_refY2_ = dalloc(static_cast<size_t>(ypitch_ / 4) * mbh_ * 4);
    segs_ = (mbw_ + h264gpu::kMaxSegMbw - 1) / h264gpu::kMaxSegMbw;
    // Latency shaping: the row kernels are serial in the MB chain of one
    // slice, and a frame whose rows alone underfill the 256-CU chip is
    // LATENCY-bound — so split each row into more slices until there is
    // ~one workgroup per CU, as long as chains keep >=20 MBs (slice
    // restarts reset intra/MVP context, so ultra-short chains waste bits
    // and pipeline warmup). 1080p: 68 rows -> 4 slices of 30 MBs.
    // Each extra slice costs a NAL (host assembly + ~0.1% bits at
    // 1080p noise). Two regimes, both measured:
    //  * latency (rows alone underfill the chip, job_cap > 1): split
    //    toward one workgroup per CU but keep total jobs <= 256 so the
    //    CAVLC kernel stays in its wide 1024-thread shape (exceeding it
    //    fell back to 512 threads and lost 20-40%), chains >= 30 MBs.
    //    1080p -> 3 slices of 40 (864 -> 2127 fps with the rest of v25+).
    //  * throughput (4K/8K: enough rows to fill the chip): the tail of
    //    each wave-round is still chain-latency-bound, so short chains
    //    win regardless of CAVLC width — cap chains at ~30 MBs, at most
    //    8 slices/row. 4K 513 -> 711 fps (8 slices of 30), 8K 193 -> 205
    //    (8 slices of 60; 12 slices measured slightly worse).
    {
      int fill = (256 + mbh_ - 1) / std::max(1, mbh_);
      int chain_cap = std::max(1, mbw_ / 30);
      int job_cap = std::max(1, 256 / std::max(1, mbh_));
      if (job_cap > 1)
        segs_ = std::max(segs_, std::min(std::min(fill, job_cap),
                                         std::min(chain_cap, 8)));
      else
        segs_ = std::max(segs_, std::min(chain_cap, 8));
      if (const char* e = std::getenv("HIPFLUX_SEGS")) {
        int v = std::atoi(e);
        if (v >= 1)
          segs_ = std::max((mbw_ + h264gpu::kMaxSegMbw - 1) /
                               h264gpu::kMaxSegMbw, v);
      }
    }
    seg_w0_ = (mbw_ + segs_ - 1) / segs_;   // widest segment
    const int max_jobs = mbh_ * segs_;
    HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&h_levels_), level_bytes,
                            hipHostMallocDefault));
    HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&h_meta_), meta_bytes,
                            hipHostMallocDefault));
    // GPU entropy buffers (one slot per SLICE = row segment)
    cpu_entropy_ = std::getenv("HIPFLUX_CPU_ENTROPY") != nullptr;
    const int nitems = h264gpu::items_per_row(seg_w0_);
    ent_stride_words_ = nitems * h264gpu::kStageWordsPerItem;
    ent_copy_words_ = 1 << 30;     // first frame after (re)alloc: full copy
    size_t stage_bytes =
        (size_t)max_jobs * nitems * h264gpu::kStageWordsPerItem * 4;
    d_stage_ = reinterpret_cast<uint32_t*>(dalloc(stage_bytes));
    d_nbits_ = reinterpret_cast<int*>(dalloc((size_t)max_jobs * nitems * 4));
    for (int i = 0; i < 2; ++i) {
      d_jobs_[i] = reinterpret_cast<h264gpu::RowJob*>(
          dalloc(sizeof(h264gpu::RowJob) * max_jobs));
      d_entout_[i] = reinterpret_cast<uint32_t*>(
          dalloc((size_t)max_jobs * ent_stride_words_ * 4));
      d_outbits_[i] = reinterpret_cast<int*>(dalloc(sizeof(int) * max_jobs));
      if (h_jobs_[i]) (void)hipHostFree(h_jobs_[i]);
      if (h_entout_[i]) (void)hipHostFree(h_entout_[i]);
      if (h_outbits_[i]) (void)hipHostFree(h_outbits_[i]);
      HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&h_jobs_[i]),
                              sizeof(h264gpu::RowJob) * max_jobs,
                              hipHostMallocDefault));
      HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&h_entout_[i]),
                              (size_t)max_jobs * ent_stride_words_ * 4,
                              hipHostMallocDefault));
      HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&h_outbits_[i]),
                              sizeof(int) * max_jobs, hipHostMallocDefault));
    }
    stripes_.assign((h + stripe_h_ - 1) / stripe_h_, StripeState{});
  }

  CaptureSettings settings_;
  ThreadPool pool_;
  hipStream_t stream_{};
  hipStream_t up_stream_{};
  hipStream_t db_stream_{};
  hipStream_t rows_stream_{};
  int stripe_h_ = 64;
  int w_ = 0, h_ = 0, mbw_ = 0, mbh_ = 0, ypitch_ = 0, cpitch_ = 0;
  int segs_ = 1, seg_w0_ = 0;     // slices per MB row, widest segment
  uint8_t* d_frame_[2] = {nullptr, nullptr};
  uint8_t *d_srcY_ = nullptr, *d_srcCb_ = nullptr,
          *d_srcCr_ = nullptr, *d_refY_ = nullptr, *d_refCb_ = nullptr,
          *d_refCr_ = nullptr, *d_curY_ = nullptr, *d_curCb_ = nullptr,
          *d_curCr_ = nullptr;
  int16_t* d_levels_[2] = {};
  size_t meta_bytes_ = 0;
  uint8_t *d_mip1_ = nullptr, *d_srcY2_ = nullptr, *d_refY2_ = nullptr;
  int* d_meta_[2] = {};
  int16_t* h_levels_ = nullptr;
  int* h_meta_ = nullptr;
  uint8_t* h_stage_ = nullptr;
  size_t h_stage_bytes_ = 0;
  // GPU entropy buffers. Buffers the host writes before (h_jobs_/d_jobs_)
  // or reads after (h_entout_/h_outbits_, d_entout_ for overflow re-copy)
  // a frame's GPU work are parity-double-buffered so depth-2 pipelining
  // can submit frame N+1 while frame N's results are still being read;
  // everything device-only is hazard-free by single-stream ordering.
  bool cpu_entropy_ = false;
  h264gpu::RowJob* d_jobs_[2] = {nullptr, nullptr};
  h264gpu::RowJob* h_jobs_[2] = {nullptr, nullptr};
  uint32_t* d_stage_ = nullptr;
  int* d_nbits_ = nullptr;
  uint32_t* d_entout_[2] = {nullptr, nullptr};
  int* d_outbits_[2] = {nullptr, nullptr};
  uint32_t* h_entout_[2] = {nullptr, nullptr};
  int* h_outbits_[2] = {nullptr, nullptr};
  hipEvent_t ev_done_[2] = {}, ev_h2d_[2] = {};
  hipEvent_t ev_rows_[2] = {}, ev_db_[2] = {};
  int depth_ = 1;                  // 1 = sync (latency mode), 2 = pipelined
  int parity_ = 0;
  Pending prev_;                   // the in-flight frame (depth 2)
  // HIPFLUX_TIMES=1 phase accumulators (microseconds)
  double tm_submit_ = 0, tm_collect_ = 0, tm_h2d_ = 0, tm_sync_ = 0,
         tm_asm_ = 0, tm_concat_ = 0;
  int tm_n_ = 0;
  int ent_stride_words_ = 0;
  int ent_copy_words_ = 1 << 30;   // adaptive D2H cap (words per row)
  bool rows_v1_ = std::getenv("HIPFLUX_ROWS_V1") != nullptr;
  std::vector<hipEvent_t> batch_events_;
  std::vector<void*> device_ptrs_;
  std::vector<StripeState> stripes_;
  std::map<void*, bool> registered_;
};

}  // namespace

std::unique_ptr<EncodePipeline> make_hip_hevc_pipeline(const CaptureSettings&);

std::unique_ptr<EncodePipeline> make_hip_pipeline(const CaptureSettings& s) {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess || n == 0) return nullptr;
  if (s.gpu_id >= n) return nullptr;
  try {
    if (s.output_mode == 0) return std::make_unique<HipJpegPipeline>(s);
    if (s.output_mode == 2) return make_hip_hevc_pipeline(s);
    if (s.video_fullcolor) {
      // Hi444 separate-colour-plane H.264 is served by the CPU
      // stripe-parallel encoder (the GPU row kernels are 4:2:0).
      std::fprintf(stderr,
                   "hipflux: h264 fullcolor -> CPU Hi444 encoder\n");
      return nullptr;
    }
    return std::make_unique<HipH264Pipeline>(s);
  } catch (const std::exception& e) {
    std::fprintf(stderr, "hipflux: HIP pipeline init failed: %s\n", e.what());
    return nullptr;
  }
}

}  // namespace hipflux
