// HIP (gfx950) HEVC encode pipeline host side.
// Upload BGRX -> CSC kernel (shared with JPEG/H.264) -> per-slice CTU
// wavefront kernel (intra + transform + quant + recon) -> per-slice CABAC
// kernel -> compacted D2H -> host NAL assembly on the pool.
// Byte-identical to CpuHevcPipeline (asserted by tests/test_gpu_hevc.py);
// HIPFLUX_CPU_HEVC_ENTROPY=1 swaps the CABAC kernel for the host entropy
// coder over D2H levels/meta to isolate rows-kernel vs entropy bugs.
#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <map>
#include <cstring>
#include <stdexcept>
#include <vector>

#include "cpu/hevc/encoder.h"       // default_slices_per_row
#include "cpu/hevc/gpu_entropy.h"
#include "cpu/hevc/tables.h"        // chroma_qp
#include "engine.h"
#include "hip/hevc_kernels.h"
#include "hip/jpeg_kernels.h"       // launch_bgrx_to_planes
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

class HipHevcPipeline : public EncodePipeline {
 public:
  explicit HipHevcPipeline(const CaptureSettings& s)
      : settings_(s),
        pool_(std::min(16u,
                       std::max(2u, std::thread::hardware_concurrency() / 2))) {
    HIP_CHECK(hipSetDevice(std::max(0, s.gpu_id)));
    HIP_CHECK(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking));
    HIP_CHECK(hipStreamCreateWithFlags(&copy_stream_, hipStreamNonBlocking));
    HIP_CHECK(hipStreamCreateWithFlags(&rows_stream_, hipStreamNonBlocking));
    for (int i = 0; i < 2; ++i)
      HIP_CHECK(hipEventCreateWithFlags(&ev_rows_[i],
                                        hipEventDisableTiming));
    cpu_entropy_ = std::getenv("HIPFLUX_CPU_HEVC_ENTROPY") != nullptr;
    timing_ = std::getenv("HIPFLUX_TIMES") != nullptr;
    for (auto& e : ev_) HIP_CHECK(hipEventCreate(&e));
    for (int i = 0; i < 2; ++i) {
      HIP_CHECK(hipEventCreateWithFlags(&ev_h2d_[i],
                                        hipEventDisableTiming));
      HIP_CHECK(hipEventCreateWithFlags(&ev_done_[i],
                                        hipEventDisableTiming));
    }
    stripe_h_ = std::max(16, s.stripe_height & ~15);
    alloc_for(s.capture_width, s.capture_height);
  }

  ~HipHevcPipeline() override {
    (void)hipStreamSynchronize(stream_);
    (void)hipStreamSynchronize(copy_stream_);
    for (auto& kv : registered_) (void)hipHostUnregister(kv.first);
    for (void* p : device_ptrs_)
      if (p) (void)hipFree(p);
    for (int i = 0; i < 2; ++i) {
      if (h_jobs_[i]) (void)hipHostFree(h_jobs_[i]);
      if (h_out_[i]) (void)hipHostFree(h_out_[i]);
      if (h_counts_[i]) (void)hipHostFree(h_counts_[i]);
      (void)hipEventDestroy(ev_h2d_[i]);
      (void)hipEventDestroy(ev_done_[i]);
    }
    if (h_levels_) (void)hipHostFree(h_levels_);
    if (h_meta_) (void)hipHostFree(h_meta_);
    if (h_stage_) (void)hipHostFree(h_stage_);
    (void)hipStreamDestroy(copy_stream_);
    (void)hipStreamSynchronize(rows_stream_);
    (void)hipStreamDestroy(rows_stream_);
    for (int i = 0; i < 2; ++i) (void)hipEventDestroy(ev_rows_[i]);
  }

  struct StripeRef {
    int y0, vis_h;      // pixels
    int job0, jobn;     // job index range
  };
  struct Pending {
    bool active = false;
    int par = 0;
    int n_jobs = 0;
    int cap = 0;        // D2H compaction cap this frame was submitted with
    uint32_t frame_id = 0;
    double ts_ms = 0;
    std::vector<StripeRef> stripes;
  };

  // Queue one frame's full GPU chain (upload, CSC, CTU rows, CABAC,
  // counts + compacted bitstream D2H) on stream_; nothing blocks here
  // beyond pageable-staging copies.
  Pending submit_frame(const RawFrame& frame, const FrameContext& ctx) {
    if (frame.width != w_ || frame.height != h_)
      alloc_for(frame.width, frame.height);
    const int qp = std::min(51, std::max(0, ctx.crf));
    const int par = parity_;

    // ---- upload + CSC
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
    // frame N+1's upload/CSC/rows run on rows_stream_, concurrent with
    // frame N's CABAC on stream_ (levels/meta are parity
    // double-buffered; the recon planes have no cross-frame reader)
    HIP_CHECK(hipMemcpyAsync(d_frame_[par], src, frame_bytes,
                             hipMemcpyHostToDevice, rows_stream_));
    HIP_CHECK(hipEventRecord(ev_h2d_[par], rows_stream_));
    launch_bgrx_to_planes(d_frame_[par], w_, h_, frame.stride / 4, d_srcY_,
                          d_srcCb_, d_srcCr_, ypitch_, cpitch_, false,
                          rows_stream_);

    // ---- job list (must mirror the CPU StripeEncoder's segmentation)
    Pending pd;
    pd.par = par;
    pd.frame_id = ctx.frame_id;
    pd.ts_ms = frame.ts_ms;
    int n_jobs = 0;
    const int spr = hevc::default_slices_per_row(w_);
    for (size_t i = 0; i < ctx.stripes.size(); ++i) {
      const auto& st = ctx.stripes[i];
      if (!st.encode) continue;
      const int vis_h = std::min(st.y1, h_) - st.y0;
      const int s_ctb_h = (vis_h + 15) / 16;
      const int n_ctb = ctbw_ * s_ctb_h;
      int addr_bits = 1;
      while ((1 << addr_bits) < n_ctb) ++addr_bits;
      const int per = (ctbw_ + spr - 1) / spr;
      StripeRef sr{st.y0, vis_h, n_jobs, 0};
      for (int r = 0; r < s_ctb_h; ++r) {
        for (int s0 = 0; s0 < ctbw_; s0 += per) {
          auto& j = h_jobs_[par][n_jobs++];
          j.ctu_row = st.y0 / 16 + r;
          j.ctu_x0 = s0;
          j.seg_w = std::min(per, ctbw_ - s0);
          j.qp = qp;
          j.qpc = hevc::chroma_qp(qp);
          j.first_slice = (r == 0 && s0 == 0) ? 1 : 0;
          j.slice_addr = r * ctbw_ + s0;
          j.addr_bits = addr_bits;
          j.last_in_pic = 0;
        }
      }
      sr.jobn = n_jobs;
      pd.stripes.push_back(sr);
    }
    if (n_jobs == 0) return pd;
    pd.n_jobs = n_jobs;
    pd.active = true;
    parity_ ^= 1;

    if (timing_) HIP_CHECK(hipEventRecord(ev_[0], rows_stream_));
    HIP_CHECK(hipMemcpyAsync(d_jobs_[par], h_jobs_[par],
                             sizeof(hevcgpu::HevcJob) * n_jobs,
                             hipMemcpyHostToDevice, rows_stream_));
    hevcgpu::launch_hevc_rows(d_srcY_, d_srcCb_, d_srcCr_, ypitch_, cpitch_,
                              w_, h_, d_curY_, d_curCb_, d_curCr_, ctbw_,
                              n_jobs, d_jobs_[par], d_levels_[par],
                              d_meta_[par], rows_stream_);
    HIP_CHECK(hipEventRecord(ev_rows_[par], rows_stream_));
    HIP_CHECK(hipStreamWaitEvent(stream_, ev_rows_[par], 0));
    if (cpu_entropy_) {
      HIP_CHECK(hipMemcpyAsync(h_levels_, d_levels_[par], levels_bytes_,
                               hipMemcpyDeviceToHost, stream_));
      HIP_CHECK(hipMemcpyAsync(h_meta_, d_meta_[par], meta_bytes_,
                               hipMemcpyDeviceToHost, stream_));
      HIP_CHECK(hipEventRecord(ev_done_[par], stream_));
      return pd;
    }
    if (timing_) HIP_CHECK(hipEventRecord(ev_[1], stream_));
    hevcgpu::launch_hevc_cabac(d_levels_[par], d_meta_[par], ctbw_, n_jobs,
                               d_jobs_[par], d_out_[par], out_stride_,
                               d_counts_[par], stream_);
    if (timing_) HIP_CHECK(hipEventRecord(ev_[2], stream_));
    HIP_CHECK(hipMemcpyAsync(h_counts_[par], d_counts_[par],
                             sizeof(int) * 3 * n_jobs,
                             hipMemcpyDeviceToHost, stream_));
    // adaptive compaction: copy only ~the used prefix of each job's
    // bytes (cap = 2x last frame's max count); overflowing jobs are
    // re-copied exactly at collect time
    pd.cap = std::min(out_stride_, copy_cap_);
    HIP_CHECK(hipMemcpy2DAsync(h_out_[par], out_stride_, d_out_[par],
                               out_stride_, pd.cap, n_jobs,
                               hipMemcpyDeviceToHost, stream_));
    HIP_CHECK(hipEventRecord(ev_done_[par], stream_));
    return pd;
  }

  // Wait for a submitted frame's D2H, assemble its NALs on the pool and
  // emit. Overflow re-copies go over copy_stream_ so they never wait on
  // the NEXT frame's kernels queued behind us on stream_.
  void collect_pending(Pending& pd, const Emit& emit) {
    if (!pd.active) return;
    pd.active = false;
    const int par = pd.par;
    HIP_CHECK(hipEventSynchronize(ev_done_[par]));
    const int n_jobs = pd.n_jobs;
    if (!cpu_entropy_) {
      int max_count = 0;
      bool fix = false;
      for (int j = 0; j < n_jobs; ++j) {
        const int cnt = h_counts_[par][j * 3];
        max_count = std::max(max_count, cnt);
        if (cnt > out_stride_)
          throw std::runtime_error("hevc cabac overflow");
        if (cnt > pd.cap) {
          HIP_CHECK(hipMemcpyAsync(
              h_out_[par] + (size_t)j * out_stride_,
              d_out_[par] + (size_t)j * out_stride_, cnt,
              hipMemcpyDeviceToHost, copy_stream_));
          fix = true;
        }
      }
      if (fix) HIP_CHECK(hipStreamSynchronize(copy_stream_));
      copy_cap_ = std::max(4096, 2 * max_count);
      if (timing_) {
        float t_rows = 0, t_cab = 0;
        (void)hipEventElapsedTime(&t_rows, ev_[0], ev_[1]);
        (void)hipEventElapsedTime(&t_cab, ev_[1], ev_[2]);
        std::fprintf(stderr, "[hevc-times] rows %.2fms cabac %.2fms\n",
                     t_rows, t_cab);
      }
    }

    // ---- host NAL assembly (parallel over stripes)
    struct SOut {
      std::vector<uint8_t> bytes;
      int y0, vis_h;
    };
    std::vector<SOut> outs(pd.stripes.size());
    for (size_t si = 0; si < pd.stripes.size(); ++si) {
      pool_.submit([&, si, par] {
        const StripeRef& sr = pd.stripes[si];
        SOut& o = outs[si];
        o.y0 = sr.y0;
        o.vis_h = sr.vis_h;
        const int coded_h = (sr.vis_h + 15) & ~15;
        hevc::write_hevc_stripe_headers(ctbw_ * 16, coded_h, w_, sr.vis_h,
                                        o.bytes);
        for (int j = sr.job0; j < sr.jobn; ++j) {
          const auto& job = h_jobs_[par][j];
          if (cpu_entropy_) {
            hevc::encode_hevc_job_nal(h_levels_, h_meta_, ctbw_,
                                      job.ctu_row, job.ctu_x0, job.seg_w,
                                      job.qp, job.first_slice != 0,
                                      job.slice_addr, job.addr_bits,
                                      o.bytes);
          } else {
            hevc::assemble_hevc_slice_nal(
                h_out_[par] + (size_t)j * out_stride_,
                h_counts_[par][j * 3], h_counts_[par][j * 3 + 1],
                h_counts_[par][j * 3 + 2], job.first_slice != 0,
                job.slice_addr, job.addr_bits, job.qp, o.bytes);
          }
        }
      });
    }
    pool_.wait_all();
    for (auto& o : outs) {
      EncodedStripe s;
      s.type = StripeType::kHevc;
      s.data = o.bytes.data();
      s.size = o.bytes.size();
      s.frame_id = pd.frame_id;
      s.capture_ts_ms = pd.ts_ms;
      s.y = o.y0;
      s.width = w_;
      s.height = o.vis_h;
      s.is_keyframe = true;
      emit(s);
    }
  }

  void encode_frame(const RawFrame& frame, const FrameContext& ctx,
                    const Emit& emit) override {
    Pending cur = submit_frame(frame, ctx);
    if (!cur.active) {
      collect_pending(prev_, emit);
      return;
    }
    if (depth_ < 2 || cpu_entropy_) {
      collect_pending(cur, emit);
      return;
    }
    // depth 2: emit LAST frame's stripes (host assembly overlaps this
    // frame's GPU chain), keep this one in flight
    collect_pending(prev_, emit);
    prev_ = std::move(cur);
    // the caller may reuse its frame buffer once we return
    HIP_CHECK(hipEventSynchronize(ev_h2d_[prev_.par]));
  }

  void flush(const Emit& emit) override { collect_pending(prev_, emit); }

  void set_pipeline_depth(int d) override {
    depth_ = std::max(1, std::min(2, d));
  }

  const char* name() const override { return "hip-hevc"; }

  bool recon_dev(void** y, int* ypitch, int* height) override {
    *y = d_curY_;
    *ypitch = ypitch_;
    *height = ctbh_ * 16;
    return d_curY_ != nullptr;
  }

  bool debug_dump(DebugDump& d) override {
    d.w = w_;
    d.h = h_;
    d.ypitch = ypitch_;
    d.cpitch = cpitch_;
    const int ch16 = (h_ + 15) & ~15;
    d.y.resize(static_cast<size_t>(ypitch_) * ch16);
    d.cb.resize(static_cast<size_t>(cpitch_) * ch16 / 2);
    d.cr.resize(static_cast<size_t>(cpitch_) * ch16 / 2);
    HIP_CHECK(hipMemcpy(d.y.data(), d_curY_, d.y.size(),
                        hipMemcpyDeviceToHost));
    HIP_CHECK(hipMemcpy(d.cb.data(), d_curCb_, d.cb.size(),
                        hipMemcpyDeviceToHost));
    HIP_CHECK(hipMemcpy(d.cr.data(), d_curCr_, d.cr.size(),
                        hipMemcpyDeviceToHost));
    d.levels.resize(levels_bytes_ / sizeof(int16_t));
    d.meta.resize(meta_bytes_ / sizeof(int));
    const int last_par = parity_ ^ 1;   // parity of the last submit
    HIP_CHECK(hipMemcpy(d.levels.data(), d_levels_[last_par],
                        levels_bytes_, hipMemcpyDeviceToHost));
    HIP_CHECK(hipMemcpy(d.meta.data(), d_meta_[last_par], meta_bytes_,
                        hipMemcpyDeviceToHost));
    return true;
  }

 private:
  template <typename T>
  T* dalloc(size_t n) {
    void* p = nullptr;
    HIP_CHECK(hipMalloc(&p, n * sizeof(T)));
    device_ptrs_.push_back(p);
    return static_cast<T*>(p);
  }

  void alloc_for(int w, int h) {
    HIP_CHECK(hipStreamSynchronize(stream_));
    prev_ = Pending{};          // in-flight frame is gone with its buffers
    parity_ = 0;
    for (void* p : device_ptrs_)
      if (p) (void)hipFree(p);
    device_ptrs_.clear();
    for (int i = 0; i < 2; ++i) {
      if (h_jobs_[i]) (void)hipHostFree(h_jobs_[i]);
      if (h_out_[i]) (void)hipHostFree(h_out_[i]);
      if (h_counts_[i]) (void)hipHostFree(h_counts_[i]);
      h_jobs_[i] = nullptr;
      h_out_[i] = nullptr;
      h_counts_[i] = nullptr;
    }
    if (h_levels_) (void)hipHostFree(h_levels_);
    if (h_meta_) (void)hipHostFree(h_meta_);
    h_levels_ = nullptr;
    h_meta_ = nullptr;

    w_ = w;
    h_ = h;
    ctbw_ = (w + 15) / 16;
    ctbh_ = (h + 15) / 16;
    ypitch_ = ctbw_ * 16;
    cpitch_ = ypitch_ / 2;
    const int ch16 = ctbh_ * 16;

    for (int i = 0; i < 2; ++i)
      d_frame_[i] =
          dalloc<uint8_t>(static_cast<size_t>(w + 64) * 4 * (h + 16));
    d_srcY_ = dalloc<uint8_t>(static_cast<size_t>(ypitch_) * ch16);
    d_srcCb_ = dalloc<uint8_t>(static_cast<size_t>(cpitch_) * ch16 / 2);
    d_srcCr_ = dalloc<uint8_t>(static_cast<size_t>(cpitch_) * ch16 / 2);
    d_curY_ = dalloc<uint8_t>(static_cast<size_t>(ypitch_) * ch16);
    d_curCb_ = dalloc<uint8_t>(static_cast<size_t>(cpitch_) * ch16 / 2);
    d_curCr_ = dalloc<uint8_t>(static_cast<size_t>(cpitch_) * ch16 / 2);

    const size_t n_ctu = static_cast<size_t>(ctbw_) * ctbh_;
    levels_bytes_ = n_ctu * hevcgpu::kHevcLevelsPerCtu * sizeof(int16_t);
    meta_bytes_ = n_ctu * hevcgpu::kHevcMetaPerCtu * sizeof(int);
    for (int i = 0; i < 2; ++i) {
      d_levels_[i] = dalloc<int16_t>(n_ctu * hevcgpu::kHevcLevelsPerCtu);
      d_meta_[i] = dalloc<int>(n_ctu * hevcgpu::kHevcMetaPerCtu);
    }

    const int spr = hevc::default_slices_per_row(w);
    max_jobs_ = ctbh_ * spr + 8;
    // worst-case CABAC bytes per CTU is bounded by the coefficient count
    // (384) x a handful of bypass bytes; 2 KB/CTU is a safe ceiling
    const int per = (ctbw_ + spr - 1) / spr;
    out_stride_ = per * 2048;
    for (int i = 0; i < 2; ++i) {
      d_jobs_[i] = dalloc<hevcgpu::HevcJob>(max_jobs_);
      d_out_[i] =
          dalloc<uint8_t>(static_cast<size_t>(max_jobs_) * out_stride_);
      d_counts_[i] = dalloc<int>(static_cast<size_t>(max_jobs_) * 3);
      HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&h_jobs_[i]),
                              sizeof(hevcgpu::HevcJob) * max_jobs_,
                              hipHostMallocDefault));
      HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&h_out_[i]),
                              static_cast<size_t>(max_jobs_) * out_stride_,
                              hipHostMallocDefault));
      HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&h_counts_[i]),
                              sizeof(int) * 3 * max_jobs_,
                              hipHostMallocDefault));
    }
    HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&h_levels_),
                            levels_bytes_, hipHostMallocDefault));
    HIP_CHECK(hipHostMalloc(reinterpret_cast<void**>(&h_meta_), meta_bytes_,
                            hipHostMallocDefault));
    copy_cap_ = out_stride_;
  }

  CaptureSettings settings_;
  ThreadPool pool_;
  hipStream_t stream_{}, copy_stream_{}, rows_stream_{};
  hipEvent_t ev_[4] = {};
  hipEvent_t ev_h2d_[2] = {}, ev_done_[2] = {};
  hipEvent_t ev_rows_[2] = {};
  bool cpu_entropy_ = false;
  bool timing_ = false;
  int stripe_h_ = 64;
  int depth_ = 1;         // 1 = sync (latency mode), 2 = pipelined
  int parity_ = 0;
  Pending prev_;
  int w_ = 0, h_ = 0, ctbw_ = 0, ctbh_ = 0, ypitch_ = 0, cpitch_ = 0;
  int max_jobs_ = 0, out_stride_ = 0, copy_cap_ = 1 << 30;
  size_t levels_bytes_ = 0, meta_bytes_ = 0;

  uint8_t* d_frame_[2] = {};
  uint8_t *d_srcY_ = nullptr, *d_srcCb_ = nullptr, *d_srcCr_ = nullptr;
  uint8_t *d_curY_ = nullptr, *d_curCb_ = nullptr, *d_curCr_ = nullptr;
  int16_t* d_levels_[2] = {};
  int* d_meta_[2] = {};
  hevcgpu::HevcJob* d_jobs_[2] = {};
  uint8_t* d_out_[2] = {};
  int* d_counts_[2] = {};

  hevcgpu::HevcJob* h_jobs_[2] = {};
  uint8_t* h_out_[2] = {};
  int* h_counts_[2] = {};
  int16_t* h_levels_ = nullptr;
  int* h_meta_ = nullptr;
  uint8_t* h_stage_ = nullptr;
  size_t h_stage_bytes_ = 0;

  std::vector<void*> device_ptrs_;
  std::map<uint8_t*, bool> registered_;
};

}  // namespace

std::unique_ptr<EncodePipeline> make_hip_hevc_pipeline(
    const CaptureSettings& s) {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess || n <= 0) return nullptr;
  try {
    return std::make_unique<HipHevcPipeline>(s);
  } catch (const std::exception& e) {
    std::fprintf(stderr, "hipflux: HEVC HIP pipeline unavailable: %s\n",
                 e.what());
    return nullptr;
  }
}

}  // namespace hipflux
