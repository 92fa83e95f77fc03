// pybind11 bindings: the `hipflux._native` module.
// API surface mirrors the pixelflux contract the reference control plane
// consumes.
#include <pybind11/functional.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "audio.h"
#include "cpu/damage.h"
#include "cpu/h264/cavlc.h"
#include "cpu/h264/encoder.h"
#include "cpu/hevc/cabac.h"
#include "cpu/hevc/encoder.h"
#include "cpu/opus/celt.h"
#include "cpu/opus/range_coder.h"
#include "cpu/jpeg_enc.h"
#include "cpu/scale.h"
#include "engine.h"
#include <hip/hip_runtime.h>

#include "rccl_comm.h"

namespace py = pybind11;
using namespace hipflux;

// Forward decl from hip_util.cpp (or stub).
namespace hipflux {
int hip_device_count();
}

PYBIND11_MODULE(_native, m) {
  m.doc() = "hipflux: MI355X-native capture+encode engine";

  py::class_<CaptureSettings>(m, "CaptureSettings")
      .def(py::init<>())
      .def_readwrite("capture_width", &CaptureSettings::capture_width)
      .def_readwrite("capture_height", &CaptureSettings::capture_height)
      .def_readwrite("capture_x", &CaptureSettings::capture_x)
      .def_readwrite("capture_y", &CaptureSettings::capture_y)
      .def_readwrite("target_fps", &CaptureSettings::target_fps)
      .def_readwrite("capture_cursor", &CaptureSettings::capture_cursor)
      .def_readwrite("output_mode", &CaptureSettings::output_mode)
      .def_readwrite("capture_scale", &CaptureSettings::capture_scale)
      .def_readwrite("capture_scale_div", &CaptureSettings::capture_scale_div)
      .def_readwrite("pipeline_depth", &CaptureSettings::pipeline_depth)
      .def_readwrite("video_fullframe", &CaptureSettings::video_fullframe)
      .def_readwrite("use_cpu", &CaptureSettings::use_cpu)
      .def_readwrite("gpu_id", &CaptureSettings::gpu_id)
      .def_readwrite("video_bitrate_kbps", &CaptureSettings::video_bitrate_kbps)
      .def_readwrite("video_crf", &CaptureSettings::video_crf)
      .def_readwrite("video_cbr_mode", &CaptureSettings::video_cbr_mode)
      .def_readwrite("video_min_qp", &CaptureSettings::video_min_qp)
      .def_readwrite("video_max_qp", &CaptureSettings::video_max_qp)
      .def_readwrite("vbv_multiplier", &CaptureSettings::vbv_multiplier)
      .def_readwrite("keyframe_interval_s", &CaptureSettings::keyframe_interval_s)
      .def_readwrite("video_streaming_mode", &CaptureSettings::video_streaming_mode)
      .def_readwrite("video_deblock", &CaptureSettings::video_deblock)
      .def_readwrite("video_fullcolor", &CaptureSettings::video_fullcolor)
      .def_readwrite("use_paint_over_quality", &CaptureSettings::use_paint_over_quality)
      .def_readwrite("paint_over_trigger_frames", &CaptureSettings::paint_over_trigger_frames)
      .def_readwrite("video_paintover_crf", &CaptureSettings::video_paintover_crf)
      .def_readwrite("video_paintover_burst_frames", &CaptureSettings::video_paintover_burst_frames)
      .def_readwrite("damage_block_threshold", &CaptureSettings::damage_block_threshold)
      .def_readwrite("damage_block_duration", &CaptureSettings::damage_block_duration)
      .def_readwrite("jpeg_quality", &CaptureSettings::jpeg_quality)
      .def_readwrite("jpeg_paintover_quality", &CaptureSettings::jpeg_paintover_quality)
      .def_readwrite("stripe_height", &CaptureSettings::stripe_height)
      .def_readwrite("omit_stripe_headers", &CaptureSettings::omit_stripe_headers)
      .def_readwrite("watermark_path", &CaptureSettings::watermark_path)
      .def_readwrite("watermark_location", &CaptureSettings::watermark_location)
      .def_readwrite("capture_backend", &CaptureSettings::capture_backend)
      .def_readwrite("display", &CaptureSettings::display)
      .def_readwrite("recording_path", &CaptureSettings::recording_path)
      .def_readwrite("debug_logging", &CaptureSettings::debug_logging);

  py::class_<ScreenCapture>(m, "ScreenCapture")
      .def(py::init<>())
      .def(
          "start_capture",
          [](ScreenCapture& self, py::function cb, const CaptureSettings& s) {
            {
              // stop any previous run, then free its callback with GIL held
              py::gil_scoped_release rel;
              self.stop_capture();
            }
            self.clear_callback();
            StripeCallback native_cb = [cb](const EncodedStripe& st) {
              py::gil_scoped_acquire gil;
              try {
                py::bytes data(reinterpret_cast<const char*>(st.data),
                               st.size);
                cb(data, st.frame_id, st.y, st.width, st.height,
                   st.is_keyframe, st.capture_ts_ms, st.encode_done_ms,
                   static_cast<int>(st.type));
              } catch (py::error_already_set& e) {
                e.discard_as_unraisable("hipflux stripe callback");
              }
            };
            py::gil_scoped_release rel;
            self.start_capture(std::move(native_cb), s);
          },
          py::arg("callback"), py::arg("settings"),
          "Start the native capture+encode thread. callback(data, frame_id, "
          "y, width, height, is_keyframe, capture_ts_ms, encode_done_ms, "
          "stripe_type) runs on the native thread.")
      .def("stop_capture", &ScreenCapture::stop_capture,
           py::call_guard<py::gil_scoped_release>())
      .def(
          "set_cursor_callback",
          [](ScreenCapture& self, py::function cb) {
            // set BEFORE start_capture (callback read on capture thread)
            self.clear_cursor_callback();
            self.set_cursor_callback(
                [cb](int w, int h, int hx, int hy, const uint32_t* argb,
                     size_t n) {
                  py::gil_scoped_acquire gil;
                  try {
                    cb(w, h, hx, hy,
                       py::bytes(reinterpret_cast<const char*>(argb),
                                 n * 4));
                  } catch (py::error_already_set& e) {
                    e.discard_as_unraisable("hipflux cursor callback");
                  }
                });
          },
          "callback(width, height, hot_x, hot_y, argb_bytes) on shape change")
      .def_property_readonly("is_capturing", &ScreenCapture::is_capturing)
      .def("request_idr_frame", &ScreenCapture::request_idr_frame)
      .def("update_framerate", &ScreenCapture::update_framerate)
      .def("update_video_bitrate", &ScreenCapture::update_video_bitrate)
      .def("update_crf", &ScreenCapture::update_crf)
      .def("update_jpeg_quality", &ScreenCapture::update_jpeg_quality)
      .def("update_vbv_multiplier", &ScreenCapture::update_vbv_multiplier)
      .def("update_capture_region", &ScreenCapture::update_capture_region)
      .def_property_readonly("frames_captured", &ScreenCapture::frames_captured)
      .def_property_readonly("frames_encoded", &ScreenCapture::frames_encoded)
      .def_property_readonly("stripes_emitted", &ScreenCapture::stripes_emitted)
      .def_property_readonly("last_encode_ms", &ScreenCapture::last_encode_ms)
      .def_property_readonly("pipeline",
                             [](ScreenCapture& s) { return s.pipeline_name(); });

  // ---- direct encode / test helpers ---------------------------------------
  m.def(
      "jpeg_encode",
      [](py::buffer bgrx, int width, int height, int quality, bool fullcolor) {
        py::buffer_info info = bgrx.request();
        if (info.size < static_cast<ssize_t>(width) * height * 4)
          throw std::runtime_error("buffer too small for WxHx4");
        std::vector<uint8_t> out;
        {
          py::gil_scoped_release rel;
          jpeg_encode_bgrx(static_cast<const uint8_t*>(info.ptr), width * 4,
                           width, height, quality, fullcolor, out);
        }
        return py::bytes(reinterpret_cast<const char*>(out.data()),
                         out.size());
      },
      py::arg("bgrx"), py::arg("width"), py::arg("height"),
      py::arg("quality") = 80, py::arg("fullcolor") = false,
      "Encode a BGRX buffer as baseline JPEG (CPU reference path).");

  m.def(
      "_jpeg_encode_restart",
      [](py::buffer bgrx, int width, int height, int quality,
         bool fullcolor) {
        py::buffer_info info = bgrx.request();
        std::vector<uint8_t> out;
        {
          py::gil_scoped_release rel;
          jpeg_encode_bgrx(static_cast<const uint8_t*>(info.ptr), width * 4,
                           width, height, quality, fullcolor, out, true);
        }
        return py::bytes(reinterpret_cast<const char*>(out.data()),
                         out.size());
      },
      py::arg("bgrx"), py::arg("width"), py::arg("height"),
      py::arg("quality") = 80, py::arg("fullcolor") = false,
      "Test hook: restart-row JPEG framing (the GPU entropy structure).");

  // ---- H.264 stripe encoder (direct access for tests + conformance) ------
  struct PyH264 {
    h264::StripeEncoder enc;
    int w, h, ypitch, cpitch;
    bool fullcolor;
    std::vector<uint8_t> yuv;
    PyH264(int width, int height, bool deblock, bool fc)
        : enc(width, height, deblock, fc), w(width), h(height),
          fullcolor(fc) {
      ypitch = (w + 15) & ~15;
      cpitch = fullcolor ? ypitch : ypitch / 2;
      int yh = (h + 15) & ~15;
      yuv.resize(static_cast<size_t>(ypitch) * yh * (fullcolor ? 3 : 2) *
                 (fullcolor ? 1 : 3) / (fullcolor ? 1 : 4));
    }
  };
  py::class_<PyH264>(m, "H264Encoder")
      .def(py::init<int, int, bool, bool>(), py::arg("width"),
           py::arg("height"), py::arg("deblock") = true,
           py::arg("fullcolor") = false)
      .def(
          "encode",
          [](PyH264& self, py::buffer bgrx, int qp, bool idr) {
            py::buffer_info info = bgrx.request();
            if (info.size < static_cast<ssize_t>(self.w) * self.h * 4)
              throw std::runtime_error("buffer too small");
            std::vector<uint8_t> out;
            h264::EncodeStats st;
            {
              py::gil_scoped_release rel;
              uint8_t* y = self.yuv.data();
              int yh = (self.h + 15) & ~15;
              uint8_t* cb = y + static_cast<size_t>(self.ypitch) * yh;
              uint8_t* cr = cb + static_cast<size_t>(self.cpitch) *
                                     (self.fullcolor ? yh : yh / 2);
              if (self.fullcolor)
                h264::bgrx_to_yuv444(static_cast<const uint8_t*>(info.ptr),
                                     self.w * 4, self.w, self.h, y, cb, cr,
                                     self.ypitch);
              else
                h264::bgrx_to_yuv420(static_cast<const uint8_t*>(info.ptr),
                                     self.w * 4, self.w, self.h, y,
                                     self.ypitch, cb, cr, self.cpitch);
              self.enc.encode_frame(y, self.ypitch, cb, cr, self.cpitch, qp,
                                    idr, out, &st);
            }
            py::dict d;
            d["data"] = py::bytes(reinterpret_cast<const char*>(out.data()),
                                  out.size());
            d["is_idr"] = st.is_idr;
            d["mb_intra"] = st.mb_intra;
            d["mb_inter"] = st.mb_inter;
            d["mb_skip"] = st.mb_skip;
            return d;
          },
          py::arg("bgrx"), py::arg("qp") = 26, py::arg("idr") = false)
      .def(
          "encode_yuv",
          [](PyH264& self, py::buffer ybuf, py::buffer cbbuf, py::buffer crbuf,
             int qp, bool idr) {
            auto yi = ybuf.request(), cbi = cbbuf.request(),
                 cri = crbuf.request();
            std::vector<uint8_t> out;
            h264::EncodeStats st;
            {
              py::gil_scoped_release rel;
              self.enc.encode_frame(static_cast<const uint8_t*>(yi.ptr),
                                    self.w, static_cast<const uint8_t*>(cbi.ptr),
                                    static_cast<const uint8_t*>(cri.ptr),
                                    (self.w + 1) / 2, qp, idr, out, &st);
            }
            return py::bytes(reinterpret_cast<const char*>(out.data()),
                             out.size());
          },
          py::arg("y"), py::arg("cb"), py::arg("cr"), py::arg("qp") = 26,
          py::arg("idr") = false,
          "Encode from tightly-packed YUV420 planes (pitch = width).")
      .def("recon",
           [](PyH264& self) {
             // returns (y, cb, cr, ypitch, cpitch) of the reference frame
             int yh = (self.h + 15) & ~15;
             int ch = self.fullcolor ? yh : yh / 2;
             py::bytes y(reinterpret_cast<const char*>(self.enc.recon_y()),
                         static_cast<size_t>(self.enc.recon_ypitch()) * yh);
             py::bytes cb(reinterpret_cast<const char*>(self.enc.recon_cb()),
                          static_cast<size_t>(self.enc.recon_cpitch()) * ch);
             py::bytes cr(reinterpret_cast<const char*>(self.enc.recon_cr()),
                          static_cast<size_t>(self.enc.recon_cpitch()) * ch);
             return py::make_tuple(y, cb, cr, self.enc.recon_ypitch(),
                                   self.enc.recon_cpitch());
           });

  // ---- HEVC CPU reference encoder (BASELINE config 3 path) --------------
  struct PyHevc {
    hevc::StripeEncoder enc;
    int w, h, ypitch, cpitch;
    std::vector<uint8_t> yuv;
    PyHevc(int width, int height, int slices_per_row)
        : enc(width, height, slices_per_row), w(width), h(height) {
      ypitch = (width + 15) & ~15;
      cpitch = ypitch / 2;
      int yh = (height + 15) & ~15;
      yuv.resize(static_cast<size_t>(ypitch) * yh * 3 / 2);
    }
  };
  py::class_<PyHevc>(m, "HevcEncoder")
      .def(py::init<int, int, int>(), py::arg("width"), py::arg("height"),
           py::arg("slices_per_row") = 1)
      .def(
          "encode",
          [](PyHevc& self, py::buffer bgrx, int qp) {
            py::buffer_info info = bgrx.request();
            if (info.size < static_cast<ssize_t>(self.w) * self.h * 4)
              throw std::runtime_error("buffer too small");
            std::vector<uint8_t> out;
            hevc::EncodeStats st;
            {
              py::gil_scoped_release rel;
              uint8_t* y = self.yuv.data();
              int yh = (self.h + 15) & ~15;
              uint8_t* cb = y + static_cast<size_t>(self.ypitch) * yh;
              uint8_t* cr = cb + static_cast<size_t>(self.cpitch) * (yh / 2);
              h264::bgrx_to_yuv420(static_cast<const uint8_t*>(info.ptr),
                                   self.w * 4, self.w, self.h, y, self.ypitch,
                                   cb, cr, self.cpitch);
              self.enc.encode_frame(y, self.ypitch, cb, cr, self.cpitch, qp,
                                    out, &st);
            }
            py::dict d;
            d["data"] = py::bytes(reinterpret_cast<const char*>(out.data()),
                                  out.size());
            d["qp"] = st.frame_qp;
            d["ctus"] = st.ctu_count;
            return d;
          },
          py::arg("bgrx"), py::arg("qp") = 30)
      .def("recon", [](PyHevc& self) {
        int yh = (self.h + 15) & ~15;
        int yp = self.enc.recon_ypitch(), cp = self.enc.recon_cpitch();
        py::bytes y(reinterpret_cast<const char*>(self.enc.recon_y()),
                    static_cast<size_t>(yp) * yh);
        py::bytes cb(reinterpret_cast<const char*>(self.enc.recon_cb()),
                     static_cast<size_t>(cp) * yh / 2);
        py::bytes cr(reinterpret_cast<const char*>(self.enc.recon_cr()),
                     static_cast<size_t>(cp) * yh / 2);
        return py::make_tuple(y, cb, cr, yp, cp);
      });

  // CABAC pair-fuzz hook: encode an explicit (kind, ctx, bin) op sequence
  // with the I-slice context bank; tests decode it back with the from-spec
  // Python CABAC decoder. kind: 0 = context bin, 1 = bypass, 2 = terminate.
  m.def(
      "_hevc_cabac_encode",
      [](const std::vector<std::tuple<int, int, int>>& ops, int qp) {
        std::vector<uint8_t> bytes;
        hevc::CabacEncoder cab(bytes);
        hevc::ContextBank bank;
        bank.init(qp);
        for (const auto& op : ops) {
          int kind = std::get<0>(op), ctx = std::get<1>(op),
              bin = std::get<2>(op);
          if (kind == 0)
            cab.encode_bin(bank.ctx[ctx % hevc::kNumContexts], bin);
          else if (kind == 1)
            cab.encode_bypass(bin);
          else
            cab.encode_terminate(bin);
        }
        auto tail = cab.finish();
        return py::make_tuple(
            py::bytes(reinterpret_cast<const char*>(bytes.data()),
                      bytes.size()),
            tail.bits, tail.nbits);
      },
      py::arg("ops"), py::arg("qp") = 30);

  // ---- Opus-framed CELT-class audio encoder -----------------------------
  py::class_<opus::CeltEncoder>(m, "OpusEncoder")
      .def(py::init<int>(), py::arg("bitrate_bps") = 96000)
      .def("set_bitrate", &opus::CeltEncoder::set_bitrate)
      .def(
          "encode",
          [](opus::CeltEncoder& e, py::buffer pcm, int channels) {
            py::buffer_info info = pcm.request();
            if (info.size <
                static_cast<ssize_t>(opus::kFrameSamples) * channels)
              throw std::runtime_error("need 960 frames of s16 PCM");
            std::vector<uint8_t> out;
            {
              py::gil_scoped_release rel;
              out = e.encode_frame(
                  static_cast<const int16_t*>(info.ptr), channels);
            }
            return py::bytes(reinterpret_cast<const char*>(out.data()),
                             out.size());
          },
          py::arg("pcm"), py::arg("channels") = 2);

  // range-coder pair-fuzz hook (tests/test_opus.py): ops =
  // (kind, a, b): 0 = bit_logp(bit=a, logp=b), 1 = uint(fl=a, ft=b),
  // 2 = raw bits(fl=a, nbits=b), 3 = icdf(sym=a over a 4-entry table)
  m.def("_opus_range_encode",
        [](const std::vector<std::tuple<int, int, int>>& ops,
           size_t capacity) {
          opus::RangeEncoder ec(capacity);
          static const uint8_t icdf4[4] = {200, 120, 40, 0};
          for (const auto& op : ops) {
            int kind = std::get<0>(op), a = std::get<1>(op),
                b = std::get<2>(op);
            if (kind == 0)
              ec.enc_bit_logp(a, static_cast<unsigned>(b));
            else if (kind == 1)
              ec.enc_uint(static_cast<uint32_t>(a),
                          static_cast<uint32_t>(b));
            else if (kind == 2)
              ec.enc_bits(static_cast<uint32_t>(a),
                          static_cast<unsigned>(b));
            else
              ec.enc_icdf(a % 4, icdf4, 8);
          }
          ec.done();
          auto s = ec.stream();
          return py::make_tuple(
              py::bytes(reinterpret_cast<const char*>(s.data()),
                        s.size()),
              ec.error());
        });

  // ---- RCCL tile-parallel collective layer (BASELINE config 5) ----------
  py::class_<TileComm>(m, "TileComm")
      .def(py::init<int, int, const std::string&, int>(), py::arg("rank"),
           py::arg("world"), py::arg("uid"), py::arg("device") = 0)
      .def_static("make_uid",
                  [] { return py::bytes(TileComm::make_uid()); })
      .def_property_readonly("rank", &TileComm::rank)
      .def_property_readonly("world", &TileComm::world)
      .def(
          "exchange",
          [](TileComm& c, uintptr_t src_dev, size_t bytes, int schedule) {
            py::gil_scoped_release rel;
            return c.exchange(src_dev, bytes, schedule);
          },
          py::arg("src_dev"), py::arg("bytes"), py::arg("schedule") = 1,
          "All-gather this rank's device payload to every rank; "
          "schedule 0 = ring (bulk), 1 = one-hop p2p (latency). "
          "Returns ms.")
      .def(
          "exchange_host",
          [](TileComm& c, py::bytes payload, int schedule) {
            std::string s = payload;
            void* d = nullptr;
            if (hipMalloc(&d, s.size()) != hipSuccess)
              throw std::runtime_error("alloc failed");
            (void)hipMemcpy(d, s.data(), s.size(), hipMemcpyHostToDevice);
            double ms;
            {
              py::gil_scoped_release rel;
              ms = c.exchange(reinterpret_cast<uintptr_t>(d), s.size(),
                              schedule);
            }
            (void)hipFree(d);
            return ms;
          },
          py::arg("payload"), py::arg("schedule") = 1,
          "Upload host bytes and exchange (tests).")
      .def("gathered",
           [](TileComm& c, int rank, size_t bytes) {
             auto v = c.gathered(rank, bytes);
             return py::bytes(reinterpret_cast<const char*>(v.data()),
                              v.size());
           })
      .def(
          "broadcast",
          [](TileComm& c, uintptr_t buf, size_t bytes, int root) {
            py::gil_scoped_release rel;
            return c.broadcast(buf, bytes, root);
          },
          py::arg("buf_dev"), py::arg("bytes"), py::arg("root") = 0,
          "Broadcast a reference-frame buffer from root. Returns ms.");

  m.def(
      "bgrx_to_yuv420",
      [](py::buffer bgrx, int width, int height) {
        py::buffer_info info = bgrx.request();
        int cw = (width + 1) / 2, ch = (height + 1) / 2;
        std::vector<uint8_t> y(static_cast<size_t>(width) * height),
            cb(static_cast<size_t>(cw) * ch), cr(static_cast<size_t>(cw) * ch);
        h264::bgrx_to_yuv420(static_cast<const uint8_t*>(info.ptr), width * 4,
                             width, height, y.data(), width, cb.data(),
                             cr.data(), cw);
        return py::make_tuple(
            py::bytes(reinterpret_cast<char*>(y.data()), y.size()),
            py::bytes(reinterpret_cast<char*>(cb.data()), cb.size()),
            py::bytes(reinterpret_cast<char*>(cr.data()), cr.size()));
      },
      py::arg("bgrx"), py::arg("width"), py::arg("height"));

  m.def(
      "_pipeline_encode",
      [](const std::string& kind, py::list frames, int w, int h, int qp,
         int stripe_h, int output_mode, bool dump, bool fullcolor,
         int pipeline_depth) -> py::object {
        CaptureSettings s;
        s.capture_width = w;
        s.capture_height = h;
        s.output_mode = output_mode;
        s.stripe_height = stripe_h;
        s.video_crf = qp;
        s.jpeg_quality = qp;
        s.video_fullcolor = fullcolor;
        s.use_cpu = kind == "cpu";
        s.gpu_id = kind == "cpu" ? -1 : 0;
        std::unique_ptr<EncodePipeline> p;
        if (kind == "gpu") {
          p = make_hip_pipeline(s);
          if (!p) throw std::runtime_error("no HIP pipeline available");
        } else {
          p = output_mode == 1   ? make_cpu_h264_pipeline(s)
              : output_mode == 2 ? make_cpu_hevc_pipeline(s)
                                 : make_cpu_jpeg_pipeline(s);
        }
        p->set_pipeline_depth(pipeline_depth);
        // emissions land in result[st.frame_id] so depth-2 pipelining
        // (stripes of frame N emitted during the call for N+1) yields the
        // same per-frame grouping as the synchronous path
        std::vector<py::list> per_frame(frames.size());
        auto sink = [&](EncodedStripe& st) {
          per_frame[st.frame_id].append(py::make_tuple(
              py::bytes(reinterpret_cast<const char*>(st.data), st.size),
              st.y, st.height, st.is_keyframe));
        };
        py::list result;
        for (size_t i = 0; i < frames.size(); ++i) {
          py::buffer buf = frames[i].cast<py::buffer>();
          py::buffer_info info = buf.request();
          if (info.size < static_cast<ssize_t>(w) * h * 4)
            throw std::runtime_error("frame buffer too small");
          RawFrame f;
          f.data = static_cast<const uint8_t*>(info.ptr);
          f.width = w;
          f.height = h;
          f.stride = w * 4;
          f.ts_ms = now_ms();
          FrameContext ctx;
          ctx.frame_id = static_cast<uint32_t>(i);
          ctx.idr = (i == 0);
          ctx.crf = qp;
          ctx.jpeg_quality = qp;
          for (int y = 0; y < h; y += stripe_h) {
            StripeJob j;
            j.y0 = y;
            j.y1 = std::min(y + stripe_h, h);
            j.encode = true;
            ctx.stripes.push_back(j);
          }
          p->encode_frame(f, ctx, sink);
        }
        p->flush(sink);
        for (auto& fo : per_frame) result.append(fo);
        if (dump) {
          EncodePipeline::DebugDump d;
          if (!p->debug_dump(d))
            throw std::runtime_error("pipeline has no debug_dump");
          py::dict dd;
          dd["w"] = d.w;
          dd["h"] = d.h;
          dd["ypitch"] = d.ypitch;
          dd["cpitch"] = d.cpitch;
          dd["y"] = py::bytes(reinterpret_cast<char*>(d.y.data()),
                              d.y.size());
          dd["cb"] = py::bytes(reinterpret_cast<char*>(d.cb.data()),
                               d.cb.size());
          dd["cr"] = py::bytes(reinterpret_cast<char*>(d.cr.data()),
                               d.cr.size());
          dd["levels"] = py::bytes(
              reinterpret_cast<char*>(d.levels.data()),
              d.levels.size() * sizeof(int16_t));
          dd["meta"] = py::bytes(reinterpret_cast<char*>(d.meta.data()),
                                 d.meta.size() * sizeof(int));
          return py::object(py::make_tuple(result, dd));
        }
        return py::object(result);
      },
      py::arg("kind"), py::arg("frames"), py::arg("w"), py::arg("h"),
      py::arg("qp") = 26, py::arg("stripe_h") = 64,
      py::arg("output_mode") = 1, py::arg("dump") = false,
      py::arg("fullcolor") = false, py::arg("pipeline_depth") = 1,
      "Test hook: run frames through a named encode pipeline.");

  // ---- persistent pipeline handle for benchmarking ------------------------
  struct BenchPipeline {
    std::unique_ptr<EncodePipeline> p;
    int w, h, qp, stripe_h;
    uint32_t frame_id = 0;
    long last_frame_id = -1;   // frame whose stripes the last call emitted
    void* d_boundary = nullptr;
    size_t boundary_cap = 0;
    BenchPipeline(const std::string& kind, int width, int height, int qp_,
                  int stripe, int output_mode, int gpu_id,
                  int pipeline_depth)
        : w(width), h(height), qp(qp_), stripe_h(stripe) {
      CaptureSettings s;
      s.capture_width = w;
      s.capture_height = h;
      s.output_mode = output_mode;
      s.stripe_height = stripe_h;
      s.video_crf = qp;
      s.jpeg_quality = qp;
      s.gpu_id = gpu_id;
      s.use_cpu = kind == "cpu";
      if (kind == "gpu") {
        p = make_hip_pipeline(s);
        if (!p) throw std::runtime_error("no HIP pipeline available");
      } else {
        p = output_mode == 1   ? make_cpu_h264_pipeline(s)
            : output_mode == 2 ? make_cpu_hevc_pipeline(s)
                               : make_cpu_jpeg_pipeline(s);
      }
      p->set_pipeline_depth(pipeline_depth);
    }
  };
  py::class_<BenchPipeline>(m, "BenchPipeline")
      .def(py::init<const std::string&, int, int, int, int, int, int,
                    int>(),
           py::arg("kind"), py::arg("width"), py::arg("height"),
           py::arg("qp") = 28, py::arg("stripe_height") = 64,
           py::arg("output_mode") = 1, py::arg("gpu_id") = 0,
           py::arg("pipeline_depth") = 1)
      .def_property_readonly("pipeline",
                             [](BenchPipeline& b) { return b.p->name(); })
      .def_readonly("last_frame_id", &BenchPipeline::last_frame_id)
      .def(
          "boundary_dev",
          [](BenchPipeline& b, int rows) {
            // pack the top and bottom `rows` recon rows into a compact
            // device buffer for the RCCL tile-boundary exchange; returns
            // (device_ptr, nbytes) or (0, 0) on CPU pipelines
            void* y = nullptr;
            int ypitch = 0, height = 0;
            if (!b.p->recon_dev(&y, &ypitch, &height)) {
              return py::make_tuple(static_cast<uintptr_t>(0),
                                    static_cast<size_t>(0));
            }
            rows = std::min(rows, height / 2);
            size_t need = static_cast<size_t>(ypitch) * rows * 2;
            if (b.boundary_cap < need) {
              if (b.d_boundary) (void)hipFree(b.d_boundary);
              if (hipMalloc(&b.d_boundary, need) != hipSuccess)
                throw std::runtime_error("boundary alloc failed");
              b.boundary_cap = need;
            }
            auto* dst = static_cast<uint8_t*>(b.d_boundary);
            auto* srcy = static_cast<uint8_t*>(y);
            size_t half = static_cast<size_t>(ypitch) * rows;
            if (hipMemcpy(dst, srcy, half, hipMemcpyDeviceToDevice) !=
                    hipSuccess ||
                hipMemcpy(dst + half,
                          srcy + static_cast<size_t>(ypitch) *
                                     (height - rows),
                          half, hipMemcpyDeviceToDevice) != hipSuccess)
              throw std::runtime_error("boundary pack failed");
            return py::make_tuple(
                reinterpret_cast<uintptr_t>(b.d_boundary),
                static_cast<size_t>(need));
          },
          py::arg("rows") = 16,
          "Pack top+bottom recon rows into a device buffer; returns "
          "(dev_ptr, nbytes) for TileComm.exchange.")
      .def(
          "resize",
          [](BenchPipeline& b, int w, int h) {
            b.w = w;
            b.h = h;
          },
          py::arg("w"), py::arg("h"),
          "Change frame dimensions (next encode reallocates; an "
          "in-flight pipelined frame is dropped, streams restart IDR).")
      .def(
          "flush",
          [](BenchPipeline& b) {
            size_t total = 0;
            int stripes = 0;
            long done = -1;
            {
              py::gil_scoped_release rel;
              b.p->flush([&](EncodedStripe& st) {
                total += st.size;
                ++stripes;
                done = st.frame_id;
              });
            }
            b.last_frame_id = done;
            return py::make_tuple(total, stripes);
          },
          "Emit any pipelined frame still in flight (depth 2); returns "
          "(bitstream_bytes, stripes).")
      .def(
          "encode",
          [](BenchPipeline& b, py::buffer bgrx, bool idr) {
            py::buffer_info info = bgrx.request();
            if (info.size < static_cast<ssize_t>(b.w) * b.h * 4)
              throw std::runtime_error("frame buffer too small");
            size_t total = 0;
            int stripes = 0;
            {
              py::gil_scoped_release rel;
              RawFrame f;
              f.data = static_cast<const uint8_t*>(info.ptr);
              f.width = b.w;
              f.height = b.h;
              f.stride = b.w * 4;
              f.ts_ms = now_ms();
              FrameContext ctx;
              ctx.frame_id = b.frame_id++;
              ctx.idr = idr;
              ctx.crf = b.qp;
              ctx.jpeg_quality = b.qp;
              for (int y = 0; y < b.h; y += b.stripe_h) {
                StripeJob j;
                j.y0 = y;
                j.y1 = std::min(y + b.stripe_h, b.h);
                j.encode = true;
                ctx.stripes.push_back(j);
              }
              long done = -1;
              b.p->encode_frame(f, ctx, [&](EncodedStripe& st) {
                total += st.size;
                ++stripes;
                done = st.frame_id;
              });
              b.last_frame_id = done;
            }
            return py::make_tuple(total, stripes);
          },
          py::arg("bgrx"), py::arg("idr") = false,
          "Encode one frame; returns (bitstream_bytes, stripes).");

  m.def(
      "_cavlc_bits",
      [](std::vector<int> zz, int nC) {
        h264::BitWriter bw;
        int tmp[16] = {};
        for (size_t i = 0; i < zz.size() && i < 16; ++i) tmp[i] = zz[i];
        h264::cavlc_residual(bw, tmp, static_cast<int>(zz.size()), nC);
        size_t nbits = bw.bit_count();
        bw.put_bit(1);
        while (bw.bit_count() % 8) bw.put_bit(0);
        std::string s;
        const auto& bytes = bw.bytes();
        for (size_t i = 0; i < nbits; ++i)
          s += ((bytes[i / 8] >> (7 - i % 8)) & 1) ? '1' : '0';
        return s;
      },
      "debug: CAVLC-encode one zigzag block, return bit string");

  // ---- audio (pcmflux contract) -------------------------------------------
  py::class_<AudioCaptureSettings>(m, "AudioCaptureSettings")
      .def(py::init<>())
      .def_readwrite("device_name", &AudioCaptureSettings::device_name)
      .def_readwrite("sample_rate", &AudioCaptureSettings::sample_rate)
      .def_readwrite("channels", &AudioCaptureSettings::channels)
      .def_readwrite("codec", &AudioCaptureSettings::codec)
      .def_readwrite("opus_bitrate", &AudioCaptureSettings::opus_bitrate)
      .def_readwrite("frame_duration_ms",
                     &AudioCaptureSettings::frame_duration_ms)
      .def_readwrite("red_distance", &AudioCaptureSettings::red_distance)
      .def_readwrite("omit_audio_header",
                     &AudioCaptureSettings::omit_audio_header)
      .def_readwrite("debug_logging", &AudioCaptureSettings::debug_logging);

  py::class_<AudioCapture>(m, "AudioCapture")
      .def(py::init<>())
      .def(
          "start_capture",
          [](AudioCapture& self, const AudioCaptureSettings& s,
             py::function cb) {
            {
              py::gil_scoped_release rel;
              self.stop_capture();
            }
            self.clear_callback();
            AudioCapture::Callback native = [cb](const AudioFrame& f) {
              py::gil_scoped_acquire gil;
              try {
                cb(py::bytes(reinterpret_cast<const char*>(f.data), f.size),
                   f.pts_ms);
              } catch (py::error_already_set& e) {
                e.discard_as_unraisable("hipflux audio callback");
              }
            };
            py::gil_scoped_release rel;
            self.start_capture(s, std::move(native));
          },
          py::arg("settings"), py::arg("callback"))
      .def("stop_capture", &AudioCapture::stop_capture,
           py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("is_capturing", &AudioCapture::is_capturing)
      .def("update_audio_bitrate", &AudioCapture::update_audio_bitrate);

  py::class_<AudioPlaybackSettings>(m, "AudioPlaybackSettings")
      .def(py::init<>())
      .def_readwrite("sample_rate", &AudioPlaybackSettings::sample_rate)
      .def_readwrite("channels", &AudioPlaybackSettings::channels)
      .def_readwrite("max_buffer_bytes",
                     &AudioPlaybackSettings::max_buffer_bytes);

  py::class_<AudioPlayback>(m, "AudioPlayback")
      .def(py::init<const AudioPlaybackSettings&>())
      .def("write",
           [](AudioPlayback& self, py::bytes data) {
             std::string s = data;
             return self.write(
                 reinterpret_cast<const uint8_t*>(s.data()), s.size());
           })
      .def("read",
           [](AudioPlayback& self, size_t n) {
             std::vector<uint8_t> out(n);
             size_t got = self.read(out.data(), n);
             return py::bytes(reinterpret_cast<char*>(out.data()), got);
           })
      .def_property_readonly("buffered", &AudioPlayback::buffered);

  m.def(
      "_bilinear_downscale",
      [](py::buffer bgrx, int w, int h, float scale) {
        py::buffer_info info = bgrx.request();
        if (info.size < static_cast<ssize_t>(w) * h * 4)
          throw std::runtime_error("buffer too small");
        std::vector<uint8_t> out;
        int ow, oh, ostride;
        bilinear_downscale_bgrx(static_cast<const uint8_t*>(info.ptr),
                                w * 4, w, h, scale, out, ow, oh, ostride);
        return py::make_tuple(
            py::bytes(reinterpret_cast<const char*>(out.data()), out.size()),
            ow, oh);
      },
      py::arg("bgrx"), py::arg("w"), py::arg("h"), py::arg("scale"),
      "Test hook: exact fixed-point bilinear downscale (engine path).");

  m.def(
      "screenshot",
      [](const std::string& backend, const std::string& display, int w,
         int h) {
        std::unique_ptr<FrameSource> src;
        {
          py::gil_scoped_release rel;
          if (backend == "x11" || (backend == "auto" && !display.empty()))
            src = make_x11_source(display, 0, 0, w, h);
          if (!src) {
            std::string pattern = "desktop";
            auto pos = backend.find(':');
            if (pos != std::string::npos) pattern = backend.substr(pos + 1);
            src = make_synthetic_source(w > 0 ? w : 1920, h > 0 ? h : 1080,
                                        pattern);
          }
        }
        RawFrame f;
        if (!src->acquire(f)) throw std::runtime_error("capture failed");
        // tightly pack
        std::vector<uint8_t> out(static_cast<size_t>(f.width) * f.height * 4);
        for (int y = 0; y < f.height; ++y)
          std::memcpy(out.data() + static_cast<size_t>(y) * f.width * 4,
                      f.data + static_cast<size_t>(y) * f.stride,
                      static_cast<size_t>(f.width) * 4);
        return py::make_tuple(
            py::bytes(reinterpret_cast<char*>(out.data()), out.size()),
            f.width, f.height);
      },
      py::arg("backend") = "auto", py::arg("display") = "",
      py::arg("w") = 0, py::arg("h") = 0,
      "One-shot framebuffer grab -> (bgrx, w, h).");

  m.def("hip_device_count", &hip_device_count,
        "Number of usable HIP devices (0 on CPU-only hosts).");
  m.def("__version__", [] { return "0.1.0"; });
}
