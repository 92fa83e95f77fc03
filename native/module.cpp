// pybind11 bindings: the `hipflux._native` module.
// API surface mirrors the pixelflux contract the reference control plane
// consumes.
#include <pybind11/functional.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "cpu/damage.h"
#include "cpu/jpeg_enc.h"
#include "engine.h"

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

  m.def("hip_device_count", &hip_device_count,
        "Number of usable HIP devices (0 on CPU-only hosts).");
  m.def("__version__", [] { return "0.1.0"; });
}
