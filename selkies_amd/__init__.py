"""selkies_amd — MI355X-native low-latency remote desktop streaming framework.

A from-scratch rebuild of the capabilities of selkies-project/selkies
(see SURVEY.md for the structural analysis of the reference) with the
capture→encode media path implemented as a C++/HIP engine ("hipflux")
containing hand-written CDNA4 (gfx950) kernels: block damage detection,
BGRX→NV12/I444 color conversion, H.264 transform/quantize, MFMA-int8
motion-estimation, in-loop deblocking and striped JPEG encode.

Layer map (mirrors reference SURVEY.md §1):
  L0  hipflux native engine            (hipflux/ — C++/HIP, pybind11)
  L1  capture orchestration            (selkies_amd/streaming.py)
  L2a WebSocket data plane             (selkies_amd/streaming.py)
  L3  input / clipboard / cursor       (selkies_amd/input_handler.py)
  L4  supervisor + HTTP control plane  (selkies_amd/stream_server.py)
  L5  config                           (selkies_amd/settings.py)
  L6  CLI                              (selkies_amd/__main__.py)
  L7  HTML5 client                     (selkies_amd/web/)
"""

__version__ = "0.1.0"
