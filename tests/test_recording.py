"""Raw-ES recording tap: per-stripe .h264 files that decode."""

import threading
import time

import pytest

hipflux = pytest.importorskip("hipflux")
if not hipflux.native_available():
    pytest.skip("hipflux native module not built", allow_module_level=True)

from h264_ref_decoder import Decoder


def test_recording_files_decode(tmp_path):
    s = hipflux.CaptureSettings()
    s.capture_width = 320
    s.capture_height = 128
    s.target_fps = 30
    s.output_mode = 1
    s.use_cpu = True
    s.gpu_id = -1
    s.capture_backend = "synthetic:noise"
    s.video_fullframe = True
    s.stripe_height = 64
    s.recording_path = str(tmp_path / "rec")
    done = threading.Event()
    n = [0]

    def cb(*a):
        n[0] += 1
        if n[0] > 10:
            done.set()

    cap = hipflux.ScreenCapture()
    cap.start_capture(cb, s)
    done.wait(5)
    cap.stop_capture()

    files = sorted(tmp_path.glob("rec.s*.h264"))
    assert [f.name for f in files] == ["rec.s0.h264", "rec.s64.h264"]
    for f in files:
        frames = Decoder().decode(f.read_bytes())
        assert len(frames) >= 3
        assert frames[0][0].shape == (64, 320)
