"""Cursor: composite into frames (capture_cursor) and shape callback."""

import threading
import time

import pytest

hipflux = pytest.importorskip("hipflux")
if not hipflux.native_available():
    pytest.skip("hipflux native module not built", allow_module_level=True)


def base_settings():
    s = hipflux.CaptureSettings()
    s.capture_width = 256
    s.capture_height = 128
    s.target_fps = 30
    s.output_mode = 0
    s.jpeg_quality = 95
    s.use_cpu = True
    s.gpu_id = -1
    s.capture_backend = "synthetic:static"
    s.stripe_height = 128
    s.damage_block_duration = 1
    s.use_paint_over_quality = False
    return s


def test_cursor_composite_causes_damage():
    """Static background + moving composited cursor => frames keep being
    encoded (the cursor motion is real damage)."""
    s = base_settings()
    s.capture_cursor = True
    n = [0]
    cap = hipflux.ScreenCapture()
    cap.start_capture(lambda *a: n.__setitem__(0, n[0] + 1), s)
    time.sleep(0.6)
    frames = cap.frames_encoded
    cap.stop_capture()
    assert frames > 5, "moving cursor should keep producing damage"


def test_cursor_shape_callback():
    s = base_settings()
    s.capture_cursor = False
    got = {}
    ev = threading.Event()

    def on_cursor(w, h, hx, hy, argb):
        got.update(w=w, h=h, hx=hx, hy=hy, argb=argb)
        ev.set()

    cap = hipflux.ScreenCapture()
    cap.set_cursor_callback(on_cursor)
    cap.start_capture(lambda *a: None, s)
    ev.wait(3)
    cap.stop_capture()
    assert got.get("w") == 12 and got.get("h") == 16
    assert len(got["argb"]) == 12 * 16 * 4
    # shape callback fires once (serial never changes on synthetic)
