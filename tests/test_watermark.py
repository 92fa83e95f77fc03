"""Watermark compositing (engine-level alpha blend)."""

import io
import struct
import threading
import time

import numpy as np
import pytest

hipflux = pytest.importorskip("hipflux")
if not hipflux.native_available():
    pytest.skip("hipflux native module not built", allow_module_level=True)

from PIL import Image


def capture_first_stripe(settings):
    got = {}
    ev = threading.Event()

    def cb(data, fid, y, w, h, key, ts, done, typ):
        if fid == 0 and y == 0 and "d" not in got:
            got["d"] = data
            ev.set()

    cap = hipflux.ScreenCapture()
    cap.start_capture(cb, settings)
    ev.wait(5)
    cap.stop_capture()
    return got.get("d")


def make_settings(tmp_path, location):
    s = hipflux.CaptureSettings()
    s.capture_width = 256
    s.capture_height = 128
    s.target_fps = 30
    s.output_mode = 0          # JPEG (easy pixel check)
    s.jpeg_quality = 95
    s.use_cpu = True
    s.gpu_id = -1
    s.capture_backend = "synthetic:static"
    s.stripe_height = 128
    if location:
        # 32x32 solid opaque white square
        wm = tmp_path / "wm.bgra"
        with open(wm, "wb") as f:
            f.write(struct.pack("<II", 32, 32))
            f.write(b"\xff\xff\xff\xff" * (32 * 32))
        s.watermark_path = str(wm)
        s.watermark_location = location
    return s


def decode(data):
    return np.asarray(Image.open(io.BytesIO(data[6:])).convert("RGB"))


def test_watermark_top_left(tmp_path):
    plain = decode(capture_first_stripe(make_settings(tmp_path, 0)))
    marked = decode(capture_first_stripe(make_settings(tmp_path, 1)))
    # inside the watermark (16..48): near-white; far away: unchanged-ish
    assert marked[20:44, 20:44].mean() > 230
    assert plain[20:44, 20:44].mean() < 200        # random background
    diff_far = np.abs(marked[80:120, 120:200].astype(int) -
                      plain[80:120, 120:200].astype(int)).mean()
    assert diff_far < 3
