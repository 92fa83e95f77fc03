"""CBR rate control: engine converges the produced bitrate toward the
target on incompressible content, stays within QP clamps, and the varying
per-frame QP still decodes (slice_qp_delta path)."""

import threading
import time

import pytest

hipflux = pytest.importorskip("hipflux")
if not hipflux.native_available():
    pytest.skip("hipflux native module not built", allow_module_level=True)

from h264_ref_decoder import Decoder


class Collector:
    def __init__(self):
        self.lock = threading.Lock()
        self.stripes = []

    def __call__(self, data, frame_id, y, *rest):
        with self.lock:
            self.stripes.append((frame_id, y, bytes(data)))


def run(bitrate_kbps, seconds=2.0, fps=30):
    s = hipflux.CaptureSettings()
    s.capture_width = 320
    s.capture_height = 192
    s.target_fps = fps
    s.output_mode = 1
    s.use_cpu = True
    s.gpu_id = -1
    s.capture_backend = "synthetic:noise"   # incompressible, 100% damage
    s.video_fullframe = True
    s.video_cbr_mode = True
    s.video_bitrate_kbps = bitrate_kbps
    s.video_crf = 28
    s.video_min_qp = 2
    s.video_max_qp = 51
    col = Collector()
    cap = hipflux.ScreenCapture()
    cap.start_capture(col, s)
    time.sleep(seconds)
    cap.stop_capture()
    return col.stripes


def measured_kbps(stripes, fps, skip_frames=15):
    frames = {}
    for fid, y, data in stripes:
        frames.setdefault(fid, 0)
        frames[fid] += len(data)
    ids = sorted(frames)
    settled = [frames[i] for i in ids[skip_frames:]]
    assert settled
    bytes_per_frame = sum(settled) / len(settled)
    return bytes_per_frame * 8 * fps / 1000


def test_cbr_converges():
    target = 4000  # kbps — noise at default QP would be far above this
    stripes = run(target)
    rate = measured_kbps(stripes, 30)
    assert 0.5 * target < rate < 1.8 * target, f"CBR rate {rate:.0f} kbps"


def test_cbr_tracks_different_targets():
    lo = measured_kbps(run(2000), 30)
    hi = measured_kbps(run(12000), 30)
    assert hi > lo * 2, f"rate control not tracking targets: {lo} vs {hi}"


def test_cbr_stream_decodes():
    stripes = run(3000, seconds=1.0)
    rows = {}
    for fid, y, data in sorted(stripes, key=lambda t: t[0]):
        rows.setdefault(y, b"")
        rows[y] += data[10:]
    for y, stream in rows.items():
        frames = Decoder().decode(stream)
        assert len(frames) > 3
