"""Audio engine: pcmflux API contract, wire framing,
RED redundancy, playback sink, server broadcast."""

import asyncio
import struct
import time

import numpy as np
import pytest

hipflux = pytest.importorskip("hipflux")
if not hipflux.native_available():
    pytest.skip("hipflux native module not built", allow_module_level=True)

from hipflux import _native


def capture_frames(settings, seconds=0.25):
    # framing/RED tests below assert raw-PCM payload sizes
    frames = []
    cap = _native.AudioCapture()
    cap.start_capture(settings, lambda d, pts: frames.append((d, pts)))
    assert cap.is_capturing
    time.sleep(seconds)
    cap.stop_capture()
    assert not cap.is_capturing
    return frames


def test_synthetic_capture_basic():
    s = _native.AudioCaptureSettings()
    s.codec = "pcm"
    frames = capture_frames(s)
    # 20ms frames over 250ms -> ~12 frames; allow scheduling slack
    assert 8 <= len(frames) <= 16
    d, pts = frames[0]
    assert d[0] == 0x01 and d[1] == 0
    # 48kHz stereo s16 20ms = 3840 bytes + 2 header
    assert len(d) == 2 + 48000 * 2 * 2 // 50
    # pts monotonically increases by frame duration
    assert frames[1][1] - frames[0][1] == 20.0


def test_red_redundancy_layout():
    s = _native.AudioCaptureSettings()
    s.codec = "pcm"
    s.red_distance = 2
    frames = capture_frames(s)
    # first frame: no history yet
    assert frames[0][0][1] == 0
    d = frames[3][0]
    assert d[1] == 2
    off = 2
    payload_len = 48000 * 2 * 2 // 50
    for _ in range(2):
        ln = struct.unpack(">H", d[off:off + 2])[0]
        assert ln == payload_len
        off += 2 + ln
    assert len(d) - off == payload_len
    # redundant payload k matches the primary payload of frame n-k
    red_newest = d[2 + 2 + payload_len + 2:2 + 2 + payload_len + 2
                   + payload_len]
    prev_primary = frames[2][0][2:]          # frame 2 had n_red=2 as well?
    prev = frames[2][0]
    prev_off = 2
    for _ in range(prev[1]):
        ln = struct.unpack(">H", prev[prev_off:prev_off + 2])[0]
        prev_off += 2 + ln
    assert red_newest == prev[prev_off:]


def test_silence_and_tone_content():
    s = _native.AudioCaptureSettings()
    s.codec = "pcm"
    s.device_name = "silence"
    frames = capture_frames(s)
    pcm = np.frombuffer(frames[0][0][2:], np.int16)
    assert not pcm.any()
    s2 = _native.AudioCaptureSettings()
    s2.codec = "pcm"
    frames2 = capture_frames(s2)
    pcm2 = np.frombuffer(frames2[1][0][2:], np.int16)
    assert np.abs(pcm2).max() > 1000      # audible tone

def test_playback_ring():
    ps = _native.AudioPlaybackSettings()
    ps.max_buffer_bytes = 64
    pb = _native.AudioPlayback(ps)
    pb.write(b"a" * 48)
    pb.write(b"b" * 48)                    # overflows: oldest dropped
    assert pb.buffered == 64
    data = pb.read(100)
    assert len(data) == 64
    assert data.endswith(b"b" * 48)
    assert pb.buffered == 0


def test_mono_and_channels():
    s = _native.AudioCaptureSettings()
    s.codec = "pcm"
    s.channels = 1
    frames = capture_frames(s)
    assert len(frames[0][0]) == 2 + 48000 * 2 // 50


def _primary_payload(d):
    off = 2
    for _ in range(d[1]):
        ln = struct.unpack(">H", d[off:off + 2])[0]
        off += 2 + ln
    return bytes(d[off:])


def test_opus_default_capture_decodes():
    """Default codec is Opus: engine frames carry Opus packets inside the
    same [0x01, n_red] RED framing; the from-spec decoder recovers the
    tone (reference selkies gst pcmflux/opusenc path)."""
    from opus_ref_decoder import OpusDecoder

    s = _native.AudioCaptureSettings()
    assert s.codec == "opus"
    frames = []
    cap = _native.AudioCapture()
    cap.start_capture(s, lambda d, pts: frames.append((d, pts)))
    deadline = time.time() + 5.0
    while len(frames) < 12 and time.time() < deadline:
        time.sleep(0.02)
    cap.stop_capture()
    assert len(frames) >= 12
    dec = OpusDecoder()
    payloads = [_primary_payload(d) for d, _ in frames]
    # CBR at the default 128 kb/s: TOC + 320-byte payload
    assert all(len(p) == 1 + 128000 // 400 for p in payloads)
    for p in payloads:
        dec.decode_packet(p)
    out = dec.samples()[960 * 2:960 * (len(payloads) - 1)]
    # synthetic device: L=440 Hz, R=554.37 Hz; the mono downmix holds
    # both — SNR against the best-fit pair of sinusoids
    t = np.arange(out.size) / 48000.0
    ref = np.vstack([np.sin(2 * np.pi * 440 * t),
                     np.cos(2 * np.pi * 440 * t),
                     np.sin(2 * np.pi * 554.37 * t),
                     np.cos(2 * np.pi * 554.37 * t)]).T
    coef, *_ = np.linalg.lstsq(ref, out, rcond=None)
    resid = out - ref @ coef
    snr = 10 * np.log10((out ** 2).sum() / max((resid ** 2).sum(), 1e-12))
    assert snr > 15.0, snr


def test_opus_red_wraps_packets():
    s = _native.AudioCaptureSettings()
    s.red_distance = 1
    frames = capture_frames(s, seconds=0.3)
    d = frames[2][0]
    assert d[0] == 0x01 and d[1] == 1
    ln = struct.unpack(">H", d[2:4])[0]
    assert d[4 + ln:] != b""
    # redundant slot equals previous frame's primary
    assert bytes(d[4:4 + ln]) == _primary_payload(frames[1][0])


def test_opus_bitrate_update_live():
    s = _native.AudioCaptureSettings()
    frames = []
    cap = _native.AudioCapture()
    cap.start_capture(s, lambda d, pts: frames.append(bytes(d)))
    deadline = time.time() + 5.0
    while len(frames) < 3 and time.time() < deadline:
        time.sleep(0.02)
    cap.update_audio_bitrate(64000)
    deadline = time.time() + 5.0
    while time.time() < deadline:
        if frames and len(_primary_payload(frames[-1])) == 1 + 64000 // 400:
            break
        time.sleep(0.02)
    cap.stop_capture()
    sizes = {len(_primary_payload(f)) for f in frames}
    assert 1 + 128000 // 400 in sizes and 1 + 64000 // 400 in sizes
