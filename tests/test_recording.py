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


def test_mp4_mux_structure_and_samples(tmp_path):
    """Wrap a recorded stripe stream in MP4 and verify the box tree:
    sample offsets/sizes point at valid AVCC NALs inside mdat, stss lines
    up with IDR samples, and the avcC carries the stream's SPS/PPS."""
    import struct
    import numpy as np
    from selkies_amd import mp4
    from hipflux import _native

    w, h, n = 192, 64, 8
    rng = np.random.default_rng(3)
    frames = [np.ascontiguousarray(rng.integers(0, 256, (h, w, 4),
                                                dtype=np.uint8))
              for _ in range(n)]
    out = _native._pipeline_encode("cpu", frames, w, h, 26, 64, 1)
    raw = tmp_path / "row.h264"
    with open(raw, "wb") as f:
        for fr in out:
            for data, y, _, _ in fr:
                if y == 0:
                    f.write(bytes(data))
    dst = tmp_path / "row.mp4"
    n_aus = mp4.mux_file(str(raw), str(dst), w, 64, fps=30)
    assert n_aus == n

    blob = dst.read_bytes()
    top = mp4.parse_boxes(blob)
    tags = [t for t, _, _ in top]
    assert tags == [b"ftyp", b"mdat", b"moov"]
    moov = dict((t, (a, b)) for t, a, b in
                mp4.parse_boxes(blob, *top[2][1:]))
    assert b"mvhd" in moov and b"trak" in moov

    def find(path, off, end):
        for tag in path:
            boxes = mp4.parse_boxes(blob, off, end)
            d = {t: (a, b) for t, a, b in boxes}
            assert tag in d, f"missing {tag} among {list(d)}"
            off, end = d[tag]
            if tag in (b"stsd",):
                off += 8   # full box: version/flags + entry count
            elif tag == b"avc1":
                off += 78  # visual sample entry header
        return off, end

    stbl_off, stbl_end = find([b"trak", b"mdia", b"minf", b"stbl"],
                              *top[2][1:])
    stbl = {t: (a, b) for t, a, b in
            mp4.parse_boxes(blob, stbl_off, stbl_end)}
    for req in (b"stsd", b"stts", b"stsc", b"stsz", b"stco", b"stss"):
        assert req in stbl

    # sample tables: every sample points at AVCC NALs inside mdat
    sz_off = stbl[b"stsz"][0] + 4
    _, count = struct.unpack_from(">II", blob, sz_off)
    assert count == n
    sizes = struct.unpack_from(f">{count}I", blob, sz_off + 8)
    co_off = stbl[b"stco"][0] + 4
    n_off = struct.unpack_from(">I", blob, co_off)[0]
    offsets = struct.unpack_from(f">{n_off}I", blob, co_off + 4)
    mdat_a, mdat_b = top[1][1], top[1][2]
    for o, s in zip(offsets, sizes):
        assert mdat_a <= o and o + s <= mdat_b
        p = o
        while p < o + s:    # AVCC walk: 4-byte length + NAL
            ln = struct.unpack_from(">I", blob, p)[0]
            assert 1 <= ln <= s
            p += 4 + ln
        assert p == o + s
    # first sample is the IDR -> listed in stss
    ss_off = stbl[b"stss"][0] + 4
    n_sync = struct.unpack_from(">I", blob, ss_off)[0]
    syncs = struct.unpack_from(f">{n_sync}I", blob, ss_off + 4)
    assert 1 in syncs
    # avcC present with our SPS
    avc1_off, avc1_end = find([b"trak", b"mdia", b"minf", b"stbl",
                               b"stsd", b"avc1"], *top[2][1:])
    sub = {t: (a, b) for t, a, b in
           mp4.parse_boxes(blob, avc1_off, avc1_end)}
    assert b"avcC" in sub
    assert blob[sub[b"avcC"][0]] == 1      # configurationVersion
