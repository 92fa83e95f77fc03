"""Wire framing round-trips (SURVEY.md §3.2 frame formats)."""

from selkies_amd import protocol as P


def test_h264_stripe_roundtrip():
    payload = b"\x00\x00\x00\x01\x65" + b"x" * 100
    buf = P.pack_h264_stripe(payload, frame_id=70000, y=128, w=1920, h=64,
                             is_keyframe=True)
    assert buf[0] == P.TAG_H264
    key, fid, y, w, h, off = P.unpack_h264_header(buf)
    assert (key, y, w, h) == (1, 128, 1920, 64)
    assert fid == 70000 & 0xFFFF          # frame ids wrap at u16
    assert buf[off:] == payload
    assert off == 10                      # reference: 10-byte 0x04 header


def test_jpeg_stripe_roundtrip():
    buf = P.pack_jpeg_stripe(b"\xff\xd8jpegdata", frame_id=3, y=256)
    flags, fid, y, off = P.unpack_jpeg_header(buf)
    assert (flags, fid, y) == (0, 3, 256)
    assert buf[off:] == b"\xff\xd8jpegdata"


def test_audio_roundtrip():
    buf = P.pack_audio(b"opusish", n_red=2)
    n_red, payload = P.unpack_audio(buf)
    assert n_red == 2 and payload == b"opusish"


def test_gzip_control():
    small = P.maybe_gzip_text("MODE,websockets")
    assert small == "MODE,websockets"
    big_text = "SETTINGS," + "x" * 5000
    framed = P.maybe_gzip_text(big_text)
    assert isinstance(framed, bytes) and framed[0] == P.TAG_GZIP
    assert P.inflate_gz_bounded(framed[1:]) == big_text


def test_control_verbs():
    msg = P.encode_control("SETTINGS", {"framerate": 60})
    verb, rest = P.parse_control(msg)
    assert verb == "SETTINGS"
    import json
    assert json.loads(rest) == {"framerate": 60}
