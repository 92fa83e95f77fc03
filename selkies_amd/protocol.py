"""Binary + text wire protocol shared by server and tests.

The wire format matches the reference WebSocket data plane so the HTML5
client contract is preserved (reference SURVEY.md §3.2; framing knowledge in
reference selkies.py:821-822 and client selkies-ws-core.js:54-100,4532-4566):

server → client binary frames
  0x01  audio    [0x01, n_red:u8] + payload            (optional pts later)
  0x03  JPEG     [0x03, flags:u8, frame_id:u16be, y:u16be] + JFIF bytes
  0x04  H.264    [0x04, keyflag:u8, frame_id:u16be, y:u16be, w:u16be,
                  h:u16be] + AnnexB NALs               (10-byte header)
  0x05  gzip'd control text (whole payload after the tag is gzip)
  0x06  HEVC     same 10-byte header layout as 0x04, AnnexB H.265 NALs
                 (this framework's extension; the reference has no HEVC)

client → server binary frames
  0x02  mic PCM s16le
  0x05  gzip'd control text

Text frames are comma-separated verbs ("SETTINGS,{json}", "CLIENT_FRAME_ACK,<id>",
input verbs "kd,<keysym>" etc. — dispatched by input_handler).
"""

from __future__ import annotations

import gzip
import json
import struct
from dataclasses import dataclass

# frame type tags
TAG_AUDIO = 0x01
TAG_MIC_PCM = 0x02
TAG_JPEG = 0x03
TAG_H264 = 0x04
TAG_GZIP = 0x05
TAG_HEVC = 0x06

H264_HEADER = struct.Struct(">BBHHHH")   # tag, keyflag, frame_id, y, w, h
JPEG_HEADER = struct.Struct(">BBHH")     # tag, flags, frame_id, y

GZIP_THRESHOLD = 512          # compress control text above this size
MAX_INFLATED = 32 * 1024 * 1024


def pack_h264_stripe(payload: bytes | memoryview, frame_id: int, y: int,
                     w: int, h: int, is_keyframe: bool) -> bytes:
    return H264_HEADER.pack(TAG_H264, 1 if is_keyframe else 0,
                            frame_id & 0xFFFF, y, w, h) + bytes(payload)


def unpack_h264_header(buf: bytes) -> tuple:
    tag, key, fid, y, w, h = H264_HEADER.unpack_from(buf)
    assert tag == TAG_H264
    return key, fid, y, w, h, H264_HEADER.size


def pack_jpeg_stripe(payload: bytes | memoryview, frame_id: int, y: int,
                     flags: int = 0) -> bytes:
    return JPEG_HEADER.pack(TAG_JPEG, flags, frame_id & 0xFFFF, y) + bytes(payload)


def unpack_jpeg_header(buf: bytes) -> tuple:
    tag, flags, fid, y = JPEG_HEADER.unpack_from(buf)
    assert tag == TAG_JPEG
    return flags, fid, y, JPEG_HEADER.size


def pack_audio(payload: bytes, n_red: int = 0) -> bytes:
    return bytes((TAG_AUDIO, n_red & 0xFF)) + payload


def unpack_audio(buf: bytes) -> tuple:
    assert buf[0] == TAG_AUDIO
    return buf[1], buf[2:]


def maybe_gzip_text(text: str, threshold: int = GZIP_THRESHOLD) -> bytes | str:
    """Return the original text, or a 0x05-tagged gzip frame if large."""
    if len(text) < threshold:
        return text
    return bytes((TAG_GZIP,)) + gzip.compress(text.encode("utf-8"), 6)


def inflate_gz_bounded(buf: bytes, limit: int = MAX_INFLATED) -> str:
    """Bounded gzip inflater for inbound 0x05 frames (reference settings.py:53)."""
    out = gzip.decompress(bytes(buf))
    if len(out) > limit:
        raise ValueError("inflated control frame exceeds limit")
    return out.decode("utf-8")


@dataclass
class StripeFrame:
    """One encoded stripe as produced by the hipflux engine callback."""
    data: memoryview            # wire-ready payload including header
    frame_id: int
    y: int
    height: int
    is_keyframe: bool
    frame_ts_ms: float          # capture timestamp (monotonic ms)


def encode_control(verb: str, *fields) -> str:
    parts = [verb]
    for f in fields:
        parts.append(json.dumps(f) if isinstance(f, (dict, list)) else str(f))
    return ",".join(parts)


def parse_control(msg: str) -> tuple[str, str]:
    """Split 'VERB,rest' -> (verb, rest)."""
    verb, _, rest = msg.partition(",")
    return verb, rest
