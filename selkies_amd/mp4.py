"""Minimal progressive MP4 (ISO BMFF) muxer for H.264 recordings.

Reference parity: the reference's recording pipeline can produce MP4
files (SURVEY.md §2.3 soak coverage: "recording socket, MP4 recorder").
This muxer wraps one H.264 elementary stream (one stripe-row sequence or
a full-frame session) into a standards-shaped `.mp4`: ftyp + streaming
`mdat` (size backpatched on finalize) + `moov` with the full sample
tables (stts/stsc/stsz/stco/stss) and an `avcC` built from the stream's
own SPS/PPS. Annex-B access units are converted to AVCC (4-byte length
prefixes).

Structure-verified by tests/test_recording.py: the box tree is re-parsed,
sample offsets/sizes are checked to point at valid AVCC NALs inside
mdat, and sync samples line up with IDR frames.
"""

from __future__ import annotations

import struct
from typing import BinaryIO, Optional


def _split_annexb(data: bytes):
    nals = []
    i = 0
    starts = []
    while True:
        j = data.find(b"\x00\x00\x01", i)
        if j < 0:
            break
        starts.append(j + 3)
        i = j + 3
    for k, s in enumerate(starts):
        e = len(data)
        if k + 1 < len(starts):
            e = starts[k + 1] - 3
            while e > s and data[e - 1] == 0:
                e -= 1
        if e > s:
            nals.append(data[s:e])
    return nals


def _box(tag: bytes, payload: bytes) -> bytes:
    return struct.pack(">I", 8 + len(payload)) + tag + payload


def _full(tag: bytes, version: int, flags: int, payload: bytes) -> bytes:
    return _box(tag, struct.pack(">I", (version << 24) | flags) + payload)


class Mp4Writer:
    """Stream H.264 access units into a progressive MP4 file."""

    def __init__(self, fp: BinaryIO, width: int, height: int,
                 fps: float = 60.0, timescale: int = 90000):
        self.fp = fp
        self.width = width
        self.height = height
        self.timescale = timescale
        self.delta = max(1, int(round(timescale / fps)))
        self.sps: Optional[bytes] = None
        self.pps: Optional[bytes] = None
        self.sizes: list[int] = []
        self.offsets: list[int] = []
        self.sync: list[int] = []      # 1-based sample numbers of IDR AUs
        fp.write(_box(b"ftyp", b"isom" + struct.pack(">I", 0x200) +
                      b"isomiso2avc1mp41"))
        self._mdat_pos = fp.tell()
        fp.write(struct.pack(">I", 0) + b"mdat")

    def add_frame(self, annexb: bytes, keyframe: bool = False) -> None:
        """One access unit of Annex-B; SPS/PPS are captured into avcC and
        also kept in-band (players accept both)."""
        nals = _split_annexb(annexb)
        avcc = bytearray()
        has_idr = False
        for n in nals:
            t = n[0] & 0x1F
            if t == 7 and self.sps is None:
                self.sps = n
            elif t == 8 and self.pps is None:
                self.pps = n
            if t == 5:
                has_idr = True
            avcc += struct.pack(">I", len(n)) + n
        self.offsets.append(self.fp.tell())
        self.sizes.append(len(avcc))
        if has_idr or keyframe:
            self.sync.append(len(self.sizes))
        self.fp.write(bytes(avcc))

    # ---- finalize -----------------------------------------------------------
    def _avc1(self) -> bytes:
        assert self.sps and self.pps, "no SPS/PPS seen"
        profile, compat, level = self.sps[1], self.sps[2], self.sps[3]
        avcc = bytes([1, profile, compat, level, 0xFF, 0xE1]) + \
            struct.pack(">H", len(self.sps)) + self.sps + bytes([1]) + \
            struct.pack(">H", len(self.pps)) + self.pps
        sample = struct.pack(">6xH", 1)                     # dref index
        sample += struct.pack(">HHIII", 0, 0, 0, 0, 0)      # pre_defined
        sample += struct.pack(">HH", self.width, self.height)
        sample += struct.pack(">IIIH", 0x480000, 0x480000, 0, 1)
        sample += b"\x00" * 32                              # compressorname
        sample += struct.pack(">Hh", 0x18, -1)
        sample += _box(b"avcC", avcc)
        return _box(b"avc1", sample)

    def finalize(self) -> None:
        n = len(self.sizes)
        dur = n * self.delta
        end = self.fp.tell()
        # backpatch mdat size
        self.fp.seek(self._mdat_pos)
        self.fp.write(struct.pack(">I", end - self._mdat_pos))
        self.fp.seek(end)

        stsd = _full(b"stsd", 0, 0, struct.pack(">I", 1) + self._avc1())
        stts = _full(b"stts", 0, 0,
                     struct.pack(">III", 1, n, self.delta))
        stsc = _full(b"stsc", 0, 0,
                     struct.pack(">IIII", 1, 1, 1, 1))
        stsz = _full(b"stsz", 0, 0,
                     struct.pack(">II", 0, n) +
                     b"".join(struct.pack(">I", s) for s in self.sizes))
        stco = _full(b"stco", 0, 0,
                     struct.pack(">I", n) +
                     b"".join(struct.pack(">I", o) for o in self.offsets))
        stss = _full(b"stss", 0, 0,
                     struct.pack(">I", len(self.sync)) +
                     b"".join(struct.pack(">I", s) for s in self.sync))
        stbl = _box(b"stbl", stsd + stts + stsc + stsz + stco + stss)
        url = _full(b"url ", 0, 1, b"")
        dref = _full(b"dref", 0, 0, struct.pack(">I", 1) + url)
        dinf = _box(b"dinf", dref)
        vmhd = _full(b"vmhd", 0, 1, struct.pack(">HHHH", 0, 0, 0, 0))
        minf = _box(b"minf", vmhd + dinf + stbl)
        hdlr = _full(b"hdlr", 0, 0,
                     struct.pack(">I", 0) + b"vide" + b"\x00" * 12 +
                     b"selkies-amd video\x00")
        mdhd = _full(b"mdhd", 0, 0,
                     struct.pack(">IIIIHH", 0, 0, self.timescale, dur,
                                 0x55C4, 0))
        mdia = _box(b"mdia", mdhd + hdlr + minf)
        tkhd = _full(b"tkhd", 0, 7,
                     struct.pack(">IIIIII", 0, 0, 1, 0, dur, 0) +
                     struct.pack(">IHHHH", 0, 0, 0, 0, 0) +
                     struct.pack(">9i", 0x10000, 0, 0, 0, 0x10000, 0, 0, 0,
                                 0x40000000) +
                     struct.pack(">II", self.width << 16,
                                 self.height << 16))
        trak = _box(b"trak", tkhd + mdia)
        mvhd = _full(b"mvhd", 0, 0,
                     struct.pack(">IIII", 0, 0, self.timescale, dur) +
                     struct.pack(">IH", 0x10000, 0x100) + b"\x00" * 10 +
                     struct.pack(">9i", 0x10000, 0, 0, 0, 0x10000, 0, 0, 0,
                                 0x40000000) + b"\x00" * 24 +
                     struct.pack(">I", 2))
        moov = _box(b"moov", mvhd + trak)
        self.fp.write(moov)


def parse_boxes(data: bytes, off: int = 0, end: Optional[int] = None):
    """Flat child-box listing [(tag, payload_off, payload_end)] — used by
    the structure tests."""
    end = len(data) if end is None else end
    out = []
    while off + 8 <= end:
        size = struct.unpack_from(">I", data, off)[0]
        tag = data[off + 4:off + 8]
        if size < 8 or off + size > end:
            break
        out.append((tag, off + 8, off + size))
        off += size
    return out


def mux_file(src_annexb: str, dst_mp4: str, width: int, height: int,
             fps: float = 60.0) -> int:
    """Wrap a raw .h264 recording (Annex-B, IDR-led) into MP4. Access
    units are split at IDR/SPS boundaries and non-IDR slice starts."""
    data = open(src_annexb, "rb").read()
    nals = _split_annexb(data)
    # group into access units: a picture is one or more slices (we emit
    # one slice per MB row); a new AU starts at an SPS or at a slice with
    # first_mb_in_slice == 0 (ue(v) '1' = first RBSP bit set)
    aus = []
    cur: list[bytes] = []
    cur_has_slice = False
    for n in nals:
        t = n[0] & 0x1F
        is_slice = t in (1, 5)
        starts_pic = is_slice and len(n) > 1 and (n[1] & 0x80)
        if (t == 7 or starts_pic) and cur_has_slice:
            aus.append(cur)
            cur = []
            cur_has_slice = False
        cur.append(n)
        if is_slice:
            cur_has_slice = True
    if cur:
        aus.append(cur)
    with open(dst_mp4, "wb") as f:
        w = Mp4Writer(f, width, height, fps)
        for au in aus:
            raw = b"".join(b"\x00\x00\x00\x01" + n for n in au)
            w.add_frame(raw)
        w.finalize()
    return len(aus)


def mux_entrypoint(argv=None) -> int:
    """`selkies-mux <rec.h264> <out.mp4> WxH [fps]` CLI."""
    import sys
    args = argv if argv is not None else sys.argv[1:]
    if len(args) < 3:
        print("usage: selkies-mux <in.h264> <out.mp4> WxH [fps]")
        return 2
    w, _, h = args[2].partition("x")
    fps = float(args[3]) if len(args) > 3 else 60.0
    n = mux_file(args[0], args[1], int(w), int(h), fps)
    print(f"muxed {n} access units -> {args[1]}")
    return 0
