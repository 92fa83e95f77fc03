"""Minimal SCTP-over-DTLS + DCEP for WebRTC data channels.

Implements the subset browsers use for `RTCDataChannel` (reliable,
ordered): RFC 4960 association setup (INIT / INIT-ACK+cookie /
COOKIE-ECHO / COOKIE-ACK), DATA/SACK with CRC32c, fragment reassembly
(B/E flags), HEARTBEAT echo, ABORT, and RFC 8832 DCEP
(DATA_CHANNEL_OPEN / ACK). Reference parity: the reference's input/control
data channel (webrtc_mode.py) rides this transport.

Pure state machine — no sockets, no asyncio: `receive()` consumes one
SCTP packet, `outbound()` drains packets to ship (through the DTLS
endpoint), `poll(now)` drives retransmission. Verified by loopback tests
(tests/test_webrtc_stack.py) driving two associations against each other,
including a lossy-link retransmit case. CRC32c is checked against the
published check value crc32c("123456789") == 0xE3069283.
"""

from __future__ import annotations

import os
import struct
from typing import Callable, Optional

# ---- CRC32c (Castagnoli, reflected poly 0x82F63B78) -----------------------
_TABLE = []
for _i in range(256):
    _c = _i
    for _ in range(8):
        _c = (_c >> 1) ^ 0x82F63B78 if _c & 1 else _c >> 1
    _TABLE.append(_c)


def crc32c(data: bytes) -> int:
    c = 0xFFFFFFFF
    for b in data:
        c = _TABLE[(c ^ b) & 0xFF] ^ (c >> 8)
    return c ^ 0xFFFFFFFF


# chunk types
CT_DATA, CT_INIT, CT_INIT_ACK, CT_SACK, CT_HEARTBEAT, CT_HEARTBEAT_ACK, \
    CT_ABORT, CT_SHUTDOWN, CT_SHUTDOWN_ACK, CT_ERROR, CT_COOKIE_ECHO, \
    CT_COOKIE_ACK = range(12)

# DCEP (RFC 8832)
PPID_DCEP = 50
PPID_STRING = 51
PPID_BINARY = 53
PPID_STRING_EMPTY = 56
PPID_BINARY_EMPTY = 57
DCEP_OPEN = 0x03
DCEP_ACK = 0x02


def _pad4(b: bytes) -> bytes:
    return b + b"\x00" * (-len(b) % 4)


def _chunk(ctype: int, flags: int, value: bytes) -> bytes:
    return _pad4(struct.pack(">BBH", ctype, flags, 4 + len(value)) + value)


def _tsn_gt(a: int, b: int) -> bool:
    return ((a - b) & 0xFFFFFFFF) < 0x80000000 and a != b


class Channel:
    def __init__(self, sid: int, label: str, protocol: str = ""):
        self.sid = sid
        self.label = label
        self.protocol = protocol
        self.open = False
        self.seq = 0


class SctpAssociation:
    """One SCTP association (the WebRTC peer connection has exactly one)."""

    RTO = 1.0
    MAX_RETRANS = 8

    def __init__(self, is_server: bool = True, port: int = 5000,
                 on_message: Optional[Callable] = None,
                 on_channel_open: Optional[Callable] = None):
        self.is_server = is_server
        self.port = port
        self.remote_port = port
        self.on_message = on_message      # (sid, ppid, bytes)
        self.on_channel_open = on_channel_open
        self.my_tag = struct.unpack(">I", os.urandom(4))[0] or 1
        self.peer_tag = 0
        self.my_tsn = struct.unpack(">I", os.urandom(4))[0]
        self.established = False
        self.channels: dict[int, Channel] = {}
        self._out: list[bytes] = []
        self._unacked: dict[int, tuple] = {}   # tsn -> (packet, sent_at, n)
        self.cum_peer_tsn: Optional[int] = None
        self._pending: dict[int, bytes] = {}   # out-of-order tsn -> chunk val
        self._reasm: dict[int, list] = {}      # sid -> [(tsn, flags, data)]
        self._cookie = os.urandom(16)
        self.a_rwnd = 1 << 20
        self.errors = 0

    # ---- packet building ---------------------------------------------------
    def _packet(self, chunks: bytes, vtag: Optional[int] = None) -> bytes:
        head = struct.pack(">HHI", self.port, self.remote_port,
                           self.peer_tag if vtag is None else vtag)
        raw = head + b"\x00\x00\x00\x00" + chunks
        crc = crc32c(raw)
        return head + struct.pack("<I", crc) + chunks

    def outbound(self) -> list[bytes]:
        out, self._out = self._out, []
        return out

    # ---- association setup -------------------------------------------------
    def start(self) -> None:
        """Client role: send INIT."""
        v = struct.pack(">IIHHI", self.my_tag, self.a_rwnd, 1024, 1024,
                        self.my_tsn)
        self._out.append(self._packet(_chunk(CT_INIT, 0, v), vtag=0))

    def open_channel(self, sid: int, label: str,
                     protocol: str = "") -> Channel:
        ch = Channel(sid, label, protocol)
        self.channels[sid] = ch
        lb, pb = label.encode(), protocol.encode()
        msg = struct.pack(">BBHIHH", DCEP_OPEN, 0x00, 0, 0, len(lb),
                          len(pb)) + lb + pb
        self._send_data(sid, PPID_DCEP, msg, ch)
        return ch

    # ---- data sending ------------------------------------------------------
    def send(self, sid: int, data, ppid: Optional[int] = None,
             now: float = 0.0) -> None:
        ch = self.channels.get(sid)
        if ch is None:
            raise ValueError(f"stream {sid} not open")
        if isinstance(data, str):
            payload = data.encode()
            p = ppid or (PPID_STRING if payload else PPID_STRING_EMPTY)
        else:
            payload = bytes(data)
            p = ppid or (PPID_BINARY if payload else PPID_BINARY_EMPTY)
        self._send_data(sid, p, payload or b"\x00", ch, now)

    def _send_data(self, sid: int, ppid: int, payload: bytes, ch: Channel,
                   now: float = 0.0) -> None:
        # fragment to ~1100-byte chunks (DTLS+IP headroom under 1500 MTU)
        MAX = 1100
        parts = [payload[i:i + MAX] for i in range(0, len(payload), MAX)] \
            or [b""]
        for i, part in enumerate(parts):
            flags = (0x02 if i == 0 else 0) | \
                    (0x01 if i == len(parts) - 1 else 0)
            tsn = self.my_tsn
            self.my_tsn = (self.my_tsn + 1) & 0xFFFFFFFF
            v = struct.pack(">IHHI", tsn, sid, ch.seq, ppid) + part
            pkt = self._packet(_chunk(CT_DATA, flags, v))
            self._unacked[tsn] = (pkt, now, 0)
            self._out.append(pkt)
        ch.seq = (ch.seq + 1) & 0xFFFF

    def poll(self, now: float) -> None:
        """Retransmit DATA unacked past RTO."""
        for tsn, (pkt, t0, n) in list(self._unacked.items()):
            if now - t0 >= self.RTO:
                if n + 1 > self.MAX_RETRANS:
                    del self._unacked[tsn]
                    self.errors += 1
                    continue
                self._unacked[tsn] = (pkt, now, n + 1)
                self._out.append(pkt)

    # ---- receive -----------------------------------------------------------
    def receive(self, packet: bytes, now: float = 0.0) -> None:
        if len(packet) < 12:
            return
        sport, dport, vtag = struct.unpack(">HHI", packet[:8])
        crc = struct.unpack("<I", packet[8:12])[0]
        if crc32c(packet[:8] + b"\x00\x00\x00\x00" + packet[12:]) != crc:
            self.errors += 1
            return
        self.remote_port = sport
        off = 12
        while off + 4 <= len(packet):
            ctype, flags, clen = struct.unpack(">BBH", packet[off:off + 4])
            if clen < 4 or off + clen > len(packet):
                break
            value = packet[off + 4:off + clen]
            self._on_chunk(ctype, flags, value, now)
            off += clen + (-clen % 4)

    def _on_chunk(self, ctype: int, flags: int, v: bytes,
                  now: float) -> None:
        if ctype == CT_INIT and len(v) >= 16:
            tag, rwnd, nout, nin, tsn = struct.unpack(">IIHHI", v[:16])
            self.peer_tag = tag
            self.cum_peer_tsn = (tsn - 1) & 0xFFFFFFFF
            ack = struct.pack(">IIHHI", self.my_tag, self.a_rwnd, 1024,
                              1024, self.my_tsn)
            ack += struct.pack(">HH", 7, 4 + len(self._cookie)) + \
                _pad4(self._cookie)
            self._out.append(self._packet(_chunk(CT_INIT_ACK, 0, ack)))
        elif ctype == CT_INIT_ACK and len(v) >= 16:
            tag, rwnd, nout, nin, tsn = struct.unpack(">IIHHI", v[:16])
            self.peer_tag = tag
            self.cum_peer_tsn = (tsn - 1) & 0xFFFFFFFF
            # find the State Cookie parameter (type 7)
            off = 16
            cookie = b""
            while off + 4 <= len(v):
                pt, pl = struct.unpack(">HH", v[off:off + 4])
                if pl < 4:
                    break
                if pt == 7:
                    cookie = v[off + 4:off + pl]
                off += pl + (-pl % 4)
            self._out.append(self._packet(_chunk(CT_COOKIE_ECHO, 0,
                                                 cookie)))
            self.established = True
        elif ctype == CT_COOKIE_ECHO:
            self._out.append(self._packet(_chunk(CT_COOKIE_ACK, 0, b"")))
            self.established = True
        elif ctype == CT_COOKIE_ACK:
            self.established = True
        elif ctype == CT_HEARTBEAT:
            self._out.append(self._packet(_chunk(CT_HEARTBEAT_ACK, 0, v)))
        elif ctype == CT_SACK and len(v) >= 12:
            cum, rwnd, ngap, ndup = struct.unpack(">IIHH", v[:12])
            for tsn in list(self._unacked):
                if not _tsn_gt(tsn, cum):
                    del self._unacked[tsn]
        elif ctype == CT_DATA and len(v) >= 12:
            self._on_data(flags, v, now)
        elif ctype == CT_ABORT:
            self.established = False

    def _on_data(self, flags: int, v: bytes, now: float) -> None:
        tsn = struct.unpack(">I", v[:4])[0]
        if self.cum_peer_tsn is None:
            self.cum_peer_tsn = (tsn - 1) & 0xFFFFFFFF
        if not _tsn_gt(tsn, self.cum_peer_tsn):
            self._sack()           # duplicate: re-SACK
            return
        self._pending[tsn] = (flags, v)
        # advance cumulative TSN over any in-order run
        while ((self.cum_peer_tsn + 1) & 0xFFFFFFFF) in self._pending:
            nxt = (self.cum_peer_tsn + 1) & 0xFFFFFFFF
            f, val = self._pending.pop(nxt)
            self.cum_peer_tsn = nxt
            self._deliver(f, val)
        self._sack()

    def _sack(self) -> None:
        gaps = b""
        sack = struct.pack(">IIHH", self.cum_peer_tsn, self.a_rwnd, 0, 0)
        self._out.append(self._packet(_chunk(CT_SACK, 0, sack + gaps)))

    def _deliver(self, flags: int, v: bytes) -> None:
        tsn, sid, sseq, ppid = struct.unpack(">IHHI", v[:12])
        payload = v[12:]
        buf = self._reasm.setdefault(sid, [])
        buf.append((flags, payload))
        if not (flags & 0x01):      # not the End fragment yet
            return
        data = b"".join(p for _, p in buf)
        self._reasm[sid] = []
        if ppid == PPID_DCEP:
            self._on_dcep(sid, data)
            return
        if ppid in (PPID_STRING_EMPTY, PPID_BINARY_EMPTY):
            data = b""
        if self.on_message:
            self.on_message(sid, ppid, data)

    def _on_dcep(self, sid: int, data: bytes) -> None:
        if not data:
            return
        if data[0] == DCEP_OPEN and len(data) >= 12:
            _, _, _, _, llen, plen = struct.unpack(">BBHIHH", data[:12])
            label = data[12:12 + llen].decode(errors="replace")
            proto = data[12 + llen:12 + llen + plen].decode(errors="replace")
            ch = Channel(sid, label, proto)
            ch.open = True
            self.channels[sid] = ch
            # DCEP ACK on the same stream
            self._send_data(sid, PPID_DCEP, bytes([DCEP_ACK]), ch)
            if self.on_channel_open:
                self.on_channel_open(ch)
        elif data[0] == DCEP_ACK:
            ch = self.channels.get(sid)
            if ch:
                ch.open = True
                if self.on_channel_open:
                    self.on_channel_open(ch)
