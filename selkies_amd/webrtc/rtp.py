"""RTP H.264 packetization (RFC 6184), RTCP SR, RTCP feedback parsing."""

from __future__ import annotations

import struct
import time

MTU_PAYLOAD = 1180   # fits in 1280-byte datagrams with SRTP+headers


def split_annexb(data: bytes) -> list[bytes]:
    """Annex-B stream -> raw NAL units (no start codes)."""
    nals = []
    i = 0
    n = len(data)
    starts = []
    while i + 2 < n:
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


class H264Packetizer:
    """NAL units -> RTP payloads (single NAL / STAP-A / FU-A)."""

    def __init__(self, ssrc: int, payload_type: int = 102):
        self.ssrc = ssrc
        self.pt = payload_type
        self.seq = 0
        # stats for RTCP SR
        self.packets_sent = 0
        self.bytes_sent = 0

    def _header(self, marker: bool, ts90k: int) -> bytes:
        b1 = self.pt | (0x80 if marker else 0)
        h = struct.pack(">BBHII", 0x80, b1, self.seq & 0xFFFF,
                        ts90k & 0xFFFFFFFF, self.ssrc)
        self.seq = (self.seq + 1) & 0xFFFF
        return h

    def packetize(self, annexb: bytes, ts90k: int) -> list[bytes]:
        """One access unit of Annex-B -> list of complete RTP packets."""
        nals = split_annexb(annexb)
        out = []
        agg: list[bytes] = []
        agg_size = 0

        def flush_agg(marker=False):
            nonlocal agg, agg_size
            if not agg:
                return
            if len(agg) == 1:
                out.append(self._header(marker, ts90k) + agg[0])
            else:
                nri = max((n[0] >> 5) & 3 for n in agg)
                body = bytes([24 | (nri << 5)])  # STAP-A
                for n in agg:
                    body += struct.pack(">H", len(n)) + n
                out.append(self._header(marker, ts90k) + body)
            agg = []
            agg_size = 0

        for idx, nal in enumerate(nals):
            last_nal = idx == len(nals) - 1
            if len(nal) <= MTU_PAYLOAD - 4:
                if agg_size + len(nal) + 2 > MTU_PAYLOAD - 1:
                    flush_agg(False)
                agg.append(nal)
                agg_size += len(nal) + 2
                if last_nal:
                    flush_agg(True)
            else:
                flush_agg(False)
                # FU-A fragmentation
                hdr = nal[0]
                nri = hdr & 0x60
                ntype = hdr & 0x1F
                payload = nal[1:]
                pos = 0
                first = True
                while pos < len(payload):
                    chunk = payload[pos:pos + MTU_PAYLOAD - 2]
                    pos += len(chunk)
                    fu_ind = 28 | nri
                    fu_hdr = ntype | (0x80 if first else 0) | \
                        (0x40 if pos >= len(payload) else 0)
                    marker = last_nal and pos >= len(payload)
                    out.append(self._header(marker, ts90k) +
                               bytes([fu_ind, fu_hdr]) + chunk)
                    first = False
        self.packets_sent += len(out)
        self.bytes_sent += sum(len(p) - 12 for p in out)
        return out


NTP_EPOCH_OFFSET = 2208988800  # 1900 -> 1970


def build_sender_report(ssrc: int, ts90k: int, packets: int,
                        octets: int) -> bytes:
    now = time.time() + NTP_EPOCH_OFFSET
    ntp_hi = int(now)
    ntp_lo = int((now - ntp_hi) * (1 << 32)) & 0xFFFFFFFF
    body = struct.pack(">IIIIII", ntp_hi, ntp_lo, ts90k & 0xFFFFFFFF,
                       packets, octets, 0)[:-4]
    # SR: V=2, PT=200, length in 32-bit words - 1
    pkt = struct.pack(">BBHI", 0x80, 200, 6, ssrc) + \
        struct.pack(">II", ntp_hi, ntp_lo) + \
        struct.pack(">III", ts90k & 0xFFFFFFFF, packets, octets)
    return pkt


def parse_rtcp(data: bytes) -> list[dict]:
    """Parse a compound RTCP packet into typed reports (subset)."""
    out = []
    off = 0
    while off + 4 <= len(data):
        v_p_rc, pt, length = struct.unpack_from(">BBH", data, off)
        size = 4 * (length + 1)
        chunk = data[off:off + size]
        if pt == 201:
            rep = {"type": "RR", "blocks": []}
            rc = v_p_rc & 0x1F
            boff = off + 8          # header + reporter SSRC
            for _ in range(rc):
                if boff + 24 > off + size:
                    break
                (ssrc, fl_cum, ext_seq, jitter, lsr,
                 dlsr) = struct.unpack_from(">IIIIII", data, boff)
                rep["blocks"].append({
                    "ssrc": ssrc,
                    "fraction_lost": (fl_cum >> 24) / 256.0,
                    "cum_lost": fl_cum & 0xFFFFFF,
                    "jitter": jitter,
                })
                boff += 24
            out.append(rep)
        elif pt == 200:
            out.append({"type": "SR"})
        elif pt == 206:  # PSFB
            fmt = v_p_rc & 0x1F
            if fmt == 1:
                out.append({"type": "PLI"})
            elif fmt == 4:
                out.append({"type": "FIR"})
            else:
                out.append({"type": f"PSFB{fmt}"})
        elif pt == 205:  # RTPFB (NACK / TWCC)
            fmt = v_p_rc & 0x1F
            out.append({"type": "NACK" if fmt == 1 else f"RTPFB{fmt}"})
        else:
            out.append({"type": f"PT{pt}"})
        if size <= 0:
            break
        off += size
    return out


class AudioPacketizer:
    """G.711 frames -> single RTP packets (RFC 3551: PCMU pt 0, 8 kHz
    clock, one 20 ms frame per packet, timestamp += sample count)."""

    def __init__(self, ssrc: int, payload_type: int = 0):
        self.ssrc = ssrc
        self.pt = payload_type
        self.seq = 0
        self.timestamp = 0
        self.packets_sent = 0
        self.bytes_sent = 0
        self._first = True

    def packetize(self, payload: bytes) -> bytes:
        marker = 0x80 if self._first else 0   # marker starts the stream
        self._first = False
        b1 = self.pt | marker
        h = struct.pack(">BBHII", 0x80, b1, self.seq & 0xFFFF,
                        self.timestamp & 0xFFFFFFFF, self.ssrc)
        self.seq = (self.seq + 1) & 0xFFFF
        self.timestamp = (self.timestamp + len(payload)) & 0xFFFFFFFF
        self.packets_sent += 1
        self.bytes_sent += len(payload)
        return h + payload
