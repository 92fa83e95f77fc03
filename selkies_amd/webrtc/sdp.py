"""SDP offer parsing + ice-lite answer building (video sendonly)."""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional


@dataclass
class MediaSection:
    kind: str
    mid: str = ""
    ice_ufrag: str = ""
    ice_pwd: str = ""
    fingerprint: str = ""
    h264_pts: list = field(default_factory=list)  # (pt, profile_level_id)
    g711_pt: int = -1          # PCMU(0)/PCMA(8) if offered
    raw: list = field(default_factory=list)


@dataclass
class Offer:
    session_fingerprint: str = ""
    media: list = field(default_factory=list)


def parse_offer(sdp: str) -> Offer:
    offer = Offer()
    cur: Optional[MediaSection] = None
    rtpmap = {}
    fmtp = {}
    for line in sdp.replace("\r\n", "\n").split("\n"):
        if line.startswith("m="):
            kind = line[2:].split()[0]
            cur = MediaSection(kind=kind)
            offer.media.append(cur)
        tgt = cur if cur is not None else None
        if line.startswith("a=fingerprint:"):
            fp = line.split(":", 1)[1].strip()
            if tgt is not None:
                tgt.fingerprint = fp
            else:
                offer.session_fingerprint = fp
        elif line.startswith("a=ice-ufrag:") and tgt is not None:
            tgt.ice_ufrag = line.split(":", 1)[1].strip()
        elif line.startswith("a=ice-pwd:") and tgt is not None:
            tgt.ice_pwd = line.split(":", 1)[1].strip()
        elif line.startswith("a=mid:") and tgt is not None:
            tgt.mid = line.split(":", 1)[1].strip()
        elif line.startswith("a=rtpmap:") and tgt is not None:
            body = line.split(":", 1)[1]
            try:
                pt, codec = body.split(" ", 1)
                rtpmap[int(pt)] = codec.strip().lower()
            except ValueError:
                pass   # client-controlled input: ignore malformed lines
        elif line.startswith("a=fmtp:") and tgt is not None:
            body = line.split(":", 1)[1]
            try:
                pt, params = body.split(" ", 1)
                fmtp[int(pt)] = params.strip()
            except ValueError:
                pass
        if tgt is not None:
            tgt.raw.append(line)
    # audio: accept G.711 (PCMU preferred over PCMA); static PTs may be
    # offered without an rtpmap line, so also scan the m= format list
    for m in offer.media:
        if m.kind != "audio":
            continue
        fmts = []
        for ln in m.raw:
            if ln.startswith("m=audio"):
                fmts = [int(f) for f in ln.split()[3:] if f.isdigit()]
        offered = {pt for pt, codec in rtpmap.items()
                   if codec.startswith(("pcmu/", "pcma/"))}
        offered |= {f for f in fmts if f in (0, 8)}
        if 0 in offered:
            m.g711_pt = 0
        elif 8 in offered:
            m.g711_pt = 8
    # pick H.264 payload types (prefer packetization-mode=1 42e01f)
    for m in offer.media:
        if m.kind != "video":
            continue
        for pt, codec in rtpmap.items():
            if codec.startswith("h264/"):
                params = fmtp.get(pt, "")
                m.h264_pts.append((pt, params))
        def score(entry):
            pt, params = entry
            s = 0
            if "packetization-mode=1" in params:
                s += 2
            if "42e01f" in params or "42001f" in params:
                s += 1
            return -s
        m.h264_pts.sort(key=score)
    return offer


def build_answer(offer: Offer, ice_ufrag: str, ice_pwd: str,
                 fingerprint: str, host_ip: str, port: int,
                 ssrc: int, cname: str = "selkies-amd",
                 audio_ssrc: int = 0) -> str:
    """ice-lite answer: video sendonly with the chosen H.264 PT, audio
    sendonly G.711 when offered (and audio_ssrc is set); every other
    m-line is rejected (port 0) but kept for BUNDLE ordering."""
    lines = [
        "v=0",
        f"o=- 0 0 IN IP4 {host_ip}",
        "s=-",
        "t=0 0",
        "a=ice-lite",
        "a=msid-semantic: WMS selkies",
    ]
    mids = [m.mid for m in offer.media]
    video = next((m for m in offer.media if m.kind == "video"), None)
    assert video is not None and video.h264_pts, "offer lacks H.264 video"
    pt = video.h264_pts[0][0]
    lines.append("a=group:BUNDLE " + " ".join(mids))
    for m in offer.media:
        if m is video:
            lines += [
                f"m=video {port} UDP/TLS/RTP/SAVPF {pt}",
                f"c=IN IP4 {host_ip}",
                f"a=mid:{m.mid}",
                f"a=ice-ufrag:{ice_ufrag}",
                f"a=ice-pwd:{ice_pwd}",
                f"a=fingerprint:sha-256 {fingerprint}",
                "a=setup:passive",
                "a=sendonly",
                "a=rtcp-mux",
                f"a=rtpmap:{pt} H264/90000",
                f"a=fmtp:{pt} level-asymmetry-allowed=1;"
                "packetization-mode=1;profile-level-id=42e01f",
                f"a=rtcp-fb:{pt} nack",
                f"a=rtcp-fb:{pt} nack pli",
                f"a=rtcp-fb:{pt} ccm fir",
                f"a=ssrc:{ssrc} cname:{cname}",
                f"a=ssrc:{ssrc} msid:selkies video0",
                f"a=candidate:1 1 udp 2130706431 {host_ip} {port} typ host",
                "a=end-of-candidates",
            ]
        elif m.kind == "application" and "webrtc-datachannel" in \
                " ".join(m.raw):
            lines += [
                f"m=application {port} UDP/DTLS/SCTP webrtc-datachannel",
                f"c=IN IP4 {host_ip}",
                f"a=mid:{m.mid}",
                f"a=ice-ufrag:{ice_ufrag}",
                f"a=ice-pwd:{ice_pwd}",
                f"a=fingerprint:sha-256 {fingerprint}",
                "a=setup:passive",
                "a=sctp-port:5000",
                "a=max-message-size:65536",
                f"a=candidate:1 1 udp 2130706431 {host_ip} {port} typ host",
                "a=end-of-candidates",
            ]
        elif m.kind == "audio" and m.g711_pt >= 0 and audio_ssrc:
            apt = m.g711_pt
            codec = "PCMU" if apt == 0 else "PCMA"
            lines += [
                f"m=audio {port} UDP/TLS/RTP/SAVPF {apt}",
                f"c=IN IP4 {host_ip}",
                f"a=mid:{m.mid}",
                f"a=ice-ufrag:{ice_ufrag}",
                f"a=ice-pwd:{ice_pwd}",
                f"a=fingerprint:sha-256 {fingerprint}",
                "a=setup:passive",
                "a=sendonly",
                "a=rtcp-mux",
                f"a=rtpmap:{apt} {codec}/8000",
                "a=ptime:20",
                f"a=ssrc:{audio_ssrc} cname:{cname}",
                f"a=ssrc:{audio_ssrc} msid:selkies audio0",
                f"a=candidate:1 1 udp 2130706431 {host_ip} {port} typ host",
                "a=end-of-candidates",
            ]
        else:
            proto = "UDP/TLS/RTP/SAVPF 0" if m.kind == "audio" else \
                "UDP/DTLS/SCTP webrtc-datachannel"
            lines += [
                f"m={m.kind} 0 {proto}",
                f"c=IN IP4 0.0.0.0",
                f"a=mid:{m.mid}",
                "a=inactive",
                f"a=ice-ufrag:{ice_ufrag}",
                f"a=ice-pwd:{ice_pwd}",
                f"a=fingerprint:sha-256 {fingerprint}",
                "a=setup:passive",
            ]
    return "\r\n".join(lines) + "\r\n"
