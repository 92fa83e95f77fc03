"""WebRTC streaming service: ICE-lite + DTLS-SRTP video transport.

The reference's WebRTC mode (SURVEY.md §2.1 webrtc_mode.py + vendored
stacks) rebuilt on this environment's primitives (ctypes-OpenSSL DTLS, own
SRTP/RTP/ICE — selkies_amd/webrtc/). Round-1 scope: H.264 video sendonly
over DTLS-SRTP with PLI/FIR-driven IDR and RTCP sender reports; signaling
is plain HTTP (POST /api/webrtc/offer); input/control ride the WebSocket
(connect with ?display=none for control-only).
"""

from __future__ import annotations

import asyncio
import logging
import secrets
import time
from typing import Optional

import hipflux

from .webrtc import dtls, ice, rtp, sdp
from .webrtc.pacer import AUDIO, VIDEO, Pacer
from .webrtc.sctp import PPID_STRING, PPID_STRING_EMPTY, SctpAssociation
from .webrtc.srtp import SrtpSession, is_rtcp

logger = logging.getLogger("selkies.webrtc")


class PeerState:
    def __init__(self, addr, role: str = "controller"):
        self.addr = addr
        self.role = role
        self.dtls: Optional[dtls.DtlsEndpoint] = None
        self.srtp_out: Optional[SrtpSession] = None
        self.srtp_in: Optional[SrtpSession] = None
        self.sctp: Optional[SctpAssociation] = None
        self.connected = False
        self.last_sr = 0.0


class WebRTCService:
    """One UDP endpoint serving all WebRTC peers (ice-lite)."""

    def __init__(self, settings, streaming):
        self.settings = settings
        self.streaming = streaming       # reuse capture orchestration
        self.cert = dtls.Certificate()
        self.ufrag, self.pwd = ice.make_ice_credentials()
        self.ssrc = secrets.randbits(31) | 1
        self.audio_ssrc = (secrets.randbits(31) | 1) ^ 0x10000
        self.packetizer = rtp.H264Packetizer(self.ssrc)
        self.audio_packetizer = rtp.AudioPacketizer(self.audio_ssrc)
        self.peers: dict[tuple, PeerState] = {}
        self.transport = None
        self.port = 0
        # NAT'd hosts advertise the configured public IP in candidates
        self.host_ip = (settings.webrtc_public_ip or
                        ice.default_host_ip())
        self.capture: Optional[hipflux.ScreenCapture] = None
        # certificate fingerprints signaled in accepted offers; DTLS clients
        # whose cert does not hash to one of these are rejected (RFC 8122)
        self.allowed_fingerprints: set[str] = set()
        self.audio_capture = None
        self.pacer = None
        self._audio_enabled = False
        self._ts_base = time.monotonic()
        self.frames_sent = 0
        self.audio_frames_sent = 0

    # ---- lifecycle ---------------------------------------------------------
    async def start(self, port: int = 0):
        loop = asyncio.get_running_loop()
        self.transport, _ = await loop.create_datagram_endpoint(
            lambda: _Proto(self), local_addr=("0.0.0.0", port))
        self.port = self.transport.get_extra_info("sockname")[1]
        # strict-priority token-bucket pacer: smooths keyframe bursts at
        # 1.25x the video target (+ audio headroom); audio skips ahead
        self.pacer = Pacer(
            lambda pkt, addr: self.transport.sendto(pkt, addr),
            rate_bytes_per_s=self.settings.video_bitrate_kbps * 125.0 *
            1.25 + 32_000)
        self.pacer.start()
        logger.info("webrtc ice-lite endpoint on %s:%d", self.host_ip,
                    self.port)

    async def stop(self):
        self.stop_video()
        self.stop_audio()
        if getattr(self, "pacer", None) is not None:
            await self.pacer.stop()
            self.pacer = None
        if self.transport:
            self.transport.close()
            self.transport = None
        for p in self.peers.values():
            if p.dtls:
                p.dtls.close()
        self.peers.clear()

    # ---- signaling -----------------------------------------------------------
    def handle_offer(self, offer_sdp: str) -> str:
        offer = sdp.parse_offer(offer_sdp)
        for fp in ([offer.session_fingerprint]
                   + [m.fingerprint for m in offer.media]):
            if fp:
                # drop the "sha-256" hash-algo prefix, keep hex only
                digest = fp.split()[-1].upper().replace(":", "")
                self.allowed_fingerprints.add(digest)
        video = next((m for m in offer.media if m.kind == "video"), None)
        if video is not None and video.h264_pts:
            self.packetizer.pt = video.h264_pts[0][0]
        audio = next((m for m in offer.media
                      if m.kind == "audio" and m.g711_pt >= 0), None)
        self._audio_enabled = audio is not None and             getattr(self.settings, "enable_audio", True)
        if audio is not None:
            self.audio_packetizer.pt = audio.g711_pt
        answer = sdp.build_answer(
            offer, self.ufrag, self.pwd, self.cert.fingerprint,
            self.host_ip, self.port, self.ssrc,
            audio_ssrc=self.audio_ssrc if self._audio_enabled else 0)
        return answer

    # ---- media ---------------------------------------------------------------
    def start_video(self):
        if self.capture is not None and self.capture.is_capturing:
            return
        loop = asyncio.get_running_loop()
        cs = self.streaming.build_capture_settings("primary")
        cs.stripe_height = ((cs.capture_height + 15) & ~15)  # full frame
        cs.output_mode = 1
        self.capture = hipflux.ScreenCapture()

        def on_stripe(data, frame_id, y, width, height, is_keyframe,
                      capture_ts_ms, encode_done_ms, stripe_type):
            payload = bytes(data[10:])  # drop the WS wire header
            try:
                loop.call_soon_threadsafe(self._send_frame, payload)
            except RuntimeError:
                pass

        self.capture.start_capture(on_stripe, cs)
        logger.info("webrtc video capture started (pipeline=%s)",
                    self.capture.pipeline)

    def stop_video(self):
        if self.capture is not None:
            self.capture.stop_capture()
            self.capture = None

    # ---- audio (G.711 over SRTP; webrtc/g711.py) ---------------------------
    def start_audio(self):
        if not self._audio_enabled or self.audio_capture is not None:
            return
        from hipflux import _native
        from .webrtc import g711
        st = self.settings
        s = _native.AudioCaptureSettings()
        s.device_name = st.audio_device if st.audio_device != "auto"             else "synthetic"
        s.channels = st.audio_channels
        s.frame_duration_ms = 20
        s.red_distance = 0           # RTP has its own loss handling
        loop = asyncio.get_running_loop()
        channels = st.audio_channels

        def on_frame(data, pts_ms):
            payload = g711.wire_frame_to_ulaw(bytes(data), channels)
            if payload:
                try:
                    loop.call_soon_threadsafe(self._send_audio, payload)
                except RuntimeError:
                    pass

        self.audio_capture = _native.AudioCapture()
        self.audio_capture.start_capture(s, on_frame)
        logger.info("webrtc audio capture started (G.711 %s)",
                    "PCMU" if self.audio_packetizer.pt == 0 else "PCMA")

    def stop_audio(self):
        if self.audio_capture is not None:
            self.audio_capture.stop_capture()
            self.audio_capture = None

    def _send_audio(self, payload: bytes):
        pkt = self.audio_packetizer.packetize(payload)
        self.audio_frames_sent += 1
        for peer in list(self.peers.values()):
            if not peer.connected:
                continue
            try:
                self.pacer.enqueue(AUDIO, peer.srtp_out.protect_rtp(pkt),
                                   peer.addr)
            except Exception as exc:
                logger.debug("srtp audio send failed: %r", exc)

    def _send_frame(self, annexb: bytes):
        if not any(p.connected for p in self.peers.values()):
            return
        ts90k = int((time.monotonic() - self._ts_base) * 90000)
        packets = self.packetizer.packetize(annexb, ts90k)
        self.frames_sent += 1
        now = time.monotonic()
        for peer in list(self.peers.values()):
            if not peer.connected:
                continue
            for pkt in packets:
                try:
                    self.pacer.enqueue(VIDEO,
                                       peer.srtp_out.protect_rtp(pkt),
                                       peer.addr)
                except Exception as exc:
                    logger.debug("srtp send failed: %r", exc)
            if now - peer.last_sr > 1.0:
                peer.last_sr = now
                sr = rtp.build_sender_report(self.ssrc, ts90k,
                                             self.packetizer.packets_sent,
                                             self.packetizer.bytes_sent)
                try:
                    self.transport.sendto(peer.srtp_out.protect_rtcp(sr),
                                          peer.addr)
                except Exception:
                    pass

    # ---- datagram demux -------------------------------------------------------
    def on_datagram(self, data: bytes, addr):
        if ice.is_stun(data):
            # ICE gate: the binding request must carry OUR
            # ufrag in USERNAME and a MESSAGE-INTEGRITY keyed with OUR
            # password (RFC 8445 §7.3) before the sender becomes a peer.
            username = ice.parse_username(data)
            if not username.startswith(self.ufrag + ":"):
                return
            if not ice.verify_request_integrity(data, self.pwd):
                return
            resp = ice.binding_response(data, addr, self.pwd)
            self.transport.sendto(resp, addr)
            if addr not in self.peers:
                role = ("controller"
                        if not any(p.role == "controller"
                                   for p in self.peers.values())
                        else "viewer")
                self.peers[addr] = PeerState(addr, role)
                logger.info("webrtc peer candidate %s (%s)", addr, role)
            return
        peer = self.peers.get(addr)
        if peer is None:
            return
        first = data[0] if data else 0
        if 20 <= first <= 63:        # DTLS record
            if peer.dtls is None:
                peer.dtls = dtls.DtlsEndpoint(self.cert, server=True)
            peer.dtls.put_datagram(data)
            for out in peer.dtls.take_datagrams():
                self.transport.sendto(out, addr)
            if peer.dtls.handshake_done:
                self._pump_sctp(peer)
            if peer.dtls.handshake_done and not peer.connected:
                # bind the DTLS client to the signaled offer: its cert must
                # hash to an a=fingerprint we answered (RFC 8122)
                fp = peer.dtls.peer_fingerprint()
                if (fp is None or
                        fp.replace(":", "") not in self.allowed_fingerprints):
                    logger.warning(
                        "webrtc peer %s rejected: certificate fingerprint "
                        "not in any signaled offer", addr)
                    peer.dtls.close()
                    self.peers.pop(addr, None)
                    return
                (ck, cs), (sk, ss) = peer.dtls.export_srtp_keys()
                # we are the DTLS server: send with server keys,
                # receive with client keys
                peer.srtp_out = SrtpSession(sk, ss)
                peer.srtp_in = SrtpSession(ck, cs)
                peer.connected = True
                logger.info("webrtc peer %s connected (DTLS-SRTP up)", addr)
                self.start_video()
                self.start_audio()
                if self.capture:
                    self.capture.request_idr_frame()
        elif first >= 128:           # SRTP/SRTCP
            if peer.connected and is_rtcp(data):
                try:
                    plain = peer.srtp_in.unprotect_rtcp(data)
                except ValueError:
                    return
                for report in rtp.parse_rtcp(plain):
                    if report["type"] in ("PLI", "FIR"):
                        if self.capture:
                            self.capture.request_idr_frame()
                    elif report["type"] == "RR" and report.get("blocks"):
                        self._on_receiver_report(report["blocks"])

    # ---- data channel (SCTP over DTLS; webrtc/sctp.py) ---------------------
    def _pump_sctp(self, peer: PeerState):
        """Feed decrypted app datagrams into the peer's SCTP association
        and flush its outbound packets back through DTLS."""
        inbound = peer.dtls.recv_app()
        if inbound and peer.sctp is None:
            peer.sctp = SctpAssociation(
                True,
                on_message=lambda sid, ppid, d, p=peer:
                    self._on_dc_message(p, sid, ppid, d),
                on_channel_open=lambda ch:
                    logger.info("datachannel '%s' open (sid %d)", ch.label,
                                ch.sid))
        if peer.sctp is None:
            return
        now = time.monotonic()
        for pkt in inbound:
            peer.sctp.receive(pkt, now)
        peer.sctp.poll(now)
        out = peer.sctp.outbound()
        if out:
            for pkt in out:
                peer.dtls.send_app(pkt)
            for rec in peer.dtls.take_datagrams():
                self.transport.sendto(rec, peer.addr)

    def _on_dc_message(self, peer: PeerState, sid: int, ppid: int,
                       data: bytes):
        """Data-channel messages carry the same text verbs as the control
        WebSocket (kd/ku/m/mb/... — docs/protocol.md)."""
        if ppid not in (PPID_STRING, PPID_STRING_EMPTY):
            return
        try:
            text = data.decode("utf-8")
        except UnicodeDecodeError:
            return
        # same role policy as the WebSocket path (streaming._on_text):
        # only the controller — or anyone when sharing is enabled — may
        # inject input (the DC bypassed this gate).
        if (peer.role != "controller"
                and not getattr(self.settings, "enable_shared", False)):
            return
        try:
            self.streaming.input.on_message(text)
        except Exception:
            logger.debug("datachannel verb failed", exc_info=True)

    # ---- loss-based congestion control (reference runs GCC/TWCC in
    # webrtc_mode._congestion_control_loop; this is the loss-driven AIMD
    # slice of it: back off multiplicatively above 2% loss, probe up
    # additively when clean) ------------------------------------------------
    LOSS_BACKOFF_AT = 0.02
    BITRATE_MIN_KBPS = 1000
    BITRATE_MAX_KBPS = 60000
    PROBE_STEP_KBPS = 1000

    def _on_receiver_report(self, blocks):
        loss = max(b.get("fraction_lost", 0.0) for b in blocks)
        now = time.monotonic()
        if now - getattr(self, "_last_cc", 0.0) < 0.5:
            return
        self._last_cc = now
        kbps = getattr(self, "_video_kbps",
                       getattr(self.settings, "video_bitrate_kbps", 16000))
        if loss > self.LOSS_BACKOFF_AT:
            kbps = max(self.BITRATE_MIN_KBPS,
                       int(kbps * (1.0 - 0.5 * min(1.0, loss * 4))))
        else:
            kbps = min(self.BITRATE_MAX_KBPS, kbps + self.PROBE_STEP_KBPS)
        if kbps != getattr(self, "_video_kbps", None):
            self._video_kbps = kbps
            if self.capture is not None:
                try:
                    self.capture.update_video_bitrate(kbps)
                except Exception:
                    pass
            if getattr(self, "pacer", None) is not None:
                self.pacer.set_rate(kbps * 125.0 * 1.25 + 32_000)

    def dc_broadcast(self, text: str):
        """Send a control message to every open data channel."""
        for peer in self.peers.values():
            if peer.sctp is None:
                continue
            for sid, ch in peer.sctp.channels.items():
                if ch.open:
                    try:
                        peer.sctp.send(sid, text, now=time.monotonic())
                    except Exception:
                        pass
            self._flush_sctp(peer)

    def _flush_sctp(self, peer: PeerState):
        for pkt in peer.sctp.outbound():
            peer.dtls.send_app(pkt)
        for rec in peer.dtls.take_datagrams():
            self.transport.sendto(rec, peer.addr)

    def stats(self) -> dict:
        return {
            "port": self.port,
            "peers": len(self.peers),
            "connected": sum(p.connected for p in self.peers.values()),
            "datachannels": sum(
                len([c for c in p.sctp.channels.values() if c.open])
                for p in self.peers.values() if p.sctp),
            "frames_sent": self.frames_sent,
            "packets_sent": self.packetizer.packets_sent,
        }


class _Proto(asyncio.DatagramProtocol):
    def __init__(self, svc: WebRTCService):
        self.svc = svc

    def datagram_received(self, data, addr):
        try:
            self.svc.on_datagram(data, addr)
        except Exception as exc:
            logger.warning("webrtc datagram error: %r", exc)
