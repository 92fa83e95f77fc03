"""Regression tests for the round-1 advisor findings.

Covers: ICE binding-request authentication, DTLS fingerprint binding,
data-channel role gating, SRTP/SRTCP replay protection and receiver ROC
estimation, relay byte accounting, scroll-count clamp, and the auth
middleware hardening.
"""

import asyncio
import secrets
import struct

import pytest

from selkies_amd.webrtc import ice
from selkies_amd.webrtc.srtp import SrtpSession


# ---- ICE gate --------------------------------------------------------------

def test_stun_requires_username_and_integrity():
    ufrag, pwd = ice.make_ice_credentials()
    good = ice.binding_request(f"{ufrag}:client", pwd)
    assert ice.parse_username(good) == f"{ufrag}:client"
    assert ice.verify_request_integrity(good, pwd)
    # wrong password fails integrity
    assert not ice.verify_request_integrity(good, "not-the-password")
    # bare request has neither
    bare = struct.pack(">HHI", 0x0001, 0, ice.MAGIC) + secrets.token_bytes(12)
    assert ice.parse_username(bare) == ""
    assert not ice.verify_request_integrity(bare, pwd)


def test_service_drops_anonymous_stun():
    from selkies_amd.webrtc_service import WebRTCService

    class _S:
        enable_audio = False
        enable_shared = False

    svc = WebRTCService.__new__(WebRTCService)
    svc.settings = _S()
    svc.ufrag, svc.pwd = ice.make_ice_credentials()
    svc.peers = {}

    sent = []

    class _T:
        def sendto(self, data, addr):
            sent.append((data, addr))

    svc.transport = _T()
    addr = ("203.0.113.9", 4242)
    bare = struct.pack(">HHI", 0x0001, 0, ice.MAGIC) + secrets.token_bytes(12)
    svc.on_datagram(bare, addr)
    assert svc.peers == {} and sent == []
    # authenticated request registers the peer and gets a response
    good = ice.binding_request(f"{svc.ufrag}:cli", svc.pwd)
    svc.on_datagram(good, addr)
    assert addr in svc.peers and len(sent) == 1
    assert svc.peers[addr].role == "controller"
    # second peer becomes a viewer
    addr2 = ("203.0.113.10", 4243)
    svc.on_datagram(ice.binding_request(f"{svc.ufrag}:c2", svc.pwd), addr2)
    assert svc.peers[addr2].role == "viewer"


def test_datachannel_role_gate():
    from selkies_amd.webrtc_service import WebRTCService, PeerState
    from selkies_amd.webrtc.sctp import PPID_STRING

    class _S:
        enable_shared = False

    seen = []

    class _In:
        def on_message(self, text):
            seen.append(text)

    class _Streaming:
        input = _In()

    svc = WebRTCService.__new__(WebRTCService)
    svc.settings = _S()
    svc.streaming = _Streaming()
    viewer = PeerState(("1.2.3.4", 1), role="viewer")
    svc._on_dc_message(viewer, 1, PPID_STRING, b"kd,65")
    assert seen == []
    controller = PeerState(("1.2.3.4", 2), role="controller")
    svc._on_dc_message(controller, 1, PPID_STRING, b"kd,65")
    assert seen == ["kd,65"]


# ---- SRTP replay / ROC -----------------------------------------------------

def _rtp(seq, ssrc=0x1234, payload=b"x" * 24):
    return struct.pack(">BBHII", 0x80, 96, seq & 0xFFFF, 1000 + seq,
                       ssrc) + payload


def _pair():
    key, salt = secrets.token_bytes(16), secrets.token_bytes(14)
    return SrtpSession(key, salt), SrtpSession(key, salt)


def test_srtp_replay_rejected():
    tx, rx = _pair()
    pkts = [tx.protect_rtp(_rtp(s)) for s in range(10)]
    for p in pkts:
        rx.unprotect_rtp(p)
    for p in pkts:                       # straight replay of all 10
        with pytest.raises(ValueError, match="replay"):
            rx.unprotect_rtp(p)


def test_srtp_reorder_within_window_ok():
    tx, rx = _pair()
    pkts = [tx.protect_rtp(_rtp(s)) for s in range(8)]
    order = [0, 2, 1, 5, 3, 4, 7, 6]
    for i in order:
        rx.unprotect_rtp(pkts[i])
    with pytest.raises(ValueError):
        rx.unprotect_rtp(pkts[3])        # but not twice


def test_srtp_old_packet_outside_window_rejected():
    tx, rx = _pair()
    first = tx.protect_rtp(_rtp(0))
    rx.unprotect_rtp(first)
    for s in range(1, 80):
        rx.unprotect_rtp(tx.protect_rtp(_rtp(s)))
    with pytest.raises(ValueError, match="old"):
        rx.unprotect_rtp(first)


def test_srtp_roc_across_seq_wrap_with_reorder():
    tx, rx = _pair()
    seqs = [0xFFFD, 0xFFFE, 0xFFFF, 0x0000, 0x0001]
    pkts = {s: tx.protect_rtp(_rtp(s)) for s in seqs}
    # deliver with the wrap-straddling packets swapped: 0x0000 before 0xFFFF
    for s in [0xFFFD, 0xFFFE, 0x0000, 0xFFFF, 0x0001]:
        out = rx.unprotect_rtp(pkts[s])
        assert out == _rtp(s)
    # post-wrap state: new packets still authenticate (ROC advanced once)
    assert rx.unprotect_rtp(tx.protect_rtp(_rtp(0x0002))) == _rtp(0x0002)


def test_srtcp_replay_rejected():
    tx, rx = _pair()
    rtcp = struct.pack(">BBHI", 0x80, 200, 1, 0xABCD) + b"\x00" * 20
    p1 = tx.protect_rtcp(rtcp)
    rx.unprotect_rtcp(p1)
    with pytest.raises(ValueError, match="replay"):
        rx.unprotect_rtcp(p1)
    rx.unprotect_rtcp(tx.protect_rtcp(rtcp))   # next index still fine


# ---- relay accounting ------------------------------------------------------

def test_relay_sent_bytes_counted_once():
    from selkies_amd.relay import VideoRelay

    async def main():
        sent = []

        async def send(payload):
            sent.append(payload)

        relay = VideoRelay(send, request_idr=lambda: None, bitrate_bps=8e6)
        relay.start()
        relay.offer(b"a" * 100, y=0, is_keyframe=True)
        relay.offer(b"b" * 50, y=0, is_keyframe=False)
        for _ in range(50):
            if relay.sent_frames == 2:
                break
            await asyncio.sleep(0.01)
        assert relay.sent_frames == 2
        assert relay.sent_bytes == 150      # was 300 with the double count
        await relay.stop()

    asyncio.run(main())


# ---- scroll clamp ----------------------------------------------------------

def test_scroll_count_clamped():
    from selkies_amd.input_handler import InputDispatcher

    events = []

    class _B:
        def mouse_button(self, b, down):
            events.append((b, down))

    h = InputDispatcher.__new__(InputDispatcher)
    h.backend = _B()
    h.enable_input = True
    h._button_mask = 0
    h.on_message("sw,u,100000000")
    assert len(events) == 200            # 100 press+release pairs max


# ---- auth middleware -------------------------------------------------------

def test_query_token_rejected_on_plain_http():
    """?token= works only for WebSocket upgrades; plain HTTP needs the
    Authorization header (tokens in URLs leak into logs/history)."""
    import aiohttp
    from test_server import make_server, start_on_free_port

    async def main():
        server = make_server(SELKIES_AUTH_TOKEN="sekrit")
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                r = await sess.get(
                    f"http://127.0.0.1:{port}/api/status?token=sekrit")
                assert r.status == 401
                r = await sess.get(
                    f"http://127.0.0.1:{port}/api/status",
                    headers={"Authorization": "Bearer sekrit"})
                assert r.status == 200
                r = await sess.get(
                    f"http://127.0.0.1:{port}/api/status",
                    headers={"Authorization": "Bearer wrong"})
                assert r.status == 401
        finally:
            await runner.cleanup()

    asyncio.run(main())
