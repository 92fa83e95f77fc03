"""End-to-end WS server test: live aiohttp server + websocket client
receiving real encoded stripes, control verbs, input round-trip, auth,
API surface. (Behavioral equivalent of the reference's integration
test_protocol.py tier — SURVEY.md §4.)"""

import asyncio
import json

import pytest

hipflux = pytest.importorskip("hipflux")
if not hipflux.native_available():
    pytest.skip("hipflux native module not built", allow_module_level=True)

import aiohttp
from aiohttp import WSMsgType

from selkies_amd.settings import load_settings
from selkies_amd.stream_server import CentralizedStreamServer
from selkies_amd.input_handler import RecordingBackend


def make_server(**env):
    base_env = {
        "SELKIES_PORT": "0",
        "SELKIES_CAPTURE_BACKEND": "synthetic:desktop",
        "SELKIES_RESOLUTION": "320x192",
        "SELKIES_FRAMERATE": "30",
        "SELKIES_USE_CPU": "true",
        "SELKIES_ENCODER": "h264enc-striped",
        "SELKIES_ENABLE_AUDIO": "false",
    }
    base_env.update(env)
    settings = load_settings(argv=[], env=base_env)
    server = CentralizedStreamServer(settings)
    # force deterministic input backend
    server.streaming.input.backend = RecordingBackend()
    return server


async def start_on_free_port(server):
    from aiohttp import web
    runner = web.AppRunner(server.app)
    await runner.setup()
    site = web.TCPSite(runner, "127.0.0.1", 0)
    await site.start()
    port = site._server.sockets[0].getsockname()[1]
    return runner, port


@pytest.fixture()
def loop():
    loop = asyncio.new_event_loop()
    yield loop
    loop.close()


def test_video_flows_and_control(loop):
    async def main():
        server = make_server()
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                # API surface
                async with sess.get(f"http://127.0.0.1:{port}/api/health") as r:
                    assert (await r.json())["ok"]
                async with sess.get(f"http://127.0.0.1:{port}/api/status") as r:
                    st = await r.json()
                    assert st["mode"] == "websockets"

                async with sess.ws_connect(
                        f"http://127.0.0.1:{port}/websockets") as ws:
                    texts, frames = [], []
                    deadline = asyncio.get_event_loop().time() + 8
                    while asyncio.get_event_loop().time() < deadline:
                        msg = await ws.receive(timeout=8)
                        if msg.type == WSMsgType.TEXT:
                            texts.append(msg.data)
                        elif msg.type == WSMsgType.BINARY:
                            frames.append(msg.data)
                            if len(frames) >= 24:
                                break
                        else:
                            break
                    # handshake pushes
                    assert any(t.startswith("MODE,") for t in texts)
                    assert any(t.startswith("SETTINGS_PAYLOAD,") for t in texts)
                    # the liveness floor the reference e2e asserts (>=24
                    # binary frames)
                    assert len(frames) >= 24
                    assert frames[0][0] == 0x04  # H.264 stripe tag
                    # ACK a frame id, send input, change a setting
                    await ws.send_str("CLIENT_FRAME_ACK,0")
                    await ws.send_str("kd,65")
                    await ws.send_str("ku,65")
                    await ws.send_str("m,10,20,0")
                    await ws.send_str('SETTINGS,{"framerate": 15}')
                    # wait for settings echo
                    deadline = asyncio.get_event_loop().time() + 5
                    got_echo = False
                    while asyncio.get_event_loop().time() < deadline:
                        msg = await ws.receive(timeout=5)
                        if msg.type == WSMsgType.TEXT and \
                                msg.data.startswith("SETTINGS_PAYLOAD,"):
                            payload = json.loads(
                                msg.data.split(",", 1)[1])
                            assert payload["framerate"]["value"] == 15
                            got_echo = True
                            break
                    assert got_echo
                ev = server.streaming.input.backend.events
                assert ("key", 65, True) in ev and ("key", 65, False) in ev
                assert ("move", 10, 20) in ev
        finally:
            server.streaming.stop_capture()
            await runner.cleanup()

    loop.run_until_complete(main())


def test_auth_token(loop):
    async def main():
        server = make_server(SELKIES_AUTH_TOKEN="sekrit")
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                async with sess.get(f"http://127.0.0.1:{port}/api/status") as r:
                    assert r.status == 401
                async with sess.get(
                        f"http://127.0.0.1:{port}/api/status",
                        headers={"Authorization": "Bearer sekrit"}) as r:
                    assert r.status == 200
                # health is always open
                async with sess.get(f"http://127.0.0.1:{port}/api/health") as r:
                    assert r.status == 200
        finally:
            await runner.cleanup()

    loop.run_until_complete(main())


def test_jpeg_mode_stripes(loop):
    async def main():
        server = make_server(SELKIES_ENCODER="jpeg")
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                async with sess.ws_connect(
                        f"http://127.0.0.1:{port}/ws") as ws:
                    deadline = asyncio.get_event_loop().time() + 8
                    got = None
                    while asyncio.get_event_loop().time() < deadline:
                        msg = await ws.receive(timeout=8)
                        if msg.type == WSMsgType.BINARY and \
                                msg.data[0] == 0x03:
                            got = msg.data
                            break
                    assert got is not None
                    # payload after the 6-byte header is a JFIF image
                    assert got[6:8] == b"\xff\xd8"
        finally:
            server.streaming.stop_capture()
            await runner.cleanup()

    loop.run_until_complete(main())


def test_audio_broadcast(loop):
    async def main():
        server = make_server(SELKIES_ENABLE_AUDIO="true",
                             SELKIES_AUDIO_RED_DISTANCE="1")
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                async with sess.ws_connect(
                        f"http://127.0.0.1:{port}/ws") as ws:
                    audio = []
                    deadline = asyncio.get_event_loop().time() + 8
                    while asyncio.get_event_loop().time() < deadline:
                        msg = await ws.receive(timeout=8)
                        if msg.type == WSMsgType.BINARY and \
                                msg.data[0] == 0x01:
                            audio.append(msg.data)
                            if len(audio) >= 5:
                                break
                    assert len(audio) >= 5
                    # RED depth reaches the configured distance
                    assert audio[-1][1] == 1
        finally:
            server.streaming.stop_capture()
            server.streaming.stop_audio()
            await runner.cleanup()

    loop.run_until_complete(main())


def test_turn_endpoint_and_credentials():
    """/api/turn mints coturn use-auth-secret credentials; falls back
    to a STUN-only config when no TURN is configured (resolution
    chain, reference webrtc_utils.get_rtc_configuration)."""
    import asyncio
    import base64
    import hashlib
    import hmac as hmac_mod
    import aiohttp
    from test_server import make_server, start_on_free_port

    async def main():
        srv = make_server()
        runner, port = await start_on_free_port(srv)
        try:
            async with aiohttp.ClientSession() as sess:
                r = await sess.get(f"http://127.0.0.1:{port}/api/turn")
                assert r.status == 200
                assert r.headers["X-RTC-Source"] == "stun"
                cfg = await r.json()
                assert cfg["iceServers"][0]["urls"][0].startswith("stun:")
        finally:
            await srv.stop()
            await runner.cleanup()

        srv = make_server(SELKIES_TURN_HOST="relay.example",
                          SELKIES_TURN_PORT="3478",
                          SELKIES_TURN_SHARED_SECRET="s3cret")
        runner, port = await start_on_free_port(srv)
        try:
            async with aiohttp.ClientSession() as sess:
                r = await sess.get(
                    f"http://127.0.0.1:{port}/api/turn?user=alice")
                assert r.status == 200
                cfg = await r.json()
        finally:
            await srv.stop()
            await runner.cleanup()
        turn = cfg["iceServers"][1]
        assert turn["urls"] == ["turn:relay.example:3478?transport=udp"]
        expiry, user = turn["username"].split(":")
        assert user == "alice" and int(expiry) > 0
        want = base64.b64encode(hmac_mod.new(
            b"s3cret", turn["username"].encode(),
            hashlib.sha1).digest()).decode()
        assert turn["credential"] == want
        assert any(u.startswith("stun:relay.example")
                   for u in cfg["iceServers"][0]["urls"])

    asyncio.new_event_loop().run_until_complete(main())


def test_webrtc_stats_recorder(tmp_path):
    """POSTed getStats rows land sanitized + ordered in the per-day CSV;
    disabled by default (404)."""
    import asyncio
    import aiohttp
    from test_server import make_server, start_on_free_port

    async def main():
        srv = make_server()
        runner, port = await start_on_free_port(srv)
        try:
            async with aiohttp.ClientSession() as sess:
                r = await sess.post(
                    f"http://127.0.0.1:{port}/api/webrtc-stats", json=[])
                assert r.status == 404
        finally:
            await srv.stop()
            await runner.cleanup()

        srv = make_server(SELKIES_ENABLE_WEBRTC_STATISTICS="true",
                          SELKIES_WEBRTC_STATISTICS_DIR=str(tmp_path))
        runner, port = await start_on_free_port(srv)
        try:
            async with aiohttp.ClientSession() as sess:
                rows = [{"id": "t1", "type": "inbound-rtp",
                         "framesDecoded": 42, "bad,key" * 40: "x",
                         "note": "a,b\nc"},
                        {"id": "t2", "type": "candidate-pair",
                         "rtt": 0.012}]
                r = await sess.post(
                    f"http://127.0.0.1:{port}/api/webrtc-stats", json=rows)
                assert r.status == 200
                assert (await r.json())["accepted"] == 2
        finally:
            await srv.stop()
            await runner.cleanup()
        files = list(tmp_path.glob("webrtc-*.csv"))
        assert len(files) == 1
        lines = files[0].read_text().splitlines()
        assert lines[0].startswith("ts,")
        assert len(lines) == 3
        assert "framesDecoded" in lines[0]
        assert "a;b c" in lines[1]          # sanitized separators
        assert "bad,key" not in lines[0]    # oversized key dropped

    asyncio.new_event_loop().run_until_complete(main())


def test_mode_switch_runtime():
    """POST /api/mode switches transports at runtime (reference
    switch_to_mode): webrtc brings the RTC stack up eagerly, websockets
    tears it down; /api/status reflects the active mode."""
    import asyncio
    import aiohttp

    async def main():
        srv = make_server()
        runner, port = await start_on_free_port(srv)
        try:
            async with aiohttp.ClientSession() as sess:
                base = f"http://127.0.0.1:{port}"
                r = await sess.post(base + "/api/mode",
                                    json={"mode": "webrtc"})
                assert r.status == 200
                body = await r.json()
                assert body == {"mode": "webrtc", "webrtc_active": True}
                assert srv.webrtc is not None
                st = await (await sess.get(base + "/api/status")).json()
                assert st["mode"] == "webrtc"

                r = await sess.post(base + "/api/mode",
                                    json={"mode": "websockets"})
                assert (await r.json())["webrtc_active"] is False
                assert srv.webrtc is None

                r = await sess.post(base + "/api/mode",
                                    json={"mode": "bogus"})
                assert r.status == 400
        finally:
            await srv.stop()
            srv.streaming.stop_capture()
            srv.streaming.stop_audio()
            await runner.cleanup()

    asyncio.new_event_loop().run_until_complete(main())


def test_glass_to_glass_stats():
    """CLIENT_FRAME_ACK drives the capture->ACK latency window exposed in
    /api/stats (the BASELINE glass-to-glass metric plumbing)."""
    async def main():
        server = make_server(SELKIES_VIDEO_FULLFRAME="true")
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                async with sess.ws_connect(
                        f"http://127.0.0.1:{port}/ws") as ws:
                    acked = 0
                    rows = {}
                    n_rows = 3  # 192 / 64
                    for _ in range(400):
                        msg = await asyncio.wait_for(ws.receive(), 10)
                        if msg.type != aiohttp.WSMsgType.BINARY:
                            continue
                        d = msg.data
                        if d[0] != 0x04:
                            continue
                        fid = (d[2] << 8) | d[3]
                        s = rows.setdefault(fid, set())
                        s.add((d[4] << 8) | d[5])
                        if len(s) == n_rows:
                            rows.pop(fid)
                            await ws.send_str(f"CLIENT_FRAME_ACK,{fid}")
                            acked += 1
                            if acked >= 5:
                                break
                    assert acked >= 5
                    # give the server a beat to process the last ack
                    await asyncio.sleep(0.1)
                r = await sess.get(f"http://127.0.0.1:{port}/api/stats")
                st = await r.json()
            g2g = st["streaming"]["glass_to_glass_ms"]
            assert g2g["n"] >= 4
            assert 0 < g2g["p50"] < 5000
            assert g2g["p95"] >= g2g["p50"]
        finally:
            server.streaming.stop_capture()
            await runner.cleanup()

    asyncio.run(main())


def test_dashboard_page_served():
    async def main():
        server = make_server()
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                r = await sess.get(f"http://127.0.0.1:{port}/dashboard")
                assert r.status == 200
                body = await r.text()
                assert "dashboard-app.js" in body
                r = await sess.get(
                    f"http://127.0.0.1:{port}/static/dashboard-app.js")
                assert r.status == 200
                assert "postMessage" in await r.text()
                r = await sess.get(
                    f"http://127.0.0.1:{port}/static/postmessage-bridge.js")
                assert r.status == 200
        finally:
            server.streaming.stop_capture()
            await runner.cleanup()

    asyncio.run(main())


def test_viewonly_password_forces_viewer(loop):
    """basic_auth_viewonly_password: valid second password authenticates
    but the WS session is pinned to the viewer role (reference
    basic_auth_viewonly_password)."""
    import base64

    async def main():
        server = make_server(SELKIES_ENABLE_BASIC_AUTH="true",
                             SELKIES_BASIC_AUTH_PASSWORD="main",
                             SELKIES_BASIC_AUTH_VIEWONLY_PASSWORD="view")
        runner, port = await start_on_free_port(server)

        def hdr(pw):
            tok = base64.b64encode(f"selkies:{pw}".encode()).decode()
            return {"Authorization": f"Basic {tok}"}

        try:
            async with aiohttp.ClientSession() as sess:
                async with sess.get(f"http://127.0.0.1:{port}/api/status",
                                    headers=hdr("wrong")) as r:
                    assert r.status == 401
                async with sess.get(f"http://127.0.0.1:{port}/api/status",
                                    headers=hdr("view")) as r:
                    assert r.status == 200
                # WS under the viewonly password: server assigns viewer
                # even though this is the first (would-be controller) client
                async with sess.ws_connect(
                        f"ws://127.0.0.1:{port}/ws",
                        headers=hdr("view")) as ws:
                    role = None
                    for _ in range(20):
                        msg = await asyncio.wait_for(ws.receive(), 5)
                        if (msg.type == WSMsgType.TEXT and
                                msg.data.startswith("ROLE,")):
                            role = msg.data.split(",", 1)[1]
                            break
                    assert role == "viewer"
        finally:
            await runner.cleanup()

    loop.run_until_complete(main())


def test_subfolder_routes(loop):
    async def main():
        server = make_server(SELKIES_SUBFOLDER="desk1")
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                for path in ("/desk1", "/desk1/static/selkies-client.js"):
                    async with sess.get(
                            f"http://127.0.0.1:{port}{path}") as r:
                        assert r.status == 200, path
        finally:
            await runner.cleanup()

    loop.run_until_complete(main())


def test_run_after_connect_and_disconnect_hooks(loop, tmp_path):
    mark_c = tmp_path / "connected"
    mark_d = tmp_path / "disconnected"

    async def main():
        server = make_server(
            SELKIES_RUN_AFTER_CONNECT=f"touch {mark_c}",
            SELKIES_RUN_AFTER_DISCONNECT=f"touch {mark_d}")
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                async with sess.ws_connect(
                        f"ws://127.0.0.1:{port}/ws") as ws:
                    await asyncio.wait_for(ws.receive(), 5)
            for _ in range(50):
                if mark_c.exists() and mark_d.exists():
                    break
                await asyncio.sleep(0.1)
            assert mark_c.exists() and mark_d.exists()
        finally:
            await runner.cleanup()

    loop.run_until_complete(main())


def test_ui_settings_in_client_payload(loop):
    async def main():
        server = make_server(SELKIES_UI_TITLE="My Desk")
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                async with sess.ws_connect(
                        f"ws://127.0.0.1:{port}/ws") as ws:
                    payload = None
                    for _ in range(20):
                        msg = await asyncio.wait_for(ws.receive(), 5)
                        if (msg.type == WSMsgType.TEXT and
                                msg.data.startswith("SETTINGS_PAYLOAD,")):
                            payload = json.loads(
                                msg.data.split(",", 1)[1])
                            break
                    assert payload is not None
                    assert payload["ui_title"]["value"] == "My Desk"
                    assert payload["ui_show_sidebar"]["value"] is True
                    assert payload["ui_sidebar_show_stats"]["value"] is True
        finally:
            await runner.cleanup()

    loop.run_until_complete(main())


def test_transfer_ui_and_file_roundtrip(loop, tmp_path):
    """The index page ships the transfer panel; upload -> list ->
    download round-trips through the API the panel drives."""
    async def main():
        server = make_server(SELKIES_UPLOAD_DIR=str(tmp_path))
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                async with sess.get(f"http://127.0.0.1:{port}/") as r:
                    page = await r.text()
                    assert "upload-btn" in page and "file-list" in page
                async with sess.post(
                        f"http://127.0.0.1:{port}/api/upload?name=a.txt",
                        data=b"hello transfers") as r:
                    assert r.status == 200
                async with sess.get(
                        f"http://127.0.0.1:{port}/api/files") as r:
                    files = await r.json()
                    assert any(f["name"] == "a.txt" for f in files)
                async with sess.get(
                        f"http://127.0.0.1:{port}/api/download?name=a.txt"
                ) as r:
                    assert await r.read() == b"hello transfers"
        finally:
            await runner.cleanup()

    loop.run_until_complete(main())


def test_mic_uplink_binary_frames(loop):
    """Client mic PCM (binary 0x02) lands in the virtual microphone sink
    when enable_microphone is on."""
    async def main():
        server = make_server(SELKIES_ENABLE_MICROPHONE="true")
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                async with sess.ws_connect(
                        f"ws://127.0.0.1:{port}/ws") as ws:
                    await asyncio.wait_for(ws.receive(), 5)
                    pcm = bytes(2048)
                    await ws.send_bytes(b"\x02" + pcm)
                    for _ in range(100):
                        sink = server.streaming.mic_sink
                        if sink is not None and sink.buffered >= 2048:
                            break
                        await asyncio.sleep(0.02)
                    assert server.streaming.mic_sink is not None
                    assert server.streaming.mic_sink.buffered >= 2048
        finally:
            await runner.cleanup()

    loop.run_until_complete(main())


def test_stream_control_verbs(loop):
    """START/STOP_VIDEO and START/STOP_AUDIO gate the per-client streams
    (reference control verbs); REQUEST_KEYFRAME aliases REQUEST_IDR."""
    async def main():
        server = make_server()
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                async with sess.ws_connect(
                        f"ws://127.0.0.1:{port}/ws") as ws:
                    # receive some video first
                    saw_video = False
                    for _ in range(100):
                        msg = await asyncio.wait_for(ws.receive(), 5)
                        if msg.type == aiohttp.WSMsgType.BINARY and \
                                msg.data[0] in (3, 4, 6):
                            saw_video = True
                            break
                    assert saw_video
                    await ws.send_str("STOP_VIDEO,")
                    # drain until the ack; then confirm video stops
                    acked = False
                    for _ in range(200):
                        msg = await asyncio.wait_for(ws.receive(), 5)
                        if msg.type == aiohttp.WSMsgType.TEXT and \
                                msg.data.startswith("VIDEO_STOPPED"):
                            acked = True
                            break
                    assert acked
                    # after the ack no more video frames arrive
                    stray = 0
                    try:
                        for _ in range(30):
                            msg = await asyncio.wait_for(ws.receive(), 0.2)
                            if (msg.type == aiohttp.WSMsgType.BINARY and
                                    msg.data[0] in (3, 4, 6)):
                                stray += 1
                    except asyncio.TimeoutError:
                        pass
                    assert stray == 0, stray
                    await ws.send_str("START_VIDEO,")
                    resumed = False
                    for _ in range(200):
                        msg = await asyncio.wait_for(ws.receive(), 5)
                        if msg.type == aiohttp.WSMsgType.BINARY and \
                                msg.data[0] in (3, 4, 6):
                            resumed = True
                            break
                    assert resumed
                    await ws.send_str("REQUEST_KEYFRAME,")
        finally:
            await runner.cleanup()

    loop.run_until_complete(main())


def test_pipeline_reset_and_display_config_notices(loop):
    """Structural setting changes broadcast PIPELINE_RESETTING before
    the restart and DISPLAY_CONFIG_UPDATE after (reference notices)."""
    async def main():
        server = make_server()
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                async with sess.ws_connect(
                        f"ws://127.0.0.1:{port}/ws") as ws:
                    await asyncio.wait_for(ws.receive(), 5)
                    await ws.send_str(
                        'SETTINGS,{"video_fullframe": false}')
                    saw_reset = saw_cfg = False
                    for _ in range(300):
                        msg = await asyncio.wait_for(ws.receive(), 5)
                        if msg.type != aiohttp.WSMsgType.TEXT:
                            continue
                        if msg.data.startswith("PIPELINE_RESETTING"):
                            saw_reset = True
                        if msg.data.startswith("DISPLAY_CONFIG_UPDATE,"):
                            saw_cfg = True
                            payload = json.loads(
                                msg.data.split(",", 1)[1])
                            assert payload["displays"][0]["width"] > 0
                        if saw_reset and saw_cfg:
                            break
                    assert saw_reset and saw_cfg
        finally:
            await runner.cleanup()

    loop.run_until_complete(main())


def test_direct_files_route(loop, tmp_path):
    async def main():
        server = make_server(SELKIES_UPLOAD_DIR=str(tmp_path))
        (tmp_path / "doc.txt").write_bytes(b"direct")
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                async with sess.get(
                        f"http://127.0.0.1:{port}/files/doc.txt") as r:
                    assert r.status == 200
                    assert await r.read() == b"direct"
                async with sess.get(
                        f"http://127.0.0.1:{port}/files/../etc/passwd"
                ) as r:
                    assert r.status in (403, 404)
        finally:
            await runner.cleanup()

    loop.run_until_complete(main())


def test_live_audio_is_opus_and_decodes(loop):
    """The live server's audio fan-out carries Opus packets by default;
    the from-spec decoder recovers audio from the wire frames."""
    import numpy as np
    from opus_ref_decoder import OpusDecoder

    async def main():
        server = make_server(SELKIES_ENABLE_AUDIO="true")
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                async with sess.ws_connect(
                        f"ws://127.0.0.1:{port}/ws") as ws:
                    pkts = []
                    for _ in range(400):
                        msg = await asyncio.wait_for(ws.receive(), 5)
                        if msg.type == aiohttp.WSMsgType.BINARY and \
                                msg.data[0] == 0x01:
                            d = msg.data
                            off = 2
                            for _ in range(d[1]):
                                ln = (d[off] << 8) | d[off + 1]
                                off += 2 + ln
                            pkts.append(bytes(d[off:]))
                            if len(pkts) >= 8:
                                break
                    assert len(pkts) >= 8
                    # TOC: CELT fullband 20 ms mono
                    assert all(p[0] >> 3 == 31 for p in pkts)
                    dec = OpusDecoder()
                    for p in pkts:
                        dec.decode_packet(p)
                    out = dec.samples()[960:]
                    assert np.abs(out).max() > 0.01   # audible tone
        finally:
            await runner.cleanup()

    loop.run_until_complete(main())
