"""ICE/RTC config resolution chain + file monitor.

Reference parity: webrtc_utils.get_rtc_configuration:1037 (JSON file ->
TURN-REST -> coturn HMAC -> legacy static -> STUN-only) and
RTCConfigFileMonitor:465 (hot reload + push to clients).
"""

import asyncio
import json

import pytest

from selkies_amd.settings import load_settings
from selkies_amd.webrtc import turn as T


def make_settings(**env):
    base = {"SELKIES_PORT": "0"}
    base.update(env)
    return load_settings(argv=[], env=base)


def test_validate_rtc_config():
    good = {"iceServers": [{"urls": ["stun:a:1"]},
                           {"urls": "turn:b:2?transport=udp",
                            "username": "u", "credential": "c"}]}
    assert T.validate_rtc_config(good)
    assert not T.validate_rtc_config({})
    assert not T.validate_rtc_config({"iceServers": []})
    assert not T.validate_rtc_config({"iceServers": [{"urls": ["http://x"]}]})
    assert not T.validate_rtc_config({"iceServers": [{"urls": []}]})


def test_chain_stun_only_fallback():
    s = make_settings()
    cfg, source = asyncio.new_event_loop().run_until_complete(
        T.resolve_rtc_config(s))
    assert source == "stun"
    assert cfg["iceServers"][0]["urls"][0].startswith("stun:")


def test_chain_hmac_beats_static():
    s = make_settings(SELKIES_TURN_HOST="relay.example",
                      SELKIES_TURN_SHARED_SECRET="s3cret",
                      SELKIES_TURN_USERNAME="u",
                      SELKIES_TURN_PASSWORD="p")
    cfg, source = asyncio.new_event_loop().run_until_complete(
        T.resolve_rtc_config(s, user="alice"))
    assert source == "hmac"
    turn_srv = cfg["iceServers"][1]
    assert turn_srv["username"].endswith(":alice")
    exp = int(turn_srv["username"].split(":")[0])
    assert T.hmac_credential("s3cret", turn_srv["username"]) == \
        turn_srv["credential"]
    assert exp > 0


def test_chain_static_without_secret():
    s = make_settings(SELKIES_TURN_HOST="relay.example",
                      SELKIES_TURN_USERNAME="user1",
                      SELKIES_TURN_PASSWORD="pass1",
                      SELKIES_TURN_TLS="true")
    cfg, source = asyncio.new_event_loop().run_until_complete(
        T.resolve_rtc_config(s))
    assert source == "static"
    turn_srv = cfg["iceServers"][1]
    assert turn_srv["username"] == "user1"
    assert turn_srv["urls"][0].startswith("turns:relay.example:")


def test_chain_file_wins(tmp_path):
    p = tmp_path / "rtc.json"
    p.write_text(json.dumps(
        {"iceServers": [{"urls": ["turn:filehost:3478"],
                         "username": "f", "credential": "f"}]}))
    s = make_settings(SELKIES_RTC_CONFIG_JSON=str(p),
                      SELKIES_TURN_HOST="relay.example",
                      SELKIES_TURN_SHARED_SECRET="x")
    cfg, source = asyncio.new_event_loop().run_until_complete(
        T.resolve_rtc_config(s))
    assert source == "file"
    assert cfg["iceServers"][0]["urls"] == ["turn:filehost:3478"]
    # malformed file falls through to the next link
    p.write_text("{nope")
    cfg, source = asyncio.new_event_loop().run_until_complete(
        T.resolve_rtc_config(s))
    assert source == "hmac"


def test_chain_rest(tmp_path):
    from aiohttp import web

    async def main():
        served = {"iceServers": [{"urls": ["turn:resthost:443"],
                                  "username": "r", "credential": "r"}]}
        seen = {}

        async def handler(request):
            seen.update(request.query)
            return web.json_response(served)

        app = web.Application()
        app.router.add_get("/turn", handler)
        runner = web.AppRunner(app)
        await runner.setup()
        site = web.TCPSite(runner, "127.0.0.1", 0)
        await site.start()
        port = runner.addresses[0][1]
        try:
            s = make_settings(
                SELKIES_TURN_REST_URI=f"http://127.0.0.1:{port}/turn",
                SELKIES_TURN_HOST="relay.example",
                SELKIES_TURN_SHARED_SECRET="x")
            cfg, source = await T.resolve_rtc_config(s, user="bob")
            assert source == "rest"
            assert cfg["iceServers"][0]["urls"] == ["turn:resthost:443"]
            assert seen == {"service": "turn", "username": "bob"}
            # invalid REST payload falls through to HMAC
            served.clear()
            cfg, source = await T.resolve_rtc_config(s)
            assert source == "hmac"
        finally:
            await runner.cleanup()

    asyncio.new_event_loop().run_until_complete(main())


def test_file_monitor_detects_change(tmp_path):
    p = tmp_path / "rtc.json"
    p.write_text(json.dumps({"iceServers": [{"urls": ["stun:a:1"]}]}))

    async def main():
        got = []
        mon = T.RTCConfigFileMonitor(str(p), got.append, interval_s=0.05)
        mon.start()
        try:
            await asyncio.sleep(0.15)
            assert got == []            # unchanged: no callback
            p.write_text(json.dumps(
                {"iceServers": [{"urls": ["stun:b:2"]}]}))
            for _ in range(100):
                if got:
                    break
                await asyncio.sleep(0.05)
            assert got and got[-1]["iceServers"][0]["urls"] == ["stun:b:2"]
            # malformed rewrite is ignored
            n = len(got)
            p.write_text("{bad")
            await asyncio.sleep(0.2)
            assert len(got) == n
        finally:
            mon.stop()

    asyncio.new_event_loop().run_until_complete(main())


def test_rtc_config_push_to_clients(tmp_path):
    """End to end: changing the rtc_config_json file pushes RTC_CONFIG
    to connected WS clients via the server's file monitor."""
    import pytest
    hipflux = pytest.importorskip("hipflux")
    if not hipflux.native_available():
        pytest.skip("hipflux native module not built")
    import aiohttp
    from aiohttp import WSMsgType
    from test_server import make_server, start_on_free_port

    p = tmp_path / "rtc.json"
    p.write_text(json.dumps({"iceServers": [{"urls": ["stun:a:1"]}]}))

    async def main():
        server = make_server(SELKIES_RTC_CONFIG_JSON=str(p))
        runner, port = await start_on_free_port(server)
        # start_on_free_port bypasses server.start(); run the monitor
        # ourselves at a test-friendly poll interval
        server._rtc_monitor = T.RTCConfigFileMonitor(
            str(p), server._on_rtc_config_change, interval_s=0.05)
        server._rtc_monitor.start()
        try:
            async with aiohttp.ClientSession() as sess:
                ws = await sess.ws_connect(
                    f"http://127.0.0.1:{port}/websockets")
                await asyncio.sleep(0.2)
                p.write_text(json.dumps(
                    {"iceServers": [{"urls": ["turn:new:5349"],
                                     "username": "u",
                                     "credential": "c"}]}))
                deadline = asyncio.get_event_loop().time() + 5
                got = None
                while asyncio.get_event_loop().time() < deadline:
                    msg = await ws.receive(timeout=5)
                    if (msg.type == WSMsgType.TEXT and
                            msg.data.startswith("RTC_CONFIG,")):
                        got = json.loads(msg.data.split(",", 1)[1])
                        break
                assert got is not None, "no RTC_CONFIG push received"
                assert got["iceServers"][0]["urls"] == ["turn:new:5349"]
                await ws.close()
        finally:
            await server.stop()
            server.streaming.stop_capture()
            server.streaming.stop_audio()
            await runner.cleanup()

    asyncio.new_event_loop().run_until_complete(main())


def test_chain_rest_headers_and_api_key():
    """TURN-REST header knobs (reference turn_rest_* settings): API key
    as Bearer, username/protocol/TLS hints in configured headers."""
    from aiohttp import web

    async def main():
        served = {"iceServers": [{"urls": ["turn:resthost:443"],
                                  "username": "r", "credential": "r"}]}
        seen = {}

        async def handler(request):
            seen.update({k: v for k, v in request.headers.items()
                         if k.startswith("X-") or k == "Authorization"})
            seen["username_q"] = request.query.get("username")
            return web.json_response(served)

        app = web.Application()
        app.router.add_get("/turn", handler)
        runner = web.AppRunner(app)
        await runner.setup()
        site = web.TCPSite(runner, "127.0.0.1", 0)
        await site.start()
        port = runner.addresses[0][1]
        try:
            s = make_settings(
                SELKIES_TURN_REST_URI=f"http://127.0.0.1:{port}/turn",
                SELKIES_TURN_REST_API_KEY="sekrit",
                SELKIES_TURN_REST_USERNAME="alice",
                SELKIES_TURN_REST_USERNAME_AUTH_HEADER="X-Auth-User",
                SELKIES_TURN_REST_PROTOCOL_HEADER="X-Turn-Protocol",
                SELKIES_TURN_REST_TLS_HEADER="X-Turn-Tls",
                SELKIES_TURN_PROTOCOL="tcp",
                SELKIES_TURN_TLS="true")
            cfg, source = await T.resolve_rtc_config(s)
            assert source == "rest"
            assert seen["Authorization"] == "Bearer sekrit"
            assert seen["X-Auth-User"] == "alice"
            assert seen["X-Turn-Protocol"] == "tcp"
            assert seen["X-Turn-Tls"] == "true"
            assert seen["username_q"] == "alice"
        finally:
            await runner.cleanup()

    asyncio.new_event_loop().run_until_complete(main())


def test_chain_cloudflare_turn():
    """Cloudflare TURN credential minting beats TURN-REST in the chain
    and maps the response into an rtc config."""
    from aiohttp import web

    async def main():
        async def handler(request):
            assert request.headers["Authorization"] == "Bearer cftok"
            body = await request.json()
            assert body["ttl"] == 86400
            return web.json_response({
                "iceServers": {
                    "urls": ["turn:turn.cloudflare.com:3478?transport=udp"],
                    "username": "cfu", "credential": "cfc"}}, status=201)

        app = web.Application()
        app.router.add_post("/gen", handler)
        runner = web.AppRunner(app)
        await runner.setup()
        site = web.TCPSite(runner, "127.0.0.1", 0)
        await site.start()
        port = runner.addresses[0][1]
        try:
            s = make_settings(
                SELKIES_ENABLE_CLOUDFLARE_TURN="true",
                SELKIES_CLOUDFLARE_TURN_TOKEN_ID="kid",
                SELKIES_CLOUDFLARE_TURN_API_TOKEN="cftok",
                SELKIES_TURN_HOST="relay.example",
                SELKIES_TURN_SHARED_SECRET="x")
            s._cloudflare_endpoint = f"http://127.0.0.1:{port}/gen"
            cfg, source = await T.resolve_rtc_config(s)
            assert source == "cloudflare"
            assert cfg["iceServers"][0]["username"] == "cfu"
            # minting failure falls through to HMAC
            s._cloudflare_endpoint = f"http://127.0.0.1:{port}/nope"
            cfg, source = await T.resolve_rtc_config(s)
            assert source == "hmac"
        finally:
            await runner.cleanup()

    asyncio.new_event_loop().run_until_complete(main())
