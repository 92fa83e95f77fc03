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
