"""Role/slot policy: one controller, viewers can't inject, promotion."""

import asyncio

import pytest

hipflux = pytest.importorskip("hipflux")
if not hipflux.native_available():
    pytest.skip("hipflux native module not built", allow_module_level=True)

import aiohttp
from aiohttp import WSMsgType

from selkies_amd.input_handler import RecordingBackend
from test_server import make_server, start_on_free_port


@pytest.fixture()
def loop():
    loop = asyncio.new_event_loop()
    yield loop
    loop.close()


async def read_role(ws, timeout=20):
    deadline = asyncio.get_event_loop().time() + timeout
    while asyncio.get_event_loop().time() < deadline:
        msg = await ws.receive(timeout=timeout)
        if msg.type == WSMsgType.TEXT and msg.data.startswith("ROLE,"):
            return msg.data.split(",", 1)[1]
    return None


def test_controller_assignment_and_promotion(loop):
    async def main():
        server = make_server()
        backend = RecordingBackend()
        server.streaming.input.backend = backend
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                url = f"http://127.0.0.1:{port}/websockets"
                ws1 = await sess.ws_connect(url)
                assert await read_role(ws1) == "controller"
                ws2 = await sess.ws_connect(url)
                assert await read_role(ws2) == "viewer"

                async def wait_event(ev, timeout=5.0):
                    deadline = asyncio.get_event_loop().time() + timeout
                    while asyncio.get_event_loop().time() < deadline:
                        if ev in backend.events:
                            return True
                        await asyncio.sleep(0.02)
                    return False

                # viewer input is ignored
                await ws2.send_str("kd,120")
                # controller input lands (also orders the viewer check:
                # by the time 121 arrived, 120 would have too)
                await ws1.send_str("kd,121")
                assert await wait_event(("key", 121, True))
                assert ("key", 120, True) not in backend.events

                # controller leaves -> viewer promoted
                await ws1.close()
                assert await read_role(ws2) == "controller"
                await ws2.send_str("kd,122")
                assert await wait_event(("key", 122, True))
                await ws2.close()
        finally:
            server.streaming.stop_capture()
            server.streaming.stop_audio()
            await runner.cleanup()

    loop.run_until_complete(main())


def test_concurrent_join_exactly_one_controller(loop):
    """Slot enforcement under a simultaneous join burst: exactly one of
    N concurrent connects wins the controller role (reference
    reconnection/slot semantics; weak spot flagged in review)."""
    async def main():
        server = make_server()
        server.streaming.input.backend = RecordingBackend()
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                url = f"http://127.0.0.1:{port}/websockets"
                wss = await asyncio.gather(
                    *[sess.ws_connect(url) for _ in range(5)])
                roles = await asyncio.gather(
                    *[read_role(ws) for ws in wss])
                assert roles.count("controller") == 1, roles
                assert roles.count("viewer") == 4
                # the controller disconnects; someone else is promoted,
                # still exactly one

                async def next_role(ws):
                    try:
                        return await read_role(ws, timeout=20)
                    except (asyncio.TimeoutError, Exception):
                        return None

                idx = roles.index("controller")
                await wss[idx].close()
                later = await asyncio.gather(
                    *[next_role(ws) for i, ws in enumerate(wss)
                      if i != idx])
                assert later.count("controller") == 1, later
                for i, ws in enumerate(wss):
                    if i != idx:
                        await ws.close()
        finally:
            server.streaming.stop_capture()
            server.streaming.stop_audio()
            await runner.cleanup()

    loop.run_until_complete(main())


def test_collab_tokens_roles_and_reconcile(loop):
    """Collab token table: per-token role assignment at connect,
    MK_ACCESS announcements, live revocation disconnects (reference
    user_tokens + reconcile_clients)."""
    import aiohttp as _aiohttp

    async def main():
        server = make_server(SELKIES_ENABLE_COLLAB="true",
                             SELKIES_MASTER_TOKEN="mt")
        server.streaming.input.backend = RecordingBackend()
        runner, port = await start_on_free_port(server)
        hdr = {"Authorization": "Bearer mt"}
        try:
            async with _aiohttp.ClientSession() as sess:
                # mint tokens
                r = await sess.post(
                    f"http://127.0.0.1:{port}/api/tokens",
                    json={"set": {"alice": {"role": "controller"},
                                  "bob": {"role": "viewer", "seat": 1}}},
                    headers=hdr)
                assert r.status == 200
                # wrong master token rejected
                r = await sess.post(
                    f"http://127.0.0.1:{port}/api/tokens", json={},
                    headers={"Authorization": "Bearer nope"})
                assert r.status == 401
                # tokenless connect rejected while the table is populated
                try:
                    ws = await sess.ws_connect(
                        f"ws://127.0.0.1:{port}/ws")
                    msg = await ws.receive(timeout=3)
                    assert msg.type in (_aiohttp.WSMsgType.CLOSE,
                                        _aiohttp.WSMsgType.CLOSED,
                                        _aiohttp.WSMsgType.ERROR)
                except _aiohttp.WSServerHandshakeError:
                    pass
                # bob (viewer token) connects FIRST but stays viewer
                wsb = await sess.ws_connect(
                    f"ws://127.0.0.1:{port}/ws?utoken=bob")
                assert await read_role(wsb) == "viewer"
                wsa = await sess.ws_connect(
                    f"ws://127.0.0.1:{port}/ws?utoken=alice")
                assert await read_role(wsa) == "controller"
                # revoke alice -> her socket is closed by reconcile
                r = await sess.post(
                    f"http://127.0.0.1:{port}/api/tokens",
                    json={"revoke": ["alice"]}, headers=hdr)
                assert r.status == 200
                closed = False
                for _ in range(100):
                    msg = await wsa.receive(timeout=5)
                    if msg.type in (_aiohttp.WSMsgType.CLOSE,
                                    _aiohttp.WSMsgType.CLOSED,
                                    _aiohttp.WSMsgType.ERROR):
                        closed = True
                        break
                assert closed
                await wsb.close()
        finally:
            server.streaming.stop_capture()
            server.streaming.stop_audio()
            await runner.cleanup()

    loop.run_until_complete(main())
