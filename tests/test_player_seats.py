"""Player gamepad seats: a viewer claims seat n (player2..4) and its
js,* verbs are remapped onto that pad slot; the controller keeps the
rest.

Reference parity: signaling player2-4 slot enables
(signaling_server.py allowed_client_slots) + dashboard
PlayerGamepadButton (a secondary client claims a gamepad).
"""

import asyncio

import pytest

hipflux = pytest.importorskip("hipflux")
if not hipflux.native_available():
    pytest.skip("hipflux native module not built", allow_module_level=True)

import aiohttp
from aiohttp import WSMsgType

from selkies_amd.input_handler import RecordingBackend
from selkies_amd.streaming import _js_index, _remap_js
from test_server import make_server, start_on_free_port
from test_roles import read_role


def test_js_verb_helpers():
    assert _js_index("js,b,2,0,1") == 2
    assert _js_index("js,c,0,TmFtZQ==,18,4") == 0
    assert _js_index("js") is None
    assert _js_index("js,b,x,0,1") is None
    assert _remap_js("js,b,0,5,1", 3) == "js,b,3,5,1"
    assert _remap_js("js,c,0,TmFtZQ==,18,4", 1) == "js,c,1,TmFtZQ==,18,4"


class RecordingHub:
    """Stands in for GamepadHub; records the verbs that reach it."""

    def __init__(self):
        self.msgs = []

    async def handle(self, msg):
        self.msgs.append(msg)


async def read_seat(ws, timeout=5):
    deadline = asyncio.get_event_loop().time() + timeout
    while asyncio.get_event_loop().time() < deadline:
        msg = await ws.receive(timeout=timeout)
        if msg.type == WSMsgType.TEXT and msg.data.startswith("SEAT,"):
            return int(msg.data.split(",", 1)[1])
    return None


@pytest.fixture()
def loop():
    loop = asyncio.new_event_loop()
    yield loop
    loop.close()


def test_seat_claim_remap_and_release(loop):
    async def main():
        server = make_server(SELKIES_ENABLE_PLAYER2="true")
        server.streaming.input.backend = RecordingBackend()
        hub = RecordingHub()
        server.streaming.gamepads = hub
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                url = f"http://127.0.0.1:{port}/websockets"
                ws1 = await sess.ws_connect(url)
                assert await read_role(ws1) == "controller"
                ws2 = await sess.ws_connect(url)
                assert await read_role(ws2) == "viewer"

                # unseated viewer gamepad input is dropped
                await ws2.send_str("js,b,0,0,1")
                await asyncio.sleep(0.2)
                assert hub.msgs == []

                # controller cannot claim a seat (it owns slot 0)
                await ws1.send_str("CLAIM_SEAT,1")
                assert await read_seat(ws1) == -1
                # enable_player3 is off -> seat 2 denied
                await ws2.send_str("CLAIM_SEAT,2")
                assert await read_seat(ws2) == -1
                # seat 1 (player2) granted
                await ws2.send_str("CLAIM_SEAT,1")
                assert await read_seat(ws2) == 1

                # the player's pads remap onto seat slot 1
                await ws2.send_str("js,c,0,UGFk,18,4")
                await ws2.send_str("js,b,0,3,1")
                deadline = asyncio.get_event_loop().time() + 5
                while (len(hub.msgs) < 2 and
                       asyncio.get_event_loop().time() < deadline):
                    await asyncio.sleep(0.02)
                assert hub.msgs == ["js,c,1,UGFk,18,4", "js,b,1,3,1"]

                # controller still drives slot 0, but claimed slot 1
                # is protected from it
                await ws1.send_str("js,b,0,0,1")
                await ws1.send_str("js,b,1,0,1")
                deadline = asyncio.get_event_loop().time() + 5
                while (hub.msgs[-1] != "js,b,0,0,1" and
                       asyncio.get_event_loop().time() < deadline):
                    await asyncio.sleep(0.02)
                assert hub.msgs[-1] == "js,b,0,0,1"

                # disconnect releases the seat and unplugs its pad
                await ws2.close()
                deadline = asyncio.get_event_loop().time() + 5
                while (hub.msgs[-1] != "js,d,1" and
                       asyncio.get_event_loop().time() < deadline):
                    await asyncio.sleep(0.05)
                assert hub.msgs[-1] == "js,d,1"
                ws3 = await sess.ws_connect(url)
                assert await read_role(ws3) == "viewer"
                await ws3.send_str("CLAIM_SEAT,1")
                assert await read_seat(ws3) == 1
                await ws3.close()
                await ws1.close()
        finally:
            server.streaming.stop_capture()
            server.streaming.stop_audio()
            await runner.cleanup()

    loop.run_until_complete(main())


def test_seat_collision_denied(loop):
    async def main():
        server = make_server(SELKIES_ENABLE_PLAYER2="true",
                             SELKIES_ENABLE_PLAYER3="true")
        server.streaming.input.backend = RecordingBackend()
        server.streaming.gamepads = RecordingHub()
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                url = f"http://127.0.0.1:{port}/websockets"
                ws1 = await sess.ws_connect(url)
                await read_role(ws1)
                ws2 = await sess.ws_connect(url)
                await read_role(ws2)
                ws3 = await sess.ws_connect(url)
                await read_role(ws3)
                await ws2.send_str("CLAIM_SEAT,1")
                assert await read_seat(ws2) == 1
                # occupied seat is denied; the free one is granted
                await ws3.send_str("CLAIM_SEAT,1")
                assert await read_seat(ws3) == -1
                await ws3.send_str("CLAIM_SEAT,2")
                assert await read_seat(ws3) == 2
                # release frees it for someone else
                await ws2.send_str("RELEASE_SEAT")
                assert await read_seat(ws2) == -1
                await ws3.send_str("CLAIM_SEAT,1")  # still holds 2? no:
                # a client holds one seat at a time; claiming again
                # while seated moves it (seat 2 freed on claim of 1)
                got = await read_seat(ws3)
                assert got == 1
                for w in (ws1, ws2, ws3):
                    await w.close()
        finally:
            server.streaming.stop_capture()
            server.streaming.stop_audio()
            await runner.cleanup()

    loop.run_until_complete(main())
