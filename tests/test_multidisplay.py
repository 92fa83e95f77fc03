"""Multi-display: per-display capture instances and fan-out isolation."""

import asyncio

import pytest

hipflux = pytest.importorskip("hipflux")
if not hipflux.native_available():
    pytest.skip("hipflux native module not built", allow_module_level=True)

import aiohttp
from aiohttp import WSMsgType

from test_server import make_server, start_on_free_port


@pytest.fixture()
def loop():
    loop = asyncio.new_event_loop()
    yield loop
    loop.close()


def test_two_displays_stream_independently(loop):
    async def main():
        server = make_server(SELKIES_RESOLUTION="320x128",
                             SELKIES_RESOLUTION2="256x64",
                             SELKIES_SECOND_DISPLAY="true")
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                ws1 = await sess.ws_connect(
                    f"http://127.0.0.1:{port}/websockets")
                ws2 = await sess.ws_connect(
                    f"http://127.0.0.1:{port}/websockets?display=display2")

                async def collect(ws, n=6, timeout=8):
                    out = []
                    deadline = asyncio.get_event_loop().time() + timeout
                    while (asyncio.get_event_loop().time() < deadline
                           and len(out) < n):
                        msg = await ws.receive(timeout=timeout)
                        if msg.type == WSMsgType.BINARY and \
                                msg.data[0] == 0x04:
                            out.append(msg.data)
                        elif msg.type not in (WSMsgType.TEXT,
                                              WSMsgType.BINARY):
                            break
                    return out

                f1, f2 = await asyncio.gather(collect(ws1), collect(ws2))
                assert len(f1) >= 6 and len(f2) >= 6
                # stripe headers carry each display's own width
                w1 = (f1[0][6] << 8) | f1[0][7]
                w2 = (f2[0][6] << 8) | f2[0][7]
                assert w1 == 320 and w2 == 256
                assert len(server.streaming.captures) == 2
                await ws2.close()
                await asyncio.sleep(0.3)
                assert "display2" not in server.streaming.captures
                assert "primary" in server.streaming.captures
                await ws1.close()
        finally:
            server.streaming.stop_capture()
            await runner.cleanup()

    loop.run_until_complete(main())
