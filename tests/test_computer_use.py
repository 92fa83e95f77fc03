"""Computer-use agent API: screenshot + actions."""

import asyncio
import io

import numpy as np
import pytest

hipflux = pytest.importorskip("hipflux")
if not hipflux.native_available():
    pytest.skip("hipflux native module not built", allow_module_level=True)

import aiohttp
from PIL import Image

from selkies_amd.input_handler import RecordingBackend
from test_server import make_server, start_on_free_port


@pytest.fixture()
def loop():
    loop = asyncio.new_event_loop()
    yield loop
    loop.close()


def test_screenshot_and_actions(loop):
    async def main():
        server = make_server()
        backend = RecordingBackend()
        server.streaming.input.backend = backend
        runner, port = await start_on_free_port(server)
        try:
            async with aiohttp.ClientSession() as sess:
                url = f"http://127.0.0.1:{port}"
                r = await sess.get(f"{url}/computer-use/screenshot")
                assert r.status == 200
                img = Image.open(io.BytesIO(await r.read()))
                assert img.size == (320, 192)
                assert np.asarray(img).std() > 10   # real content

                r = await sess.post(f"{url}/computer-use/action",
                                    json={"action": "click", "x": 10,
                                          "y": 20})
                assert (await r.json())["ok"]
                r = await sess.post(f"{url}/computer-use/action",
                                    json={"action": "type", "text": "hi"})
                assert (await r.json())["ok"]
                r = await sess.post(f"{url}/computer-use/action",
                                    json={"action": "bogus"})
                assert r.status == 400
        finally:
            await runner.cleanup()
        ev = backend.events
        assert ("move", 10, 20) in ev
        assert ("btn", 1, True) in ev and ("btn", 1, False) in ev
        assert ("key", ord("h"), True) in ev and ("key", ord("i"), True) in ev

    loop.run_until_complete(main())
