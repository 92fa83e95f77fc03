"""VideoRelay semantics: byte budget, keyframe exemption, row chain repair,
stalled-send reaping (reference contract surveyed at selkies.py:733-890)."""

import asyncio

import pytest

from selkies_amd.relay import VideoRelay


def run(coro):
    return asyncio.get_event_loop_policy().new_event_loop().run_until_complete(coro)


@pytest.fixture()
def loop():
    loop = asyncio.new_event_loop()
    yield loop
    loop.close()


def test_budget_drop_breaks_row_and_requests_idr(loop):
    async def main():
        sent = []
        idr_requests = []
        gate = asyncio.Event()

        async def send(b):
            await gate.wait()
            sent.append(b)

        r = VideoRelay(send, lambda: idr_requests.append(1),
                       bitrate_bps=8 * 1000, budget_seconds=1.0)  # tiny budget
        r.start()
        big = b"x" * 200 * 1024
        assert r.offer(big, y=0, is_keyframe=True)        # keyframes exempt
        assert r.offer(big, y=64, is_keyframe=True)
        # budget (max(256KB, 1KB)) is now exceeded by backlog
        assert not r.offer(b"y" * 300 * 1024, y=0, is_keyframe=False)
        assert idr_requests, "dropping a delta stripe must request an IDR"
        # row 0 chain now broken: deltas for row 0 rejected even when small
        assert not r.offer(b"z", y=0, is_keyframe=False)
        # other rows unaffected (within budget after drain)
        gate.set()
        await asyncio.sleep(0.05)
        assert r.offer(b"z", y=64, is_keyframe=False)
        # keyframe for row 0 repairs the chain
        assert r.offer(b"k", y=0, is_keyframe=True)
        assert r.offer(b"z", y=0, is_keyframe=False)
        await asyncio.sleep(0.05)
        assert r.sent_frames >= 4
        await r.stop()

    loop.run_until_complete(main())


def test_stalled_send_marks_dead(loop):
    async def main():
        async def send(b):
            await asyncio.sleep(10)

        died = []
        r = VideoRelay(send, lambda: None, stall_timeout_s=0.1,
                       on_dead=lambda: died.append(1))
        r.start()
        r.offer(b"k", y=0, is_keyframe=True)
        await asyncio.sleep(0.3)
        assert r.dead and died
        assert not r.offer(b"k2", y=0, is_keyframe=True)
        await r.stop()

    loop.run_until_complete(main())


def test_fifo_delivery(loop):
    async def main():
        sent = []

        async def send(b):
            sent.append(bytes(b))

        r = VideoRelay(send, lambda: None)
        r.start()
        for i in range(10):
            r.offer(bytes([i]), y=0, is_keyframe=(i == 0))
        await asyncio.sleep(0.05)
        assert sent == [bytes([i]) for i in range(10)]
        await r.stop()

    loop.run_until_complete(main())
