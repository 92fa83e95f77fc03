"""Strict-priority token-bucket pacer (reference webrtc/pacer.py
semantics): audio skips ahead of video, video is rate-limited with a
bounded queue that drops oldest on overflow."""

import asyncio

import pytest

from selkies_amd.webrtc.pacer import AUDIO, VIDEO, Pacer


class Clock:
    def __init__(self):
        self.t = 0.0

    def __call__(self):
        return self.t


def test_audio_strict_priority():
    sent = []
    clk = Clock()
    p = Pacer(lambda pkt, addr: sent.append(pkt), 1_000_000, clock=clk)
    for i in range(5):
        p.enqueue(VIDEO, b"v" * 1000, None)
    p.enqueue(AUDIO, b"a" * 100, None)
    p.drain_once()
    # audio went out first even though video was enqueued earlier
    assert sent[0] == b"a" * 100


def test_video_rate_limited_and_resumes():
    sent = []
    clk = Clock()
    # 100 kB/s, 15 ms burst -> 1500 B of tokens at t=0
    p = Pacer(lambda pkt, addr: sent.append(pkt), 100_000, clock=clk)
    for _ in range(10):
        p.enqueue(VIDEO, b"v" * 1000, None)
    pending = p.drain_once()
    assert pending and len(sent) == 1          # one packet fits the burst
    clk.t += 0.05                              # refill (capped at burst)
    p.drain_once()
    assert len(sent) == 2                      # burst cap: no catch-up spike
    # steady state: 100 kB/s = one 1000-B packet per 10 ms tick
    for _ in range(8):
        clk.t += 0.01
        p.drain_once()
    assert len(sent) == 10


def test_audio_sends_even_at_token_deficit():
    sent = []
    clk = Clock()
    p = Pacer(lambda pkt, addr: sent.append(pkt), 100_000, clock=clk)
    p.enqueue(VIDEO, b"v" * 1500, None)
    p.drain_once()                             # consumes the whole burst
    p.enqueue(AUDIO, b"a" * 200, None)
    p.enqueue(VIDEO, b"v" * 1000, None)
    p.drain_once()
    assert b"a" * 200 in sent                  # audio never waits
    assert b"v" * 1000 not in sent             # video does


def test_video_queue_bounded_drop_oldest():
    sent = []
    clk = Clock()
    p = Pacer(lambda pkt, addr: sent.append(pkt), 100_000,
              max_queue_s=0.1, clock=clk)      # 10 kB queue bound
    first = b"F" * 1000
    p.enqueue(VIDEO, first, None)
    for _ in range(12):
        p.enqueue(VIDEO, b"v" * 1000, None)
    assert p.dropped_packets >= 3
    clk.t += 10.0
    p.drain_once()
    assert first not in sent                   # oldest was dropped


def test_async_loop_drains():
    async def main():
        sent = []
        p = Pacer(lambda pkt, addr: sent.append(pkt), 10_000_000)
        p.start()
        for _ in range(20):
            p.enqueue(VIDEO, b"v" * 1200, None)
        for _ in range(100):
            if len(sent) == 20:
                break
            await asyncio.sleep(0.01)
        await p.stop()
        assert len(sent) == 20

    asyncio.new_event_loop().run_until_complete(main())
