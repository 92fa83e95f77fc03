"""Strict-priority token-bucket RTP pacer.

Smooths encoder bursts (a 4K keyframe can be hundreds of packets in one
event-loop tick) onto the wire at a configured rate so routers/reorder
buffers downstream never see line-rate spikes, while audio always skips
ahead of video (reference selkies webrtc/pacer.py:5-37 semantics: strict
priority classes over one token bucket).

* Tokens accrue at ``rate_bytes_per_s`` up to ``burst_ms`` worth.
* ``AUDIO`` drains fully before ``VIDEO`` each tick (5 ms cadence).
* The video queue is bounded by bytes (default one second at rate);
  overflow drops the OLDEST video packets — the per-row IDR repair path
  recovers the affected rows. Audio is never dropped by the pacer.
* ``send_now`` bypasses the queue for tiny control traffic (RTCP).

Pure asyncio, clock-injectable for tests.
"""

from __future__ import annotations

import asyncio
import logging
import time
from collections import deque
from typing import Callable, Optional, Tuple

logger = logging.getLogger(__name__)

AUDIO = 0
VIDEO = 1


class Pacer:
    TICK_S = 0.005

    def __init__(self, send: Callable[[bytes, Tuple], None],
                 rate_bytes_per_s: float,
                 burst_ms: float = 15.0,
                 max_queue_s: float = 1.0,
                 clock: Callable[[], float] = time.monotonic):
        self._send = send
        self.rate = max(16_000.0, float(rate_bytes_per_s))
        self.burst_ms = burst_ms
        self.max_queue_s = max_queue_s
        self._clock = clock
        self._queues = (deque(), deque())      # AUDIO, VIDEO
        self._queued_video_bytes = 0
        self._tokens = self._burst_bytes()
        self._last_refill = clock()
        self._task: Optional[asyncio.Task] = None
        self._wake: Optional[asyncio.Event] = None
        self._stopped = False
        self.sent_packets = 0
        self.dropped_packets = 0

    def _burst_bytes(self) -> float:
        return self.rate * self.burst_ms / 1000.0

    def set_rate(self, rate_bytes_per_s: float):
        self.rate = max(16_000.0, float(rate_bytes_per_s))

    def start(self):
        self._stopped = False
        self._wake = asyncio.Event()
        self._task = asyncio.get_running_loop().create_task(self._run())

    async def stop(self):
        self._stopped = True
        if self._wake is not None:
            self._wake.set()
        t, self._task = self._task, None
        if t is not None:
            t.cancel()
            try:
                await asyncio.wait_for(asyncio.gather(
                    t, return_exceptions=True), timeout=1.0)
            except (asyncio.TimeoutError, asyncio.CancelledError):
                pass

    def send_now(self, payload: bytes, addr):
        """Unpaced control traffic (RTCP SR/PLI — tens of bytes)."""
        try:
            self._send(payload, addr)
        except Exception as exc:
            logger.debug("pacer direct send failed: %r", exc)

    def enqueue(self, prio: int, payload: bytes, addr):
        q = self._queues[prio]
        q.append((payload, addr))
        if prio == VIDEO:
            self._queued_video_bytes += len(payload)
            limit = self.rate * self.max_queue_s
            while self._queued_video_bytes > limit and q:
                old, _ = q.popleft()
                self._queued_video_bytes -= len(old)
                self.dropped_packets += 1
        if self._wake is not None:
            self._wake.set()

    def _refill(self):
        now = self._clock()
        self._tokens = min(self._burst_bytes(),
                           self._tokens + (now - self._last_refill) *
                           self.rate)
        self._last_refill = now

    def drain_once(self) -> bool:
        """One pacing tick: refill, then send strictly by priority while
        tokens last. Returns True if anything is still queued."""
        self._refill()
        for prio in (AUDIO, VIDEO):
            q = self._queues[prio]
            while q:
                payload, addr = q[0]
                if prio == VIDEO and self._tokens < len(payload):
                    return True
                q.popleft()
                if prio == VIDEO:
                    self._queued_video_bytes -= len(payload)
                # audio sends even at a momentary token deficit (strict
                # priority: it is tiny and latency-critical); the deficit
                # still counts against video
                self._tokens -= len(payload)
                self.sent_packets += 1
                try:
                    self._send(payload, addr)
                except Exception as exc:
                    logger.debug("paced send failed: %r", exc)
        return bool(self._queues[AUDIO] or self._queues[VIDEO])

    async def _run(self):
        while not self._stopped:
            pending = self.drain_once()
            if pending:
                await asyncio.sleep(self.TICK_S)
            else:
                self._wake.clear()
                await self._wake.wait()
