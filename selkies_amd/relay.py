"""Per-client bounded video relay with reference-chain repair.

Fresh implementation of the reference `_VideoRelay` contract
(reference selkies.py:733-890 semantics; byte budget + keyframe exemption
:108-127; 1 s stalled-send bound :103-107; per-row IDR gating after drops
:750-765):

* Each connected client gets one relay task draining a bounded backlog.
* Backlog is bounded by BYTES (a time budget of the stream bitrate), not
  item count; keyframe stripes are exempt from the budget so recovery
  data is never the thing dropped.
* When a non-key stripe for stripe-row y is dropped, that row's reference
  chain is broken: subsequent delta stripes for y are discarded until a
  keyframe stripe for y arrives, and an IDR is requested upstream once
  (collapsed across rows/clients by the engine's request flag).
* A send that stalls longer than `stall_timeout_s` marks the client dead;
  the owner closes the socket.
"""

from __future__ import annotations

import asyncio
import logging
import time
from typing import Awaitable, Callable, Optional

logger = logging.getLogger("selkies.relay")

DEFAULT_BUDGET_SECONDS = 2.0
DEFAULT_STALL_TIMEOUT = 1.0


class VideoRelay:
    def __init__(self,
                 send: Callable[[bytes], Awaitable[None]],
                 request_idr: Callable[[], None],
                 bitrate_bps: float = 16_000_000.0,
                 budget_seconds: float = DEFAULT_BUDGET_SECONDS,
                 stall_timeout_s: float = DEFAULT_STALL_TIMEOUT,
                 on_dead: Optional[Callable[[], None]] = None):
        self._send = send
        self._request_idr = request_idr
        self._budget_seconds = budget_seconds
        self._stall_timeout_s = stall_timeout_s
        self._on_dead = on_dead
        self.set_bitrate(bitrate_bps)

        self._backlog: asyncio.Queue = asyncio.Queue()
        self._backlog_bytes = 0
        self._broken_rows: set[int] = set()
        self._task: Optional[asyncio.Task] = None
        self.dead = False
        # stats
        self.sent_frames = 0
        self.sent_bytes = 0
        self.dropped_frames = 0

    def set_bitrate(self, bitrate_bps: float) -> None:
        self._byte_budget = max(256 * 1024,
                                int(bitrate_bps / 8 * self._budget_seconds))

    def start(self) -> None:
        if self._task is None:
            self._task = asyncio.get_running_loop().create_task(self._run())

    async def stop(self) -> None:
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except (asyncio.CancelledError, Exception):
                pass
            self._task = None

    # -- producer side (called on the event loop) --------------------------
    def offer(self, payload: bytes, y: int, is_keyframe: bool) -> bool:
        """Queue one wire-ready stripe. Returns False if dropped."""
        if self.dead:
            return False
        if not is_keyframe:
            if y in self._broken_rows:
                self.dropped_frames += 1
                return False
            if self._backlog_bytes + len(payload) > self._byte_budget:
                # over budget: drop this delta stripe, break the row chain
                self._broken_rows.add(y)
                self.dropped_frames += 1
                self._request_idr()
                return False
        else:
            self._broken_rows.discard(y)
        self._backlog_bytes += len(payload)
        self._backlog.put_nowait((payload, y))
        return True

    # -- consumer task ------------------------------------------------------
    async def _run(self) -> None:
        while True:
            payload, y = await self._backlog.get()
            self._backlog_bytes -= len(payload)
            t0 = time.monotonic()
            try:
                await asyncio.wait_for(self._send(payload), self._stall_timeout_s)
            except asyncio.TimeoutError:
                logger.warning("relay send stalled > %.1fs; marking client dead",
                               self._stall_timeout_s)
                self._mark_dead()
                return
            except asyncio.CancelledError:
                raise
            except Exception as exc:
                logger.debug("relay send failed: %r", exc)
                self._mark_dead()
                return
            self.sent_frames += 1
            self.sent_bytes += len(payload)
            # long (but sub-timeout) sends show congestion; let backpressure
            # logic observe it via stats rather than acting here
            _ = t0

    def _mark_dead(self) -> None:
        self.dead = True
        if self._on_dead:
            try:
                self._on_dead()
            except Exception:
                pass

    @property
    def backlog_bytes(self) -> int:
        return self._backlog_bytes
