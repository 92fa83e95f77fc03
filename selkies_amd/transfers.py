"""File transfer with stream-protective pacing.

Re-implements the reference transfer behaviors (SURVEY.md §2.1
stream_server.py:138-384): a token-bucket `TransferPacer` shared by all
transfers so uploads/downloads never starve the media stream, uploads that
stage to hidden siblings and rename atomically, stale-part reaping, and a
path-jailed browse/download API."""

from __future__ import annotations

import asyncio
import logging
import os
import time
from typing import Optional

logger = logging.getLogger("selkies.transfers")


class TransferPacer:
    """Token bucket shared across all concurrent transfers."""

    def __init__(self, rate_bytes_per_s: float = 12_500_000,
                 burst_bytes: Optional[float] = None):
        self.rate = rate_bytes_per_s
        self.burst = burst_bytes or rate_bytes_per_s / 4
        self._tokens = self.burst
        self._last = time.monotonic()
        self._lock = asyncio.Lock()

    def set_rate(self, rate_bytes_per_s: float):
        self.rate = max(64 * 1024, rate_bytes_per_s)

    async def consume(self, nbytes: int):
        async with self._lock:
            while True:
                now = time.monotonic()
                self._tokens = min(self.burst,
                                   self._tokens + (now - self._last) *
                                   self.rate)
                self._last = now
                if self._tokens >= nbytes:
                    self._tokens -= nbytes
                    return
                deficit = nbytes - self._tokens
                await asyncio.sleep(min(0.25, deficit / self.rate))


class UplinkAllowance:
    """Adaptive transfer budget that protects the media stream.

    Mirrors the reference's upload-pacing intent (transfers
    must never starve video): AIMD on the shared :class:`TransferPacer`
    rate, driven by the observed video send rate and the clients' frame-ACK
    RTT. When RTT inflates above its rolling floor while transfers run, the
    link is saturated -> multiplicative back-off; while RTT stays near the
    floor the allowance creeps back up additively.

    Pure-python and clock-injectable for unit tests.
    """

    RTT_INFLATE = 1.8          # x floor considered congested
    BACKOFF = 0.5
    CREEP_BYTES = 1_000_000    # +1 MB/s per healthy observation

    def __init__(self, pacer: TransferPacer,
                 floor_rate: float = 256_000.0,
                 cap_rate: float = 100_000_000.0):
        self.pacer = pacer
        self.floor_rate = floor_rate
        self.cap_rate = cap_rate
        self.rate = pacer.rate
        self._rtt_floor: Optional[float] = None

    def observe(self, video_bps: float, rtt_ms: Optional[float]) -> float:
        """Feed one sampling interval; returns the new transfer rate."""
        if rtt_ms is not None and rtt_ms > 0:
            self._rtt_floor = (rtt_ms if self._rtt_floor is None
                               else min(self._rtt_floor * 1.02, rtt_ms))
        congested = (rtt_ms is not None and self._rtt_floor is not None
                     and rtt_ms > self._rtt_floor * self.RTT_INFLATE)
        if congested:
            self.rate = max(self.floor_rate, self.rate * self.BACKOFF)
        else:
            self.rate = min(self.cap_rate, self.rate + self.CREEP_BYTES)
        # regardless of probing, never budget into the video's share:
        # assume the link is at least video + current allowance when healthy
        if video_bps > 0:
            headroom_cap = max(self.floor_rate,
                               (video_bps + self.rate) * 0.35)
            self.rate = min(self.rate, headroom_cap)
        self.pacer.set_rate(self.rate)
        return self.rate


class TransferManager:
    PART_TTL_S = 3600          # stale staging parts reaped after 1 h
    CHUNK = 256 * 1024

    def __init__(self, root: str, pacer: Optional[TransferPacer] = None,
                 allow_upload: bool = True, allow_download: bool = True):
        self.root = os.path.realpath(os.path.expanduser(root))
        self.pacer = pacer or TransferPacer()
        self.allow_upload = allow_upload
        self.allow_download = allow_download

    # ---- path jail ---------------------------------------------------------
    def resolve(self, rel: str) -> str:
        rel = (rel or "").lstrip("/")
        p = os.path.realpath(os.path.join(self.root, rel))
        if p != self.root and not p.startswith(self.root + os.sep):
            raise PermissionError(f"path escapes transfer root: {rel!r}")
        return p

    # ---- upload ------------------------------------------------------------
    async def upload(self, rel_name: str, reader) -> dict:
        """reader: async iterator of byte chunks. Stages to a hidden sibling
        and renames atomically on success."""
        if not self.allow_upload:
            raise PermissionError("uploads disabled")
        final = self.resolve(rel_name)
        os.makedirs(os.path.dirname(final), exist_ok=True)
        part = os.path.join(os.path.dirname(final),
                            "." + os.path.basename(final) + ".part")
        total = 0
        try:
            with open(part, "wb") as f:
                async for chunk in reader:
                    if not chunk:
                        continue
                    await self.pacer.consume(len(chunk))
                    f.write(chunk)
                    total += len(chunk)
            os.replace(part, final)
        except BaseException:
            try:
                os.unlink(part)
            except OSError:
                pass
            raise
        return {"name": rel_name, "bytes": total}

    # ---- download ----------------------------------------------------------
    async def stream_file(self, rel_name: str):
        if not self.allow_download:
            raise PermissionError("downloads disabled")
        path = self.resolve(rel_name)
        with open(path, "rb") as f:
            while True:
                chunk = f.read(self.CHUNK)
                if not chunk:
                    return
                await self.pacer.consume(len(chunk))
                yield chunk

    # ---- browse / maintenance ----------------------------------------------
    def listdir(self, rel: str = "") -> list[dict]:
        path = self.resolve(rel)
        out = []
        try:
            names = sorted(os.listdir(path))
        except OSError:
            return out
        for n in names:
            if n.startswith(".") and n.endswith(".part"):
                continue
            full = os.path.join(path, n)
            try:
                st = os.stat(full)
            except OSError:
                continue
            out.append({"name": n, "dir": os.path.isdir(full),
                        "size": st.st_size, "mtime": int(st.st_mtime)})
        return out

    def reap_stale_parts(self):
        now = time.time()
        for dirpath, _dirs, files in os.walk(self.root):
            for n in files:
                if n.startswith(".") and n.endswith(".part"):
                    full = os.path.join(dirpath, n)
                    try:
                        if now - os.path.getmtime(full) > self.PART_TTL_S:
                            os.unlink(full)
                    except OSError:
                        pass
