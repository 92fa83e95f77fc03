"""G.711 audio for the WebRTC audio m-line.

The browsers' mandatory-to-implement audio codecs are Opus and G.711
(PCMU/PCMA, static payload types 0/8, RFC 3551). There is no Opus encoder
in this image, so the WebRTC audio sender uses G.711 µ-law: trivially
implementable from the ITU definition, verified in tests against the
stdlib ``audioop`` reference, and decodable by every browser.

Pipeline: 48 kHz s16 interleaved capture -> mono mix -> 6:1 decimation
(box filter) to 8 kHz -> µ-law. 20 ms frames = 160 output samples.

Reference parity note: the reference sends Opus over WebRTC
(selkies webrtc_mode.py); G.711 is the dependency-free stand-in — the
codec stage is pluggable when an Opus encoder is available.
"""

from __future__ import annotations

import numpy as np

# ITU G.711 14-bit-domain constants (the audioop/spandsp formulation)
_BIAS14 = 0x84 >> 2          # 33
_CLIP14 = 8159
_SEG_UEND = np.array([0x3F, 0x7F, 0xFF, 0x1FF, 0x3FF, 0x7FF, 0xFFF,
                      0x1FFF], np.int32)


def ulaw_encode(pcm: np.ndarray) -> bytes:
    """s16 numpy array -> µ-law bytes (exact integer port of the ITU
    segment-search encoder; bit-identical to stdlib audioop)."""
    x = pcm.astype(np.int32) >> 2          # 14-bit domain
    mask = np.where(x < 0, 0x7F, 0xFF)
    mag = np.where(x < 0, -x, x)
    mag = np.minimum(mag, _CLIP14) + _BIAS14
    seg = np.searchsorted(_SEG_UEND, mag, side="left").astype(np.int32)
    in_range = seg < 8
    seg_c = np.minimum(seg, 7)
    uval = (seg_c << 4) | ((mag >> (seg_c + 1)) & 0xF)
    u = np.where(in_range, uval ^ mask, 0x7F ^ mask) & 0xFF
    return u.astype(np.uint8).tobytes()


def ulaw_decode(data: bytes) -> np.ndarray:
    """µ-law bytes -> s16 numpy array (for tests/loopback)."""
    u = (~np.frombuffer(data, np.uint8).astype(np.int32)) & 0xFF
    t = ((u & 0x0F) << 3) + 0x84
    t = t << ((u & 0x70) >> 4)
    v = np.where(u & 0x80, 0x84 - t, t - 0x84)
    return v.clip(-32768, 32767).astype(np.int16)


def downmix_8k(pcm_s16: bytes, channels: int, rate: int = 48000
               ) -> np.ndarray:
    """Interleaved s16 -> mono 8 kHz (box-filter decimation)."""
    a = np.frombuffer(pcm_s16, np.int16)
    if channels > 1:
        n = (len(a) // channels) * channels
        a = a[:n].reshape(-1, channels).mean(axis=1)
    factor = max(1, rate // 8000)
    n = (len(a) // factor) * factor
    if n == 0:
        return np.zeros(0, np.int16)
    return (a[:n].reshape(-1, factor).mean(axis=1)
            .round().clip(-32768, 32767).astype(np.int16))


def wire_frame_to_ulaw(frame: bytes, channels: int,
                       rate: int = 48000) -> bytes:
    """hipflux audio wire frame ([0x01, n_red] + redundant + primary PCM)
    -> µ-law 8 kHz mono payload of the PRIMARY frame only."""
    if len(frame) < 2 or frame[0] != 0x01:
        return b""
    n_red = frame[1]
    off = 2
    for _ in range(n_red):
        if off + 2 > len(frame):
            return b""
        ln = int.from_bytes(frame[off:off + 2], "little")
        off += 2 + ln
    pcm = frame[off:]
    return ulaw_encode(downmix_8k(pcm, channels, rate))
