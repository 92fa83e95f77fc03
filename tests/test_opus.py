"""Opus-framed CELT-class audio codec: RFC 6716 range-coder pair fuzz
(C++ encoder vs the from-spec Python decoder) and full codec
round-trips (PSNR on tones, bitrate obeys the knob, packets carry the
correct TOC)."""

import math

import numpy as np
import pytest

hipflux = pytest.importorskip("hipflux")
if not hipflux.native_available():
    pytest.skip("hipflux native module not built", allow_module_level=True)

from hipflux import _native
import opus_ref_decoder as ord_
from opus_ref_decoder import OpusDecoder, RangeDecoder


def test_range_coder_pair_fuzz():
    """Every op kind (bit_logp / uint / raw bits / icdf) must decode back
    exactly: the C++ encoder is the spec anchor, the Python decoder is
    the independent §4.1.4 implementation."""
    rng = np.random.default_rng(11)
    icdf4 = [200, 120, 40, 0]
    for trial in range(60):
        n = int(rng.integers(1, 300))
        ops = []
        for _ in range(n):
            kind = int(rng.integers(0, 4))
            if kind == 0:
                ops.append((0, int(rng.integers(0, 2)),
                            int(rng.integers(1, 15))))
            elif kind == 1:
                ft = int(rng.integers(2, 1 << 20))
                ops.append((1, int(rng.integers(0, ft)), ft))
            elif kind == 2:
                b = int(rng.integers(1, 25))
                ops.append((2, int(rng.integers(0, 1 << b)), b))
            else:
                ops.append((3, int(rng.integers(0, 4)), 0))
        data, err = _native._opus_range_encode(ops, 1275)
        assert not err, f"trial {trial}: encoder overflow"
        dec = RangeDecoder(data)
        for i, (kind, a, b) in enumerate(ops):
            if kind == 0:
                got = dec.dec_bit_logp(b)
                assert got == a, f"trial {trial} op {i} bit_logp"
            elif kind == 1:
                got = dec.dec_uint(b)
                assert got == a, f"trial {trial} op {i} uint({b})"
            elif kind == 2:
                got = dec.dec_bits(b)
                assert got == a, f"trial {trial} op {i} bits"
            else:
                got = dec.dec_icdf(icdf4, 8)
                assert got == a % 4, f"trial {trial} op {i} icdf"


def tone(freq, n, amp=0.5):
    t = np.arange(n) / 48000.0
    return (amp * 32767 * np.sin(2 * np.pi * freq * t)).astype(np.int16)


def encode_stream(pcm_mono, bitrate=96000):
    enc = _native.OpusEncoder(bitrate)
    pkts = []
    n = len(pcm_mono) // 960 * 960
    stereo = np.repeat(pcm_mono[:n], 2).astype(np.int16)
    for i in range(0, n, 960):
        pkts.append(enc.encode(stereo[i * 2:(i + 960) * 2].tobytes(), 2))
    return pkts


def test_packet_framing_and_bitrate():
    pcm = tone(440, 960 * 25)
    for bitrate in (48000, 96000, 192000):
        pkts = encode_stream(pcm, bitrate)
        assert all(p[0] >> 3 == 31 for p in pkts)      # CELT FB 20 ms TOC
        total_bits = sum(len(p) * 8 for p in pkts)
        achieved = total_bits / (len(pkts) * 0.02)
        assert abs(achieved - bitrate) / bitrate < 0.15, (bitrate, achieved)


def roundtrip_snr(freq, bitrate):
    n_frames = 30
    pcm = tone(freq, 960 * (n_frames + 1))
    pkts = encode_stream(pcm, bitrate)
    dec = OpusDecoder()
    for p in pkts:
        dec.decode_packet(p)
    out = dec.samples() * 32768.0
    # compare frames 4.. (skip MDCT warmup), delayed by one frame
    a = pcm[960 * 3:960 * n_frames].astype(np.float64)
    b = out[960 * 4:960 * (n_frames + 1)]
    noise = a - b
    return 10 * math.log10((a ** 2).sum() / max((noise ** 2).sum(), 1e-9))


@pytest.mark.parametrize("freq,floor", [(440, 20), (3000, 12),
                                        (7000, 14)])
def test_tone_roundtrip_snr(freq, floor):
    """Pure tones must survive the codec with real fidelity at the
    default 96 kb/s."""
    snr = roundtrip_snr(freq, 96000)
    assert snr > floor, f"{freq} Hz: SNR {snr:.1f} dB"


def test_snr_scales_with_bitrate():
    """Doubling the bitrate must buy measurable quality (the surplus
    allocation really reaches the band shapes)."""
    for freq in (3000, 7000):
        lo = roundtrip_snr(freq, 64000)
        hi = roundtrip_snr(freq, 192000)
        assert hi > lo + 5, (freq, lo, hi)
    assert roundtrip_snr(7000, 192000) > 25


def test_silence_is_cheap_energy():
    """Digital silence stays silent after decode."""
    pcm = np.zeros(960 * 6, np.int16)
    pkts = encode_stream(pcm, 96000)
    dec = OpusDecoder()
    for p in pkts:
        dec.decode_packet(p)
    assert np.abs(dec.samples()).max() < 2e-3
