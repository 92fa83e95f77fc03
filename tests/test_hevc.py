"""HEVC encoder conformance vs the from-spec Python reference decoder.

Mirrors tests/test_h264.py: the decoder (tests/hevc_ref_decoder.py) parses
the full bitstream (NALs, SPS, slice headers, CABAC, transforms) and must
reproduce the encoder's reconstruction BIT-EXACTLY. The CABAC engine pair
is additionally round-trip fuzzed bin-by-bin."""

import math

import numpy as np
import pytest

hipflux = pytest.importorskip("hipflux")
if not hipflux.native_available():
    pytest.skip("hipflux native module not built", allow_module_level=True)

import hevc_ref_decoder as hrd
from hevc_ref_decoder import Decoder


def psnr(a, b):
    mse = ((a.astype(np.int64) - b.astype(np.int64)) ** 2).mean()
    return 10 * math.log10(255 * 255 / max(mse, 1e-12))


def recon_planes(enc, w, h):
    y, cb, cr, yp, cp = enc.recon()
    yh = (h + 15) & ~15
    ya = np.frombuffer(y, np.uint8).reshape(yh, yp)[:h, :w]
    cba = np.frombuffer(cb, np.uint8).reshape(yh // 2, cp)[:(h + 1) // 2,
                                                           :(w + 1) // 2]
    cra = np.frombuffer(cr, np.uint8).reshape(yh // 2, cp)[:(h + 1) // 2,
                                                           :(w + 1) // 2]
    return ya, cba, cra


def noise_frame(rng, w, h):
    img = rng.integers(0, 256, (h, w, 4), dtype=np.uint8)
    img[:, :, 3] = 255
    return img


def gradient_frame(w, h):
    img = np.zeros((h, w, 4), np.uint8)
    img[:, :, 0] = np.linspace(0, 255, w, dtype=np.uint8)[None, :]
    img[:, :, 1] = np.linspace(0, 255, h, dtype=np.uint8)[:, None]
    img[:, :, 2] = 96
    img[:, :, 3] = 255
    return img


# ---- CABAC engine pair fuzz ------------------------------------------------

def cabac_roundtrip(ops, qp):
    data, tail_bits, tail_n = hipflux._native._hevc_cabac_encode(ops, qp)
    # append tail bits + a stop bit so the decoder never runs dry
    buf = bytearray(data)
    acc = (tail_bits << (8 - tail_n)) | (1 << (7 - tail_n)) if tail_n < 8 \
        else tail_bits
    buf.append(acc & 0xFF)
    buf.append(0)
    br = hrd.BitReader(bytes(buf))
    cab = hrd.Cabac(br)
    # flat bank in the same order as native/cpu/hevc/tables.h CtxOffset
    flat = (list(hrd.INIT["split_cu"]) + list(hrd.INIT["prev_intra"])
            + list(hrd.INIT["chroma_mode"]) + list(hrd.INIT["cbf_luma"])
            + list(hrd.INIT["cbf_chroma"]) + list(hrd.INIT["last_x"])
            + list(hrd.INIT["last_y"]) + list(hrd.INIT["csbf"])
            + list(hrd.INIT["sig"]) + list(hrd.INIT["gt1"])
            + list(hrd.INIT["gt2"]))
    assert len(flat) == hevc_num_ctx()
    ctxs = [hrd.init_ctx(v, qp) for v in flat]
    out = []
    for kind, ctx, bin_ in ops:
        if kind == 0:
            out.append(cab.decision(ctxs[ctx % len(ctxs)]))
        elif kind == 1:
            out.append(cab.bypass())
        else:
            out.append(cab.terminate())
    return out


def hevc_num_ctx():
    return 123


def test_cabac_pair_fuzz():
    """Encoder bins must decode back exactly under the spec decoder for
    random context/bypass/terminate mixes."""
    rng = np.random.default_rng(7)
    for trial in range(40):
        qp = int(rng.integers(0, 52))
        n = int(rng.integers(1, 400))
        ops = []
        for _ in range(n):
            kind = int(rng.choice([0, 0, 0, 1, 1, 2],
                                  p=[.35, .2, .15, .14, .14, .02]))
            ctx = int(rng.integers(0, hevc_num_ctx()))
            # terminate bins must be 0 until the final one
            b = int(rng.integers(0, 2)) if kind != 2 else 0
            ops.append((kind, ctx, b))
        ops.append((2, 0, 1))  # final terminate = 1
        got = cabac_roundtrip(ops, qp)
        want = [b for _, _, b in ops]
        assert got == want, f"trial {trial}: CABAC desync"


# ---- full bitstream round-trips -------------------------------------------

@pytest.mark.parametrize("qp", [10, 22, 30, 40])
def test_idr_bit_exact_noise(qp):
    w, h = 128, 64
    rng = np.random.default_rng(3)
    img = noise_frame(rng, w, h)
    enc = hipflux._native.HevcEncoder(w, h)
    r = enc.encode(img.tobytes(), qp=qp)
    frames = Decoder().decode(r["data"])
    assert len(frames) == 1
    dy, dcb, dcr = frames[0]
    ey, ecb, ecr = recon_planes(enc, w, h)
    assert np.array_equal(dy, ey), "luma recon mismatch"
    assert np.array_equal(dcb, ecb), "cb recon mismatch"
    assert np.array_equal(dcr, ecr), "cr recon mismatch"


def test_idr_bit_exact_gradient():
    w, h = 96, 48
    img = gradient_frame(w, h)
    enc = hipflux._native.HevcEncoder(w, h)
    r = enc.encode(img.tobytes(), qp=26)
    frames = Decoder().decode(r["data"])
    dy, dcb, dcr = frames[0]
    ey, ecb, ecr = recon_planes(enc, w, h)
    assert np.array_equal(dy, ey)
    assert np.array_equal(dcb, ecb)
    assert np.array_equal(dcr, ecr)


def test_psnr_vs_source():
    """Recon must actually resemble the source (catches decoder+encoder
    agreeing on garbage)."""
    w, h = 128, 64
    img = gradient_frame(w, h)
    enc = hipflux._native.HevcEncoder(w, h)
    r = enc.encode(img.tobytes(), qp=22)
    dy, _, _ = Decoder().decode(r["data"])[0]
    ysrc, _, _ = hipflux._native.bgrx_to_yuv420(img.tobytes(), w, h)
    ysrc = np.frombuffer(ysrc, np.uint8).reshape(h, w)
    assert psnr(dy, ysrc) > 40, psnr(dy, ysrc)


def test_qp_monotonic_size():
    w, h = 128, 64
    rng = np.random.default_rng(11)
    img = noise_frame(rng, w, h)
    sizes = []
    for qp in (14, 26, 38, 48):
        enc = hipflux._native.HevcEncoder(w, h)
        sizes.append(len(enc.encode(img.tobytes(), qp=qp)["data"]))
    assert sizes == sorted(sizes, reverse=True), sizes


def test_odd_dimensions_cropping():
    w, h = 130, 54   # crops via conformance window
    rng = np.random.default_rng(5)
    img = noise_frame(rng, w, h)
    enc = hipflux._native.HevcEncoder(w, h)
    r = enc.encode(img.tobytes(), qp=28)
    dy, dcb, dcr = Decoder().decode(r["data"])[0]
    assert dy.shape == (h, w)
    ey, ecb, ecr = recon_planes(enc, w, h)
    assert np.array_equal(dy, ey)
    assert np.array_equal(dcb, ecb)


def test_multiple_slices_per_row():
    w, h = 160, 32
    rng = np.random.default_rng(9)
    img = noise_frame(rng, w, h)
    enc = hipflux._native.HevcEncoder(w, h, 3)   # 3 slice segments per row
    r = enc.encode(img.tobytes(), qp=30)
    dy, _, _ = Decoder().decode(r["data"])[0]
    ey, _, _ = recon_planes(enc, w, h)
    assert np.array_equal(dy, ey)
    # and the stream really contains more slice NALs
    enc1 = hipflux._native.HevcEncoder(w, h, 1)
    r1 = enc1.encode(img.tobytes(), qp=30)
    def count_idr(b):
        return sum(1 for n in hrd.split_nals(bytes(b))
                   if (n[0] >> 1) & 0x3F == 19)
    assert count_idr(r["data"]) == 3 * count_idr(r1["data"])


def test_multi_frame_stream():
    w, h = 64, 32
    rng = np.random.default_rng(21)
    enc = hipflux._native.HevcEncoder(w, h)
    blobs = b""
    recons = []
    for _ in range(3):
        img = noise_frame(rng, w, h)
        blobs += enc.encode(img.tobytes(), qp=30)["data"]
        recons.append(recon_planes(enc, w, h)[0].copy())
    frames = Decoder().decode(blobs)
    assert len(frames) == 3
    for i, (dy, _, _) in enumerate(frames):
        assert np.array_equal(dy, recons[i]), f"frame {i}"
