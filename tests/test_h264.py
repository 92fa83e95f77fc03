"""H.264 encoder conformance vs the from-spec Python reference decoder.

The decoder (tests/h264_ref_decoder.py) parses the full bitstream (NALs,
SPS/PPS, slice headers, CAVLC, transforms) and must reproduce the encoder's
reconstruction BIT-EXACTLY — any entropy or reconstruction mismatch
desyncs the parse or corrupts pixels."""

import math

import numpy as np
import pytest

hipflux = pytest.importorskip("hipflux")
if not hipflux.native_available():
    pytest.skip("hipflux native module not built", allow_module_level=True)

from h264_ref_decoder import Decoder


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


@pytest.mark.parametrize("qp", [10, 26, 40])
def test_idr_bit_exact(qp):
    w, h = 128, 64
    rng = np.random.default_rng(3)
    img = noise_frame(rng, w, h)
    enc = hipflux.H264Encoder(w, h)
    r = enc.encode(img.tobytes(), qp=qp, idr=True)
    frames = Decoder().decode(r["data"])
    assert len(frames) == 1
    dy, dcb, dcr = frames[0]
    ry, rcb, rcr = recon_planes(enc, w, h)
    assert np.array_equal(dy, ry), "luma mismatch decoder vs encoder recon"
    assert np.array_equal(dcb, rcb), "Cb mismatch"
    assert np.array_equal(dcr, rcr), "Cr mismatch"


def test_p_sequence_bit_exact_and_modes():
    w, h = 192, 96
    rng = np.random.default_rng(5)
    base = gradient_frame(w, h)
    enc = hipflux.H264Encoder(w, h)
    stream = b""
    recons = []
    stats = []
    for i in range(5):
        img = base.copy()
        # moving box forces inter/intra work; rest of frame should skip
        x = 16 + i * 16
        img[32:64, x:x + 32, 0] = 255
        img[32:64, x:x + 32, 2] = 0
        r = enc.encode(img.tobytes(), qp=24, idr=(i == 0))
        stream += r["data"]
        recons.append(recon_planes(enc, w, h))
        stats.append(r)
    frames = Decoder().decode(stream)
    assert len(frames) == 5
    for i, (dec, rec) in enumerate(zip(frames, recons)):
        for d, r, name in zip(dec, rec, "y cb cr".split()):
            assert np.array_equal(d, r), f"frame {i} plane {name} mismatch"
    # mode accounting: later frames should use skip for the static area
    assert stats[0]["mb_intra"] == (w // 16) * (h // 16)
    assert sum(s["mb_skip"] for s in stats[1:]) > 0


def test_psnr_vs_source():
    w, h = 256, 128
    img = gradient_frame(w, h)
    enc = hipflux.H264Encoder(w, h)
    r = enc.encode(img.tobytes(), qp=18, idr=True)
    dy, _, _ = Decoder().decode(r["data"])[0]
    # compare luma vs the CSC'd source
    y, cb, cr = hipflux.bgrx_to_yuv420(img.tobytes(), w, h)
    sy = np.frombuffer(y, np.uint8).reshape(h, w)
    p = psnr(dy, sy)
    assert p > 42, f"IDR luma PSNR {p:.1f} too low at QP18"


def test_qp_monotonic_size():
    w, h = 256, 128
    rng = np.random.default_rng(11)
    img = noise_frame(rng, w, h)
    sizes = []
    for qp in (12, 26, 40):
        enc = hipflux.H264Encoder(w, h)
        sizes.append(len(enc.encode(img.tobytes(), qp=qp, idr=True)["data"]))
    assert sizes[0] > sizes[1] > sizes[2]


def test_odd_dimensions_cropping():
    w, h = 100, 52
    rng = np.random.default_rng(13)
    img = noise_frame(rng, w, h)
    enc = hipflux.H264Encoder(w, h)
    r = enc.encode(img.tobytes(), qp=30, idr=True)
    dy, dcb, dcr = Decoder().decode(r["data"])[0]
    assert dy.shape == (52, 100)
    ry, rcb, rcr = recon_planes(enc, w, h)
    assert np.array_equal(dy, ry)
    assert np.array_equal(dcb, rcb)


def test_static_content_all_skip():
    w, h = 128, 64
    img = gradient_frame(w, h)
    enc = hipflux.H264Encoder(w, h)
    enc.encode(img.tobytes(), qp=24, idr=True)
    r2 = enc.encode(img.tobytes(), qp=24)
    nmb = (w // 16) * (h // 16)
    assert r2["mb_skip"] == nmb, f"static frame: {r2['mb_skip']}/{nmb} skip"
    assert len(r2["data"]) < 60  # a few bytes of slice headers only


def test_inter_residuals_bit_exact_and_efficient():
    """A global small brightness step is too large to skip but an excellent
    MC candidate: every MB must be coded P_L0_16x16 WITH residual (not
    intra), the decode must match the encoder recon bit-exactly, and the
    P frame must be far smaller than an intra refresh of the same change."""
    w, h = 192, 96
    rng = np.random.default_rng(21)
    base = noise_frame(rng, w, h)
    lifted = base.copy()
    lifted[:, :, :3] = np.clip(base[:, :, :3].astype(np.int16) + 6,
                               0, 255).astype(np.uint8)
    enc = hipflux.H264Encoder(w, h)
    r0 = enc.encode(base.tobytes(), qp=28, idr=True)
    r1 = enc.encode(lifted.tobytes(), qp=28)
    rec1 = recon_planes(enc, w, h)
    nmb = (w // 16) * (h // 16)
    assert r1["mb_inter"] == nmb, \
        f"expected all-inter: {r1['mb_inter']}/{nmb} (skip {r1['mb_skip']}," \
        f" intra {r1['mb_intra']})"
    frames = Decoder().decode(r0["data"] + r1["data"])
    assert len(frames) == 2
    for d, r, name in zip(frames[1], rec1, "y cb cr".split()):
        assert np.array_equal(d, r), f"P resid frame plane {name} mismatch"
    # the inter-coded step costs a small fraction of the IDR
    assert len(r1["data"]) < len(r0["data"]) * 0.35, \
        f"P {len(r1['data'])} vs IDR {len(r0['data'])}"
    # quality must track the IDR's (both bounded by the QP28 quant noise;
    # without coded residuals the +6 step alone would cap PSNR at ~32.5)
    y0, _, _ = hipflux.bgrx_to_yuv420(base.tobytes(), w, h)
    y1, _, _ = hipflux.bgrx_to_yuv420(lifted.tobytes(), w, h)
    p_idr = psnr(frames[0][0], np.frombuffer(y0, np.uint8).reshape(h, w))
    p_p = psnr(frames[1][0], np.frombuffer(y1, np.uint8).reshape(h, w))
    assert p_p > p_idr - 2.0, \
        f"P-frame luma PSNR {p_p:.1f} far below IDR {p_idr:.1f}"


def test_inter_residual_moving_content_chain():
    """Rolling noise: MC aligns perfectly after the shift, small residuals
    at block seams. Chain of 6 P frames must stay bit-exact vs recon."""
    w, h = 160, 96
    rng = np.random.default_rng(23)
    base = noise_frame(rng, w, h)
    enc = hipflux.H264Encoder(w, h)
    stream = b""
    recons = []
    inter_seen = 0
    for i in range(6):
        img = np.roll(base, 2 * i, axis=1)
        # add a small local perturbation so MC is imperfect
        img[40:48, 40:48, :3] = np.clip(
            img[40:48, 40:48, :3].astype(np.int16) + 12, 0, 255
        ).astype(np.uint8)
        r = enc.encode(img.tobytes(), qp=26, idr=(i == 0))
        stream += r["data"]
        recons.append(recon_planes(enc, w, h))
        if i:
            inter_seen += r["mb_inter"]
    assert inter_seen > 0
    frames = Decoder().decode(stream)
    assert len(frames) == 6
    for i, (dec, rec) in enumerate(zip(frames, recons)):
        for d, r, name in zip(dec, rec, "y cb cr".split()):
            assert np.array_equal(d, r), f"frame {i} plane {name} mismatch"


def test_halfpel_and_odd_integer_mvs():
    """Shift by 1 px (odd integer -> chroma bilinear) and by a half pixel
    (box-blur of adjacent columns -> luma 6-tap): both must stay bit-exact
    decoder-vs-recon, and the parsed MVs must show the sub-grid in use."""
    from h264_ref_decoder import Decoder
    w, h = 192, 96
    rng = np.random.default_rng(31)
    base = rng.integers(0, 256, (h, w, 4), dtype=np.uint8)
    # smooth it so sub-pel interpolation is a good predictor
    f = base.astype(np.float32)
    for _ in range(2):
        f = (f + np.roll(f, 1, 1) + np.roll(f, -1, 1) +
             np.roll(f, 1, 0) + np.roll(f, -1, 0)) / 5
    f0 = f.astype(np.uint8)
    f0[:, :, 3] = 255
    f1 = np.roll(f0, 1, axis=1)                       # 1 px right
    fh = ((f0.astype(np.uint16) + np.roll(f0, 1, 1)) // 2).astype(np.uint8)
    fh[:, :, 3] = 255                                  # ~half-pel right

    class MvDecoder(Decoder):
        def __init__(self):
            super().__init__()
            self.mvs = []

        def decode_p16(self, br, mbx, mby, ctx, qp):
            info = super().decode_p16(br, mbx, mby, ctx, qp)
            self.mvs.append(tuple(ctx["left_mv"]))
            return info

    enc = hipflux.H264Encoder(w, h)
    stream = b""
    recons = []
    for img in (f0, f1, fh):
        r = enc.encode(np.ascontiguousarray(img).tobytes(), qp=22,
                       idr=(img is f0))
        stream += r["data"]
        recons.append(recon_planes(enc, w, h))
    d = MvDecoder()
    frames = d.decode(stream)
    assert len(frames) == 3
    for i, (dec, rec) in enumerate(zip(frames, recons)):
        for a, b, name in zip(dec, rec, "y cb cr".split()):
            assert np.array_equal(a, b), f"frame {i} plane {name} mismatch"
    odd = [mv for mv in d.mvs if (mv[0] // 4) % 2 == 1 and mv[0] % 4 == 0]
    half = [mv for mv in d.mvs if mv[0] % 4 == 2 or mv[1] % 4 == 2]
    assert odd, "no odd-integer MVs coded for the 1-px shift"
    assert half, "no half-pel MVs coded for the half-shifted frame"


def test_quarterpel_mvs():
    """A ~quarter-pixel shift (3:1 blend of adjacent columns) must produce
    quarter-pel MVs (mv & 3 odd) and stay bit-exact decoder-vs-recon."""
    from h264_ref_decoder import Decoder
    w, h = 192, 96
    rng = np.random.default_rng(41)
    base = rng.integers(0, 256, (h, w, 4), dtype=np.uint8)
    f = base.astype(np.float32)
    for _ in range(2):
        f = (f + np.roll(f, 1, 1) + np.roll(f, -1, 1) +
             np.roll(f, 1, 0) + np.roll(f, -1, 0)) / 5
    f0 = f.astype(np.uint8)
    f0[:, :, 3] = 255
    fq = ((3 * f0.astype(np.uint16) + np.roll(f0, 1, 1)) // 4).astype(
        np.uint8)
    fq[:, :, 3] = 255

    class MvDecoder(Decoder):
        def __init__(self):
            super().__init__()
            self.mvs = []

        def decode_p16(self, br, mbx, mby, ctx, qp):
            info = super().decode_p16(br, mbx, mby, ctx, qp)
            self.mvs.append(tuple(ctx["left_mv"]))
            return info

    enc = hipflux.H264Encoder(w, h)
    stream = b""
    recons = []
    for img in (f0, fq):
        r = enc.encode(np.ascontiguousarray(img).tobytes(), qp=22,
                       idr=(img is f0))
        stream += r["data"]
        recons.append(recon_planes(enc, w, h))
    d = MvDecoder()
    frames = d.decode(stream)
    assert len(frames) == 2
    for i, (dec, rec) in enumerate(zip(frames, recons)):
        for a, b, name in zip(dec, rec, "y cb cr".split()):
            assert np.array_equal(a, b), f"frame {i} plane {name} mismatch"
    quarter = [mv for mv in d.mvs if (mv[0] & 3) in (1, 3)
               or (mv[1] & 3) in (1, 3)]
    assert quarter, "no quarter-pel MVs coded for the quarter shift"



def test_fast_scroll_motion_lock():
    """12 px/frame scroll (beyond the ±8 fine grid): the ±16 coarse
    acquisition finds it and the temporal predictor/hint keeps tracking;
    P frames must be mostly inter with mv x = -48 quarter-pel, stay
    bit-exact, and cost far less than intra refreshes. (Acquisition range
    is ±16 like the reference x264 default; a resolution pyramid for
    faster motion is round-2 — profiles/NOTES.md.)"""
    from h264_ref_decoder import Decoder
    w, h, n = 320, 96, 4
    rng = np.random.default_rng(61)
    base = rng.integers(0, 256, (h, w, 4), dtype=np.uint8).astype(
        np.float32)
    for _ in range(2):
        base = (base + np.roll(base, 1, 1) + np.roll(base, -1, 1) +
                np.roll(base, 1, 0) + np.roll(base, -1, 0)) / 5
    f0 = base.astype(np.uint8)
    f0[:, :, 3] = 255

    class MvDecoder(Decoder):
        def __init__(self):
            super().__init__()
            self.mvs = []

        def decode_p16(self, br, mbx, mby, ctx, qp):
            info = super().decode_p16(br, mbx, mby, ctx, qp)
            self.mvs.append(tuple(ctx["left_mv"]))
            return info

    enc = hipflux.H264Encoder(w, h)
    stream = b""
    sizes = []
    recons = []
    for i in range(n):
        img = np.ascontiguousarray(np.roll(f0, 12 * i, axis=1))
        r = enc.encode(img.tobytes(), qp=24, idr=(i == 0))
        stream += r["data"]
        sizes.append(len(r["data"]))
        recons.append(recon_planes(enc, w, h))
    d = MvDecoder()
    frames = d.decode(stream)
    assert len(frames) == n
    for i, (dec, rec) in enumerate(zip(frames, recons)):
        for a, b, name in zip(dec, rec, "y cb cr".split()):
            assert np.array_equal(a, b), f"frame {i} plane {name} mismatch"
    locked = [mv for mv in d.mvs if mv[0] == -48]
    assert len(locked) > len(d.mvs) * 0.4, \
        f"scroll not locked: {len(locked)}/{len(d.mvs)} at -48"
    assert max(sizes[2:]) < sizes[0] * 0.6, f"P frames too large: {sizes}"


def test_deblock_changes_recon_and_signals_idc2():
    """The in-loop deblocking filter must actually fire (recon differs
    from the unfiltered encode) and be signaled via
    disable_deblocking_filter_idc = 2; both variants stay bit-exact
    under the from-spec decoder."""
    w, h = 128, 64
    img = gradient_frame(w, h)
    # blocky content: coarse quantization creates block-edge steps
    on = hipflux.H264Encoder(w, h)
    off = hipflux.H264Encoder(w, h, deblock=False)
    r_on = on.encode(img.tobytes(), qp=40, idr=True)
    r_off = off.encode(img.tobytes(), qp=40, idr=True)
    y_on, _, _ = recon_planes(on, w, h)
    y_off, _, _ = recon_planes(off, w, h)
    assert not np.array_equal(y_on, y_off), "deblock had no effect"
    # decoder agrees with both (parses idc and filters only when 2)
    dy_on = Decoder().decode(r_on["data"])[0][0]
    dy_off = Decoder().decode(r_off["data"])[0][0]
    assert np.array_equal(dy_on, y_on)
    assert np.array_equal(dy_off, y_off)


def test_deblock_p_frame_chain_bit_exact():
    """Deblocked recon feeds the reference chain: a multi-frame P
    sequence with motion must stay bit-exact (decoder MC reads the
    deblocked reference exactly like the encoder)."""
    w, h = 128, 64
    rng = np.random.default_rng(17)
    base = rng.integers(0, 256, (h + 32, w, 4), dtype=np.uint8)
    enc = hipflux.H264Encoder(w, h)
    stream = b""
    recons = []
    for i in range(4):
        img = np.ascontiguousarray(base[i * 8:i * 8 + h])
        r = enc.encode(img.tobytes(), qp=32, idr=(i == 0))
        stream += r["data"]
        recons.append(recon_planes(enc, w, h)[0].copy())
    frames = Decoder().decode(stream)
    assert len(frames) == 4
    for i, (dy, _, _) in enumerate(frames):
        assert np.array_equal(dy, recons[i]), f"frame {i} recon mismatch"


# ---- Hi444 fullcolor (separate colour planes) -----------------------------

def recon_planes_444(enc, w, h):
    y, cb, cr, yp, cp = enc.recon()
    yh = (h + 15) & ~15
    ya = np.frombuffer(y, np.uint8).reshape(yh, yp)[:h, :w]
    cba = np.frombuffer(cb, np.uint8).reshape(yh, cp)[:h, :w]
    cra = np.frombuffer(cr, np.uint8).reshape(yh, cp)[:h, :w]
    return ya, cba, cra


@pytest.mark.parametrize("qp", [10, 26, 40])
def test_fullcolor_idr_bit_exact(qp):
    """Hi444 separate-colour-plane IDR: the from-spec decoder reproduces
    all three full-resolution planes bit-exactly (profile 244,
    colour_plane_id slices; reference pixelflux h264_fullcolor mode)."""
    w, h = 120, 52          # odd MB alignment exercises CropUnit=1
    rng = np.random.default_rng(11)
    img = noise_frame(rng, w, h)
    enc = hipflux.H264Encoder(w, h, fullcolor=True)
    r = enc.encode(img.tobytes(), qp=qp, idr=True)
    frames = Decoder().decode(r["data"])
    assert len(frames) == 1
    dy, dcb, dcr = frames[0]
    assert dy.shape == (h, w) and dcb.shape == (h, w)
    ry, rcb, rcr = recon_planes_444(enc, w, h)
    assert np.array_equal(dy, ry)
    assert np.array_equal(dcb, rcb)
    assert np.array_equal(dcr, rcr)


def test_fullcolor_p_chain_bit_exact():
    """Moving content over a fullcolor IDR+P chain stays bit-exact (skip,
    zero-residual inter, and the mono I16x16 residual fallback)."""
    w, h = 96, 64
    rng = np.random.default_rng(12)
    base = gradient_frame(w, h)
    enc = hipflux.H264Encoder(w, h, fullcolor=True)
    stream = b""
    for i in range(4):
        img = np.roll(base, i * 6, axis=1).copy()
        img[40:48, 40:48, :3] = rng.integers(0, 256, (8, 8, 3))
        r = enc.encode(img.tobytes(), qp=24, idr=(i == 0))
        stream += r["data"]
    frames = Decoder().decode(stream)
    assert len(frames) == 4
    dy, dcb, dcr = frames[-1]
    ry, rcb, rcr = recon_planes_444(enc, w, h)
    assert np.array_equal(dy, ry)
    assert np.array_equal(dcb, rcb)
    assert np.array_equal(dcr, rcr)


def test_fullcolor_chroma_fidelity_beats_420():
    """The point of 4:4:4: saturated single-pixel chroma detail (colored
    text) survives; 4:2:0 subsampling blurs it."""
    w, h = 64, 64
    img = np.zeros((h, w, 4), np.uint8)
    img[:, :, 3] = 255
    img[:, ::2, 2] = 255          # alternating red columns
    img[:, 1::2, 0] = 255         # and blue columns
    e444 = hipflux.H264Encoder(w, h, fullcolor=True)
    e420 = hipflux.H264Encoder(w, h)
    r444 = e444.encode(img.tobytes(), qp=12, idr=True)
    r420 = e420.encode(img.tobytes(), qp=12, idr=True)
    _, cb444, _ = Decoder().decode(r444["data"])[0]
    _, cb420, _ = Decoder().decode(r420["data"])[0]
    # 444 keeps the per-column chroma alternation; 420 averages it away
    col_swing_444 = np.abs(cb444[:, ::2].astype(int).mean() -
                           cb444[:, 1::2].astype(int).mean())
    up420 = np.repeat(cb420, 2, axis=1)[:, :w]
    col_swing_420 = np.abs(up420[:, ::2].astype(int).mean() -
                           up420[:, 1::2].astype(int).mean())
    assert col_swing_444 > 100
    assert col_swing_420 < 10
