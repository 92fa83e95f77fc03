"""CPU JPEG encoder vs an independent decoder (PIL/libjpeg)."""

import io
import math

import numpy as np
import pytest

hipflux = pytest.importorskip("hipflux")
if not hipflux.native_available():
    pytest.skip("hipflux native module not built", allow_module_level=True)

from PIL import Image


def psnr(a, b):
    mse = ((a.astype(np.int32) - b.astype(np.int32)) ** 2).mean()
    return 10 * math.log10(255 * 255 / max(mse, 1e-9))


def make_gradient(w, h):
    img = np.zeros((h, w, 4), np.uint8)
    img[:, :, 0] = np.linspace(0, 255, w, dtype=np.uint8)[None, :]
    img[:, :, 1] = np.linspace(0, 255, h, dtype=np.uint8)[:, None]
    img[:, :, 2] = 128
    img[:, :, 3] = 255
    return img


def decode(jpg):
    return np.asarray(Image.open(io.BytesIO(jpg)).convert("RGB"))


def bgrx_to_rgb(img):
    return np.stack([img[:, :, 2], img[:, :, 1], img[:, :, 0]], -1)


@pytest.mark.parametrize("fullcolor", [False, True])
def test_gradient_roundtrip(fullcolor):
    w, h = 320, 128
    img = make_gradient(w, h)
    jpg = hipflux.jpeg_encode(img.tobytes(), w, h, 90, fullcolor)
    dec = decode(jpg)
    assert dec.shape == (h, w, 3)
    p = psnr(dec, bgrx_to_rgb(img))
    assert p > 40, f"PSNR too low: {p:.1f} dB"


def test_noise_decodable():
    rng = np.random.default_rng(7)
    w, h = 256, 64
    img = rng.integers(0, 256, (h, w, 4), dtype=np.uint8)
    jpg = hipflux.jpeg_encode(img.tobytes(), w, h, 80, False)
    dec = decode(jpg)
    assert dec.shape == (h, w, 3)


def test_odd_dimensions():
    # non multiple-of-16 dims must still encode (edge replication)
    img = make_gradient(123, 45)
    jpg = hipflux.jpeg_encode(img.tobytes(), 123, 45, 85, False)
    dec = decode(jpg)
    assert dec.shape == (45, 123, 3)
    assert psnr(dec, bgrx_to_rgb(img)) > 35


def test_quality_monotonic():
    img = make_gradient(256, 64)
    sizes = [len(hipflux.jpeg_encode(img.tobytes(), 256, 64, q, False))
             for q in (30, 60, 90)]
    assert sizes[0] < sizes[1] < sizes[2]


def test_fullcolor_better_chroma():
    # saturated red/blue checkerboard: 4:4:4 should beat 4:2:0 clearly
    w, h = 128, 64
    img = np.zeros((h, w, 4), np.uint8)
    checker = (np.add.outer(np.arange(h), np.arange(w)) % 2).astype(bool)
    img[:, :, 2][checker] = 255      # red squares
    img[:, :, 0][~checker] = 255     # blue squares
    j420 = hipflux.jpeg_encode(img.tobytes(), w, h, 95, False)
    j444 = hipflux.jpeg_encode(img.tobytes(), w, h, 95, True)
    ref = bgrx_to_rgb(img)
    assert psnr(decode(j444), ref) > psnr(decode(j420), ref) + 3


def test_restart_interval_stream_decodes():
    """Restart-row framing (DRI + RSTn per MCU row — the GPU entropy
    kernel's structure) must produce streams PIL decodes identically to
    the continuous form."""
    import io
    import numpy as np
    from PIL import Image
    from hipflux import _native

    w, h = 128, 96
    rng = np.random.default_rng(17)
    img = rng.integers(0, 256, (h, w, 4), dtype=np.uint8)
    img[:, :, 3] = 255
    # run both entropy framings over the same quantized blocks via the
    # native test hook (encode once per mode through the CPU pipeline
    # equivalence: jpeg_encode is continuous; _jpeg_restart is row-reset)
    cont = _native.jpeg_encode(img.tobytes(), w, h, 80, False)
    rst = _native._jpeg_encode_restart(img.tobytes(), w, h, 80, False)
    a = np.asarray(Image.open(io.BytesIO(bytes(cont))).convert("L"),
                   dtype=np.int16)
    b = np.asarray(Image.open(io.BytesIO(bytes(rst))).convert("L"),
                   dtype=np.int16)
    assert b.shape == (h, w)
    # same quantized coefficients -> identical pixels either framing
    assert np.abs(a - b).max() <= 1
    assert b"\xff\xdd" in bytes(rst)      # DRI present
    assert b"\xff\xd0" in bytes(rst)      # first RST marker
