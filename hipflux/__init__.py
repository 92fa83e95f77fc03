"""hipflux — MI355X-native capture + encode engine (python wrapper).

The native module implements the capture-engine API contract the reference
control plane consumes from pixelflux: `CaptureSettings`,
`ScreenCapture.start_capture(cb, settings)` / `stop_capture` /
`is_capturing`, `request_idr_frame`, live tunables, striped JPEG/H.264
encoders emitting the 0x03/0x04 wire format.

On a machine with an AMD GPU the HIP (gfx950) pipeline is mandatory: if the
native extension is missing we raise instead of silently falling back, so a
mis-built deployment can't masquerade as GPU-accelerated.
"""

from __future__ import annotations

import os
from dataclasses import dataclass

try:
    from hipflux._native import (  # noqa: F401
        CaptureSettings,
        H264Encoder,
        ScreenCapture,
        bgrx_to_yuv420,
        hip_device_count,
        jpeg_encode,
    )
    _NATIVE_OK = True
    _NATIVE_ERR = None
except ImportError as exc:  # pragma: no cover - exercised only when unbuilt
    _NATIVE_OK = False
    _NATIVE_ERR = exc

    def _gpu_present() -> bool:
        try:
            return len(os.listdir("/sys/class/kfd/kfd/topology/nodes")) > 1
        except OSError:
            return False

    if _gpu_present():
        raise ImportError(
            "hipflux native extension is not built but this host has an AMD "
            "GPU. Build it with `make -C native` (gfx950). Refusing to fall "
            f"back silently. Original error: {exc}"
        ) from exc


def native_available() -> bool:
    return _NATIVE_OK


@dataclass
class Stripe:
    """Decoded form of the stripe callback arguments."""
    data: bytes
    frame_id: int
    y: int
    width: int
    height: int
    is_keyframe: bool
    capture_ts_ms: float
    encode_done_ms: float
    stripe_type: int  # 0x03 JPEG, 0x04 H.264


def make_stripe(*args) -> Stripe:
    return Stripe(*args)

# ---- Wayland surface (pixelflux module-function contract; SURVEY.md
# §2.3: ensure_wayland_display / get_wayland_display_name). The
# compositor is pure control-plane Python (selkies_amd.wayland).
def ensure_wayland_display(name="selkies-wl-0", width=1920, height=1080):
    from selkies_amd.wayland import ensure_wayland_display as _e
    return _e(name, width, height)


def get_wayland_display_name():
    from selkies_amd.wayland import get_wayland_display_name as _g
    return _g()


def probe_wayland_gpu():
    """Headless compositor needs no GBM/EGL: software path always works;
    report the GPU count so entrypoints can prefer the HIP encoder."""
    try:
        return {"ok": True, "gpus": hip_device_count()}
    except Exception:
        return {"ok": True, "gpus": 0}
