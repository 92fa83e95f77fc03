"""Wayland backend (phase 1): in-tree headless compositor + the
pixelflux Wayland control contract. No wayland-server
library exists in this environment, so the wire protocol is implemented
directly (wire.py) — real clients connect over WAYLAND_DISPLAY."""

from .compositor import (Compositor, ensure_wayland_display,
                         get_compositor, get_wayland_display_name,
                         shutdown_wayland_display)

__all__ = ["Compositor", "ensure_wayland_display", "get_compositor",
           "get_wayland_display_name", "shutdown_wayland_display"]
