"""Display plumbing: resolution management, DPI, layout math.

Re-implements the reference display layer's behavioral core (SURVEY.md §2.1
display_utils.py: resize_display/ensure_mode/align_dims_16/compute_dual_
layout/set_dpi) for X11 via the xrandr/xrdb CLIs (subprocess — the
reference's own fallback path, display_utils.py:12-15), with the pure
layout math factored out so it is testable without a display.
"""

from __future__ import annotations

import logging
import os
import re
import shutil
import subprocess
from dataclasses import dataclass
from typing import Optional

logger = logging.getLogger("selkies.display")


def align_dims_16(w: int, h: int) -> tuple[int, int]:
    """Round dimensions down to the 16-px macroblock grid (min 16).
    (Reference align_dims_16, display_utils.py:559.)"""
    return max(16, w & ~15), max(16, h & ~15)


def cvt_modeline(w: int, h: int, refresh: float = 60.0) -> tuple[str, str]:
    """Compute a CVT reduced-blanking modeline (VESA CVT 1.2 RB) for
    arbitrary resolutions — what `cvt -r` would print, computed natively so
    no external tool is needed."""
    # CVT-RB: hbp=80, hsync=32, hfp=48 -> hblank = 160; vblank rules below
    hblank = 160
    htotal = w + hblank
    vbi = 460.0 / 1000.0  # min vblank interval us -> 460us
    # lines: vfp=3, vsync(by aspect, use 4), vbp>=6
    vsync = 4
    min_vblank_lines = 3 + vsync + 6
    # iterate: vtotal from required vblank time
    period_est = (1.0 / refresh)
    hperiod_est = (period_est - vbi * 1e-3 * 0) / (h + min_vblank_lines)
    vblank_lines = max(min_vblank_lines, int(vbi * 1e-3 / hperiod_est) + 1)
    vtotal = h + vblank_lines
    pclk = htotal * vtotal * refresh / 1e6  # MHz
    pclk = round(pclk * 4) / 4  # quantize to 0.25MHz
    name = f"{w}x{h}_{refresh:.2f}"
    mode = (f"{pclk:.2f} {w} {w + 48} {w + 48 + 32} {htotal} "
            f"{h} {h + 3} {h + 3 + vsync} {vtotal} +hsync -vsync")
    return name, mode


@dataclass
class Monitor:
    name: str
    x: int
    y: int
    width: int
    height: int
    primary: bool = False


def compute_dual_layout(w1: int, h1: int, w2: int, h2: int,
                        position: str = "right") -> list[Monitor]:
    """Logical monitor layout for the extended (second) display
    (reference compute_dual_layout, display_utils.py:447)."""
    w1, h1 = align_dims_16(w1, h1)
    w2, h2 = align_dims_16(w2, h2)
    if position == "right":
        return [Monitor("primary", 0, 0, w1, h1, True),
                Monitor("display2", w1, 0, w2, h2)]
    if position == "left":
        return [Monitor("primary", w2, 0, w1, h1, True),
                Monitor("display2", 0, 0, w2, h2)]
    if position == "below":
        return [Monitor("primary", 0, 0, w1, h1, True),
                Monitor("display2", 0, h1, w2, h2)]
    return [Monitor("primary", 0, 0, w1, h1, True),
            Monitor("display2", 0, -h2, w2, h2)]


def framebuffer_bounds(monitors: list[Monitor]) -> tuple[int, int]:
    """Total framebuffer size enclosing all monitors (after normalizing
    negative origins)."""
    min_x = min(m.x for m in monitors)
    min_y = min(m.y for m in monitors)
    for m in monitors:
        m.x -= min_x
        m.y -= min_y
    return (max(m.x + m.width for m in monitors),
            max(m.y + m.height for m in monitors))


# ---- xrandr-backed operations (graceful when no display) -------------------

def _run(cmd: list[str], display: Optional[str]) -> Optional[str]:
    env = dict(os.environ)
    if display:
        env["DISPLAY"] = display
    try:
        out = subprocess.run(cmd, capture_output=True, text=True, env=env,
                             timeout=10)
    except (OSError, subprocess.TimeoutExpired) as exc:
        logger.debug("%s failed: %r", cmd[0], exc)
        return None
    if out.returncode != 0:
        logger.debug("%s rc=%d: %s", cmd[0], out.returncode,
                     out.stderr.strip()[:200])
        return None
    return out.stdout


def have_xrandr() -> bool:
    return shutil.which("xrandr") is not None


def list_outputs(display: Optional[str] = None) -> list[dict]:
    out = _run(["xrandr", "--query"], display)
    if out is None:
        return []
    outputs = []
    for line in out.splitlines():
        m = re.match(r"^(\S+) (connected|disconnected)(?: primary)?"
                     r"(?: (\d+)x(\d+)\+(\d+)\+(\d+))?", line)
        if m:
            outputs.append({
                "name": m.group(1),
                "connected": m.group(2) == "connected",
                "width": int(m.group(3)) if m.group(3) else 0,
                "height": int(m.group(4)) if m.group(4) else 0,
                "x": int(m.group(5)) if m.group(5) else 0,
                "y": int(m.group(6)) if m.group(6) else 0,
            })
    return outputs


def ensure_mode(w: int, h: int, refresh: float = 60.0,
                output: Optional[str] = None,
                display: Optional[str] = None) -> Optional[str]:
    """Create (if needed) and return the name of a mode WxH@refresh
    (reference ensure_mode, display_utils.py:295)."""
    name, modeline = cvt_modeline(w, h, refresh)
    q = _run(["xrandr", "--query"], display) or ""
    if name not in q:
        if _run(["xrandr", "--newmode", name] + modeline.split(),
                display) is None:
            return None
    if output:
        _run(["xrandr", "--addmode", output, name], display)
    return name


def resize_display(w: int, h: int, display: Optional[str] = None,
                   output: Optional[str] = None) -> bool:
    """Resize the (first connected) output to WxH, creating the mode when
    missing (reference resize_display, display_utils.py:1295)."""
    w, h = align_dims_16(w, h)
    if not have_xrandr():
        return False
    if output is None:
        outs = [o for o in list_outputs(display) if o["connected"]]
        if not outs:
            # headless Xvfb-style: resize the fb directly
            return _run(["xrandr", "--fb", f"{w}x{h}"], display) is not None
        output = outs[0]["name"]
    mode = ensure_mode(w, h, 60.0, output, display)
    if mode is None:
        return False
    return _run(["xrandr", "--output", output, "--mode", mode],
                display) is not None


def set_dpi(dpi: int, display: Optional[str] = None) -> bool:
    """Apply DPI via the xrdb Xft.dpi ladder (reference set_dpi,
    display_utils.py:1997)."""
    if shutil.which("xrdb") is None:
        return False
    env = dict(os.environ)
    if display:
        env["DISPLAY"] = display
    try:
        p = subprocess.run(["xrdb", "-merge"], input=f"Xft.dpi: {dpi}\n",
                           text=True, capture_output=True, env=env,
                           timeout=10)
        return p.returncode == 0
    except (OSError, subprocess.TimeoutExpired):
        return False


def cursor_size_for_dpi(dpi: float, base_size: int = 24) -> int:
    """Scale the cursor theme size with DPI (reference
    cursor_size_for_dpi, display_utils.py:553)."""
    return max(1, int(round(float(dpi) / 96.0 * base_size)))


def monitor_geometry(w: int, h: int, x: int, y: int,
                     dpi: float = 96.0) -> str:
    """RandR --setmonitor geometry `w/mmWxh/mmH+x+y` with physical mm
    derived from the session DPI (reference builds mm the same way,
    display_utils.py:352-358)."""
    mm_w = max(1, round(w * 25.4 / dpi))
    mm_h = max(1, round(h * 25.4 / dpi))
    return f"{w}/{mm_w}x{h}/{mm_h}+{x}+{y}"


def set_logical_monitor(name: str, w: int, h: int, x: int, y: int,
                        output: str = "none", dpi: float = 96.0,
                        display: Optional[str] = None) -> bool:
    """Create/replace a RandR 1.5 logical monitor (reference
    display_utils.py:1001-1010: xrandr --setmonitor fallback path). Used
    to present dual logical displays on a single large framebuffer."""
    geom = monitor_geometry(w, h, x, y, dpi)
    return _run(["xrandr", "--setmonitor", name, geom, output],
                display) is not None


def delete_logical_monitor(name: str,
                           display: Optional[str] = None) -> bool:
    """Remove a logical monitor (reference display_utils.py:1022)."""
    return _run(["xrandr", "--delmonitor", name], display) is not None


def apply_logical_dual_layout(w1: int, h1: int, w2: int, h2: int,
                              orientation: str = "right",
                              dpi: float = 96.0,
                              display: Optional[str] = None) -> bool:
    """Expose the dual-display layout (compute_dual_layout) as two RandR
    logical monitors on the shared framebuffer."""
    m1, m2 = compute_dual_layout(w1, h1, w2, h2, orientation)
    fb_w, fb_h = framebuffer_bounds([m1, m2])
    if _run(["xrandr", "--fb", f"{fb_w}x{fb_h}"], display) is None:
        return False
    ok1 = set_logical_monitor("selkies-0", m1.width, m1.height, m1.x, m1.y,
                              dpi=dpi, display=display)
    ok2 = set_logical_monitor("selkies-1", m2.width, m2.height, m2.x, m2.y,
                              dpi=dpi, display=display)
    return ok1 and ok2


def _persist_xresources_dpi(dpi: int, path: Optional[str] = None) -> bool:
    """Rewrite only the Xft.dpi resource in ~/.Xresources (reference
    _write_xresources_dpi, display_utils.py:1574-1590)."""
    path = path or os.path.expanduser("~/.Xresources")
    try:
        lines: list[str] = []
        if os.path.exists(path):
            with open(path, "r", encoding="utf-8", errors="replace") as f:
                lines = [ln for ln in f.read().splitlines()
                         if not re.match(r"^\s*Xft\.dpi\s*:", ln)]
        lines.append(f"Xft.dpi:   {dpi}")
        tmp = path + ".tmp"
        with open(tmp, "w", encoding="utf-8") as f:
            f.write("\n".join(lines) + "\n")
        os.replace(tmp, path)
        return True
    except OSError:
        return False


def apply_dpi_ladder(dpi: int, display: Optional[str] = None,
                     xresources_path: Optional[str] = None) -> dict:
    """Apply DPI through every available desktop-environment channel
    (reference ladder: xrdb live merge, Xresources persistence,
    xfconf-query for XFCE, gsettings for MATE/GNOME — display_utils.py:
    1767-1891). Returns which channels were applied."""
    applied = {"xrdb": set_dpi(dpi, display),
               "xresources": _persist_xresources_dpi(dpi, xresources_path)}
    cursor = cursor_size_for_dpi(dpi)
    if shutil.which("xfconf-query"):
        applied["xfconf"] = (
            _run(["xfconf-query", "-c", "xsettings", "-p", "/Xft/DPI",
                  "-n", "-t", "int", "-s", str(dpi)], display) is not None)
        _run(["xfconf-query", "-c", "xsettings", "-p",
              "/Gtk/CursorThemeSize", "-n", "-t", "int", "-s",
              str(cursor)], display)
    if shutil.which("gsettings"):
        scale = dpi / 96.0
        whole = int(scale) if float(scale).is_integer() else 1
        text_scale = scale / whole
        applied["gsettings"] = (
            _run(["gsettings", "set", "org.gnome.desktop.interface",
                  "text-scaling-factor", f"{text_scale:.4f}"],
                 display) is not None)
        _run(["gsettings", "set", "org.gnome.desktop.interface",
              "cursor-size", str(cursor)], display)
    return applied


def resize_entrypoint(argv=None) -> int:
    """`selkies-resize WxH` CLI (reference entrypoint,
    display_utils.py:2200)."""
    import sys
    args = argv if argv is not None else sys.argv[1:]
    if not args:
        print("usage: selkies-resize WxH [display]")
        return 2
    w, _, h = args[0].partition("x")
    display = args[1] if len(args) > 1 else None
    ok = resize_display(int(w), int(h), display)
    print("ok" if ok else "failed")
    return 0 if ok else 1


def parse_dri_node_to_index(path: str) -> int:
    """'/dev/dri/renderD129' -> 1 (reference parse_dri_node_to_index,
    display_utils.py:2223)."""
    m = re.search(r"renderD(\d+)$", path or "")
    if not m:
        return -1
    return max(0, int(m.group(1)) - 128)
