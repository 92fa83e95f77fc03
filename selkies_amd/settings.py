"""Declarative settings schema + resolution for selkies_amd.

Re-implements the reference's config layer behavior (see SURVEY.md §5.6;
reference src/selkies/settings.py: SETTING_DEFINITIONS at :83, precedence
rules at :7-11, override grammar at :17-32) as a fresh design:

* One declarative list of `SettingDef`s (name, type, default, range/allowed,
  client exposure metadata).
* Resolution precedence: CLI flag  >  SELKIES_<NAME> env  >  fallback env
  names  >  default.
* Operator override grammar for deployment-time narrowing:
    - enum/list settings:  "a,b,c"  narrows the allowed set (first = default)
    - bool settings:       "true|locked"  pins the value and locks it
    - range settings:      "init,lo-hi"   sets default + clamps the range;
                           a degenerate span (lo == hi) locks the value.
* Client contract: `build_client_settings_payload()` publishes value +
  allowed/locked per client-exposed setting; `sanitize_client_setting()`
  validates every client proposal server-side (both transports share it).
"""

from __future__ import annotations

import argparse
import os
from dataclasses import dataclass, field
from typing import Any, Callable, Optional, Sequence

__all__ = [
    "SettingDef",
    "AppSettings",
    "SETTING_DEFINITIONS",
    "load_settings",
]


@dataclass
class SettingDef:
    name: str                       # snake_case identifier
    type: type                      # bool / int / float / str
    default: Any
    help: str = ""
    # numeric settings: inclusive (lo, hi); enforced on every assignment
    value_range: Optional[tuple] = None
    # enum settings: permitted values; first entry of an operator override
    # becomes the new default
    allowed: Optional[Sequence[str]] = None
    # exposed to (and settable by) the browser client
    client: bool = False
    # additional env var names honored for backwards compat
    fallback_env: Sequence[str] = field(default_factory=tuple)
    # operator lock: client/API writes rejected
    locked: bool = False
    # optional post-parse normalization
    normalize: Optional[Callable[[Any], Any]] = None

    @property
    def env_name(self) -> str:
        return "SELKIES_" + self.name.upper()

    @property
    def cli_name(self) -> str:
        return "--" + self.name.replace("_", "-")


def _parse_bool(raw: str) -> bool:
    return str(raw).strip().lower() in ("1", "true", "yes", "on")


# Wire names follow the reference encoder menu (reference selkies.py:144
# PIXELFLUX_VIDEO_ENCODERS: h264enc, h264enc-striped, jpeg, openh264enc).
# Ours: h264enc (full-frame H.264), h264enc-striped (stripe-parallel
# independent bitstreams), jpeg (striped MJPEG). All three run on the
# hipflux HIP path on MI355X with a CPU fallback of the same bitstream.
VIDEO_ENCODERS = ("h264enc", "h264enc-striped", "hevcenc-striped", "jpeg")

SETTING_DEFINITIONS: list[SettingDef] = [
    # ---- process / transport ----
    SettingDef("addr", str, "0.0.0.0", "Bind address for the HTTP/WS server."),
    SettingDef("port", int, 8080, "TCP port for the single-port HTTP/WS server.",
               value_range=(1, 65535)),
    SettingDef("unix_socket", str, "", "Serve on a unix socket path instead of TCP."),
    SettingDef("mode", str, "websockets", "Streaming transport mode.",
               allowed=("websockets", "webrtc")),
    SettingDef("enable_basic_auth", bool, False, "Require HTTP basic auth."),
    SettingDef("basic_auth_user", str, "selkies", "Basic auth username."),
    SettingDef("basic_auth_password", str, "", "Basic auth password."),
    SettingDef("auth_token", str, "", "Shared bearer token for API/WS auth."),
    SettingDef("enable_https", bool, False, "Serve TLS (requires cert+key)."),
    SettingDef("https_cert", str, "", "TLS certificate path (hot-reloaded)."),
    SettingDef("https_key", str, "", "TLS key path (hot-reloaded)."),
    SettingDef("allowed_ws_origins", str, "", "Comma list of allowed WS origins ('' = any)."),
    SettingDef("enable_metrics_http", bool, False, "Expose Prometheus /metrics."),
    SettingDef("webrtc_udp_port", int, 0,
               "UDP port for the WebRTC ICE-lite endpoint (0 = ephemeral).",
               value_range=(0, 65535)),
    SettingDef("metrics_http_token", str, "", "Bearer token guarding /metrics."),
    # TURN relay (coturn use-auth-secret scheme; reference settings.py:774-800)
    SettingDef("turn_host", str, "", "TURN server hostname/IP for clients."),
    SettingDef("turn_port", int, 3478, "TURN server port.",
               value_range=(1, 65535)),
    SettingDef("turn_shared_secret", str, "",
               "coturn static-auth-secret for HMAC short-term credentials."),
    SettingDef("turn_protocol", str, "udp", "TURN transport for clients.",
               allowed=("udp", "tcp")),
    SettingDef("turn_tls", bool, False, "Use turns: (TLS) TURN URLs."),
    SettingDef("turn_username", str, "",
               "Legacy long-term TURN username (when no shared secret)."),
    SettingDef("turn_password", str, "",
               "Legacy long-term TURN password (when no shared secret)."),
    SettingDef("turn_rest_uri", str, "",
               "TURN-REST service URI for fetched ICE configs."),
    SettingDef("turn_rest_api_key", str, "",
               "Bearer token for the TURN-REST service."),
    SettingDef("turn_rest_username", str, "",
               "Username requested from the TURN-REST service."),
    SettingDef("turn_rest_username_auth_header", str, "",
               "Header name carrying the username to TURN-REST."),
    SettingDef("turn_rest_protocol_header", str, "",
               "Header name carrying the TURN protocol hint."),
    SettingDef("turn_rest_tls_header", str, "",
               "Header name carrying the TURN TLS hint."),
    SettingDef("enable_cloudflare_turn", bool, False,
               "Mint short-lived TURN credentials from Cloudflare."),
    SettingDef("cloudflare_turn_token_id", str, "",
               "Cloudflare TURN key id."),
    SettingDef("cloudflare_turn_api_token", str, "",
               "Cloudflare API token for TURN credential minting."),
    SettingDef("rtc_config_json", str, "",
               "Path to a JSON RTC config file (highest-priority ICE "
               "source; hot-reloaded on change)."),
    SettingDef("stun_host", str, "", "Extra STUN host to advertise first."),
    SettingDef("stun_port", int, 3478, "Port for stun_host.",
               value_range=(1, 65535)),
    SettingDef("enable_webrtc_statistics", bool, False,
               "Record client-reported WebRTC getStats payloads to CSV."),
    SettingDef("webrtc_statistics_dir", str, "/tmp/selkies-webrtc-stats",
               "Directory for the per-day WebRTC statistics CSV files."),
    SettingDef("capture_scale", float, 1.0,
               "Fractional capture downscale (0.25-1.0); overrides "
               "capture_scale_div when below 1.",
               value_range=(0.25, 1.0), client=True),
    SettingDef("capture_scale_div", int, 1,
               "Integer capture downscale: encode at capture/div "
               "(2 = 4K capture -> 1080p stream).",
               value_range=(1, 4), client=True),

    # ---- display / capture ----
    SettingDef("display", str, ":0", "X DISPLAY to capture/inject into.",
               fallback_env=("DISPLAY",)),
    SettingDef("capture_backend", str, "auto",
               "Framebuffer source: auto (x11 if DISPLAY reachable else "
               "synthetic), x11, synthetic, or synthetic:<pattern>.",
               normalize=lambda v: v if (
                   v in ("auto", "x11", "synthetic")
                   or str(v).startswith("synthetic:")) else "auto"),
    SettingDef("resolution", str, "1920x1080", "Initial capture WxH.", client=True),
    SettingDef("resolution2", str, "1280x720",
               "Second (extended) display WxH.", client=True),
    SettingDef("framerate", int, 60, "Target capture/encode fps.",
               value_range=(1, 240), client=True),
    SettingDef("capture_cursor", bool, True, "Composite the cursor into frames.", client=True),
    SettingDef("second_display", bool, False, "Enable the extended (second) display."),

    # ---- encoder ----
    SettingDef("encoder", str, "h264enc-striped", "Video encoder.",
               allowed=VIDEO_ENCODERS, client=True),
    SettingDef("gpu_id", int, -1, "HIP device ordinal for encode; -1 = CPU/software.",
               value_range=(-1, 63)),
    SettingDef("use_cpu", bool, False, "Force the CPU encode path.", client=True),
    SettingDef("video_bitrate_kbps", int, 16000, "Target bitrate for CBR mode.",
               value_range=(100, 800000), client=True),
    SettingDef("video_crf", int, 25, "CRF quality for capped-quality mode.",
               value_range=(0, 51), client=True),
    SettingDef("video_cbr_mode", bool, False, "CBR (true) vs CRF (false) rate control.",
               client=True),
    SettingDef("video_min_qp", int, 2, "Minimum QP clamp.", value_range=(0, 51)),
    SettingDef("video_max_qp", int, 48, "Maximum QP clamp.", value_range=(0, 51)),
    SettingDef("video_vbv_multiplier", float, 1.5, "VBV buffer size as x of bitrate/fps.",
               value_range=(0.2, 10.0)),
    SettingDef("keyframe_interval_s", float, 0.0,
               "Seconds between forced IDR frames; 0 = infinite GOP (on-demand IDR only).",
               value_range=(0.0, 600.0)),
    SettingDef("video_fullcolor", bool, False, "4:4:4 chroma (I444) instead of 4:2:0.",
               client=True),
    SettingDef("video_pipeline_depth", int, 1,
               "Encoder frame pipelining depth: 1 = synchronous (lowest "
               "interactive latency), 2 = one frame in flight (throughput "
               "mode for recording/transcode; emission lags one frame).",
               value_range=(1, 2)),
    SettingDef("video_fullframe", bool, False,
               "Always encode the full frame (disable damage gating).", client=True),
    SettingDef("video_deblock", bool, True,
               "H.264 in-loop deblocking filter (within-slice edges).",
               client=True),
    SettingDef("video_streaming_mode", bool, False,
               "Motion-optimized mode: bias rate control for smooth motion.", client=True),
    SettingDef("jpeg_quality", int, 80, "JPEG stripe quality.", value_range=(1, 100),
               client=True),
    SettingDef("use_paint_over_quality", bool, True,
               "Refine static screens with a high-quality pass.", client=True),
    SettingDef("paint_over_trigger_frames", int, 15,
               "Consecutive still frames before the paint-over pass.",
               value_range=(1, 1000)),
    SettingDef("video_paintover_crf", int, 18, "CRF used for paint-over refinement.",
               value_range=(0, 51)),
    SettingDef("video_paintover_burst_frames", int, 5,
               "Frames re-sent at paint-over quality.", value_range=(1, 60)),
    SettingDef("damage_block_threshold", int, 15,
               "Per-16px-block diff threshold that marks a block damaged.",
               value_range=(0, 255)),
    SettingDef("damage_block_duration", int, 30,
               "Frames a block stays 'damaged' after its last change.",
               value_range=(1, 600)),
    SettingDef("stripe_height", int, 64,
               "Stripe height (16-aligned) for striped encoders.",
               value_range=(16, 1088)),
    SettingDef("watermark_path", str, "", "PNG watermark path ('' = none)."),
    SettingDef("watermark_location", int, 0,
               "0=none,1=TL,2=TR,3=BL,4=BR,5=center,6=animated.",
               value_range=(0, 6)),

    # ---- audio ----
    SettingDef("enable_audio", bool, True, "Capture + stream audio.", client=True),
    SettingDef("audio_device", str, "auto", "Audio source (auto/synthetic/none)."),
    SettingDef("audio_codec", str, "opus",
               "Audio codec: opus (CELT-class, in-tree) or pcm.",
               allowed=("opus", "pcm"), client=True),
    SettingDef("audio_bitrate", int, 128000, "Audio codec bitrate.",
               value_range=(16000, 512000), client=True),
    SettingDef("audio_channels", int, 2, "Channel count.", value_range=(1, 6)),
    SettingDef("audio_frame_duration_ms", int, 20, "Audio frame duration.",
               value_range=(2, 60)),
    SettingDef("audio_red_distance", int, 2,
               "Redundant-audio (RED) depth for loss resilience.",
               value_range=(0, 5)),
    SettingDef("enable_microphone", bool, False, "Accept client mic uplink.", client=True),

    # ---- input ----
    SettingDef("enable_input", bool, True, "Inject keyboard/mouse input."),
    SettingDef("enable_clipboard", bool, True, "Bidirectional clipboard sync.",
               client=True),
    SettingDef("enable_gamepad", bool, True, "Gamepad passthrough (interposer/uinput)."),
    SettingDef("enable_collab", bool, False,
               "Per-user collab tokens (role/seat table with live "
               "reconciliation)."),
    SettingDef("master_token", str, "",
               "Admin token guarding the /api/tokens collab table."),
    SettingDef("enable_shared", bool, False,
               "Allow multiple controlling clients (shared input)."),
    SettingDef("enable_player2", bool, False,
               "Allow a second client to claim gamepad seat 1.", client=True),
    SettingDef("enable_player3", bool, False,
               "Allow a third client to claim gamepad seat 2.", client=True),
    SettingDef("enable_player4", bool, False,
               "Allow a fourth client to claim gamepad seat 3.", client=True),
    SettingDef("enable_command_input", bool, False,
               "Allow the 'cmd' wire verb to run shell commands."),

    # ---- behavior ----
    SettingDef("enable_resize", bool, True, "Resize server display to client window.",
               client=True),
    SettingDef("dpi", int, 96, "Server display DPI.", value_range=(48, 384), client=True),
    SettingDef("cursor_size", int, 24, "Server cursor size.", value_range=(8, 256)),
    SettingDef("enable_binary_clipboard", bool, False,
               "Allow binary clipboard payloads.", client=True),
    SettingDef("file_transfers", str, "upload,download",
               "Enabled transfer directions (comma list; '' = disabled)."),
    SettingDef("upload_dir", str, "~/Desktop", "Directory receiving uploads."),
    SettingDef("ws_max_message_mb", int, 16, "Inbound WS message ceiling.",
               value_range=(1, 256)),
    SettingDef("debug", bool, False, "Verbose logging."),
    # ---- app lifecycle (reference app_ready/run_after hooks) ----
    SettingDef("use_wayland", bool, False,
               "Capture from the in-tree headless Wayland compositor "
               "instead of X11 (engine shm capture seam)."),
    SettingDef("wayland_display", str, "selkies-wl-0",
               "WAYLAND_DISPLAY name the in-tree compositor binds."),
    SettingDef("app_ready_file", str, "",
               "Path whose existence marks the session app as ready."),
    SettingDef("app_wait_ready", bool, False,
               "Delay capture start until app_ready_file exists."),
    SettingDef("run_after_connect", str, "",
               "Shell command run when the first client connects."),
    SettingDef("run_after_disconnect", str, "",
               "Shell command run when the last client disconnects."),
    # ---- auth extras ----
    SettingDef("basic_auth_viewonly_password", str, "",
               "Second basic-auth password granting view-only role."),
    # ---- infra knobs ----
    SettingDef("cert_reload_interval", int, 60,
               "Seconds between TLS cert/key change polls.",
               value_range=(5, 3600)),
    SettingDef("file_transfer_limit_mbps", float, 0.0,
               "Hard ceiling on file-transfer bandwidth (0 = adaptive "
               "pacer only).", value_range=(0.0, 10000.0)),
    SettingDef("subfolder", str, "",
               "URL prefix to serve under (reverse-proxy subpath)."),
    SettingDef("web_root", str, "",
               "Override directory for the static web client ('' = "
               "bundled)."),
    SettingDef("js_socket_path", str, "/tmp/selkies_js",
               "Base path for gamepad interposer unix sockets."),
    SettingDef("webrtc_public_ip", str, "",
               "Advertise this IP in ICE candidates (NAT'd hosts)."),
    # ---- display/cursor extras ----
    SettingDef("force_aligned_resolution", bool, True,
               "Round client-requested resolutions to even dimensions.",
               client=True),
    SettingDef("enable_cursors", bool, True,
               "Push server cursor shapes to clients.", client=True),
    SettingDef("use_css_scaling", bool, False,
               "Scale the remote canvas with CSS instead of matching "
               "the stream resolution 1:1 (sharper text off, fit-to-"
               "window on).", client=True),
    SettingDef("use_browser_cursors", bool, True,
               "Render the cursor via CSS on the client instead of "
               "compositing it into the video.", client=True),
    SettingDef("debug_cursors", bool, False, "Log cursor shape updates."),
    SettingDef("paint_over_jpeg_quality", int, 95,
               "JPEG quality for static paint-over passes.",
               value_range=(10, 100), client=True),
    # ---- web UI contract (client-side visibility; reference ui_*) ----
    SettingDef("ui_title", str, "Selkies", "Browser tab / UI title.",
               client=True),
    SettingDef("ui_show_logo", bool, True, "Show the logo in the UI.",
               client=True),
    SettingDef("ui_show_sidebar", bool, True, "Show the sidebar.",
               client=True),
    SettingDef("ui_show_core_buttons", bool, True,
               "Show fullscreen/settings core buttons.", client=True),
    SettingDef("ui_sidebar_show_apps", bool, True, "", client=True),
    SettingDef("ui_sidebar_show_audio_settings", bool, True, "",
               client=True),
    SettingDef("ui_sidebar_show_clipboard", bool, True, "", client=True),
    SettingDef("ui_sidebar_show_files", bool, True, "", client=True),
    SettingDef("ui_sidebar_show_fullscreen", bool, True, "", client=True),
    SettingDef("ui_sidebar_show_gamepads", bool, True, "", client=True),
    SettingDef("ui_sidebar_show_gaming_mode", bool, True, "", client=True),
    SettingDef("ui_sidebar_show_keyboard_button", bool, True, "",
               client=True),
    SettingDef("ui_sidebar_show_screen_settings", bool, True, "",
               client=True),
    SettingDef("ui_sidebar_show_sharing", bool, True, "", client=True),
    SettingDef("ui_sidebar_show_shortcuts", bool, True, "", client=True),
    SettingDef("ui_sidebar_show_soft_buttons", bool, True, "",
               client=True),
    SettingDef("ui_sidebar_show_stats", bool, True, "", client=True),
    SettingDef("ui_sidebar_show_trackpad", bool, True, "", client=True),
    SettingDef("ui_sidebar_show_video_settings", bool, True, "",
               client=True),

    # ---- multi-GPU / scaling ----
    SettingDef("session_gpus", str, "",
               "Comma list of HIP device ordinals for session placement "
               "('' = round-robin over visible devices)."),
]

_DEFS_BY_NAME = {d.name: d for d in SETTING_DEFINITIONS}


class AppSettings:
    """Resolved settings bag.  Attribute access returns the resolved value."""

    def __init__(self, definitions: Sequence[SettingDef] = SETTING_DEFINITIONS,
                 argv: Optional[Sequence[str]] = None,
                 env: Optional[dict] = None):
        self._defs = {d.name: d for d in definitions}
        self._values: dict[str, Any] = {}
        self._locked: set[str] = set()
        env = os.environ if env is None else env
        cli = self._parse_cli(definitions, argv)
        for d in definitions:
            raw, source = None, "default"
            if cli.get(d.name) is not None:
                raw, source = cli[d.name], "cli"
            elif d.env_name in env:
                raw, source = env[d.env_name], "env"
            else:
                for fb in d.fallback_env:
                    if fb in env:
                        raw, source = env[fb], "fallback_env"
                        break
            if raw is None:
                value = d.default
                if d.locked:
                    self._locked.add(d.name)
            else:
                value = self._coerce(d, raw, allow_grammar=(source != "cli"))
            self._values[d.name] = self._validate(d, value)

    # -- parsing helpers ---------------------------------------------------
    @staticmethod
    def _parse_cli(definitions, argv) -> dict:
        p = argparse.ArgumentParser(prog="selkies", add_help=True, allow_abbrev=False)
        for d in definitions:
            if d.type is bool:
                p.add_argument(d.cli_name, dest=d.name, default=None,
                               type=_parse_bool, metavar="BOOL", help=d.help)
            else:
                p.add_argument(d.cli_name, dest=d.name, default=None,
                               type=str, help=d.help)
        ns, _unknown = p.parse_known_args(argv)
        return vars(ns)

    def _coerce(self, d: SettingDef, raw: Any, allow_grammar: bool) -> Any:
        """Apply the operator override grammar, then coerce to d.type."""
        if isinstance(raw, d.type) and not isinstance(raw, str):
            return raw
        s = str(raw)
        if allow_grammar:
            # bool "<value>|locked"
            if d.type is bool and "|" in s:
                val, _, flag = s.partition("|")
                if flag.strip().lower() == "locked":
                    self._locked.add(d.name)
                return _parse_bool(val)
            # enum narrowing "a,b,c"
            if d.allowed is not None and "," in s:
                opts = [o.strip() for o in s.split(",") if o.strip()]
                bad = [o for o in opts if o not in d.allowed]
                if bad:
                    raise ValueError(f"{d.name}: unknown option(s) {bad}")
                d.allowed = tuple(opts)
                if len(opts) == 1:
                    self._locked.add(d.name)
                return opts[0]
            # range "init,lo-hi"
            if d.value_range is not None and "," in s and d.type in (int, float):
                init_s, _, span = s.partition(",")
                lo_s, _, hi_s = span.partition("-")
                conv = d.type
                lo, hi = conv(lo_s), conv(hi_s)
                d.value_range = (lo, hi)
                if lo == hi:
                    self._locked.add(d.name)
                return conv(init_s)
        if d.type is bool:
            return _parse_bool(s)
        return d.type(s)

    @staticmethod
    def _validate(d: SettingDef, value: Any) -> Any:
        if d.normalize:
            value = d.normalize(value)
        if d.type in (int, float) and d.value_range is not None:
            lo, hi = d.value_range
            value = min(max(d.type(value), lo), hi)
        if d.allowed is not None and value not in d.allowed:
            raise ValueError(f"{d.name}: {value!r} not in allowed set {d.allowed}")
        return value

    # -- access ------------------------------------------------------------
    def __getattr__(self, name: str) -> Any:
        try:
            return self.__dict__["_values"][name]
        except KeyError:
            raise AttributeError(name) from None

    def get(self, name: str, default=None):
        return self._values.get(name, default)

    def set(self, name: str, value: Any) -> Any:
        d = self._defs[name]
        if name in self._locked:
            raise PermissionError(f"setting {name} is locked")
        value = self._validate(d, value if not isinstance(value, str)
                               else self._coerce(d, value, allow_grammar=False))
        self._values[name] = value
        return value

    def is_locked(self, name: str) -> bool:
        return name in self._locked

    @property
    def resolution_wh(self) -> tuple[int, int]:
        w, _, h = self.resolution.lower().partition("x")
        return int(w), int(h)

    # -- client contract ----------------------------------------------------
    def build_client_settings_payload(self) -> dict:
        """Server→client settings contract: value + allowed/range + locked for
        every client-exposed setting (reference settings.py:1648)."""
        out = {}
        for d in self._defs.values():
            if not d.client:
                continue
            entry = {"value": self._values[d.name], "locked": d.name in self._locked}
            if d.allowed is not None:
                entry["allowed"] = list(d.allowed)
            if d.value_range is not None:
                entry["range"] = list(d.value_range)
            out[d.name] = entry
        return out

    def sanitize_client_setting(self, name: str, value: Any):
        """Validate a client proposal; returns the sanitized value or raises.
        Both transports share this path (reference settings.py:1698)."""
        d = self._defs.get(name)
        if d is None or not d.client:
            raise KeyError(f"unknown or non-client setting: {name}")
        if d.name in self._locked:
            raise PermissionError(f"setting {name} is locked")
        if d.type is bool and isinstance(value, str):
            value = _parse_bool(value)
        return self._validate(d, d.type(value))


def load_settings(argv: Optional[Sequence[str]] = None,
                  env: Optional[dict] = None) -> AppSettings:
    import copy
    return AppSettings([copy.copy(d) for d in SETTING_DEFINITIONS], argv=argv, env=env)
