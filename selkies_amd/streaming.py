"""WebSocket data plane: capture orchestration + per-client fan-out.

Fresh implementation of the reference WS streaming service behavior
(SURVEY.md §2.1 selkies.py DataStreamingServer / §3.2 hot loop):

* one hipflux ScreenCapture per display; the native stripe callback never
  touches asyncio state — it trampolines via call_soon_threadsafe (the
  reference's thread→loop rule, selkies.py:15-22);
* every client gets a bounded VideoRelay (byte budget, keyframe exemption,
  per-row IDR chain repair — relay.py);
* text control verbs (SETTINGS / CLIENT_FRAME_ACK / input protocol),
  binary 0x02 mic + 0x05 gzip'd control;
* frame-ACK backpressure: if the newest acked frame falls more than
  ~2s+RTT behind, fan-out to that client pauses until it catches up and
  an IDR repairs the chain (reference selkies.py:2261).
"""

from __future__ import annotations

import asyncio
import json
import logging
import time
from typing import Optional

from aiohttp import WSMsgType, web

import hipflux

from . import protocol as P
from .input_handler import InputDispatcher, make_backend
from .relay import VideoRelay
from .settings import AppSettings

logger = logging.getLogger("selkies.streaming")


def _js_index(text: str) -> Optional[int]:
    """Pad index of a js,* wire verb (None if malformed)."""
    parts = text.split(",", 3)
    if len(parts) < 3:
        return None
    try:
        return int(parts[2])
    except ValueError:
        return None


def _remap_js(text: str, seat: int) -> str:
    """Rewrite the pad index of a js,* verb to the holder's seat slot."""
    parts = text.split(",")
    if len(parts) >= 3:
        parts[2] = str(seat)
    return ",".join(parts)


class ClientState:
    def __init__(self, ws, relay: VideoRelay, display: str = "primary",
                 role: str = "viewer"):
        self.ws = ws
        self.relay = relay
        self.display = display
        self.role = role
        # gamepad seat held by this client: None, or 1..3 (player2..4).
        # The controller implicitly owns seat 0. (Reference: signaling
        # player2-4 slots, signaling_server.py allowed_client_slots.)
        self.player_seat: Optional[int] = None
        self.token: Optional[str] = None
        self.last_acked_frame = -1
        self.last_sent_frame = -1
        self.ack_rtt_ms = 50.0
        self.paused = False
        # per-client stream gates (reference START/STOP_VIDEO/AUDIO verbs)
        self.video_stopped = False
        self.audio_stopped = False
        self._sent_ts: dict[int, float] = {}

    def note_sent(self, frame_id: int):
        self.last_sent_frame = frame_id
        if frame_id not in self._sent_ts:
            self._sent_ts[frame_id] = time.monotonic()
            if len(self._sent_ts) > 512:
                for k in sorted(self._sent_ts)[:256]:
                    self._sent_ts.pop(k, None)

    def note_ack(self, frame_id: int):
        self.last_acked_frame = max(self.last_acked_frame, frame_id)
        ts = self._sent_ts.pop(frame_id, None)
        if ts is not None:
            rtt = (time.monotonic() - ts) * 1e3
            self.ack_rtt_ms = 0.8 * self.ack_rtt_ms + 0.2 * rtt


class StreamingService:
    """The websockets-mode streaming service."""

    def __init__(self, settings: AppSettings):
        self.settings = settings
        self.clients: dict[object, ClientState] = {}
        self.captures: dict[str, hipflux.ScreenCapture] = {}
        self.loop: Optional[asyncio.AbstractEventLoop] = None
        from .clipboard import make_clipboard
        self.clipboard = make_clipboard(settings.display)
        self.input = InputDispatcher(
            make_backend(settings.display if settings.enable_input else None),
            on_resize=self._on_resize,
            on_dpi=self._on_dpi,
            on_bitrate=lambda kbps: self._tune("video_bitrate_kbps", kbps),
            on_audio_bitrate=lambda bps: self._tune("audio_bitrate", bps),
            on_clipboard=self.clipboard.write,
            clipboard_read=self.clipboard.read,
            enable_input=settings.enable_input,
            enable_clipboard=settings.enable_clipboard,
            enable_binary_clipboard=settings.enable_binary_clipboard,
        )
        self._frame_clock = 0
        self._stats_task: Optional[asyncio.Task] = None
        self._watchdog_task: Optional[asyncio.Task] = None
        self.frames_relayed = 0
        # glass-to-glass accounting: frame id -> capture ts (native
        # steady-clock ms) and the rolling capture->ACK sample window
        self._frame_capture_ts: dict[int, float] = {}
        self._g2g_ms: list[float] = []
        self.audio: Optional[object] = None
        self._audio_queue: Optional[asyncio.Queue] = None
        self._audio_task: Optional[asyncio.Task] = None
        self.mic_sink: Optional[object] = None
        # collab token table: token -> {"role": ..., "seat": int|None}
        # (reference user_tokens + reconcile_clients)
        self.user_tokens: dict[str, dict] = {}
        self.gamepads = None
        if settings.enable_gamepad:
            from .gamepad import GamepadHub
            self.gamepads = GamepadHub(
                socket_dir=settings.js_socket_path)

    @property
    def capture(self):
        """Primary display capture (back-compat accessor)."""
        return self.captures.get("primary")

    # ---- capture lifecycle -------------------------------------------------
    def build_capture_settings(self, display: str = "primary"
                               ) -> "hipflux.CaptureSettings":
        s = self.settings
        cs = hipflux.CaptureSettings()
        if display == "primary":
            w, h = s.resolution_wh
        else:
            w, _, h = s.resolution2.lower().partition("x")
            w, h = int(w), int(h)
            # extended desktop: second display sits right of the primary
            cs.capture_x = s.resolution_wh[0]
        cs.capture_width = w
        cs.capture_height = h
        cs.target_fps = float(s.framerate)
        cs.output_mode = (0 if s.encoder == "jpeg"
                          else 2 if "hevc" in s.encoder else 1)
        cs.use_cpu = bool(s.use_cpu)
        # per-session GPU placement: explicit session_gpus list round-robins
        # displays/sessions over devices (SURVEY.md §5.8 multi-GPU facility)
        gpus = [int(g) for g in s.session_gpus.split(",") if g.strip()]
        if gpus:
            idx = 0 if display == "primary" else 1
            cs.gpu_id = gpus[idx % len(gpus)]
        else:
            cs.gpu_id = s.gpu_id if s.gpu_id >= 0 else 0
        if s.use_wayland:
            # in-tree compositor -> seqlock shm file -> native engine
            # (reference use_wayland capture contract)
            cs.capture_backend = "shm:" + self._ensure_wayland_shm()
        else:
            cs.capture_backend = s.capture_backend
        cs.display = s.display
        cs.video_bitrate_kbps = s.video_bitrate_kbps
        cs.video_crf = s.video_crf
        cs.video_cbr_mode = s.video_cbr_mode
        cs.video_min_qp = s.video_min_qp
        cs.video_max_qp = s.video_max_qp
        cs.vbv_multiplier = s.video_vbv_multiplier
        cs.keyframe_interval_s = s.keyframe_interval_s
        cs.video_fullcolor = s.video_fullcolor
        cs.video_deblock = s.video_deblock
        cs.video_fullframe = s.video_fullframe
        cs.capture_scale = s.capture_scale
        cs.capture_scale_div = s.capture_scale_div
        cs.pipeline_depth = s.video_pipeline_depth
        cs.use_paint_over_quality = s.use_paint_over_quality
        cs.paint_over_trigger_frames = s.paint_over_trigger_frames
        cs.video_paintover_crf = s.video_paintover_crf
        cs.video_paintover_burst_frames = s.video_paintover_burst_frames
        cs.jpeg_paintover_quality = s.paint_over_jpeg_quality
        cs.damage_block_threshold = s.damage_block_threshold
        cs.damage_block_duration = s.damage_block_duration
        cs.jpeg_quality = s.jpeg_quality
        cs.stripe_height = s.stripe_height
        cs.capture_cursor = s.capture_cursor
        if s.watermark_path and s.watermark_location > 0:
            raw = self._watermark_to_bgra(s.watermark_path)
            if raw:
                cs.watermark_path = raw
                cs.watermark_location = s.watermark_location
        return cs

    @staticmethod
    def _watermark_to_bgra(path: str) -> Optional[str]:
        """Convert a PNG/image watermark to the engine's raw .bgra format
        (u32 w, u32 h, BGRA pixels) via PIL."""
        if path.endswith(".bgra"):
            return path
        try:
            import struct
            from PIL import Image
            img = Image.open(path).convert("RGBA")
            out = path + ".bgra"
            import numpy as np
            arr = np.asarray(img)
            bgra = arr[:, :, [2, 1, 0, 3]].tobytes()
            with open(out, "wb") as f:
                f.write(struct.pack("<II", img.width, img.height))
                f.write(bgra)
            return out
        except Exception as exc:
            logger.warning("watermark conversion failed: %r", exc)
            return None

    def _ensure_wayland_shm(self) -> str:
        if getattr(self, "_wl_shm_path", None):
            return self._wl_shm_path
        import tempfile
        from .wayland import compositor as wl
        w, h = self.settings.resolution_wh
        wl.ensure_wayland_display(self.settings.wayland_display, w, h)
        comp = wl.get_compositor()
        path = os.path.join(tempfile.gettempdir(),
                            f"selkies-wl-fb-{os.getpid()}.shm")
        comp.start_shm_publisher(path, fps=float(self.settings.framerate))
        self._wl_shm_path = path
        return path

    @staticmethod
    def _run_hook(cmd: str):
        """Fire-and-forget session lifecycle hook (reference
        run_after_connect / run_after_disconnect)."""
        if not cmd:
            return
        import subprocess
        try:
            subprocess.Popen(cmd, shell=True,
                             stdout=subprocess.DEVNULL,
                             stderr=subprocess.DEVNULL)
        except Exception as exc:
            logger.warning("lifecycle hook failed: %r", exc)

    def start_capture(self, display: str = "primary"):
        cap = self.captures.get(display)
        if cap is not None and cap.is_capturing:
            return
        self.loop = asyncio.get_running_loop()
        if (self.settings.app_wait_ready and self.settings.app_ready_file
                and not os.path.exists(self.settings.app_ready_file)):
            # poll until the session app marks itself ready (reference
            # app_ready_file/app_wait_ready contract)
            logger.info("waiting for app ready file %s",
                        self.settings.app_ready_file)
            self.loop.call_later(0.5, self.start_capture, display)
            return
        cap = hipflux.ScreenCapture()
        self.captures[display] = cap
        loop = self.loop

        def on_stripe(data, frame_id, y, width, height, is_keyframe,
                      capture_ts_ms, encode_done_ms, stripe_type,
                      _display=display):
            # native thread -> loop (the only allowed crossing). The loop
            # can close while the capture thread drains its last frames
            # (shutdown); those stripes are simply dropped.
            try:
                loop.call_soon_threadsafe(self._fanout, _display, data,
                                          frame_id, y, is_keyframe,
                                          capture_ts_ms)
            except RuntimeError:
                pass

        if (display == "primary" and not self.settings.capture_cursor
                and self.settings.enable_cursors):
            # client-side cursor rendering: push shape updates over control
            def on_cursor(w, h, hx, hy, argb):
                if self.settings.debug_cursors:
                    logger.info("cursor shape %dx%d hot(%d,%d)", w, h, hx,
                                hy)
                try:
                    loop.call_soon_threadsafe(self._broadcast_cursor, w, h,
                                              hx, hy, argb)
                except RuntimeError:
                    pass
            cap.set_cursor_callback(on_cursor)
        cap.start_capture(on_stripe, self.build_capture_settings(display))
        logger.info("capture started for %s (pipeline=%s)", display,
                    cap.pipeline)

    def stop_capture(self, display: Optional[str] = None):
        names = [display] if display else list(self.captures)
        for name in names:
            cap = self.captures.pop(name, None)
            if cap is not None:
                cap.stop_capture()

    # ---- audio (shared encode, per-client delivery; reference
    # _pcmflux_audio_callback drop-oldest contract, selkies.py:1613) --------
    def start_audio(self):
        if self.audio is not None or not self.settings.enable_audio:
            return
        from hipflux import _native
        s = _native.AudioCaptureSettings()
        s.device_name = self.settings.audio_device if \
            self.settings.audio_device != "auto" else "synthetic"
        s.channels = self.settings.audio_channels
        s.codec = self.settings.audio_codec
        s.opus_bitrate = self.settings.audio_bitrate
        s.frame_duration_ms = self.settings.audio_frame_duration_ms
        s.red_distance = self.settings.audio_red_distance
        loop = asyncio.get_running_loop()
        self._audio_queue = asyncio.Queue(maxsize=16)

        def on_frame(data, pts_ms):
            def put():
                q = self._audio_queue
                if q is None:
                    return
                if q.full():
                    try:
                        q.get_nowait()      # drop oldest
                    except asyncio.QueueEmpty:
                        pass
                q.put_nowait(data)
            try:
                loop.call_soon_threadsafe(put)
            except RuntimeError:
                pass

        self.audio = _native.AudioCapture()
        self.audio.start_capture(s, on_frame)
        self._audio_task = loop.create_task(self._audio_sender())

    def stop_audio(self):
        if self.audio is not None:
            self.audio.stop_capture()
            self.audio = None
        if self._audio_task is not None:
            self._audio_task.cancel()
            self._audio_task = None
        self._audio_queue = None

    async def _audio_sender(self):
        while True:
            data = await self._audio_queue.get()
            for cs in list(self.clients.values()):
                if cs.relay.dead or cs.audio_stopped:
                    continue
                try:
                    await asyncio.wait_for(cs.ws.send_bytes(data), 1.0)
                except Exception:
                    pass

    def request_idr(self, display: Optional[str] = None):
        for name, cap in self.captures.items():
            if display is None or name == display:
                cap.request_idr_frame()

    # ---- fan-out -----------------------------------------------------------
    def _fanout(self, display: str, data: bytes, frame_id: int, y: int,
                is_keyframe: bool, capture_ts_ms: float = 0.0):
        self.frames_relayed += 1
        # capture timestamp per frame id (native steady_clock == Python
        # time.monotonic epoch) for glass-to-glass accounting at ACK time
        if capture_ts_ms and display == "primary":
            fid = frame_id & 0xFFFF
            if fid not in self._frame_capture_ts:
                self._frame_capture_ts[fid] = capture_ts_ms
                if len(self._frame_capture_ts) > 512:
                    for k in sorted(self._frame_capture_ts)[:256]:
                        self._frame_capture_ts.pop(k, None)
        for cs in list(self.clients.values()):
            if cs.relay.dead or cs.display != display or cs.video_stopped:
                continue
            if cs.paused and not is_keyframe:
                continue
            if cs.relay.offer(data, y, is_keyframe):
                cs.note_sent(frame_id)

    async def _notify_pipeline_reset(self, displays):
        """PIPELINE_RESETTING <display>: clients drop their per-row
        decoders and frame-id state before the encoder restarts
        (reference selkies.py:1955)."""
        for cs in list(self.clients.values()):
            for d in displays:
                try:
                    await cs.ws.send_str(f"PIPELINE_RESETTING {d}")
                except Exception:
                    pass

    async def _notify_display_config(self):
        """DISPLAY_CONFIG_UPDATE,{json}: current display layout pushed
        after reconfiguration (reference selkies.py:1580)."""
        displays = [{"display_id": name,
                     "width": self.settings.resolution_wh[0],
                     "height": self.settings.resolution_wh[1]}
                    for name in (self.captures or {"primary": None})]
        payload = json.dumps({"type": "display_config_update",
                              "displays": displays})
        for cs in list(self.clients.values()):
            try:
                await cs.ws.send_str("DISPLAY_CONFIG_UPDATE," + payload)
            except Exception:
                pass

    async def reconcile_clients(self):
        """Walk live clients against the collab token table: disconnect
        revoked tokens / role changes, push ROLE_UPDATE for seat-only
        changes, re-announce MK_ACCESS (reference reconcile_clients,
        selkies.py:5963)."""
        for cs in list(self.clients.values()):
            if cs.token is None:
                continue
            perms = self.user_tokens.get(cs.token)
            if perms is None or perms.get("role") != cs.role:
                try:
                    await cs.ws.close(code=4002,
                                      message=b"permissions changed")
                except Exception:
                    pass
                continue
            new_seat = perms.get("seat")
            if new_seat != cs.player_seat:
                cs.player_seat = new_seat
                try:
                    await cs.ws.send_str("ROLE_UPDATE," + json.dumps(
                        {"role": cs.role, "slot": new_seat}))
                except Exception:
                    pass
            try:
                await cs.ws.send_str(
                    "MK_ACCESS," +
                    ("1" if cs.role == "controller" else "0"))
            except Exception:
                pass

    def _broadcast_cursor(self, w, h, hx, hy, argb):
        import base64
        msg = P.encode_control("CURSOR", {
            "width": w, "height": h, "hot_x": hx, "hot_y": hy,
            "argb_b64": base64.b64encode(argb).decode()})
        for cs in list(self.clients.values()):
            try:
                asyncio.get_running_loop().create_task(cs.ws.send_str(msg))
            except Exception:
                pass

    # ---- backpressure (reference selkies.py:2261 behavior) -----------------
    def _backpressure_tick(self):
        budget_frames = max(4, 2 * self.settings.framerate)
        for cs in self.clients.values():
            behind = cs.last_sent_frame - cs.last_acked_frame
            if not cs.paused and behind > budget_frames:
                cs.paused = True
                self.request_idr()
            elif cs.paused and behind <= max(2, budget_frames // 4):
                cs.paused = False
                self.request_idr()

    # ---- ws handler ---------------------------------------------------------
    async def ws_handler(self, request: web.Request) -> web.WebSocketResponse:
        ws = web.WebSocketResponse(
            max_msg_size=self.settings.ws_max_message_mb * 1024 * 1024,
            heartbeat=30)
        await ws.prepare(request)

        display = request.query.get("display", "primary")
        if display not in ("primary", "display2", "none"):
            display = "primary"
        # role/slot policy (reference roles: one controller, shared viewers;
        # signaling_server.py allowed_client_slots behavior)
        want_role = request.query.get("role", "")
        if request.get("forced_role") == "viewer":
            want_role = "viewer"
        user_token = None
        if self.settings.enable_collab:
            t = request.query.get("utoken", "")
            perms = self.user_tokens.get(t) if t else None
            if self.user_tokens and perms is None:
                # a populated table makes tokens mandatory
                import aiohttp.web as _web
                raise _web.HTTPForbidden(reason="collab token required")
            if perms is not None:
                user_token = t
                want_role = perms.get("role", "viewer")
        has_controller = any(c.role == "controller"
                             for c in self.clients.values())
        if want_role == "viewer":
            role = "viewer"
        elif not has_controller:
            role = "controller"
        elif self.settings.enable_shared and want_role == "controller":
            role = "controller"
        else:
            role = "viewer"
        if display == "display2" and not self.settings.second_display:
            self.settings.set("second_display", True)
        relay = VideoRelay(
            send=ws.send_bytes,
            request_idr=lambda d=display: self.request_idr(d),
            bitrate_bps=self.settings.video_bitrate_kbps * 1000.0)
        relay.start()
        state = ClientState(ws, relay, display, role)
        state.token = user_token
        first_client = not self.clients
        self.clients[ws] = state
        if first_client:
            self._run_hook(self.settings.run_after_connect)

        try:
            await ws.send_str("AUTH_SUCCESS," + json.dumps(
                {"role": role, "slot": state.player_seat}))
            await ws.send_str(P.encode_control("MODE", "websockets"))
            await ws.send_str(P.encode_control("ROLE", role))
            await ws.send_str("MK_ACCESS," +
                              ("1" if role == "controller" else "0"))
            await ws.send_str(P.encode_control(
                "SETTINGS_PAYLOAD",
                self.settings.build_client_settings_payload()))
            if display != "none":   # "none" = control-only (WebRTC video)
                self.start_capture(display)
                self.request_idr(display)
            self.start_audio()
            if self._stats_task is None or self._stats_task.done():
                self._stats_task = asyncio.get_running_loop().create_task(
                    self._stats_pusher())
            if self._watchdog_task is None or self._watchdog_task.done():
                self._watchdog_task = asyncio.get_running_loop().create_task(
                    self._capture_watchdog())

            async for msg in ws:
                if msg.type == WSMsgType.TEXT:
                    reply = await self._on_text(state, msg.data)
                    if reply:
                        await ws.send_str(reply)
                elif msg.type == WSMsgType.BINARY:
                    self._on_binary(state, msg.data)
                elif msg.type in (WSMsgType.ERROR, WSMsgType.CLOSE):
                    break
        finally:
            self.clients.pop(ws, None)
            if not self.clients:
                self._run_hook(self.settings.run_after_disconnect)
            await relay.stop()
            if state.player_seat is not None and self.gamepads is not None:
                # free the seat and unplug its pad slot
                try:
                    await self.gamepads.handle(f"js,d,{state.player_seat}")
                except Exception:
                    pass
            # promote the oldest remaining client when the controller leaves
            if state.role == "controller" and self.clients:
                nxt = next(iter(self.clients.values()))
                nxt.role = "controller"
                try:
                    await nxt.ws.send_str(P.encode_control("ROLE",
                                                           "controller"))
                except Exception:
                    pass
            if not any(c.display == display for c in self.clients.values()):
                self.stop_capture(display)
            if not self.clients:
                self.stop_capture()
                self.stop_audio()
        return ws

    async def _on_text(self, state: ClientState, text: str) -> Optional[str]:
        verb, rest = P.parse_control(text)
        if verb == "CLIENT_FRAME_ACK":
            fid = int(rest) & 0xFFFF
            state.note_ack(fid)
            # glass-to-glass: framebuffer capture -> client ACK received
            # (the BASELINE latency metric; reference frame-ACK plumbing
            # selkies.py:2369,3612). Includes encode, WS delivery and the
            # client's ack turnaround on the measurement path.
            cap_ts = self._frame_capture_ts.pop(fid, None)
            if cap_ts is not None:
                g2g = time.monotonic() * 1e3 - cap_ts
                if 0 <= g2g < 10_000:
                    self._g2g_ms.append(g2g)
                    if len(self._g2g_ms) > 2048:
                        del self._g2g_ms[:1024]
            self._backpressure_tick()
            return None
        if verb == "SETTINGS":
            try:
                changes = json.loads(rest)
            except json.JSONDecodeError:
                return P.encode_control("ERROR", "bad SETTINGS payload")
            await self._apply_client_settings(changes)
            return P.encode_control(
                "SETTINGS_PAYLOAD",
                self.settings.build_client_settings_payload())
        if verb in ("REQUEST_IDR", "REQUEST_KEYFRAME"):
            self.request_idr()
            return None
        if verb == "STOP_VIDEO":
            state.video_stopped = True
            return P.encode_control("VIDEO_STOPPED", "")
        if verb == "START_VIDEO":
            state.video_stopped = False
            self.request_idr(state.display)
            return P.encode_control("VIDEO_STARTED", "")
        if verb == "STOP_AUDIO":
            state.audio_stopped = True
            return P.encode_control("AUDIO_STOPPED", "")
        if verb == "START_AUDIO":
            state.audio_stopped = False
            self.start_audio()
            return P.encode_control("AUDIO_STARTED", "")
        if verb == "REQUEST_CLIPBOARD":
            reply = self.input.on_message("cr,")
            return reply
        if verb == "SET_NATIVE_CURSOR_RENDERING":
            # 1 = composite the cursor into the video (native), 0 = push
            # shapes for client-side rendering (reference verb)
            want = rest.strip() in ("1", "true")
            if want != self.settings.capture_cursor:
                await self._apply_client_settings(
                    {"capture_cursor": want})
            return None
        if verb == "CLAIM_SEAT":
            # a viewer claims gamepad seat n (1..3 = player2..4); granted
            # when enable_player{n+1} is on and the seat is free
            try:
                n = int(rest)
            except (TypeError, ValueError):
                return P.encode_control("SEAT", "-1")
            free = not any(c.player_seat == n
                           for c in self.clients.values())
            allowed = (1 <= n <= 3 and free and state.role != "controller"
                       and getattr(self.settings,
                                   f"enable_player{n + 1}", False))
            if allowed:
                state.player_seat = n
            return P.encode_control("SEAT", str(n if allowed else -1))
        if verb == "RELEASE_SEAT":
            state.player_seat = None
            return P.encode_control("SEAT", "-1")
        if verb == "js" and self.gamepads is not None:
            if state.player_seat is not None:
                # a seated player's pads all map onto their one seat slot
                await self.gamepads.handle(
                    _remap_js(text, state.player_seat))
            elif state.role == "controller" or self.settings.enable_shared:
                claimed = {c.player_seat for c in self.clients.values()
                           if c.player_seat is not None}
                if _js_index(text) not in claimed:
                    await self.gamepads.handle(text)
            return None
        if state.role != "controller" and not self.settings.enable_shared:
            return None          # viewers cannot inject input
        return self.input.on_message(text)

    def _on_binary(self, state: ClientState, data: bytes):
        if not data:
            return
        if data[0] == P.TAG_GZIP:
            try:
                text = P.inflate_gz_bounded(data[1:])
            except Exception:
                return
            asyncio.get_running_loop().create_task(
                self._handle_inflated(state, text))
        elif data[0] == P.TAG_MIC_PCM:
            # client mic PCM -> virtual microphone sink
            if self.settings.enable_microphone:
                if self.mic_sink is None:
                    from hipflux import _native
                    ps = _native.AudioPlaybackSettings()
                    self.mic_sink = _native.AudioPlayback(ps)
                self.mic_sink.write(data[1:])

    async def _handle_inflated(self, state: ClientState, text: str):
        reply = await self._on_text(state, text)
        if reply:
            await state.ws.send_str(reply)

    # ---- settings application ----------------------------------------------
    STRUCTURAL = {"encoder", "resolution", "video_fullcolor", "use_cpu",
                  "capture_scale",
                  "video_fullframe"}

    async def _apply_client_settings(self, changes: dict):
        structural = False
        for name, value in changes.items():
            try:
                value = self.settings.sanitize_client_setting(name, value)
            except Exception as exc:
                logger.info("rejected client setting %s: %r", name, exc)
                continue
            self.settings.set(name, value)
            structural |= name in self.STRUCTURAL
            if not structural:
                for cap in self.captures.values():
                    if name == "framerate":
                        cap.update_framerate(float(value))
                    elif name == "video_bitrate_kbps":
                        cap.update_video_bitrate(int(value))
                    elif name == "video_crf":
                        cap.update_crf(int(value))
                    elif name == "jpeg_quality":
                        cap.update_jpeg_quality(int(value))
        if structural and self.captures:
            logger.info("structural setting changed; restarting captures")
            displays = list(self.captures)
            await self._notify_pipeline_reset(displays)
            self.stop_capture()
            for d in displays:
                self.start_capture(d)
            self.request_idr()
            await self._notify_display_config()

    def _tune(self, name: str, value):
        try:
            self.settings.set(name, value)
        except Exception:
            return
        if name == "video_bitrate_kbps":
            for cap in self.captures.values():
                cap.update_video_bitrate(int(value))

    def _on_resize(self, w: int, h: int):
        if not self.settings.enable_resize:
            return
        from . import display_utils
        if self.settings.force_aligned_resolution:
            w, h = display_utils.align_dims_16(w, h)
        else:
            w, h = max(2, w & ~1), max(2, h & ~1)  # even at minimum
        self.settings.set("resolution", f"{w}x{h}")
        # resize the real display when capturing X11
        backend = self.settings.capture_backend
        if backend == "x11" or (backend == "auto" and self.settings.display):
            display_utils.resize_display(w, h, self.settings.display)
        if self.captures:
            displays = list(self.captures)
            loop = asyncio.get_event_loop()
            loop.create_task(self._notify_pipeline_reset(displays))
            self.stop_capture()
            for d in displays:
                self.start_capture(d)
            self.request_idr()

    def _on_dpi(self, dpi: int):
        try:
            self.settings.set("dpi", dpi)
        except Exception:
            return
        from . import display_utils
        display_utils.set_dpi(dpi, self.settings.display)

    async def _stats_pusher(self):
        """Periodic system/GPU stats pushed to clients (reference
        _collect_system_stats_ws / _collect_gpu_stats_ws, selkies.py:5676)."""
        import psutil
        from .stream_server import gpu_stats_snapshot
        while self.clients:
            payload = {
                "cpu_percent": psutil.cpu_percent(interval=None),
                "mem_percent": psutil.virtual_memory().percent,
                "gpus": gpu_stats_snapshot(),
                "stream": {
                    "fps_target": self.settings.framerate,
                    "encoder": self.settings.encoder,
                    "last_encode_ms":
                        self.capture.last_encode_ms if self.capture else 0,
                },
            }
            msg = P.encode_control("SYSTEM_STATS", payload)
            for cs in list(self.clients.values()):
                try:
                    await cs.ws.send_str(msg)
                except Exception:
                    pass
            await asyncio.sleep(2.0)

    async def _capture_watchdog(self, interval: float = 5.0):
        """Rebuild stale capture instances (reference behavior: stale
        captures detected via is_capturing and rebuilt, selkies.py:
        5167-5190). A capture thread that died (encoder fault, display
        loss) is restarted as long as clients are connected."""
        while True:
            await asyncio.sleep(interval)
            if not self.clients:
                continue
            displays = {c.display for c in self.clients.values()
                        if c.display != "none"}
            for d in displays:
                cap = self.captures.get(d)
                if cap is not None and not cap.is_capturing:
                    logger.warning("capture for %s died; rebuilding", d)
                    self.stop_capture(d)
                    try:
                        self.start_capture(d)
                        self.request_idr(d)
                    except Exception:
                        logger.exception("capture rebuild failed for %s", d)

    # ---- stats --------------------------------------------------------------
    def g2g_percentiles(self) -> dict:
        """Capture->ACK latency percentiles over the rolling window."""
        if not self._g2g_ms:
            return {"p50": None, "p95": None, "n": 0}
        s = sorted(self._g2g_ms)
        return {"p50": round(s[len(s) // 2], 3),
                "p95": round(s[int(len(s) * 0.95)], 3),
                "n": len(s)}

    def stats(self) -> dict:
        cap = self.capture
        return {
            "clients": len(self.clients),
            "capturing": bool(cap and cap.is_capturing),
            "pipeline": cap.pipeline if cap else None,
            "frames_captured": cap.frames_captured if cap else 0,
            "frames_encoded": cap.frames_encoded if cap else 0,
            "stripes_emitted": cap.stripes_emitted if cap else 0,
            "last_encode_ms": cap.last_encode_ms if cap else 0.0,
            "glass_to_glass_ms": self.g2g_percentiles(),
            "relays": [
                {"backlog": c.relay.backlog_bytes,
                 "sent": c.relay.sent_frames,
                 "dropped": c.relay.dropped_frames,
                 "acked": c.last_acked_frame,
                 "rtt_ms": round(c.ack_rtt_ms, 1),
                 "paused": c.paused}
                for c in self.clients.values()
            ],
        }
