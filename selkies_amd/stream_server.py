"""Service supervisor + HTTP control plane (L4).

Fresh implementation of the reference supervisor behavior (SURVEY.md §2.1
stream_server.py CentralizedStreamServer): one aiohttp app serving
  * the WS data plane at /websockets (and /ws alias),
  * /api/status, /api/health, /api/stats, /api/settings,
  * Prometheus /metrics (opt-in, bearer-token gated),
  * the static HTML5 client,
with basic-auth/token middleware, WS origin checks, and TLS with
certificate hot-reload.
"""

from __future__ import annotations

import asyncio
import base64
import hmac
import json
import logging
import os
import ssl
import time
from typing import Optional

from aiohttp import web

from . import __version__
from .settings import AppSettings
from .streaming import StreamingService

logger = logging.getLogger("selkies.server")


class CentralizedStreamServer:
    def __init__(self, settings: AppSettings):
        self.settings = settings
        self.streaming = StreamingService(settings)
        from .transfers import TransferManager
        dirs = [d.strip() for d in settings.file_transfers.split(",")]
        self.transfers = TransferManager(
            settings.upload_dir,
            allow_upload="upload" in dirs,
            allow_download="download" in dirs)
        from .transfers import UplinkAllowance
        limit = settings.file_transfer_limit_mbps
        self.allowance = UplinkAllowance(
            self.transfers.pacer,
            cap_rate=(limit * 125_000.0 if limit > 0 else 100_000_000.0))
        self._uplink_task = None
        from concurrent.futures import ThreadPoolExecutor
        # single worker => rows land in arrival order (reference contract)
        self._stats_writer = ThreadPoolExecutor(max_workers=1)
        self._last_video_bytes = 0
        self.app = web.Application(middlewares=[self._auth_middleware])
        self.webrtc = None
        self.started_at = time.time()
        self._runner: Optional[web.AppRunner] = None
        self._ssl_ctx: Optional[ssl.SSLContext] = None
        self._cert_mtimes = (0.0, 0.0)
        self._register_routes()

    # ---- routes -------------------------------------------------------------
    async def _uplink_loop(self, interval: float = 2.0):
        """Sample video throughput + client ACK RTT; adapt the transfer
        pacer so file transfers only use spare uplink (UplinkAllowance)."""
        while True:
            await asyncio.sleep(interval)
            try:
                clients = list(self.streaming.clients.values())
                total = sum(c.relay.sent_bytes for c in clients)
                video_bps = max(0.0,
                                (total - self._last_video_bytes) / interval)
                self._last_video_bytes = total
                rtts = [c.ack_rtt_ms for c in clients if c.ack_rtt_ms > 0]
                self.allowance.observe(video_bps,
                                       max(rtts) if rtts else None)
            except Exception:
                logger.debug("uplink sampling failed", exc_info=True)

    def _register_routes(self):
        app = self.app
        app.router.add_get("/websockets", self._ws_entry)
        app.router.add_get("/ws", self._ws_entry)
        sub_ws = self.settings.subfolder.strip("/")
        if sub_ws:
            app.router.add_get(f"/{sub_ws}/websockets", self._ws_entry)
            app.router.add_get(f"/{sub_ws}/ws", self._ws_entry)
        app.router.add_get("/api/status", self.handle_status)
        app.router.add_get("/api/health", self.handle_health)
        app.router.add_get("/api/stats", self.handle_stats)
        app.router.add_get("/api/settings", self.handle_settings)
        app.router.add_get("/metrics", self.handle_metrics)
        app.router.add_get("/api/turn", self.handle_turn)
        app.router.add_post("/api/webrtc-stats", self.handle_webrtc_stats)
        app.router.add_post("/api/mode", self.handle_mode)
        app.router.add_post("/api/tokens", self.handle_tokens)
        app.router.add_get("/api/tokens", self.handle_tokens_list)
        from .computer_use import ComputerUseAPI
        ComputerUseAPI(self.settings, self.streaming.input).register(app)
        app.router.add_post("/api/webrtc/offer", self.handle_webrtc_offer)
        app.router.add_post("/api/upload", self.handle_upload)
        app.router.add_get("/api/download", self.handle_download)
        # direct-download style (reference '/files/<name>')
        app.router.add_get("/files/{name:.+}", self.handle_file_direct)
        app.router.add_get("/api/files", self.handle_files)
        web_dir = self.settings.web_root or os.path.join(
            os.path.dirname(__file__), "web")
        if os.path.isdir(web_dir):
            app.router.add_get("/", self._index)
            app.router.add_get("/dashboard", self._dashboard)
            app.router.add_static("/static", web_dir)
        # reverse-proxy subpath: mirror the app under /<subfolder>/
        sub = self.settings.subfolder.strip("/")
        if sub:
            app.router.add_get(f"/{sub}", self._index)
            app.router.add_get(f"/{sub}/", self._index)
            if os.path.isdir(web_dir):
                app.router.add_static(f"/{sub}/static", web_dir)

    def _web_dir(self):
        return self.settings.web_root or os.path.join(
            os.path.dirname(__file__), "web")

    async def _index(self, request):
        return web.FileResponse(os.path.join(self._web_dir(),
                                             "index.html"))

    async def _dashboard(self, request):
        return web.FileResponse(os.path.join(self._web_dir(),
                                             "dashboard.html"))

    async def _ws_entry(self, request):
        if not self._ws_origin_allowed(request):
            raise web.HTTPForbidden(reason="origin not allowed")
        return await self.streaming.ws_handler(request)

    def _ws_origin_allowed(self, request) -> bool:
        allowed = [o.strip() for o in
                   self.settings.allowed_ws_origins.split(",") if o.strip()]
        if not allowed:
            return True
        origin = request.headers.get("Origin", "")
        return origin in allowed

    # ---- auth ---------------------------------------------------------------
    @web.middleware
    async def _auth_middleware(self, request, handler):
        s = self.settings
        if request.path == "/api/health":
            return await handler(request)
        token = s.auth_token
        if token:
            supplied = request.headers.get("Authorization", "")
            ok = hmac.compare_digest(supplied, f"Bearer {token}")
            if not ok:
                # query-parameter tokens leak into logs/history; accept
                # them only where headers are impossible: the WS upgrade
                is_ws = (request.headers.get("Upgrade", "").lower()
                         == "websocket")
                qtoken = request.query.get("token", "") if is_ws else ""
                ok = bool(qtoken) and hmac.compare_digest(qtoken, token)
            if not ok:
                raise web.HTTPUnauthorized(reason="token required")
        elif s.enable_basic_auth:
            hdr = request.headers.get("Authorization", "")
            ok = False
            if hdr.startswith("Basic "):
                try:
                    user, _, pw = base64.b64decode(
                        hdr[6:]).decode().partition(":")
                    ok = (hmac.compare_digest(user, s.basic_auth_user) and
                          hmac.compare_digest(pw, s.basic_auth_password))
                    # second password grants view-only access (reference
                    # basic_auth_viewonly_password)
                    if (not ok and s.basic_auth_viewonly_password and
                            hmac.compare_digest(
                                user, s.basic_auth_user) and
                            hmac.compare_digest(
                                pw, s.basic_auth_viewonly_password)):
                        ok = True
                        request["forced_role"] = "viewer"
                except Exception:
                    ok = False
            if not ok:
                raise web.HTTPUnauthorized(
                    reason="auth required",
                    headers={"WWW-Authenticate": 'Basic realm="selkies"'})
        return await handler(request)

    # ---- API handlers -------------------------------------------------------
    async def handle_status(self, request):
        return web.json_response({
            "version": __version__,
            "mode": self.settings.mode,
            "uptime_s": round(time.time() - self.started_at, 1),
            "encoder": self.settings.encoder,
            "resolution": self.settings.resolution,
            "clients": len(self.streaming.clients),
        })

    async def handle_health(self, request):
        return web.json_response({"ok": True})

    async def handle_stats(self, request):
        stats = {"streaming": self.streaming.stats(),
                 "gpu": gpu_stats_snapshot()}
        return web.json_response(stats)

    async def handle_settings(self, request):
        return web.json_response(
            self.settings.build_client_settings_payload())

    async def handle_metrics(self, request):
        """Prometheus exposition (reference: /api/metrics gated by
        --enable-metrics-http + bearer token)."""
        s = self.settings
        if not s.enable_metrics_http:
            raise web.HTTPNotFound()
        if s.metrics_http_token:
            if request.headers.get("Authorization") != \
                    f"Bearer {s.metrics_http_token}":
                raise web.HTTPUnauthorized()
        from prometheus_client import (CollectorRegistry, Counter, Gauge,
                                       generate_latest)
        st = self.streaming.stats()
        reg = CollectorRegistry()
        g = Gauge("selkies_clients", "connected clients", registry=reg)
        g.set(st["clients"])
        c = Counter("selkies_frames_encoded", "encoded frames", registry=reg)
        c.inc(st["frames_encoded"])
        c2 = Counter("selkies_stripes_emitted", "emitted stripes",
                     registry=reg)
        c2.inc(st["stripes_emitted"])
        g2 = Gauge("selkies_last_encode_ms", "last frame encode ms",
                   registry=reg)
        g2.set(st["last_encode_ms"])
        g3 = Gauge("selkies_transfer_rate_bytes", "file-transfer budget",
                   registry=reg)
        g3.set(self.transfers.pacer.rate)
        if self.webrtc is not None:
            ws = self.webrtc.stats()
            g4 = Gauge("selkies_webrtc_peers", "connected WebRTC peers",
                       registry=reg)
            g4.set(ws.get("connected", 0))
            g5 = Gauge("selkies_webrtc_video_kbps",
                       "congestion-controlled video bitrate", registry=reg)
            g5.set(getattr(self.webrtc, "_video_kbps", 0) or 0)
        for gpu in gpu_stats_snapshot():
            gg = Gauge(f"selkies_gpu_busy_percent_{gpu['card']}",
                       "amdgpu busy", registry=reg)
            gg.set(gpu.get("busy_percent", 0))
        return web.Response(body=generate_latest(reg),
                            content_type="text/plain; version=0.0.4")

    async def handle_mode(self, request):
        """Runtime transport-mode switch (reference switch_to_mode,
        stream_server.py:1300). POST {"mode": "websockets"|"webrtc"}:
        switching to webrtc eagerly brings the RTC stack up (UDP
        endpoint live before clients arrive); switching back tears it
        down and drops its peers. The WS plane always stays up — it
        carries control/signaling for both modes."""
        try:
            body = await request.json()
            mode = body["mode"]
        except Exception:
            raise web.HTTPBadRequest(reason='JSON {"mode": ...} required')
        if mode not in ("websockets", "webrtc"):
            raise web.HTTPBadRequest(reason="unknown mode")
        if mode == "webrtc":
            if self.webrtc is None:
                from .webrtc_service import WebRTCService
                self.webrtc = WebRTCService(self.settings, self.streaming)
                await self.webrtc.start(self.settings.webrtc_udp_port)
        else:
            if self.webrtc is not None:
                await self.webrtc.stop()
                self.webrtc = None
        self.settings.mode = mode
        return web.json_response({"mode": mode,
                                  "webrtc_active": self.webrtc is not None})

    def _check_master_token(self, request):
        s = self.settings
        if not (s.enable_collab and s.master_token):
            raise web.HTTPNotFound()
        supplied = request.headers.get("Authorization", "")
        if not hmac.compare_digest(supplied,
                                   f"Bearer {s.master_token}"):
            raise web.HTTPUnauthorized(reason="master token required")

    async def handle_tokens(self, request):
        """Replace/patch the collab token table and reconcile live
        clients (reference user-token table + reconcile_clients). Body:
        {"set": {token: {"role": "...", "seat": n|null}},
         "revoke": [token, ...]}  — roles: controller|viewer."""
        self._check_master_token(request)
        try:
            body = await request.json()
        except Exception:
            raise web.HTTPBadRequest(reason="JSON body required")
        table = self.streaming.user_tokens
        for tok, perms in (body.get("set") or {}).items():
            role = perms.get("role", "viewer")
            if role not in ("controller", "viewer"):
                raise web.HTTPBadRequest(reason=f"bad role {role!r}")
            seat = perms.get("seat")
            if seat is not None and not (1 <= int(seat) <= 3):
                raise web.HTTPBadRequest(reason="seat must be 1..3")
            table[str(tok)] = {"role": role,
                               "seat": int(seat) if seat else None}
        for tok in body.get("revoke") or []:
            table.pop(str(tok), None)
        await self.streaming.reconcile_clients()
        return web.json_response({"tokens": len(table)})

    async def handle_tokens_list(self, request):
        self._check_master_token(request)
        return web.json_response(
            {t: p for t, p in self.streaming.user_tokens.items()})

    # ---- WebRTC signaling ---------------------------------------------------
    async def handle_webrtc_offer(self, request):
        """HTTP signaling: browser POSTs its SDP offer, gets the ice-lite
        answer. (Both transports are always live — WS is the default,
        WebRTC the alternative; reference keeps them at feature parity.)"""
        if self.webrtc is None:
            from .webrtc_service import WebRTCService
            self.webrtc = WebRTCService(self.settings, self.streaming)
            await self.webrtc.start(self.settings.webrtc_udp_port)
        try:
            body = await request.json()
            offer_sdp = body["sdp"]
        except Exception:
            raise web.HTTPBadRequest(reason="JSON {sdp} required")
        try:
            answer = self.webrtc.handle_offer(offer_sdp)
        except Exception as exc:
            raise web.HTTPBadRequest(reason=f"offer rejected: {exc}")
        return web.json_response({"type": "answer", "sdp": answer})

    # ---- file transfers -----------------------------------------------------
    async def handle_upload(self, request):
        name = request.query.get("name", "")
        if not name:
            raise web.HTTPBadRequest(reason="name required")

        async def chunks():
            async for chunk in request.content.iter_chunked(256 * 1024):
                yield chunk

        try:
            result = await self.transfers.upload(name, chunks())
        except PermissionError as exc:
            raise web.HTTPForbidden(reason=str(exc))
        return web.json_response(result)

    async def handle_download(self, request):
        name = request.query.get("name", "")
        try:
            path = self.transfers.resolve(name)
            if not os.path.isfile(path):
                raise web.HTTPNotFound()
            resp = web.StreamResponse(headers={
                "Content-Disposition":
                    f'attachment; filename="{os.path.basename(path)}"'})
            await resp.prepare(request)
            async for chunk in self.transfers.stream_file(name):
                await resp.write(chunk)
            await resp.write_eof()
            return resp
        except PermissionError as exc:
            raise web.HTTPForbidden(reason=str(exc))

    async def handle_file_direct(self, request):
        """GET /files/<name>: direct download path (reference '/files/'
        surface); same traversal-safe resolution as /api/download."""
        name = request.match_info.get("name", "")
        try:
            path = self.transfers.resolve(name)
        except PermissionError as exc:
            raise web.HTTPForbidden(reason=str(exc))
        if not os.path.isfile(path):
            raise web.HTTPNotFound()
        return web.FileResponse(path)

    async def handle_files(self, request):
        try:
            return web.json_response(
                self.transfers.listdir(request.query.get("path", "")))
        except PermissionError as exc:
            raise web.HTTPForbidden(reason=str(exc))

    # ---- TLS hot reload -----------------------------------------------------
    def _build_ssl(self) -> Optional[ssl.SSLContext]:
        s = self.settings
        if not (s.enable_https and s.https_cert and s.https_key):
            return None
        ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
        ctx.load_cert_chain(s.https_cert, s.https_key)
        try:
            self._cert_mtimes = (os.path.getmtime(s.https_cert),
                                 os.path.getmtime(s.https_key))
        except OSError:
            pass
        return ctx

    async def _watch_certs(self):
        while True:
            await asyncio.sleep(
                max(5, int(self.settings.cert_reload_interval)))
            s = self.settings
            if not (s.enable_https and s.https_cert):
                continue
            try:
                mt = (os.path.getmtime(s.https_cert),
                      os.path.getmtime(s.https_key))
            except OSError:
                continue
            if mt != self._cert_mtimes and self._ssl_ctx is not None:
                logger.info("TLS certificates changed; reloading")
                try:
                    self._ssl_ctx.load_cert_chain(s.https_cert, s.https_key)
                    self._cert_mtimes = mt
                except Exception as exc:
                    logger.error("cert reload failed: %r", exc)

    async def handle_webrtc_stats(self, request):
        """Append client-reported getStats rows to a per-day CSV
        (reference webrtc_utils.py:1201-1533: sanitized, ordered via a
        dedicated single-worker pool)."""
        s = self.settings
        if not s.enable_webrtc_statistics:
            return web.json_response({"error": "disabled"}, status=404)
        try:
            entries = await request.json()
        except Exception:
            raise web.HTTPBadRequest(text="invalid json")
        if not isinstance(entries, list):
            raise web.HTTPBadRequest(text="expected a list")
        rows = []
        for e in entries[:512]:
            if not isinstance(e, dict):
                continue
            row = {}
            for k, v in list(e.items())[:48]:
                if not isinstance(k, str) or len(k) > 64:
                    continue
                if isinstance(v, bool):
                    v = int(v)
                if isinstance(v, (int, float)):
                    row[k] = v
                elif isinstance(v, str) and len(v) <= 128:
                    # CSV-safe: no separators/newlines survive
                    row[k] = v.replace(",", ";").replace("\n", " ")
            if row:
                rows.append(row)
        if rows:
            loop = asyncio.get_running_loop()
            await loop.run_in_executor(self._stats_writer,
                                       self._append_stats_rows, rows)
        return web.json_response({"accepted": len(rows)})

    def _append_stats_rows(self, rows):
        import csv
        import datetime
        s = self.settings
        os.makedirs(s.webrtc_statistics_dir, exist_ok=True)
        day = datetime.date.today().isoformat()
        path = os.path.join(s.webrtc_statistics_dir, f"webrtc-{day}.csv")
        keys = sorted({k for r in rows for k in r})
        new = not os.path.exists(path)
        with open(path, "a", newline="") as f:
            w = csv.writer(f)
            if new:
                w.writerow(["ts"] + keys)
            ts = time.time()
            for r in rows:
                w.writerow([f"{ts:.3f}"] + [r.get(k, "") for k in keys])

    async def handle_turn(self, request):
        """RTC config via the resolution chain (reference
        webrtc_utils.get_rtc_configuration: JSON file -> TURN-REST ->
        coturn HMAC -> legacy static -> STUN-only)."""
        from .webrtc.turn import resolve_rtc_config
        cfg, source = await resolve_rtc_config(
            self.settings, request.query.get("user"))
        return web.json_response(cfg, headers={"X-RTC-Source": source})

    # ---- lifecycle ----------------------------------------------------------
    async def start(self):
        self._runner = web.AppRunner(self.app)
        await self._runner.setup()
        self._ssl_ctx = self._build_ssl()
        s = self.settings
        if s.unix_socket:
            site = web.UnixSite(self._runner, s.unix_socket,
                                ssl_context=self._ssl_ctx)
        else:
            site = web.TCPSite(self._runner, s.addr, s.port,
                               ssl_context=self._ssl_ctx)
        await site.start()
        if self._ssl_ctx is not None:
            asyncio.get_running_loop().create_task(self._watch_certs())
        self._uplink_task = asyncio.get_running_loop().create_task(
            self._uplink_loop())
        if s.rtc_config_json:
            # hot-reload the file-sourced ICE config and push it to
            # connected clients (reference RTCConfigFileMonitor)
            from .webrtc.turn import RTCConfigFileMonitor
            self._rtc_monitor = RTCConfigFileMonitor(
                s.rtc_config_json, self._on_rtc_config_change)
            self._rtc_monitor.start()
        logger.info("serving on %s:%s (mode=%s)", s.addr, s.port, s.mode)

    async def _on_rtc_config_change(self, cfg):
        payload = "RTC_CONFIG," + json.dumps(cfg)
        for state in list(self.streaming.clients.values()):
            try:
                await state.ws.send_str(payload)
            except Exception:
                pass

    async def stop(self):
        if getattr(self, "_rtc_monitor", None) is not None:
            self._rtc_monitor.stop()
            self._rtc_monitor = None
        if self._uplink_task is not None:
            self._uplink_task.cancel()
        self.streaming.stop_capture()
        self.streaming.stop_audio()
        if self.webrtc is not None:
            await self.webrtc.stop()
        if self._runner is not None:
            await self._runner.cleanup()


def gpu_stats_snapshot() -> list[dict]:
    """AMD GPU load/memory via amdgpu sysfs (works without ROCm tools —
    the reference's sysfs backfill approach, gpu_stats.py:5-14)."""
    out = []
    base = "/sys/class/drm"
    try:
        cards = sorted(c for c in os.listdir(base)
                       if c.startswith("card") and c[4:].isdigit())
    except OSError:
        return out
    for card in cards:
        dev = os.path.join(base, card, "device")
        entry = {"card": card}

        def read(name):
            try:
                with open(os.path.join(dev, name)) as f:
                    return f.read().strip()
            except OSError:
                return None

        busy = read("gpu_busy_percent")
        if busy is None:
            continue
        entry["busy_percent"] = int(busy)
        vu = read("mem_info_vram_used")
        vt = read("mem_info_vram_total")
        if vu and vt:
            entry["vram_used_mb"] = int(vu) // (1 << 20)
            entry["vram_total_mb"] = int(vt) // (1 << 20)
        out.append(entry)
    return out
