"""Computer-use agent API (reference pixelflux start_computer_use,
SURVEY.md §2.3: an HTTP server exposing screenshot + input actions for
automation agents).

Routes (mounted under /computer-use by the supervisor):
  GET  /computer-use/screenshot           -> JPEG of the current display
  POST /computer-use/action               -> {"action": ..., ...}
      click   {x, y, button=1}            double_click {x, y}
      move    {x, y}                      type  {text}
      key     {keysym, down?}             scroll {direction, count}
"""

from __future__ import annotations

import base64
import logging

from aiohttp import web

logger = logging.getLogger("selkies.computer_use")


class ComputerUseAPI:
    def __init__(self, settings, input_dispatcher):
        self.settings = settings
        self.input = input_dispatcher

    def register(self, app: web.Application):
        app.router.add_get("/computer-use/screenshot", self.screenshot)
        app.router.add_post("/computer-use/action", self.action)

    async def screenshot(self, request: web.Request):
        import hipflux
        w, h = self.settings.resolution_wh
        backend = self.settings.capture_backend
        display = self.settings.display
        bgrx, fw, fh = hipflux._native.screenshot(backend, display, w, h)
        quality = int(request.query.get("quality", "85"))
        jpg = hipflux.jpeg_encode(bgrx, fw, fh, quality, False)
        return web.Response(body=jpg, content_type="image/jpeg")

    async def action(self, request: web.Request):
        try:
            body = await request.json()
        except Exception:
            raise web.HTTPBadRequest(reason="JSON body required")
        act = body.get("action", "")
        d = self.input
        if act == "move":
            d.on_message(f"m,{int(body['x'])},{int(body['y'])},0")
        elif act in ("click", "double_click"):
            x, y, btn = int(body["x"]), int(body["y"]), \
                int(body.get("button", 1))
            mask = {1: 1, 2: 2, 3: 4}.get(btn, 1)
            reps = 2 if act == "double_click" else 1
            for _ in range(reps):
                d.on_message(f"m,{x},{y},{mask}")
                d.on_message(f"m,{x},{y},0")
        elif act == "type":
            payload = base64.b64encode(
                str(body.get("text", "")).encode()).decode()
            d.on_message(f"co,{payload}")
        elif act == "key":
            keysym = int(body["keysym"])
            if body.get("down") is None:
                d.on_message(f"kd,{keysym}")
                d.on_message(f"ku,{keysym}")
            else:
                d.on_message(f"{'kd' if body['down'] else 'ku'},{keysym}")
        elif act == "scroll":
            d.on_message(
                f"sw,{body.get('direction', 'd')[0]},"
                f"{int(body.get('count', 1))}")
        else:
            raise web.HTTPBadRequest(reason=f"unknown action {act!r}")
        return web.json_response({"ok": True, "action": act})
