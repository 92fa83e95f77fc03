"""coturn-compatible TURN credential minting (use-auth-secret scheme).

The reference serves clients an RTC config with short-term HMAC TURN
credentials (webrtc_utils.py:160-215): username = "<expiry>:<user>",
credential = base64(HMAC-SHA1(shared_secret, username)) — the exact
scheme coturn implements as `use-auth-secret`/`static-auth-secret`.
"""

from __future__ import annotations

import base64
import hashlib
import hmac
import time
from typing import Optional

DEFAULT_STUN = ["stun:stun.l.google.com:19302"]


def hmac_credential(shared_secret: str, username: str) -> str:
    digest = hmac.new(shared_secret.encode(), username.encode(),
                      hashlib.sha1).digest()
    return base64.b64encode(digest).decode()


def generate_rtc_config(turn_host: str, turn_port: int, shared_secret: str,
                        user: Optional[str] = None, protocol: str = "udp",
                        turn_tls: bool = False,
                        stun_host: Optional[str] = None,
                        stun_port: int = 3478,
                        ttl_s: int = 24 * 3600,
                        now: Optional[float] = None) -> dict:
    """Build the client RTC configuration with minted TURN credentials."""
    user = (user or "selkies").replace(":", "")
    exp = int(now if now is not None else time.time()) + ttl_s
    username = f"{exp}:{user}"
    credential = hmac_credential(shared_secret, username)
    stun_urls = []
    if stun_host:
        stun_urls.append(f"stun:{stun_host}:{stun_port}")
    stun_urls.append(f"stun:{turn_host}:{turn_port}")
    for d in DEFAULT_STUN:
        if d not in stun_urls:
            stun_urls.append(d)
    scheme = "turns" if turn_tls else "turn"
    return {
        "lifetimeDuration": f"{ttl_s}s",
        "iceServers": [
            {"urls": stun_urls},
            {
                "urls": [f"{scheme}:{turn_host}:{turn_port}"
                         f"?transport={protocol}"],
                "username": username,
                "credential": credential,
            },
        ],
        "blockStatus": "allowed",
        "iceTransportPolicy": "all",
    }


# ---- resolution chain (reference webrtc_utils.get_rtc_configuration:1037:
# JSON file -> REST -> HMAC -> legacy static -> STUN-only) ------------------

def stun_only_config(stun_host: Optional[str] = None,
                     stun_port: int = 3478) -> dict:
    urls = []
    if stun_host:
        urls.append(f"stun:{stun_host}:{stun_port}")
    urls.extend(u for u in DEFAULT_STUN if u not in urls)
    return {"lifetimeDuration": "86400s",
            "iceServers": [{"urls": urls}],
            "blockStatus": "allowed", "iceTransportPolicy": "all"}


def static_rtc_config(turn_host: str, turn_port: int, username: str,
                      password: str, protocol: str = "udp",
                      turn_tls: bool = False,
                      stun_host: Optional[str] = None,
                      stun_port: int = 3478) -> dict:
    """Legacy long-term-credential TURN config (fixed user/pass)."""
    base = stun_only_config(stun_host or turn_host,
                            stun_port if stun_host else turn_port)
    scheme = "turns" if turn_tls else "turn"
    base["iceServers"].append({
        "urls": [f"{scheme}:{turn_host}:{turn_port}?transport={protocol}"],
        "username": username,
        "credential": password,
    })
    return base


def validate_rtc_config(cfg) -> bool:
    """Shape check for externally-sourced configs (file / REST)."""
    if not isinstance(cfg, dict):
        return False
    servers = cfg.get("iceServers")
    if not isinstance(servers, list) or not servers:
        return False
    for s in servers:
        if not isinstance(s, dict):
            return False
        urls = s.get("urls")
        if isinstance(urls, str):
            urls = [urls]
        if not isinstance(urls, list) or not urls:
            return False
        if not all(isinstance(u, str) and
                   u.split(":", 1)[0] in ("stun", "stuns", "turn", "turns")
                   for u in urls):
            return False
    return True


def load_rtc_config_json(path: str) -> Optional[dict]:
    """RTC config from a JSON file (reference: rtc-config-file chain
    link + watchdog RTCConfigFileMonitor). Returns None when missing
    or malformed."""
    import json
    import os
    try:
        if not path or not os.path.isfile(path):
            return None
        with open(path, "r", encoding="utf-8") as f:
            cfg = json.load(f)
    except (OSError, ValueError):
        return None
    return cfg if validate_rtc_config(cfg) else None


async def fetch_rest_config(uri: str, user: str = "selkies",
                            timeout_s: float = 5.0,
                            api_key: str = "",
                            username_header: str = "",
                            protocol: str = "",
                            protocol_header: str = "",
                            tls: bool = False,
                            tls_header: str = "") -> Optional[dict]:
    """RTC config from a TURN-REST service (reference RESTRTCMonitor):
    GET <uri>?service=turn&username=<user> returning an iceServers doc.
    The reference's header knobs are honored: an API key rides as a
    Bearer token, and the username/protocol/TLS hints go in the
    service's configured header names."""
    import aiohttp
    headers = {}
    if api_key:
        headers["Authorization"] = f"Bearer {api_key}"
    if username_header:
        headers[username_header] = user
    if protocol_header and protocol:
        headers[protocol_header] = protocol
    if tls_header:
        headers[tls_header] = "true" if tls else "false"
    try:
        async with aiohttp.ClientSession(
                timeout=aiohttp.ClientTimeout(total=timeout_s)) as sess:
            async with sess.get(uri, params={"service": "turn",
                                             "username": user},
                                headers=headers) as resp:
                if resp.status != 200:
                    return None
                cfg = await resp.json(content_type=None)
    except Exception:
        return None
    return cfg if validate_rtc_config(cfg) else None


async def fetch_cloudflare_turn(token_id: str, api_token: str,
                                ttl_s: int = 86400,
                                timeout_s: float = 5.0,
                                endpoint: str = "") -> Optional[dict]:
    """Short-lived TURN credentials from Cloudflare's TURN service
    (reference enable_cloudflare_turn): POST to the credentials endpoint
    with the API token; returns an rtc-config iceServers doc."""
    import aiohttp
    url = endpoint or (
        "https://rtc.live.cloudflare.com/v1/turn/keys/"
        f"{token_id}/credentials/generate")
    try:
        async with aiohttp.ClientSession(
                timeout=aiohttp.ClientTimeout(total=timeout_s)) as sess:
            async with sess.post(
                    url, json={"ttl": ttl_s},
                    headers={"Authorization": f"Bearer {api_token}"}
            ) as resp:
                if resp.status != 201 and resp.status != 200:
                    return None
                body = await resp.json(content_type=None)
    except Exception:
        return None
    ice = body.get("iceServers") or {}
    urls = ice.get("urls") or []
    if not urls:
        return None
    cfg = {"lifetimeDuration": f"{ttl_s}s",
           "iceServers": [{"urls": urls,
                           "username": ice.get("username", ""),
                           "credential": ice.get("credential", "")}]}
    return cfg if validate_rtc_config(cfg) else None


async def resolve_rtc_config(s, user: Optional[str] = None):
    """Walk the chain with the given settings object; returns
    (config, source) where source names the chain link that won."""
    user = user or "selkies"
    cfg = load_rtc_config_json(getattr(s, "rtc_config_json", ""))
    if cfg is not None:
        return cfg, "file"
    if getattr(s, "enable_cloudflare_turn", False) and \
            getattr(s, "cloudflare_turn_token_id", "") and \
            getattr(s, "cloudflare_turn_api_token", ""):
        cfg = await fetch_cloudflare_turn(
            s.cloudflare_turn_token_id, s.cloudflare_turn_api_token,
            endpoint=getattr(s, "_cloudflare_endpoint", ""))
        if cfg is not None:
            return cfg, "cloudflare"
    rest = getattr(s, "turn_rest_uri", "")
    if rest:
        cfg = await fetch_rest_config(
            rest, getattr(s, "turn_rest_username", "") or user,
            api_key=getattr(s, "turn_rest_api_key", ""),
            username_header=getattr(s, "turn_rest_username_auth_header",
                                    ""),
            protocol=getattr(s, "turn_protocol", ""),
            protocol_header=getattr(s, "turn_rest_protocol_header", ""),
            tls=getattr(s, "turn_tls", False),
            tls_header=getattr(s, "turn_rest_tls_header", ""))
        if cfg is not None:
            return cfg, "rest"
    if s.turn_host and s.turn_shared_secret:
        return generate_rtc_config(
            s.turn_host, s.turn_port, s.turn_shared_secret, user=user,
            protocol=s.turn_protocol, turn_tls=s.turn_tls,
            stun_host=s.stun_host or None, stun_port=s.stun_port), "hmac"
    if s.turn_host and getattr(s, "turn_username", "") and \
            getattr(s, "turn_password", ""):
        return static_rtc_config(
            s.turn_host, s.turn_port, s.turn_username, s.turn_password,
            protocol=s.turn_protocol, turn_tls=s.turn_tls,
            stun_host=s.stun_host or None, stun_port=s.stun_port), "static"
    return stun_only_config(s.stun_host or None, s.stun_port), "stun"


class RTCConfigFileMonitor:
    """mtime-polling watcher for the JSON config file (the reference
    uses watchdog; a poll task avoids the extra dependency and works on
    every filesystem). Calls `on_change(config)` with each new valid
    config."""

    def __init__(self, path: str, on_change, interval_s: float = 2.0):
        self.path = path
        self.on_change = on_change
        self.interval_s = interval_s
        self._task = None
        self._mtime = None

    def _stat(self):
        import os
        try:
            return os.stat(self.path).st_mtime_ns
        except OSError:
            return None

    async def _run(self):
        import asyncio
        self._mtime = self._stat()
        while True:
            await asyncio.sleep(self.interval_s)
            m = self._stat()
            if m != self._mtime:
                self._mtime = m
                cfg = load_rtc_config_json(self.path)
                if cfg is not None:
                    res = self.on_change(cfg)
                    if hasattr(res, "__await__"):
                        await res

    def start(self):
        import asyncio
        if self._task is None:
            self._task = asyncio.get_event_loop().create_task(self._run())

    def stop(self):
        if self._task is not None:
            self._task.cancel()
            self._task = None
