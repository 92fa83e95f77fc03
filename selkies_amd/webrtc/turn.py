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
