"""ICE-lite STUN responder (RFC 5389/8445 subset).

An ice-lite endpoint only answers binding requests on its host candidate;
the browser does the connectivity checking. MESSAGE-INTEGRITY uses the
local ice-pwd; FINGERPRINT is CRC32 ^ 0x5354554e.
"""

from __future__ import annotations

import hmac
import hashlib
import os
import secrets
import struct
import zlib

MAGIC = 0x2112A442

ATTR_USERNAME = 0x0006
ATTR_MESSAGE_INTEGRITY = 0x0008
ATTR_XOR_MAPPED_ADDRESS = 0x0020
ATTR_FINGERPRINT = 0x8028
ATTR_USE_CANDIDATE = 0x0025
ATTR_ICE_CONTROLLING = 0x802A
ATTR_PRIORITY = 0x0024


def make_ice_credentials():
    ufrag = secrets.token_urlsafe(3)[:4]
    pwd = secrets.token_urlsafe(18)[:24]
    return ufrag, pwd


def is_stun(data: bytes) -> bool:
    return len(data) >= 20 and data[0] < 4 and \
        struct.unpack_from(">I", data, 4)[0] == MAGIC


def parse_attrs(data: bytes) -> dict:
    attrs = {}
    off = 20
    while off + 4 <= len(data):
        atype, alen = struct.unpack_from(">HH", data, off)
        attrs[atype] = data[off + 4:off + 4 + alen]
        off += 4 + ((alen + 3) & ~3)
    return attrs


def _attr(atype: int, value: bytes) -> bytes:
    pad = (-len(value)) % 4
    return struct.pack(">HH", atype, len(value)) + value + b"\x00" * pad


def binding_response(request: bytes, addr, local_pwd: str) -> bytes:
    """Build a success response with XOR-MAPPED-ADDRESS + MI + FINGERPRINT."""
    txid = request[8:20]
    ip, port = addr[0], addr[1]
    ip_bytes = bytes(int(x) for x in ip.split("."))
    xport = port ^ (MAGIC >> 16)
    xip = bytes(b ^ m for b, m in zip(ip_bytes,
                                      struct.pack(">I", MAGIC)))
    attrs = _attr(ATTR_XOR_MAPPED_ADDRESS,
                  struct.pack(">BBH", 0, 1, xport) + xip)

    def hdr(length):
        return struct.pack(">HHI", 0x0101, length, MAGIC) + txid

    # MESSAGE-INTEGRITY over header(with adjusted len)+attrs
    mi_len = len(attrs) + 24
    mi = hmac.new(local_pwd.encode(), hdr(mi_len) + attrs,
                  hashlib.sha1).digest()
    attrs += _attr(ATTR_MESSAGE_INTEGRITY, mi)
    fp_len = len(attrs) + 8
    crc = (zlib.crc32(hdr(fp_len) + attrs) ^ 0x5354554E) & 0xFFFFFFFF
    attrs += _attr(ATTR_FINGERPRINT, struct.pack(">I", crc))
    return hdr(len(attrs)) + attrs


def binding_request(username: str, remote_pwd: str,
                    txid: bytes = b"") -> bytes:
    """Build an authenticated binding request (USERNAME + MESSAGE-INTEGRITY
    + FINGERPRINT) as an ICE client would send it (RFC 8445 §7.2.2)."""
    import secrets
    txid = txid or secrets.token_bytes(12)
    attrs = _attr(ATTR_USERNAME, username.encode())

    def hdr(length):
        return struct.pack(">HHI", 0x0001, length, MAGIC) + txid

    mi = hmac.new(remote_pwd.encode(), hdr(len(attrs) + 24) + attrs,
                  hashlib.sha1).digest()
    attrs += _attr(ATTR_MESSAGE_INTEGRITY, mi)
    crc = (zlib.crc32(hdr(len(attrs) + 8) + attrs) ^ 0x5354554E) & 0xFFFFFFFF
    attrs += _attr(ATTR_FINGERPRINT, struct.pack(">I", crc))
    return hdr(len(attrs)) + attrs


def parse_username(request: bytes) -> str:
    """The USERNAME attribute of a binding request ('' if absent)."""
    attrs_raw = request[20:]
    off = 0
    while off + 4 <= len(attrs_raw):
        atype, alen = struct.unpack_from(">HH", attrs_raw, off)
        if atype == ATTR_USERNAME:
            try:
                return attrs_raw[off + 4:off + 4 + alen].decode()
            except UnicodeDecodeError:
                return ""
        off += 4 + ((alen + 3) & ~3)
    return ""


def verify_request_integrity(request: bytes, local_pwd: str) -> bool:
    """Check the browser's binding-request MESSAGE-INTEGRITY."""
    attrs_raw = request[20:]
    off = 0
    while off + 4 <= len(attrs_raw):
        atype, alen = struct.unpack_from(">HH", attrs_raw, off)
        if atype == ATTR_MESSAGE_INTEGRITY:
            mi = attrs_raw[off + 4:off + 4 + alen]
            covered_len = off + 24
            covered = struct.pack(">HHI", struct.unpack_from(">H", request)[0],
                                  covered_len, MAGIC) + request[8:20] + \
                attrs_raw[:off]
            expect = hmac.new(local_pwd.encode(), covered,
                              hashlib.sha1).digest()
            return hmac.compare_digest(mi, expect)
        off += 4 + ((alen + 3) & ~3)
    return False


def default_host_ip() -> str:
    """Pick the host's primary outbound IP (no traffic actually sent)."""
    import socket
    s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    try:
        s.connect(("10.255.255.255", 1))
        return s.getsockname()[0]
    except OSError:
        return "127.0.0.1"
    finally:
        s.close()
