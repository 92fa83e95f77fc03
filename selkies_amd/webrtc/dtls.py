"""DTLS 1.2 endpoint on ctypes-OpenSSL with SRTP keying (RFC 5764).

Drives the handshake through memory BIOs so the caller owns the UDP
socket: feed inbound datagrams with `put_datagram()`, ship outbound ones
from `take_datagrams()`. After `handshake_done`, `export_srtp_keys()`
returns the client/server SRTP key+salt per the use_srtp extension.
"""

from __future__ import annotations

import ctypes
import ctypes.util
import hashlib
import os
import subprocess
import tempfile
from typing import Optional

_ssl = ctypes.CDLL(ctypes.util.find_library("ssl") or "libssl.so.3")
_crypto = ctypes.CDLL(ctypes.util.find_library("crypto") or "libcrypto.so.3")


class _API:
    """Configured function table (one canonical wrapper per symbol —
    dlsym on libssl resolves libcrypto symbols through dependencies, so
    configuring `_ssl.X` and calling `_crypto.X` would mix wrappers)."""


api = _API()

for name, restype, argtypes in [
    ("DTLS_method", ctypes.c_void_p, []),
    ("SSL_CTX_new", ctypes.c_void_p, [ctypes.c_void_p]),
    ("SSL_CTX_free", None, [ctypes.c_void_p]),
    ("SSL_CTX_use_certificate_file", ctypes.c_int,
     [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_int]),
    ("SSL_CTX_use_PrivateKey_file", ctypes.c_int,
     [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_int]),
    ("SSL_CTX_set_tlsext_use_srtp", ctypes.c_int,
     [ctypes.c_void_p, ctypes.c_char_p]),
    ("SSL_CTX_set_verify", None,
     [ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p]),
    ("SSL_CTX_set_cipher_list", ctypes.c_int,
     [ctypes.c_void_p, ctypes.c_char_p]),
    ("SSL_new", ctypes.c_void_p, [ctypes.c_void_p]),
    ("SSL_free", None, [ctypes.c_void_p]),
    ("SSL_set_bio", None, [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p]),
    ("SSL_set_accept_state", None, [ctypes.c_void_p]),
    ("SSL_set_connect_state", None, [ctypes.c_void_p]),
    ("SSL_do_handshake", ctypes.c_int, [ctypes.c_void_p]),
    ("SSL_get_error", ctypes.c_int, [ctypes.c_void_p, ctypes.c_int]),
    ("SSL_is_init_finished", ctypes.c_int, [ctypes.c_void_p]),
    ("SSL_export_keying_material", ctypes.c_int,
     [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_size_t, ctypes.c_char_p,
      ctypes.c_size_t, ctypes.c_char_p, ctypes.c_size_t, ctypes.c_int]),
    ("SSL_get_selected_srtp_profile", ctypes.c_void_p, [ctypes.c_void_p]),
    ("SSL_read", ctypes.c_int,
     [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_int]),
    ("SSL_write", ctypes.c_int,
     [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_int]),
    ("BIO_new", ctypes.c_void_p, [ctypes.c_void_p]),
    ("BIO_s_mem", ctypes.c_void_p, []),
    ("BIO_write", ctypes.c_int,
     [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_int]),
    ("BIO_read", ctypes.c_int,
     [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_int]),
    ("BIO_ctrl_pending", ctypes.c_size_t, [ctypes.c_void_p]),
    ("BIO_set_mem_eof_return", None, []),
    # peer-certificate fingerprint (OpenSSL 3.x name first, 1.1 fallback)
    ("SSL_get1_peer_certificate", ctypes.c_void_p, [ctypes.c_void_p]),
    ("SSL_get_peer_certificate", ctypes.c_void_p, [ctypes.c_void_p]),
    ("i2d_X509", ctypes.c_int, [ctypes.c_void_p, ctypes.c_void_p]),
    ("X509_free", None, [ctypes.c_void_p]),
]:
    fn = None
    for lib in (_ssl, _crypto):
        try:
            fn = getattr(lib, name)
            break
        except AttributeError:
            continue
    if fn is None:
        continue
    fn.restype = restype
    fn.argtypes = argtypes
    setattr(api, name, fn)

SSL_ERROR_WANT_READ = 2
SSL_VERIFY_PEER = 0x01
SSL_VERIFY_FAIL_IF_NO_PEER_CERT = 0x02
SRTP_PROFILE = b"SRTP_AES128_CM_SHA1_80"

# verify callback that accepts any chain: WebRTC certs are self-signed and
# authenticated by comparing the cert fingerprint against the signaled
# a=fingerprint (RFC 8122), which peer_fingerprint() enables after the
# handshake. The callback only forces the peer to PRESENT a certificate.
_VERIFY_CB_T = ctypes.CFUNCTYPE(ctypes.c_int, ctypes.c_int, ctypes.c_void_p)
_accept_any_cert_cb = _VERIFY_CB_T(lambda preverify, store_ctx: 1)
BIO_CTRL_DGRAM_SET_MTU = 42  # not required; memory BIOs fragment for us


class SrtpProfileStruct(ctypes.Structure):
    _fields_ = [("name", ctypes.c_char_p), ("id", ctypes.c_ulong)]


class Certificate:
    """Self-signed cert+key on disk plus its SDP sha-256 fingerprint."""

    def __init__(self, directory: Optional[str] = None):
        self.dir = directory or tempfile.mkdtemp(prefix="selkies-dtls-")
        self.cert = os.path.join(self.dir, "cert.pem")
        self.key = os.path.join(self.dir, "key.pem")
        if not os.path.exists(self.cert):
            subprocess.run(
                ["openssl", "req", "-x509", "-newkey", "ec",
                 "-pkeyopt", "ec_paramgen_curve:prime256v1",
                 "-keyout", self.key, "-out", self.cert, "-days", "365",
                 "-nodes", "-subj", "/CN=selkies-amd"],
                check=True, capture_output=True)
        der = subprocess.run(
            ["openssl", "x509", "-in", self.cert, "-outform", "DER"],
            check=True, capture_output=True).stdout
        digest = hashlib.sha256(der).hexdigest().upper()
        self.fingerprint = ":".join(digest[i:i + 2]
                                    for i in range(0, len(digest), 2))


class DtlsEndpoint:
    """One DTLS association (server by default, client for tests)."""

    def __init__(self, cert: Certificate, server: bool = True):
        self.server = server
        self._ctx = api.SSL_CTX_new(api.DTLS_method())
        assert self._ctx, "SSL_CTX_new failed"
        assert api.SSL_CTX_use_certificate_file(
            self._ctx, cert.cert.encode(), 1) == 1
        assert api.SSL_CTX_use_PrivateKey_file(
            self._ctx, cert.key.encode(), 1) == 1
        assert api.SSL_CTX_set_tlsext_use_srtp(self._ctx, SRTP_PROFILE) == 0
        # Require the peer to present a certificate so its fingerprint can
        # be checked against the SDP a=fingerprint (servers with
        # verify off accepted any client). Chain validation stays off — the
        # cert is self-signed by design.
        api.SSL_CTX_set_verify(
            self._ctx, SSL_VERIFY_PEER | SSL_VERIFY_FAIL_IF_NO_PEER_CERT,
            _accept_any_cert_cb)

        self._ssl = api.SSL_new(self._ctx)
        self._rbio = api.BIO_new(api.BIO_s_mem())
        self._wbio = api.BIO_new(api.BIO_s_mem())
        api.SSL_set_bio(self._ssl, self._rbio, self._wbio)
        if server:
            api.SSL_set_accept_state(self._ssl)
        else:
            api.SSL_set_connect_state(self._ssl)
        self.handshake_done = False

    # -- datagram plumbing ---------------------------------------------------
    def put_datagram(self, data: bytes) -> None:
        api.BIO_write(self._rbio, data, len(data))
        self._pump()

    def take_datagrams(self) -> list[bytes]:
        out = []
        while api.BIO_ctrl_pending(self._wbio):
            buf = ctypes.create_string_buffer(4096)
            n = api.BIO_read(self._wbio, buf, 4096)
            if n <= 0:
                break
            out.append(buf.raw[:n])
        return out

    def start(self) -> None:
        """Client side: kick off the handshake (emits ClientHello)."""
        self._pump()

    def _pump(self) -> None:
        if self.handshake_done:
            return
        rc = api.SSL_do_handshake(self._ssl)
        if rc == 1:
            self.handshake_done = True
        else:
            err = api.SSL_get_error(self._ssl, rc)
            if err != SSL_ERROR_WANT_READ:
                raise RuntimeError(f"DTLS handshake error {err}")

    # -- application data (SCTP rides inside DTLS, RFC 8261) -----------------
    def send_app(self, data: bytes) -> None:
        """Encrypt one application datagram (drain with take_datagrams)."""
        rc = api.SSL_write(self._ssl, data, len(data))
        if rc <= 0:
            err = api.SSL_get_error(self._ssl, rc)
            raise RuntimeError(f"SSL_write error {err}")

    def recv_app(self) -> list:
        """Decrypted application datagrams received so far."""
        out = []
        while True:
            buf = ctypes.create_string_buffer(8192)
            n = api.SSL_read(self._ssl, buf, 8192)
            if n <= 0:
                break
            out.append(buf.raw[:n])
        return out

    # -- SRTP keying (RFC 5764 §4.2) -----------------------------------------
    def export_srtp_keys(self):
        """Returns ((client_key, client_salt), (server_key, server_salt))."""
        assert self.handshake_done
        prof = api.SSL_get_selected_srtp_profile(self._ssl)
        assert prof, "no SRTP profile negotiated"
        name = ctypes.cast(prof, ctypes.POINTER(SrtpProfileStruct))[0].name
        assert name == SRTP_PROFILE, name
        key_len, salt_len = 16, 14
        total = 2 * (key_len + salt_len)
        out = ctypes.create_string_buffer(total)
        rc = api.SSL_export_keying_material(
            self._ssl, out, total, b"EXTRACTOR-dtls_srtp", 19, None, 0, 0)
        assert rc == 1, "SSL_export_keying_material failed"
        m = out.raw
        ck = m[0:16]
        sk = m[16:32]
        cs = m[32:46]
        ss = m[46:60]
        return (ck, cs), (sk, ss)

    def peer_fingerprint(self) -> Optional[str]:
        """SHA-256 fingerprint of the peer's certificate, in the SDP
        colon-separated uppercase form, or None if no cert was presented."""
        get = getattr(api, "SSL_get1_peer_certificate",
                      getattr(api, "SSL_get_peer_certificate", None))
        if get is None or not self._ssl:
            return None
        x509 = get(self._ssl)
        if not x509:
            return None
        try:
            n = api.i2d_X509(x509, None)
            if n <= 0:
                return None
            buf = ctypes.create_string_buffer(n)
            p = ctypes.c_void_p(ctypes.addressof(buf))
            api.i2d_X509(x509, ctypes.byref(p))
            digest = hashlib.sha256(buf.raw[:n]).hexdigest().upper()
            return ":".join(digest[i:i + 2] for i in range(0, len(digest), 2))
        finally:
            api.X509_free(x509)

    def close(self):
        if self._ssl:
            api.SSL_free(self._ssl)   # frees the BIOs too
            self._ssl = None
        if self._ctx:
            api.SSL_CTX_free(self._ctx)
            self._ctx = None
