"""SRTP / SRTCP per RFC 3711 — AES-128-CM + HMAC-SHA1-80.

AES blocks come from OpenSSL EVP via ctypes (AES-128-ECB as the counter-mode
block function); HMAC-SHA1 from hashlib/hmac. The key-derivation function
and counter-mode layout are verified against the RFC 3711 appendix test
vectors in tests/test_webrtc_stack.py.
"""

from __future__ import annotations

import ctypes
import ctypes.util
import hmac
import hashlib
import struct

_crypto = ctypes.CDLL(ctypes.util.find_library("crypto") or "libcrypto.so.3")
for n, res, args in [
    ("EVP_CIPHER_CTX_new", ctypes.c_void_p, []),
    ("EVP_CIPHER_CTX_free", None, [ctypes.c_void_p]),
    ("EVP_aes_128_ecb", ctypes.c_void_p, []),
    ("EVP_EncryptInit_ex", ctypes.c_int,
     [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_char_p,
      ctypes.c_char_p]),
    ("EVP_EncryptUpdate", ctypes.c_int,
     [ctypes.c_void_p, ctypes.c_char_p, ctypes.POINTER(ctypes.c_int),
      ctypes.c_char_p, ctypes.c_int]),
    ("EVP_CIPHER_CTX_set_padding", ctypes.c_int,
     [ctypes.c_void_p, ctypes.c_int]),
]:
    fn = getattr(_crypto, n)
    fn.restype = res
    fn.argtypes = args


class _AesEcb:
    """Stateless AES-128 block encryptor (the CM counter core)."""

    def __init__(self, key: bytes):
        assert len(key) == 16
        self._ctx = _crypto.EVP_CIPHER_CTX_new()
        assert _crypto.EVP_EncryptInit_ex(self._ctx,
                                          _crypto.EVP_aes_128_ecb(), None,
                                          key, None) == 1
        _crypto.EVP_CIPHER_CTX_set_padding(self._ctx, 0)

    def blocks(self, data: bytes) -> bytes:
        out = ctypes.create_string_buffer(len(data) + 16)
        n = ctypes.c_int(0)
        assert _crypto.EVP_EncryptUpdate(self._ctx, out, ctypes.byref(n),
                                         data, len(data)) == 1
        return out.raw[:n.value]

    def __del__(self):
        if getattr(self, "_ctx", None):
            _crypto.EVP_CIPHER_CTX_free(self._ctx)


def _keystream(aes: _AesEcb, iv16: bytes, nbytes: int) -> bytes:
    """AES-CM keystream: AES(iv + i) for i = 0.. (counter in last 16 bits)."""
    nblocks = (nbytes + 15) // 16
    base = int.from_bytes(iv16, "big")
    ctr = b"".join((base + i).to_bytes(16, "big") for i in range(nblocks))
    return aes.blocks(ctr)[:nbytes]


def srtp_kdf(master_key: bytes, master_salt: bytes, label: int,
             length: int) -> bytes:
    """RFC 3711 §4.3 key derivation (key_derivation_rate = 0)."""
    assert len(master_salt) == 14
    x = bytearray(master_salt)
    x[7] ^= label           # key_id = label || r(=0), right-aligned
    iv = bytes(x) + b"\x00\x00"
    return _keystream(_AesEcb(master_key), iv, length)


class SrtpSession:
    """One direction of SRTP+SRTCP (sender or receiver use the same math)."""

    def __init__(self, master_key: bytes, master_salt: bytes):
        self.rtp_key = srtp_kdf(master_key, master_salt, 0, 16)
        self.rtp_auth = srtp_kdf(master_key, master_salt, 1, 20)
        self.rtp_salt = srtp_kdf(master_key, master_salt, 2, 14)
        self.rtcp_key = srtp_kdf(master_key, master_salt, 3, 16)
        self.rtcp_auth = srtp_kdf(master_key, master_salt, 4, 20)
        self.rtcp_salt = srtp_kdf(master_key, master_salt, 5, 14)
        self._rtp_aes = _AesEcb(self.rtp_key)
        self._rtcp_aes = _AesEcb(self.rtcp_key)
        # sender state per SSRC
        self.roc: dict[int, int] = {}
        self.last_seq: dict[int, int] = {}
        self.rtcp_index = 0
        # receiver state per SSRC: [roc, s_l, window_mask] where bit i of
        # window_mask marks index (roc<<16|s_l) - i as already received
        # (RFC 3711 §3.3.2 replay list, window REPLAY_WINDOW wide)
        self._rx: dict[int, list] = {}
        self._rtcp_rx_high = -1
        self._rtcp_rx_mask = 0

    REPLAY_WINDOW = 64

    # ---- RTP ---------------------------------------------------------------
    @staticmethod
    def _payload_offset(pkt: bytes) -> int:
        cc = pkt[0] & 0x0F
        off = 12 + 4 * cc
        if pkt[0] & 0x10:  # header extension
            ext_len = struct.unpack_from(">H", pkt, off + 2)[0]
            off += 4 + 4 * ext_len
        return off

    def _rtp_iv(self, ssrc: int, index: int) -> bytes:
        salt = int.from_bytes(self.rtp_salt + b"\x00\x00", "big")
        return (salt ^ (ssrc << 64) ^ (index << 16)).to_bytes(16, "big")

    def _index_for(self, ssrc: int, seq: int) -> int:
        """Sender-side packet index (monotone local counter)."""
        roc = self.roc.get(ssrc, 0)
        last = self.last_seq.get(ssrc)
        if last is not None and seq < last:  # wrapped
            roc += 1
            self.roc[ssrc] = roc
        self.last_seq[ssrc] = seq
        return (roc << 16) | seq

    def _rx_estimate(self, ssrc: int, seq: int):
        """RFC 3711 Appendix A index estimation; returns (v, index) without
        mutating state (state commits only after authentication)."""
        st = self._rx.get(ssrc)
        if st is None:
            return 0, seq
        roc, s_l, _ = st
        if s_l < 0x8000:
            v = roc - 1 if seq - s_l > 0x8000 and roc > 0 else roc
        else:
            v = roc + 1 if s_l - 0x8000 > seq else roc
        return v, (v << 16) | seq

    def _rx_replay_check(self, ssrc: int, index: int):
        """Raise if index is outside the window or already received."""
        st = self._rx.get(ssrc)
        if st is None:
            return
        roc, s_l, mask = st
        highest = (roc << 16) | s_l
        if index <= highest:
            delta = highest - index
            if delta >= self.REPLAY_WINDOW:
                raise ValueError("SRTP replay: index too old")
            if (mask >> delta) & 1:
                raise ValueError("SRTP replay: duplicate index")

    def _rx_commit(self, ssrc: int, v: int, seq: int, index: int):
        """Record a successfully authenticated index (§3.3.2 step 5)."""
        st = self._rx.get(ssrc)
        if st is None:
            self._rx[ssrc] = [v, seq, 1]
            return
        roc, s_l, mask = st
        highest = (roc << 16) | s_l
        if index > highest:
            shift = index - highest
            mask = ((mask << shift) | 1) & ((1 << self.REPLAY_WINDOW) - 1)
            st[0], st[1], st[2] = v, seq, mask
        else:
            st[2] = mask | (1 << (highest - index))

    def protect_rtp(self, pkt: bytes) -> bytes:
        seq = struct.unpack_from(">H", pkt, 2)[0]
        ssrc = struct.unpack_from(">I", pkt, 8)[0]
        index = self._index_for(ssrc, seq)
        off = self._payload_offset(pkt)
        ks = _keystream(self._rtp_aes, self._rtp_iv(ssrc, index),
                        len(pkt) - off)
        enc = pkt[:off] + bytes(a ^ b for a, b in zip(pkt[off:], ks))
        roc = index >> 16
        tag = hmac.new(self.rtp_auth, enc + struct.pack(">I", roc),
                       hashlib.sha1).digest()[:10]
        return enc + tag

    def unprotect_rtp(self, pkt: bytes) -> bytes:
        if len(pkt) < 22:
            raise ValueError("short SRTP packet")
        body, tag = pkt[:-10], pkt[-10:]
        seq = struct.unpack_from(">H", body, 2)[0]
        ssrc = struct.unpack_from(">I", body, 8)[0]
        v, index = self._rx_estimate(ssrc, seq)
        self._rx_replay_check(ssrc, index)
        roc = index >> 16
        expect = hmac.new(self.rtp_auth, body + struct.pack(">I", roc),
                          hashlib.sha1).digest()[:10]
        if not hmac.compare_digest(tag, expect):
            raise ValueError("SRTP auth failed")
        self._rx_commit(ssrc, v, seq, index)
        off = self._payload_offset(body)
        ks = _keystream(self._rtp_aes, self._rtp_iv(ssrc, index),
                        len(body) - off)
        return body[:off] + bytes(a ^ b for a, b in zip(body[off:], ks))

    # ---- RTCP --------------------------------------------------------------
    def _rtcp_iv(self, ssrc: int, index: int) -> bytes:
        salt = int.from_bytes(self.rtcp_salt + b"\x00\x00", "big")
        return (salt ^ (ssrc << 64) ^ (index << 16)).to_bytes(16, "big")

    def protect_rtcp(self, pkt: bytes) -> bytes:
        ssrc = struct.unpack_from(">I", pkt, 4)[0]
        index = self.rtcp_index
        self.rtcp_index = (self.rtcp_index + 1) & 0x7FFFFFFF
        ks = _keystream(self._rtcp_aes, self._rtcp_iv(ssrc, index),
                        len(pkt) - 8)
        enc = pkt[:8] + bytes(a ^ b for a, b in zip(pkt[8:], ks))
        e_index = struct.pack(">I", 0x80000000 | index)
        tag = hmac.new(self.rtcp_auth, enc + e_index,
                       hashlib.sha1).digest()[:10]
        return enc + e_index + tag

    def unprotect_rtcp(self, pkt: bytes) -> bytes:
        if len(pkt) < 8 + 4 + 10:
            raise ValueError("short SRTCP packet")
        body, e_index, tag = pkt[:-14], pkt[-14:-10], pkt[-10:]
        expect = hmac.new(self.rtcp_auth, body + e_index,
                          hashlib.sha1).digest()[:10]
        if not hmac.compare_digest(tag, expect):
            raise ValueError("SRTCP auth failed")
        idx = struct.unpack(">I", e_index)[0]
        index = idx & 0x7FFFFFFF
        # SRTCP replay list (§3.3.2 applies to SRTCP via the explicit index)
        if index <= self._rtcp_rx_high:
            delta = self._rtcp_rx_high - index
            if delta >= self.REPLAY_WINDOW or (self._rtcp_rx_mask >> delta) & 1:
                raise ValueError("SRTCP replay")
            self._rtcp_rx_mask |= 1 << delta
        else:
            shift = index - self._rtcp_rx_high
            self._rtcp_rx_mask = ((self._rtcp_rx_mask << shift) | 1) \
                & ((1 << self.REPLAY_WINDOW) - 1)
            self._rtcp_rx_high = index
        if not idx & 0x80000000:
            return body      # unencrypted SRTCP
        ssrc = struct.unpack_from(">I", body, 4)[0]
        ks = _keystream(self._rtcp_aes, self._rtcp_iv(ssrc, index),
                        len(body) - 8)
        return body[:8] + bytes(a ^ b for a, b in zip(body[8:], ks))


def is_rtcp(data: bytes) -> bool:
    """Demux RTP vs RTCP on the payload-type range (RFC 5761)."""
    return len(data) >= 2 and 192 <= data[1] <= 223
