"""MI355X-native WebRTC transport (video over DTLS-SRTP, ICE-lite).

The reference vendors a full aiortc fork. This image has
none of its crypto dependencies (no pyopenssl/pylibsrtp), so this package
implements the transport directly on what EXISTS here:

  dtls.py  — DTLS 1.2 over ctypes-OpenSSL (memory BIOs, use_srtp
             extension, exported keying material)
  srtp.py  — RFC 3711 SRTP/SRTCP (AES-128-CM keystream via OpenSSL EVP,
             HMAC-SHA1-80 auth, AES-CM key derivation)
  rtp.py   — RTP H.264 packetization (STAP-A / FU-A), RTCP SR builder,
             RTCP parser (PLI/FIR -> keyframe requests)
  ice.py   — ICE-lite UDP endpoint (STUN binding responses with
             MESSAGE-INTEGRITY + FINGERPRINT)
  sdp.py   — offer parsing + ice-lite answer building

Scope (round 1): video sendonly per display, input/control over the
signaling WebSocket; audio m-line and SCTP data channels are round 2.
"""
