"""From-spec decoder for the hipflux CELT-class Opus-framed audio codec
(native/cpu/opus/celt.{h,cpp} documents the conformance ledger: TOC +
range coder are RFC 6716-exact; coarse energy / allocation-row / PVQ
split layers are the in-tree profile).

The range DECODER follows RFC 6716 §4.1.2-§4.1.4 independently; the
round-trip fuzz in tests/test_opus.py anchors it against the C++
encoder bin by bin."""

import math

import numpy as np

EC_SYM_BITS = 8
EC_SYM_MAX = 0xFF
EC_CODE_BITS = 32
EC_CODE_TOP = 1 << 31
EC_CODE_BOT = EC_CODE_TOP >> EC_SYM_BITS
EC_CODE_EXTRA = (EC_CODE_BITS - 2) % EC_SYM_BITS + 1    # 7
EC_UINT_BITS = 8

FRAME = 960
OVERLAP = 120
NBANDS = 21
BAND_BINS = [0, 8, 16, 24, 32, 40, 48, 56, 64, 80, 96, 112, 128, 160,
             192, 224, 272, 320, 384, 480, 624, 800]
EMEANS = [6.4375, 6.25, 5.75, 5.3125, 5.0625, 4.8125, 4.5, 4.375,
          4.875, 4.6875, 4.5625, 4.4375, 4.875, 4.625, 4.3125, 4.5,
          4.375, 4.625, 4.75, 4.4375, 3.75]
ALLOC = [
    [0] * 21,
    [90, 80, 75, 69, 63, 56, 49, 40, 34, 29, 20, 18, 10, 0, 0, 0, 0, 0,
     0, 0, 0],
    [110, 100, 90, 84, 78, 71, 65, 58, 51, 45, 39, 32, 26, 20, 12, 0, 0,
     0, 0, 0, 0],
    [118, 110, 103, 93, 86, 80, 75, 70, 65, 59, 53, 47, 40, 31, 23, 15,
     4, 0, 0, 0, 0],
    [126, 119, 112, 104, 95, 89, 83, 78, 72, 66, 60, 54, 47, 39, 32, 25,
     17, 12, 1, 0, 0],
    [134, 127, 120, 114, 103, 97, 91, 85, 78, 72, 66, 60, 54, 47, 41,
     35, 29, 23, 16, 10, 1],
    [144, 137, 130, 124, 113, 107, 101, 95, 88, 82, 76, 70, 64, 57, 51,
     45, 39, 33, 26, 15, 1],
    [152, 145, 138, 132, 123, 117, 111, 105, 98, 92, 86, 80, 74, 67, 61,
     55, 49, 43, 36, 20, 1],
    [162, 155, 148, 142, 133, 127, 121, 115, 108, 102, 96, 90, 84, 77,
     71, 65, 59, 53, 46, 30, 1],
    [172, 165, 158, 152, 143, 137, 131, 125, 118, 112, 106, 100, 94, 87,
     81, 75, 69, 63, 56, 45, 20],
    [200] * 21,
]


def ec_ilog(v):
    return v.bit_length()


class RangeDecoder:
    """RFC 6716 §4.1 range decoder with raw bits from the buffer end."""

    def __init__(self, buf: bytes):
        self.buf = buf
        self.storage = len(buf)
        self.offs = 0
        self.end_offs = 0
        self.end_window = 0
        self.nend_bits = 0
        self.nbits_total = EC_CODE_BITS + 1 - (
            (EC_CODE_BITS - EC_CODE_EXTRA) // EC_SYM_BITS) * EC_SYM_BITS
        self.rng = 1 << EC_CODE_EXTRA
        self.rem = self._read_byte()
        self.val = self.rng - 1 - (self.rem >> (EC_SYM_BITS -
                                                EC_CODE_EXTRA))
        self.error = False
        self._normalize()

    def _read_byte(self):
        if self.offs < self.storage:
            b = self.buf[self.offs]
            self.offs += 1
            return b
        return 0

    def _read_byte_from_end(self):
        if self.end_offs < self.storage:
            self.end_offs += 1
            return self.buf[self.storage - self.end_offs]
        return 0

    def _normalize(self):
        while self.rng <= EC_CODE_BOT:
            self.nbits_total += EC_SYM_BITS
            self.rng = (self.rng << EC_SYM_BITS) & 0xFFFFFFFF
            sym = self.rem
            self.rem = self._read_byte()
            sym = ((sym << EC_SYM_BITS) | self.rem) >> (
                EC_SYM_BITS - EC_CODE_EXTRA)
            self.val = ((self.val << EC_SYM_BITS) +
                        (EC_SYM_MAX & ~sym)) & (EC_CODE_TOP - 1)

    def decode(self, ft):
        self.ext = self.rng // ft
        s = self.val // self.ext
        return ft - min(s + 1, ft)

    def dec_update(self, fl, fh, ft):
        s = self.ext * (ft - fh)
        self.val -= s
        self.rng = self.ext * (fh - fl) if fl > 0 else self.rng - s
        self._normalize()

    def dec_bit_logp(self, logp):
        r = self.rng
        d = self.val
        s = r >> logp
        ret = 1 if d < s else 0
        if not ret:
            self.val = d - s
        self.rng = s if ret else r - s
        self._normalize()
        return ret

    def dec_icdf(self, icdf, ftb):
        s = self.rng
        d = self.val
        r = s >> ftb
        ret = -1
        while True:
            ret += 1
            t = s
            s = r * icdf[ret]
            if d >= s:
                break
        self.val = d - s
        self.rng = t - s
        self._normalize()
        return ret

    def dec_uint(self, ft):
        ft -= 1
        ftb = ec_ilog(ft)
        if ftb > EC_UINT_BITS:
            ftb -= EC_UINT_BITS
            fth = (ft >> ftb) + 1
            s = self.decode(fth)
            self.dec_update(s, s + 1, fth)
            t = (s << ftb) | self.dec_bits(ftb)
            if t <= ft:
                return t
            self.error = True
            return ft
        s = self.decode(ft + 1)
        self.dec_update(s, s + 1, ft + 1)
        return s

    def dec_bits(self, bits):
        window = self.end_window
        available = self.nend_bits
        while available < bits:
            window |= self._read_byte_from_end() << available
            available += EC_SYM_BITS
        ret = window & ((1 << bits) - 1)
        window >>= bits
        available -= bits
        self.end_window = window
        self.nend_bits = available
        self.nbits_total += bits
        return ret

    def tell(self):
        return self.nbits_total - ec_ilog(self.rng)


# ---- PVQ (mirrors the textbook CWRS in celt.cpp) --------------------------

_V_CACHE = {}


def pvq_v(n, k):
    if k == 0:
        return 1
    if n == 0:
        return 0
    key = (n, k)
    if key in _V_CACHE:
        return _V_CACHE[key]
    cur = [1] + [2] * k
    for d in range(2, n + 1):
        prev = cur[:]
        cur = [1] * (k + 1)
        for i in range(1, k + 1):
            cur[i] = prev[i] + cur[i - 1] + prev[i - 1]
    _V_CACHE[key] = cur[k]
    return cur[k]


def pvq_unindex(idx, n, k):
    """Inverse of celt.cpp pvq_index."""
    y = [0] * n
    kleft = k
    for i in range(n):
        if kleft == 0:
            break
        dims = n - 1 - i
        a = 0
        while True:
            cnt = pvq_v(dims, kleft - a)
            width = cnt if a == 0 else 2 * cnt
            if idx < width:
                break
            idx -= width
            a += 1
        neg = False
        if a > 0:
            cnt = pvq_v(dims, kleft - a)
            if idx >= cnt:
                idx -= cnt
                neg = True
        y[i] = -a if neg else a
        kleft -= a
    return y


def decode_band_pvq(dec, n, k):
    if k == 0 or n == 0:
        return [0] * n
    if n > 2 and pvq_v(n, k) >= (1 << 60):
        h = n // 2
        kl = dec.dec_uint(k + 1)
        return (decode_band_pvq(dec, h, kl) +
                decode_band_pvq(dec, n - h, k - kl))
    total = pvq_v(n, k)
    if total > (1 << 30):
        hi = dec.dec_uint((total >> 30) + 1)
        lo = dec.dec_bits(30)
        idx = (hi << 30) | lo
    else:
        idx = dec.dec_uint(total)
    return pvq_unindex(idx, n, k)


def pvq_bits_frac(n, k):
    if k == 0:
        return 0
    return math.ceil(8.0 * math.log2(pvq_v(n, k))) + 16


# ---- IMDCT ----------------------------------------------------------------

_WINDOW = None


def mdct_window():
    global _WINDOW
    if _WINDOW is None:
        n = FRAME
        w = np.zeros(2 * n)
        z = (n - OVERLAP) // 2
        for i in range(OVERLAP):
            t = math.sin(0.5 * math.pi * (i + 0.5) / OVERLAP)
            r = math.sin(0.5 * math.pi * t * t)
            w[z + i] = r
            w[2 * n - 1 - z - i] = r
        w[z + OVERLAP:2 * n - z - OVERLAP] = 1.0
        _WINDOW = w
    return _WINDOW


def imdct(bins):
    n = FRAME
    j = np.arange(2 * n)
    k = np.arange(n)
    c = math.pi / n
    basis = np.cos(c * np.outer(j + 0.5 + n / 2.0, k + 0.5))
    return basis @ bins


class OpusDecoder:
    """Streaming decoder: feed packets, collect 48 kHz mono float PCM
    (one frame of algorithmic delay from the MDCT overlap-add)."""

    def __init__(self):
        self.prev_tail = np.zeros(FRAME)
        self.pcm = []

    def decode_packet(self, pkt: bytes):
        toc = pkt[0]
        assert toc >> 3 == 31, "CELT fullband 20 ms expected"
        assert toc & 0x4 == 0, "mono expected"
        assert toc & 0x3 == 0, "one frame per packet"
        dec = RangeDecoder(pkt[1:])

        assert dec.dec_bit_logp(15) == 0      # silence
        assert dec.dec_bit_logp(1) == 0       # postfilter
        assert dec.dec_bit_logp(3) == 0       # transient
        assert dec.dec_bit_logp(3) == 1       # intra

        energy = np.zeros(NBANDS)
        for b in range(NBANDS):
            qi = dec.dec_uint(64) - 16
            energy[b] = EMEANS[b] + qi

        q = dec.dec_uint(11)
        total_frac = len(pkt[1:]) * 8 * 8
        fine_bits = []
        shape_frac = []
        for b in range(NBANDS):
            nb = BAND_BINS[b + 1] - BAND_BINS[b]
            frac = ALLOC[q][b] * nb // 4
            fb = max(0, min(7, frac // 160))
            fine_bits.append(fb)
            shape_frac.append(max(0, frac - fb * 64))
        for b in range(NBANDS):
            if fine_bits[b] <= 0:
                continue
            fq = dec.dec_bits(fine_bits[b])
            energy[b] += (fq + 0.5) / (1 << fine_bits[b]) - 0.5

        bins = np.zeros(FRAME)
        for b in range(NBANDS):
            n0, nb = BAND_BINS[b], BAND_BINS[b + 1] - BAND_BINS[b]
            coded = dec.dec_bit_logp(1)
            if not coded:
                continue
            k = dec.dec_uint(256)
            y = np.array(decode_band_pvq(dec, nb, k), dtype=np.float64)
            norm = math.sqrt(float((y * y).sum())) or 1.0
            g = 2.0 ** energy[b]
            bins[n0:n0 + nb] = y / norm * g * self._band_scale(nb, k)
        # time domain
        t = imdct(bins) * mdct_window()
        out = self.prev_tail + t[:FRAME]
        self.prev_tail = t[FRAME:]
        self.pcm.append(out)
        return out

    @staticmethod
    def _band_scale(nb, k):
        return 1.0

    def samples(self):
        return np.concatenate(self.pcm) if self.pcm else np.zeros(0)
