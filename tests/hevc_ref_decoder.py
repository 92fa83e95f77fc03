"""From-spec HEVC (ITU-T H.265) decoder for the subset the hipflux encoder
emits: Main profile, all-intra IDR, CTU16 / CU16 / TB16 luma + TB8 chroma,
CABAC, modes Planar/DC/H/V, SAO+deblock off, 4:2:0, 8-bit.

Written against the 04/2013 spec text (§7 syntax, §8 decode processes,
§9.3 CABAC) as the independent check of the C++ encoder: every bin is
decoded with the standard arithmetic decoder and the reconstruction must
match the encoder's recon planes bit-exactly. Section numbers cited
inline. Deliberately structured as a decoder (parse-driven), not a
mirror of the encoder.
"""

import numpy as np

# ---- CABAC tables (Tables 9-46/47/48 + context initValues, initType 0) ----

RANGE_TAB_LPS = [
    [128, 176, 208, 240], [128, 167, 197, 227], [128, 158, 187, 216],
    [123, 150, 178, 205], [116, 142, 169, 195], [111, 135, 160, 185],
    [105, 128, 152, 175], [100, 122, 144, 166], [95, 116, 137, 158],
    [90, 110, 130, 150], [85, 104, 123, 142], [81, 99, 117, 135],
    [77, 94, 111, 128], [73, 89, 105, 122], [69, 85, 100, 116],
    [66, 80, 95, 110], [62, 76, 90, 104], [59, 72, 86, 99],
    [56, 69, 81, 94], [53, 65, 77, 89], [51, 62, 73, 85],
    [48, 59, 69, 80], [46, 56, 66, 76], [43, 53, 63, 72],
    [41, 50, 59, 69], [39, 48, 56, 65], [37, 45, 54, 62],
    [35, 43, 51, 59], [33, 41, 48, 56], [32, 39, 46, 53],
    [30, 37, 43, 50], [29, 35, 41, 48], [27, 33, 39, 45],
    [26, 31, 37, 43], [24, 30, 35, 41], [23, 28, 33, 39],
    [22, 27, 32, 37], [21, 26, 30, 35], [20, 24, 29, 33],
    [19, 23, 27, 31], [18, 22, 26, 30], [17, 21, 25, 28],
    [16, 20, 23, 27], [15, 19, 22, 25], [14, 18, 21, 24],
    [14, 17, 20, 23], [13, 16, 19, 22], [12, 15, 18, 21],
    [12, 14, 17, 20], [11, 14, 16, 19], [11, 13, 15, 18],
    [10, 12, 15, 17], [10, 12, 14, 16], [9, 11, 13, 15],
    [9, 11, 12, 14], [8, 10, 12, 14], [8, 9, 11, 13],
    [7, 9, 11, 12], [7, 9, 10, 12], [7, 8, 10, 11],
    [6, 8, 9, 11], [6, 7, 9, 10], [6, 7, 8, 9], [2, 2, 2, 2]]

TRANS_IDX_LPS = [
    0, 0, 1, 2, 2, 4, 4, 5, 6, 7, 8, 9, 9, 11, 11, 12,
    13, 13, 15, 15, 16, 16, 18, 18, 19, 19, 21, 21, 22, 22, 23, 24,
    24, 25, 26, 26, 27, 27, 28, 29, 29, 30, 30, 30, 31, 32, 32, 33,
    33, 33, 34, 34, 35, 35, 35, 36, 36, 36, 37, 37, 37, 38, 38, 63]

TRANS_IDX_MPS = list(range(1, 63)) + [62, 63]

INIT = {
    "split_cu": [139, 141, 157],
    "prev_intra": [184],
    "chroma_mode": [63],
    "cbf_luma": [111, 141],
    "cbf_chroma": [94, 138, 182, 154],
    "last_x": [110, 110, 124, 125, 140, 153, 125, 127, 140,
               109, 111, 143, 127, 111, 79, 108, 123, 63],
    "last_y": [110, 110, 124, 125, 140, 153, 125, 127, 140,
               109, 111, 143, 127, 111, 79, 108, 123, 63],
    "csbf": [91, 171, 134, 141],
    "sig": [111, 111, 125, 110, 110, 94, 124, 108, 124, 107, 125, 141,
            179, 153, 125, 107, 125, 141, 179, 153, 125, 107, 125, 141,
            179, 153, 125, 140, 139, 182, 182, 152, 136, 152, 136, 153,
            136, 139, 111, 136, 139, 111],
    "gt1": [140, 92, 137, 138, 140, 152, 138, 139, 153, 74, 149, 92,
            139, 107, 122, 152, 140, 179, 166, 182, 140, 227, 122, 197],
    "gt2": [138, 153, 136, 167, 152, 152],
}

# transform matrices (§8.6.4.2)
T8 = np.array([
    [64, 64, 64, 64, 64, 64, 64, 64],
    [89, 75, 50, 18, -18, -50, -75, -89],
    [83, 36, -36, -83, -83, -36, 36, 83],
    [75, -18, -89, -50, 50, 89, 18, -75],
    [64, -64, -64, 64, 64, -64, -64, 64],
    [50, -89, 18, 75, -75, -18, 89, -50],
    [36, -83, 83, -36, -36, 83, -83, 36],
    [18, -50, 75, -89, 89, -75, 50, -18]], dtype=np.int64)

T16 = np.array([
    [64] * 16,
    [90, 87, 80, 70, 57, 43, 25, 9, -9, -25, -43, -57, -70, -80, -87, -90],
    [89, 75, 50, 18, -18, -50, -75, -89, -89, -75, -50, -18, 18, 50, 75, 89],
    [87, 57, 9, -43, -80, -90, -70, -25, 25, 70, 90, 80, 43, -9, -57, -87],
    [83, 36, -36, -83, -83, -36, 36, 83, 83, 36, -36, -83, -83, -36, 36, 83],
    [80, 9, -70, -87, -25, 57, 90, 43, -43, -90, -57, 25, 87, 70, -9, -80],
    [75, -18, -89, -50, 50, 89, 18, -75, -75, 18, 89, 50, -50, -89, -18, 75],
    [70, -43, -87, 9, 90, 25, -80, -57, 57, 80, -25, -90, -9, 87, 43, -70],
    [64, -64, -64, 64, 64, -64, -64, 64, 64, -64, -64, 64, 64, -64, -64, 64],
    [57, -80, -25, 90, -9, -87, 43, 70, -70, -43, 87, 9, -90, 25, 80, -57],
    [50, -89, 18, 75, -75, -18, 89, -50, -50, 89, -18, -75, 75, 18, -89, 50],
    [43, -90, 57, 25, -87, 70, 9, -80, 80, -9, -70, 87, -25, -57, 90, -43],
    [36, -83, 83, -36, -36, 83, -83, 36, 36, -83, 83, -36, -36, 83, -83, 36],
    [25, -70, 90, -80, 43, 9, -57, 87, -87, 57, -9, -43, 80, -90, 70, -25],
    [18, -50, 75, -89, 89, -75, 50, -18, -18, 50, -75, 89, -89, 75, -50, 18],
    [9, -25, 43, -57, 70, -80, 87, -90, 90, -87, 80, -70, 57, -43, 25, -9]],
    dtype=np.int64)

DEQUANT_SCALE = [40, 45, 51, 57, 64, 72]


def diag_scan(size):
    """§6.5.3 up-right diagonal scan: list of (x, y) in coding order."""
    out = []
    x = y = 0
    while len(out) < size * size:
        while y >= 0:
            if x < size and y < size:
                out.append((x, y))
            y -= 1
            x += 1
        y = x
        x = 0
    return out

SCAN4 = diag_scan(4)
SCAN2 = diag_scan(2)


def chroma_qp(qp):
    if qp < 30:
        return qp
    if qp > 43:
        return qp - 6
    return [29, 30, 31, 32, 33, 33, 34, 34, 35, 35, 36, 36, 37, 37][qp - 30]


def split_nals(data):
    """Annex-B start-code scan + emulation-prevention removal."""
    nals = []
    i = 0
    n = len(data)
    starts = []
    while i + 3 <= n:
        if data[i] == 0 and data[i + 1] == 0 and data[i + 2] == 1:
            starts.append(i + 3)
            i += 3
        else:
            i += 1
    for k, s in enumerate(starts):
        e = (starts[k + 1] - 3) if k + 1 < len(starts) else n
        while e > s and data[e - 1] == 0:
            e -= 1
        raw = data[s:e]
        rbsp = bytearray()
        zeros = 0
        j = 0
        while j < len(raw):
            b = raw[j]
            if zeros >= 2 and b == 3:
                zeros = 0
                j += 1
                continue
            rbsp.append(b)
            zeros = zeros + 1 if b == 0 else 0
            j += 1
        nals.append(bytes(rbsp))
    return nals


class BitReader:
    def __init__(self, data):
        self.d = data
        self.pos = 0  # bit position

    def u(self, n):
        v = 0
        for _ in range(n):
            byte = self.d[self.pos >> 3]
            v = (v << 1) | ((byte >> (7 - (self.pos & 7))) & 1)
            self.pos += 1
        return v

    def ue(self):
        zeros = 0
        while self.u(1) == 0:
            zeros += 1
            assert zeros < 32
        return (1 << zeros) - 1 + (self.u(zeros) if zeros else 0)

    def se(self):
        k = self.ue()
        return (k + 1) >> 1 if k & 1 else -(k >> 1)

    def byte_align(self):
        assert self.u(1) == 1  # alignment_bit_equal_to_one
        while self.pos & 7:
            assert self.u(1) == 0


class Cabac:
    """Arithmetic decoding engine, §9.3.4.3."""

    def __init__(self, br):
        self.br = br
        self.range = 510
        self.offset = br.u(9)

    def _renorm(self):
        while self.range < 256:
            self.range <<= 1
            self.offset = (self.offset << 1) | self.br.u(1)

    def decision(self, ctx):
        state, mps = ctx
        lps = RANGE_TAB_LPS[state][(self.range >> 6) & 3]
        self.range -= lps
        if self.offset >= self.range:
            bin_ = 1 - mps
            self.offset -= self.range
            self.range = lps
            if state == 0:
                mps = 1 - mps
            state = TRANS_IDX_LPS[state]
        else:
            bin_ = mps
            state = TRANS_IDX_MPS[state]
        ctx[0], ctx[1] = state, mps
        self._renorm()
        return bin_

    def bypass(self):
        self.offset = (self.offset << 1) | self.br.u(1)
        if self.offset >= self.range:
            self.offset -= self.range
            return 1
        return 0

    def bypass_bits(self, n):
        v = 0
        for _ in range(n):
            v = (v << 1) | self.bypass()
        return v

    def terminate(self):
        self.range -= 2
        if self.offset >= self.range:
            return 1
        self._renorm()
        return 0


def init_ctx(init_value, qp):
    """§9.3.2.2."""
    slope = (init_value >> 4) * 5 - 45
    offset = ((init_value & 15) << 3) - 16
    pre = ((slope * min(max(qp, 0), 51)) >> 4) + offset
    pre = min(max(pre, 1), 126)
    if pre <= 63:
        return [63 - pre, 0]
    return [pre - 64, 1]


class Contexts:
    def __init__(self, qp):
        for name, vals in INIT.items():
            setattr(self, name, [init_ctx(v, qp) for v in vals])


def inv_transform(coef, n):
    """§8.6.4.2 two-stage inverse with 16-bit intermediate clip."""
    t = T8 if n == 8 else T16
    tmp = np.clip((t.T @ coef + 64) >> 7, -32768, 32767)
    return np.clip((tmp @ t + 2048) >> 12, -32768, 32767)


def dequant(level, qp, n):
    """§8.6.3, flat scaling list (m = 16)."""
    log2n = 3 if n == 8 else 4
    bd_shift = 8 + log2n - 5
    scale = (DEQUANT_SCALE[qp % 6] << (qp // 6)) * 16
    d = (level.astype(np.int64) * scale + (1 << (bd_shift - 1))) >> bd_shift
    return np.clip(d, -32768, 32767)


def build_refs(plane, x0, y0, n, avail):
    """§8.4.4.2.2: reference array + substitution. Returns (corner,
    top[2n], left[2n])."""
    h, w = plane.shape
    # linear order: below-left bottom .. left top, corner, top .. top-right
    lin = np.zeros(4 * n + 1, np.int32)
    have = np.zeros(4 * n + 1, bool)
    for i in range(n):  # below-left (bottom to top)
        yy = y0 + 2 * n - 1 - i
        if avail["below_left"] and yy < h:
            lin[i] = plane[yy, x0 - 1]
            have[i] = True
    for i in range(n):  # left (bottom to top)
        if avail["left"]:
            lin[n + i] = plane[y0 + n - 1 - i, x0 - 1]
            have[n + i] = True
    if avail["corner"]:
        lin[2 * n] = plane[y0 - 1, x0 - 1]
        have[2 * n] = True
    for i in range(n):  # top
        if avail["top"]:
            lin[2 * n + 1 + i] = plane[y0 - 1, x0 + i]
            have[2 * n + 1 + i] = True
    for i in range(n):  # top-right
        xx = x0 + n + i
        if avail["top_right"] and xx < w:
            lin[3 * n + 1 + i] = plane[y0 - 1, xx]
            have[3 * n + 1 + i] = True
    if not have.any():
        lin[:] = 128
    else:
        if not have[0]:
            lin[0] = lin[np.argmax(have)]
        for i in range(1, 4 * n + 1):
            if not have[i]:
                lin[i] = lin[i - 1]
    corner = lin[2 * n]
    top = lin[2 * n + 1:]
    left = lin[2 * n - 1::-1]  # top-to-bottom
    return corner, top.copy(), left.copy()


def filter_refs(corner, top, left):
    """§8.4.4.2.3 [1 2 1]."""
    n2 = len(top)
    lin = np.concatenate([left[::-1], [corner], top]).astype(np.int32)
    f = lin.copy()
    f[1:-1] = (lin[:-2] + 2 * lin[1:-1] + lin[2:] + 2) >> 2
    return f[n2], f[n2 + 1:], f[n2 - 1::-1]


def predict(mode, corner, top, left, n, cidx):
    """§8.4.4.2.4-7 for Planar/DC/H/V."""
    p = np.zeros((n, n), np.int32)
    log2n = {4: 2, 8: 3, 16: 4}[n]
    if mode == 0:  # planar
        x = np.arange(n)
        y = np.arange(n)
        p = ((n - 1 - x)[None, :] * left[:n][:, None]
             + (x + 1)[None, :] * top[n]
             + (n - 1 - y)[:, None] * top[:n][None, :]
             + (y + 1)[:, None] * left[n] + n) >> (log2n + 1)
    elif mode == 1:  # DC
        dc = (int(top[:n].sum()) + int(left[:n].sum()) + n) >> (log2n + 1)
        p[:, :] = dc
        if cidx == 0 and n < 32:
            p[0, 0] = (left[0] + 2 * dc + top[0] + 2) >> 2
            p[0, 1:] = (top[1:n] + 3 * dc + 2) >> 2
            p[1:, 0] = (left[1:n] + 3 * dc + 2) >> 2
    elif mode == 26:  # vertical
        p[:, :] = top[:n][None, :]
        if cidx == 0 and n < 32:
            p[:, 0] = np.clip(top[0] + ((left[:n] - corner) >> 1), 0, 255)
    elif mode == 10:  # horizontal
        p[:, :] = left[:n][:, None]
        if cidx == 0 and n < 32:
            p[0, :] = np.clip(left[0] + ((top[:n] - corner) >> 1), 0, 255)
    else:
        raise ValueError(f"unsupported intra mode {mode}")
    return p


def decode_residual(cab, ctxs, n, cidx):
    """§7.3.8.11 residual_coding -> level array (n x n)."""
    log2n = 3 if n == 8 else 4
    level = np.zeros((n, n), np.int16)
    sb_scan = SCAN2 if n == 8 else SCAN4
    n_sb = (n // 4) ** 2

    # last_sig position
    if cidx == 0:
        ctx_off, ctx_shift = 3 * (log2n - 2) + ((log2n - 1) >> 2), \
            (log2n + 1) >> 2
        bank_x, bank_y = ctxs.last_x, ctxs.last_y
    else:
        ctx_off, ctx_shift = 15, log2n - 2
        bank_x, bank_y = ctxs.last_x, ctxs.last_y
    g_max = (log2n << 1) - 1

    def read_prefix(bank):
        p = 0
        while p < g_max and cab.decision(bank[ctx_off + (p >> ctx_shift)]):
            p += 1
        return p

    px = read_prefix(bank_x)
    py = read_prefix(bank_y)
    if px > 3:
        nbits = (px >> 1) - 1
        last_x = ((2 + (px & 1)) << nbits) + cab.bypass_bits(nbits)
    else:
        last_x = px
    if py > 3:
        nbits = (py >> 1) - 1
        last_y = ((2 + (py & 1)) << nbits) + cab.bypass_bits(nbits)
    else:
        last_y = py

    # locate the sub-block / in-block scan position of last
    last_sb = last_pos = None
    for i, (sx, sy) in enumerate(sb_scan):
        for k, (x4, y4) in enumerate(SCAN4):
            if sx * 4 + x4 == last_x and sy * 4 + y4 == last_y:
                last_sb, last_pos = i, k
    assert last_sb is not None

    csbf = [False] * n_sb
    csbf[last_sb] = True
    csbf[0] = True
    prev_g1_zero = None

    for i in range(last_sb, -1, -1):
        sxy = sb_scan[i]
        sx, sy = sxy[0] * 4, sxy[1] * 4
        explicit = False
        if 0 < i < last_sb:
            right = below = 0
            for j in range(n_sb):
                if sb_scan[j] == (sxy[0] + 1, sxy[1]):
                    right = int(csbf[j])
                if sb_scan[j] == (sxy[0], sxy[1] + 1):
                    below = int(csbf[j])
            ctx = min(1, right + below) + (2 if cidx else 0)
            csbf[i] = bool(cab.decision(ctxs.csbf[ctx]))
            explicit = True
        if not csbf[i]:
            continue
        infer_dc = explicit

        sig = [False] * 16
        if i == last_sb:
            sig[last_pos] = True
            start = last_pos - 1
        else:
            start = 15
        for k in range(start, -1, -1):
            x = sx + SCAN4[k][0]
            y = sy + SCAN4[k][1]
            if k > 0 or not infer_dc:
                if x == 0 and y == 0:
                    sig_ctx = 0
                else:
                    right = below = 0
                    for j in range(n_sb):
                        if sb_scan[j] == (sxy[0] + 1, sxy[1]):
                            right = int(csbf[j])
                        if sb_scan[j] == (sxy[0], sxy[1] + 1):
                            below = int(csbf[j])
                    prev = right + (below << 1)
                    xp, yp = x & 3, y & 3
                    if prev == 0:
                        sig_ctx = 2 if xp + yp == 0 else 1 if xp + yp < 3 else 0
                    elif prev == 1:
                        sig_ctx = 2 if yp == 0 else 1 if yp == 1 else 0
                    elif prev == 2:
                        sig_ctx = 2 if xp == 0 else 1 if xp == 1 else 0
                    else:
                        sig_ctx = 2
                    if cidx == 0:
                        if (x >> 2) + (y >> 2) > 0:
                            sig_ctx += 3
                        sig_ctx += 9 if log2n == 3 else 21
                    else:
                        sig_ctx += 9 if log2n == 3 else 12
                ctx = (27 if cidx else 0) + sig_ctx
                s = bool(cab.decision(ctxs.sig[ctx]))
                if s:
                    infer_dc = False
                sig[k] = s
            else:
                sig[k] = True  # inferred DC significance

        positions = [k for k in range(15, -1, -1) if sig[k]]
        if not positions:
            continue

        # greater1 / greater2
        ctx_set = 0 if (i == 0 or cidx > 0) else 2
        if i != last_sb and prev_g1_zero:
            ctx_set += 1
        g1_ctx = 1
        first_g1 = -1
        abs_lvl = []
        for idx, k in enumerate(positions):
            if idx < 8:
                ctx = ctx_set * 4 + min(3, g1_ctx) + (16 if cidx else 0)
                g1 = cab.decision(ctxs.gt1[ctx])
                if g1:
                    g1_ctx = 0
                    if first_g1 < 0:
                        first_g1 = idx
                elif 0 < g1_ctx < 3:
                    g1_ctx += 1
                abs_lvl.append(2 if g1 else 1)
            else:
                abs_lvl.append(1)
        prev_g1_zero = (g1_ctx == 0)
        if first_g1 >= 0:
            g2 = cab.decision(ctxs.gt2[ctx_set + (4 if cidx else 0)])
            abs_lvl[first_g1] += g2
        signs = [cab.bypass() for _ in positions]
        rice = 0
        for idx, k in enumerate(positions):
            base = (3 if idx == first_g1 else 2) if idx < 8 else 1
            if abs_lvl[idx] == base:
                # coeff_abs_level_remaining (§9.3.3.13)
                prefix = 0
                while cab.bypass():
                    prefix += 1
                if prefix <= 3:
                    rem = (prefix << rice) + cab.bypass_bits(rice)
                else:
                    # exp-Golomb escape: base covers all shorter codes
                    nb = prefix - 3 + rice
                    base_v = 3 << rice
                    for p in range(prefix - 3):
                        base_v += 1 << (rice + p)
                    rem = base_v + cab.bypass_bits(nb)
                abs_lvl[idx] += rem
                if abs_lvl[idx] > (3 << rice) and rice < 4:
                    rice += 1
            x = sx + SCAN4[k][0]
            y = sy + SCAN4[k][1]
            level[y, x] = -abs_lvl[idx] if signs[idx] else abs_lvl[idx]
    return level


class Decoder:
    """Decode one independent HEVC stream (VPS/SPS/PPS + IDR pictures)."""

    def __init__(self):
        self.sps = None
        self.frames = []

    def decode(self, data):
        self.frames = []
        pic = None
        for nal in split_nals(bytes(data)):
            ntype = (nal[0] >> 1) & 0x3F
            body = nal[2:]
            if ntype == 33:
                self._parse_sps(BitReader(body))
            elif ntype in (32, 34):
                continue  # VPS / PPS: fixed-layout in this subset
            elif ntype == 19:  # IDR_W_RADL
                pic = self._decode_slice(BitReader(body), pic)
        if pic is not None:
            self._emit(pic)
        return self.frames

    def _emit(self, pic):
        y, cb, cr = pic
        s = self.sps
        self.frames.append((
            y[:s["h"], :s["w"]].astype(np.uint8),
            cb[:(s["h"] + 1) // 2, :(s["w"] + 1) // 2].astype(np.uint8),
            cr[:(s["h"] + 1) // 2, :(s["w"] + 1) // 2].astype(np.uint8)))

    def _parse_ptl(self, br):
        br.u(2 + 1 + 5)
        br.u(32)
        br.u(4)
        br.u(22)
        br.u(22)
        br.u(8)

    def _parse_sps(self, br):
        br.u(4)       # vps id
        br.u(3)       # max_sub_layers
        br.u(1)       # nesting
        self._parse_ptl(br)
        assert br.ue() == 0          # sps id
        assert br.ue() == 1          # chroma_format: 4:2:0
        cw = br.ue()
        ch = br.ue()
        w, h = cw, ch
        if br.u(1):                  # conformance window
            left = br.ue()
            right = br.ue()
            top = br.ue()
            bottom = br.ue()
            assert left == 0 and top == 0
            w = cw - 2 * right
            h = ch - 2 * bottom
        assert br.ue() == 0 and br.ue() == 0   # bit depths: 8
        br.ue()                      # log2_max_poc_lsb
        if br.u(1):                  # sub_layer_ordering_info_present
            br.ue(), br.ue(), br.ue()
        min_cb = br.ue() + 3
        ctb = min_cb + br.ue()
        min_tb = br.ue() + 2
        max_tb = min_tb + br.ue()
        assert ctb == 4 and max_tb == 4, "decoder subset: CTU16/TB16"
        br.ue(), br.ue()             # transform hierarchy depths
        assert br.u(1) == 0          # scaling lists
        br.u(1)                      # amp
        assert br.u(1) == 0          # sao
        assert br.u(1) == 0          # pcm
        assert br.ue() == 0          # num_short_term_rps
        self.sps = {"cw": cw, "ch": ch, "w": w, "h": h,
                    "ctb_w": cw // 16, "ctb_h": ch // 16}

    def _decode_slice(self, br, pic):
        s = self.sps
        first = br.u(1)
        br.u(1)                      # no_output_of_prior_pics
        assert br.ue() == 0          # pps id
        n_ctb = s["ctb_w"] * s["ctb_h"]
        addr = 0
        if not first:
            bits = max(1, (n_ctb - 1).bit_length())
            addr = br.u(bits)
        assert br.ue() == 2          # slice_type I
        qp = 26 + br.se()
        br.byte_align()

        if first:
            if pic is not None:
                self._emit(pic)
            pic = (np.zeros((s["ch"], s["cw"]), np.int32),
                   np.zeros((s["ch"] // 2, s["cw"] // 2), np.int32),
                   np.zeros((s["ch"] // 2, s["cw"] // 2), np.int32))

        cab = Cabac(br)
        ctxs = Contexts(qp)
        qpc = chroma_qp(qp)
        ctb_w = s["ctb_w"]
        caddr = addr
        left_mode = None   # luma mode of the CTU to the left (same slice)
        while True:
            cx, cy = caddr % ctb_w, caddr // ctb_w
            left_mode = self._decode_ctu(cab, ctxs, pic, cx, cy, qp, qpc,
                                         left_mode)
            end = cab.terminate()
            caddr += 1
            if end:
                break
            if caddr % ctb_w == 0:
                left_mode = None   # encoder slices never span rows
        return pic

    def _decode_ctu(self, cab, ctxs, pic, cx, cy, qp, qpc, left_mode):
        ry, rcb, rcr = pic
        x0, y0 = cx * 16, cy * 16
        assert cab.decision(ctxs.split_cu[0]) == 0, "subset: no CU split"
        # luma intra mode (§8.4.2 MPM)
        prev = cab.decision(ctxs.prev_intra[0])
        cand_a = left_mode if left_mode is not None else 1
        cand_b = 1  # above PU is in another CTU row -> DC
        if cand_a == cand_b:
            if cand_a < 2:
                lst = [0, 1, 26]
            else:
                lst = [cand_a, 2 + ((cand_a + 29) % 32),
                       2 + ((cand_a - 2 + 1) % 32)]
        else:
            lst = [cand_a, cand_b,
                   0 if cand_a != 0 and cand_b != 0 else
                   1 if cand_a != 1 and cand_b != 1 else 26]
        if prev:
            idx = 0
            if cab.bypass():
                idx = 1 + cab.bypass()
            mode = lst[idx]
        else:
            rem = cab.bypass_bits(5)
            for c in sorted(lst):
                if rem >= c:
                    rem += 1
            mode = rem
        assert cab.decision(ctxs.chroma_mode[0]) == 0, "subset: DM chroma"

        cbf_cb = cab.decision(ctxs.cbf_chroma[0])
        cbf_cr = cab.decision(ctxs.cbf_chroma[0])
        cbf_y = cab.decision(ctxs.cbf_luma[1])

        avail = {"left": left_mode is not None, "top": False,
                 "top_right": False, "below_left": False, "corner": False}

        # luma
        corner, top, left = build_refs(ry, x0, y0, 16, avail)
        if mode == 0:
            corner, top, left = filter_refs(corner, top, left)
        p = predict(mode, corner, top, left, 16, 0)
        if cbf_y:
            lv = decode_residual(cab, ctxs, 16, 0)
            res = inv_transform(dequant(lv, qp, 16), 16)
            ry[y0:y0 + 16, x0:x0 + 16] = np.clip(p + res, 0, 255)
        else:
            ry[y0:y0 + 16, x0:x0 + 16] = p

        # chroma (DM)
        cx0, cy0 = x0 // 2, y0 // 2
        for plane, cbf, cidx in ((rcb, cbf_cb, 1), (rcr, cbf_cr, 2)):
            corner, top, left = build_refs(plane, cx0, cy0, 8, avail)
            p = predict(mode, corner, top, left, 8, cidx)
            if cbf:
                lv = decode_residual(cab, ctxs, 8, cidx)
                res = inv_transform(dequant(lv, qpc, 8), 8)
                plane[cy0:cy0 + 8, cx0:cx0 + 8] = np.clip(p + res, 0, 255)
            else:
                plane[cy0:cy0 + 8, cx0:cx0 + 8] = p
        return mode
