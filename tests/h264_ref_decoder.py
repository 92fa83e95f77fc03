"""Reference H.264 decoder for the subset emitted by hipflux (pure Python).

Implements Annex-B parsing, SPS/PPS/slice headers, CAVLC residual decoding
(tables loaded from tests/data/cavlc_tables.json — same verified source as
the C++ header), I16x16 (H/DC) + chroma intra, P_Skip / P_L0_16x16, inverse
transforms and reconstruction, for streams with:
  Constrained Baseline, CAVLC, frame_mbs_only, one slice per MB row,
  pic_order_cnt_type=2, 1 reference frame, deblocking disabled.

Used by tests to verify that decoding an encoder-produced stream yields a
frame BIT-EXACTLY equal to the encoder's own reconstruction — which
exercises every entropy/ transform/prediction path end to end.
"""

from __future__ import annotations

import json
import os

import numpy as np

_TAB = json.load(open(os.path.join(os.path.dirname(__file__), "data",
                                   "cavlc_tables.json")))

ZZ4 = [0, 1, 4, 8, 5, 2, 3, 6, 9, 12, 13, 10, 7, 11, 14, 15]

QUANT_V = [[10, 16, 13], [11, 18, 14], [13, 20, 16],
           [14, 23, 18], [16, 25, 20], [18, 29, 23]]

CHROMA_QP = {30: 29, 31: 30, 32: 31, 33: 32, 34: 32, 35: 33, 36: 34, 37: 34,
             38: 35, 39: 35, 40: 36, 41: 36, 42: 37, 43: 37, 44: 37, 45: 38,
             46: 38, 47: 38, 48: 39, 49: 39, 50: 39, 51: 39}


def chroma_qp(qp):
    return qp if qp < 30 else CHROMA_QP[qp]


def coeff_class(i, j):
    ei, ej = (i & 1) == 0, (j & 1) == 0
    return 0 if (ei and ej) else (1 if (not ei and not ej) else 2)


def _tap6(a, b, c, d, e, f):
    return a - 5 * b + 20 * c + 20 * d - 5 * e + f


def luma_mc(ref, x0, y0, mvx, mvy):
    """16x16 luma MC at any quarter-pel mv: 6-tap half samples
    (8.4.2.2.1) and rounded-average quarter samples (8.4.2.2.2,
    Table 8-12)."""
    ix, iy = mvx >> 2, mvy >> 2
    fx, fy = mvx & 3, mvy & 3
    bx, by = x0 + ix, y0 + iy
    if fx == 0 and fy == 0:
        return ref[by:by + 16, bx:bx + 16].astype(np.int64)
    win = ref[by - 2:by + 20, bx - 2:bx + 20].astype(np.int64)  # 22x22

    def htap(w):          # horizontal 6-tap sums: (rows, cols-5)
        return _tap6(w[:, 0:-5], w[:, 1:-4], w[:, 2:-3], w[:, 3:-2],
                     w[:, 4:-1], w[:, 5:])

    def vtap(w):          # vertical 6-tap sums: (rows-5, cols)
        return _tap6(w[0:-5], w[1:-4], w[2:-3], w[3:-2], w[4:-1], w[5:])

    G = win[2:19, 2:19]                                     # 17x17 ints
    b = np.clip((htap(win)[2:19] + 16) >> 5, 0, 255)        # 17x17
    h = np.clip((vtap(win)[:, 2:19] + 16) >> 5, 0, 255)     # 17x17
    j = np.clip((vtap(htap(win)) + 512) >> 10, 0, 255)      # 17x17
    S = slice(0, 16)
    G0, Gx, Gy = G[S, S], G[S, 1:17], G[1:17, S]
    b0, s_ = b[S, S], b[1:17, S]        # s = b one row below
    h0, m_ = h[S, S], h[S, 1:17]        # m = h one column right
    j0 = j[S, S]

    def avg(a, c):
        return (a + c + 1) >> 1

    if fy == 0:
        return b0 if fx == 2 else avg(G0 if fx == 1 else Gx, b0)
    if fx == 0:
        return h0 if fy == 2 else avg(G0 if fy == 1 else Gy, h0)
    if fx == 2 and fy == 2:
        return j0
    if fx == 2:                          # f / q
        return avg(b0 if fy == 1 else s_, j0)
    if fy == 2:                          # i / k
        return avg(h0 if fx == 1 else m_, j0)
    return avg(b0 if fy == 1 else s_, h0 if fx == 1 else m_)


def chroma_mc(ref, cx0, cy0, mvx, mvy):
    """8x8 chroma MC: spec bilinear with eighth-pel weights."""
    cix, ciy = mvx >> 3, mvy >> 3
    dx, dy = mvx & 7, mvy & 7
    bx, by = cx0 + cix, cy0 + ciy
    h_ext = 9 if dy else 8
    w_ext = 9 if dx else 8
    W = ref[by:by + h_ext, bx:bx + w_ext].astype(np.int64)
    a = W[0:8, 0:8]
    b = W[0:8, 1:9] if dx else a
    c = W[1:9, 0:8] if dy else a
    d = (W[1:9, 1:9] if dx else c) if dy else b
    return ((8 - dx) * (8 - dy) * a + dx * (8 - dy) * b +
            (8 - dx) * dy * c + dx * dy * d + 32) >> 6


def build_decode_tree(entries):
    """entries: list of (symbol, len, bits) -> dict[(len,bits)] = symbol"""
    return {(ln, bits): sym for sym, ln, bits in entries}


def _ct_entries(arr):
    out = []
    for tc in range(len(arr)):
        for t1 in range(4):
            ln, bits = arr[tc][t1]
            if ln:
                out.append(((tc, t1), ln, bits))
    return out


CT_TABLES = [build_decode_tree(_ct_entries(a)) for a in _TAB["coeff_token"]]
CT_CDC = build_decode_tree(_ct_entries(_TAB["coeff_token_cdc"]))
TZ_TABLES = {int(k): build_decode_tree([(i, ln, bits)
                                        for i, (ln, bits) in enumerate(v)])
             for k, v in _TAB["total_zeros"].items()}
TZ_CDC = {int(k): build_decode_tree([(i, ln, bits)
                                     for i, (ln, bits) in enumerate(v)])
          for k, v in _TAB["total_zeros_cdc"].items()}
RB_TABLES = {int(k): build_decode_tree([(i, ln, bits)
                                        for i, (ln, bits) in enumerate(v)
                                        if ln])
             for k, v in _TAB["run_before"].items()}


class BitReader:
    def __init__(self, data: bytes):
        self.data = data
        self.pos = 0  # bit position

    def bit(self) -> int:
        b = (self.data[self.pos >> 3] >> (7 - (self.pos & 7))) & 1
        self.pos += 1
        return b

    def u(self, n: int) -> int:
        v = 0
        for _ in range(n):
            v = (v << 1) | self.bit()
        return v

    def ue(self) -> int:
        zeros = 0
        while self.bit() == 0:
            zeros += 1
            if zeros > 31:
                raise ValueError("bad exp-golomb")
        return (1 << zeros) - 1 + (self.u(zeros) if zeros else 0)

    def se(self) -> int:
        cn = self.ue()
        return (cn + 1) // 2 if cn & 1 else -(cn // 2)

    def vlc(self, tree: dict):
        ln, bits = 0, 0
        while ln < 20:
            bits = (bits << 1) | self.bit()
            ln += 1
            if (ln, bits) in tree:
                return tree[(ln, bits)]
        raise ValueError(f"VLC decode failed at bit {self.pos}")

    def more_rbsp_data(self) -> bool:
        # true if there are bits beyond the rbsp_stop_bit
        total = len(self.data) * 8
        if self.pos >= total:
            return False
        # find last 1-bit in the stream (the stop bit)
        last = total - 1
        while last >= 0:
            if (self.data[last >> 3] >> (7 - (last & 7))) & 1:
                break
            last -= 1
        return self.pos < last


def split_nals(data: bytes):
    """Annex-B -> list of NAL payloads (emulation prevention removed)."""
    nals = []
    i = 0
    starts = []
    while True:
        j = data.find(b"\x00\x00\x01", i)
        if j < 0:
            break
        starts.append(j + 3)
        i = j + 3
    for k, s in enumerate(starts):
        e = len(data)
        if k + 1 < len(starts):
            e = starts[k + 1] - 3
            while e > s and data[e - 1] == 0:
                e -= 1
        raw = data[s:e]
        # remove emulation prevention
        out = bytearray()
        zeros = 0
        for b in raw:
            if zeros >= 2 and b == 3:
                zeros = 0
                continue
            out.append(b)
            zeros = zeros + 1 if b == 0 else 0
        nals.append(bytes(out))
    return nals


def residual_cavlc(br: BitReader, n: int, nC: int):
    """Decode one CAVLC residual block -> (zigzag list of n coeffs, tc)."""
    if nC == -1:
        tc, t1 = br.vlc(CT_CDC)
    elif nC < 2:
        tc, t1 = br.vlc(CT_TABLES[0])
    elif nC < 4:
        tc, t1 = br.vlc(CT_TABLES[1])
    elif nC < 8:
        tc, t1 = br.vlc(CT_TABLES[2])
    else:
        v = br.u(6)
        if v == 3:
            tc, t1 = 0, 0
        else:
            tc, t1 = (v >> 2) + 1, v & 3
    zz = [0] * n
    if tc == 0:
        return zz, 0
    levels = []
    for _ in range(t1):
        levels.append(1 if br.bit() == 0 else -1)
    suffix_len = 1 if (tc > 10 and t1 < 3) else 0
    for idx in range(tc - t1):
        # level_prefix
        prefix = 0
        while br.bit() == 0:
            prefix += 1
            if prefix > 18:
                raise ValueError("bad level prefix")
        if suffix_len == 0:
            if prefix < 14:
                code = prefix
            elif prefix == 14:
                code = 14 + br.u(4)
            else:
                code = 30 + br.u(12)
        else:
            if prefix < 15:
                code = (prefix << suffix_len) | br.u(suffix_len)
            else:
                code = (15 << suffix_len) + br.u(12)
        level = (code >> 1) + 1 if (code & 1) == 0 else -((code + 1) >> 1)
        if idx == 0 and t1 < 3:
            level = level + 1 if level > 0 else level - 1
        levels.append(level)
        if suffix_len == 0:
            suffix_len = 1
        if abs(level) > (3 << (suffix_len - 1)) and suffix_len < 6:
            suffix_len += 1
    # levels[] is high-frequency-first
    if tc < n:
        if nC == -1:
            total_zeros = br.vlc(TZ_CDC[tc])
        else:
            total_zeros = br.vlc(TZ_TABLES[tc])
    else:
        total_zeros = 0
    runs = []
    zeros_left = total_zeros
    for k in range(tc - 1):
        if zeros_left > 0:
            run = br.vlc(RB_TABLES[min(zeros_left, 7)])
            zeros_left -= run
        else:
            run = 0
        runs.append(run)
    runs.append(zeros_left)  # run before the lowest-frequency coefficient
    pos = -1
    # place from lowest frequency: reconstruct positions
    # levels are hf-first; runs are hf-first (gap below each coeff)
    coeffs_lf = list(reversed(levels))       # low-frequency first
    runs_lf = list(reversed(runs))
    for c, r in zip(coeffs_lf, runs_lf):
        pos += r + 1
        zz[pos] = c
    return zz, tc


def idct4(blk):
    out = np.zeros(16, dtype=np.int64)
    tmp = np.zeros(16, dtype=np.int64)
    for i in range(4):
        a, b, c, d = blk[4 * i:4 * i + 4]
        e0, e1 = a + c, a - c
        e2, e3 = (b >> 1) - d, b + (d >> 1)
        tmp[4 * i:4 * i + 4] = [e0 + e3, e1 + e2, e1 - e2, e0 - e3]
    for j in range(4):
        a, b, c, d = tmp[j], tmp[4 + j], tmp[8 + j], tmp[12 + j]
        e0, e1 = a + c, a - c
        e2, e3 = (b >> 1) - d, b + (d >> 1)
        out[j] = (e0 + e3 + 32) >> 6
        out[4 + j] = (e1 + e2 + 32) >> 6
        out[8 + j] = (e1 - e2 + 32) >> 6
        out[12 + j] = (e0 - e3 + 32) >> 6
    return out


def hadamard4_inv(blk):
    tmp = [0] * 16
    out = [0] * 16
    for i in range(4):
        r = blk[4 * i:4 * i + 4]
        s03, d03 = r[0] + r[3], r[0] - r[3]
        s12, d12 = r[1] + r[2], r[1] - r[2]
        tmp[4 * i:4 * i + 4] = [s03 + s12, d03 + d12, s03 - s12, d03 - d12]
    for j in range(4):
        a, b, c, d = tmp[j], tmp[4 + j], tmp[8 + j], tmp[12 + j]
        s03, d03 = a + d, a - d
        s12, d12 = b + c, b - c
        out[j] = s03 + s12
        out[4 + j] = d03 + d12
        out[8 + j] = s03 - s12
        out[12 + j] = d03 - d12
    return out


def dequant_ac(level, qp, cls):
    return (level * QUANT_V[qp % 6][cls]) << (qp // 6)


def dequant_luma_dc(c, qp):
    v = QUANT_V[qp % 6][0]
    if qp >= 12:
        return (c * v) << (qp // 6 - 2)
    return (c * v + (1 << (1 - qp // 6))) >> (2 - qp // 6)


def dequant_chroma_dc(c, qp):
    v = QUANT_V[qp % 6][0]
    if qp >= 6:
        return (c * v) << (qp // 6 - 1)
    return (c * v) >> 1


def blk_xy(blk):
    bx = 2 * ((blk >> 2) & 1) + (blk & 1)
    by = 2 * (blk >> 3) + ((blk >> 1) & 1)
    return bx, by



# ---- in-loop deblocking filter (§8.7) --------------------------------------
# disable_deblocking_filter_idc == 2: slice-boundary edges are skipped, so
# with one-slice-per-MB-row only row-internal edges filter.

DB_ALPHA = [0]*16 + [4, 4, 5, 6, 7, 8, 9, 10, 12, 13, 15, 17, 20, 22, 25, 28,
                     32, 36, 40, 45, 50, 56, 63, 71, 80, 90, 101, 113, 127,
                     144, 162, 182, 203, 226, 255, 255]
DB_BETA = [0]*16 + [2, 2, 2, 3, 3, 3, 3, 4, 4, 4, 6, 6, 7, 7, 8, 8,
                    9, 9, 10, 10, 11, 11, 12, 12, 13, 13, 14, 14, 15, 15,
                    16, 16, 17, 17, 18, 18]
DB_TC0 = ([ (0,0,0) ]*17 + [(0,0,1)]*4 + [(0,1,1)]*2 + [(1,1,1)]*4 +
          [(1,1,2)]*4 + [(1,2,3)]*2 + [(2,2,3),(2,2,4),(2,3,4),(2,3,4),
           (3,3,5),(3,4,6),(3,4,6),(4,5,7),(4,5,8),(5,6,9),(6,7,10),
           (6,8,11),(7,9,12),(8,10,13),(9,12,15),(10,13,17),(11,16,20),
           (13,18,23),(14,20,25)])


def _db_filter_lines(bs, alpha, beta, tc0, P, Q, chroma):
    """Filter a batch of independent lines. P/Q: arrays [4][n] (or [2][n]
    for chroma) of samples away from the edge; returns modified copies."""
    p = [x.astype(np.int32) for x in P]
    q = [x.astype(np.int32) for x in Q]
    fl = ((np.abs(p[0] - q[0]) < alpha) & (np.abs(p[1] - p[0]) < beta)
          & (np.abs(q[1] - q[0]) < beta))
    if chroma:
        if bs < 4:
            tc = tc0 + 1
            d = np.clip(((q[0] - p[0]) * 4 + (p[1] - q[1]) + 4) >> 3, -tc, tc)
            p0n = np.clip(p[0] + d, 0, 255)
            q0n = np.clip(q[0] - d, 0, 255)
        else:
            p0n = (2 * p[1] + p[0] + q[1] + 2) >> 2
            q0n = (2 * q[1] + q[0] + p[1] + 2) >> 2
        p[0] = np.where(fl, p0n, p[0])
        q[0] = np.where(fl, q0n, q[0])
        return p, q
    ap = np.abs(p[2] - p[0]) < beta
    aq = np.abs(q[2] - q[0]) < beta
    if bs < 4:
        tc = tc0 + ap.astype(np.int32) + aq.astype(np.int32)
        d = np.clip(((q[0] - p[0]) * 4 + (p[1] - q[1]) + 4) >> 3, -tc, tc)
        p0n = np.clip(p[0] + d, 0, 255)
        q0n = np.clip(q[0] - d, 0, 255)
        p1n = p[1] + np.clip((p[2] + ((p[0] + q[0] + 1) >> 1) - 2 * p[1])
                             >> 1, -tc0, tc0)
        q1n = q[1] + np.clip((q[2] + ((p[0] + q[0] + 1) >> 1) - 2 * q[1])
                             >> 1, -tc0, tc0)
        p[1] = np.where(fl & ap, p1n, p[1])
        q[1] = np.where(fl & aq, q1n, q[1])
        p[0] = np.where(fl, p0n, p[0])
        q[0] = np.where(fl, q0n, q[0])
        return p, q
    small = np.abs(p[0] - q[0]) < ((alpha >> 2) + 2)
    strong_p = fl & ap & small
    strong_q = fl & aq & small
    p0s = (p[2] + 2 * p[1] + 2 * p[0] + 2 * q[0] + q[1] + 4) >> 3
    p1s = (p[2] + p[1] + p[0] + q[0] + 2) >> 2
    p2s = (2 * p[3] + 3 * p[2] + p[1] + p[0] + q[0] + 4) >> 3
    p0w = (2 * p[1] + p[0] + q[1] + 2) >> 2
    q0s = (q[2] + 2 * q[1] + 2 * q[0] + 2 * p[0] + p[1] + 4) >> 3
    q1s = (q[2] + q[1] + q[0] + p[0] + 2) >> 2
    q2s = (2 * q[3] + 3 * q[2] + q[1] + q[0] + p[0] + 4) >> 3
    q0w = (2 * q[1] + q[0] + p[1] + 2) >> 2
    newp0 = np.where(strong_p, p0s, np.where(fl, p0w, p[0]))
    newp1 = np.where(strong_p, p1s, p[1])
    newp2 = np.where(strong_p, p2s, p[2])
    newq0 = np.where(strong_q, q0s, np.where(fl, q0w, q[0]))
    newq1 = np.where(strong_q, q1s, q[1])
    newq2 = np.where(strong_q, q2s, q[2])
    p[0], p[1], p[2] = newp0, newp1, newp2
    q[0], q[1], q[2] = newq0, newq1, newq2
    return p, q


def _db_bs(mp, pblk, mq, qblk, mb_edge):
    if mp["intra"] or mq["intra"]:
        return 4 if mb_edge else 3
    if ((mp["nz"] >> pblk) & 1) or ((mq["nz"] >> qblk) & 1):
        return 2
    if abs(mp["mv"][0] - mq["mv"][0]) >= 4 or             abs(mp["mv"][1] - mq["mv"][1]) >= 4:
        return 1
    return 0


def deblock_segment_py(Y, Cb, Cr, mb_row, mbx0, infos, qp, qpc):
    """Independent from-spec implementation of the segment deblock (the
    encoder's deblock.h counterpart; equality asserted by the tests)."""
    iA = min(max(qp, 0), 51)
    alpha, beta = DB_ALPHA[iA], DB_BETA[iA]
    iAc = min(max(qpc, 0), 51)
    alpha_c, beta_c = DB_ALPHA[iAc], DB_BETA[iAc]
    y0, cy0 = mb_row * 16, mb_row * 8
    for m, cur in enumerate(infos):
        left = infos[m - 1] if m > 0 else None
        x0 = (mbx0 + m) * 16
        # vertical luma edges
        for e in range(4):
            if e == 0 and left is None:
                continue
            ex = x0 + e * 4
            for br4 in range(4):       # per 4-row group (bS can differ)
                rows = slice(y0 + br4 * 4, y0 + br4 * 4 + 4)
                pblk = br4 * 4 + (3 if e == 0 else e - 1)
                qblk = br4 * 4 + e
                bs = _db_bs(left if e == 0 else cur, pblk, cur, qblk,
                            e == 0)
                if bs == 0:
                    continue
                tc0 = DB_TC0[iA][bs - 1] if bs < 4 else 0
                P = [Y[rows, ex - 1 - i].copy() for i in range(4)]
                Q = [Y[rows, ex + i].copy() for i in range(4)]
                P, Q = _db_filter_lines(bs, alpha, beta, tc0, P, Q, False)
                for i in range(3):
                    Y[rows, ex - 1 - i] = P[i]
                    Y[rows, ex + i] = Q[i]
        # horizontal luma edges (y=4,8,12)
        for e in range(1, 4):
            ey = y0 + e * 4
            for bc in range(4):
                cols = slice(x0 + bc * 4, x0 + bc * 4 + 4)
                pblk = (e - 1) * 4 + bc
                qblk = e * 4 + bc
                bs = _db_bs(cur, pblk, cur, qblk, False)
                if bs == 0:
                    continue
                tc0 = DB_TC0[iA][bs - 1] if bs < 4 else 0
                P = [Y[ey - 1 - i, cols].copy() for i in range(4)]
                Q = [Y[ey + i, cols].copy() for i in range(4)]
                P, Q = _db_filter_lines(bs, alpha, beta, tc0, P, Q, False)
                for i in range(3):
                    Y[ey - 1 - i, cols] = P[i]
                    Y[ey + i, cols] = Q[i]
        # chroma (skipped for monochrome planes: ChromaArrayType == 0)
        if Cb is None:
            continue
        cx0 = (mbx0 + m) * 8
        for C in (Cb, Cr):
            for e in range(2):         # vertical, chroma x = 0, 4
                if e == 0 and left is None:
                    continue
                ex = cx0 + e * 4
                for cy in range(2):    # 4-chroma-row groups -> luma rows
                    rows = slice(cy0 + cy * 4, cy0 + cy * 4 + 4)
                    # luma block row for chroma rows cy*4..cy*4+3 varies
                    # per line; split into 2-line halves (luma rows 0-3 /
                    # 4-7 per half)
                    for half in range(2):
                        r2 = slice(cy0 + cy * 4 + half * 2,
                                   cy0 + cy * 4 + half * 2 + 2)
                        lum_brow = cy * 2 + half
                        pblk = lum_brow * 4 + (3 if e == 0 else 1)
                        qblk = lum_brow * 4 + (0 if e == 0 else 2)
                        bs = _db_bs(left if e == 0 else cur, pblk, cur,
                                    qblk, e == 0)
                        if bs == 0:
                            continue
                        tc0 = DB_TC0[iAc][bs - 1] if bs < 4 else 0
                        P = [C[r2, ex - 1 - i].copy() for i in range(2)]
                        Q = [C[r2, ex + i].copy() for i in range(2)]
                        P, Q = _db_filter_lines(bs, alpha_c, beta_c, tc0,
                                                P, Q, True)
                        C[r2, ex - 1] = P[0]
                        C[r2, ex] = Q[0]
            # horizontal, chroma y = 4
            ey = cy0 + 4
            for bc in range(2):
                for half in range(2):
                    c2 = slice(cx0 + bc * 4 + half * 2,
                               cx0 + bc * 4 + half * 2 + 2)
                    lum_bcol = bc * 2 + half
                    pblk = 1 * 4 + lum_bcol
                    qblk = 2 * 4 + lum_bcol
                    bs = _db_bs(cur, pblk, cur, qblk, False)
                    if bs == 0:
                        continue
                    tc0 = DB_TC0[iAc][bs - 1] if bs < 4 else 0
                    P = [C[ey - 1 - i, c2].copy() for i in range(2)]
                    Q = [C[ey + i, c2].copy() for i in range(2)]
                    P, Q = _db_filter_lines(bs, alpha_c, beta_c, tc0, P, Q,
                                            True)
                    C[ey - 1, c2] = P[0]
                    C[ey, c2] = Q[0]


class Decoder:
    def __init__(self):
        self.sps = None
        self.pps = None
        self.y = self.cb = self.cr = None
        self.ref_y = self.ref_cb = self.ref_cr = None
        self.planes = [None, None, None]       # Hi444 separate planes
        self.ref_planes = [None, None, None]
        self.frames = []  # decoded (cropped) frames as (y, cb, cr)

    # ---- headers -----------------------------------------------------------
    def parse_sps(self, br: BitReader):
        s = {}
        s["profile_idc"] = br.u(8)
        br.u(8)  # constraint flags + reserved
        s["level_idc"] = br.u(8)
        assert br.ue() == 0
        s["separate"] = False
        if s["profile_idc"] in (100, 110, 122, 244):
            cfi = br.ue()           # chroma_format_idc
            assert cfi == 3, "subset: high profiles only as Hi444"
            s["separate"] = br.u(1) == 1
            assert s["separate"], "subset: separate_colour_plane only"
            assert br.ue() == 0     # bit_depth_luma_minus8
            assert br.ue() == 0     # bit_depth_chroma_minus8
            assert br.u(1) == 0     # qpprime_y_zero_transform_bypass
            assert br.u(1) == 0     # seq_scaling_matrix_present
        s["log2_max_frame_num"] = br.ue() + 4
        s["poc_type"] = br.ue()
        assert s["poc_type"] == 2
        s["max_num_ref_frames"] = br.ue()
        br.u(1)
        s["mbw"] = br.ue() + 1
        s["mbh"] = br.ue() + 1
        assert br.u(1) == 1  # frame_mbs_only
        br.u(1)  # direct_8x8
        if br.u(1):  # cropping
            cl, cr_, ct, cb_ = br.ue(), br.ue(), br.ue(), br.ue()
            # CropUnit is 1 for ChromaArrayType==0 (separate planes), 2
            # for 4:2:0
            u = 1 if s["separate"] else 2
            s["crop"] = (cl * u, cr_ * u, ct * u, cb_ * u)
        else:
            s["crop"] = (0, 0, 0, 0)
        self.sps = s

    def parse_pps(self, br: BitReader):
        p = {}
        assert br.ue() == 0 and br.ue() == 0
        assert br.u(1) == 0  # CAVLC
        br.u(1)
        assert br.ue() == 0  # one slice group
        p["num_ref_idx_l0"] = br.ue() + 1
        br.ue()
        br.u(1)
        br.u(2)
        p["pic_init_qp"] = br.se() + 26
        br.se()
        p["chroma_qp_offset"] = br.se()
        p["deblocking_control"] = br.u(1)
        assert br.u(1) == 0  # constrained_intra_pred
        br.u(1)
        self.pps = p

    # ---- slice -------------------------------------------------------------
    def decode_slice(self, br: BitReader, nal_type: int):
        s = self.sps
        mbw, mbh = s["mbw"], s["mbh"]
        W, H = mbw * 16, mbh * 16
        idr = nal_type == 5
        first_mb = br.ue()
        slice_type = br.ue()
        is_i = slice_type in (2, 7)
        is_p = slice_type in (0, 5)
        assert is_i or is_p
        assert br.ue() == 0  # pps id
        cp = br.u(2) if s["separate"] else None   # colour_plane_id
        br.u(s["log2_max_frame_num"])  # frame_num
        if idr:
            br.ue()  # idr_pic_id
        if is_p:
            if br.u(1):  # num_ref_idx_override
                br.ue()
            assert br.u(1) == 0  # no ref list modification
        # dec_ref_pic_marking
        if idr:
            br.u(1)
            br.u(1)
        else:
            assert br.u(1) == 0
        qp = self.pps["pic_init_qp"] + br.se()
        dbf = 1
        if self.pps["deblocking_control"]:
            dbf = br.ue()
            assert dbf in (1, 2), \
                "decoder subset: deblocking off or within-slice only"

        if cp is not None:
            # route the shared mono machinery at this colour plane:
            # self.y is the plane, chroma disabled (ChromaArrayType==0)
            if idr and first_mb == 0:
                self.planes[cp] = np.zeros((H, W), np.int32)
            self.y = self.planes[cp]
            self.cb = self.cr = None
            self.ref_y = self.ref_planes[cp]
        elif idr and first_mb == 0:
            self.y = np.zeros((H, W), np.int32)
            self.cb = np.zeros((H // 2, W // 2), np.int32)
            self.cr = np.zeros((H // 2, W // 2), np.int32)
        if self.y is None:
            raise ValueError("P slice before any IDR")

        mb_row = first_mb // mbw

        ctx = {
            "left_avail": False,
            "left_inter": False,
            "left_mv": (0, 0),
            "left_luma_nc": [0] * 4,
            "left_cb_nc": [0] * 2,
            "left_cr_nc": [0] * 2,
        }
        mbx = first_mb % mbw        # slices may start mid-row (segments)
        db_infos = []               # per-MB info for the deblocking filter
        skip_left = 0
        if is_p:
            skip_left = br.ue()
        while mbx < mbw:
            if is_p and skip_left > 0:
                self.decode_skip(mbx, mb_row)
                ctx.update(left_avail=True, left_inter=True, left_mv=(0, 0),
                           left_luma_nc=[0] * 4, left_cb_nc=[0] * 2,
                           left_cr_nc=[0] * 2)
                db_infos.append({"intra": 0, "mv": (0, 0), "nz": 0})
                skip_left -= 1
                mbx += 1
                continue
            if not br.more_rbsp_data():
                break  # end of this slice (trailing skips / segment end)
            mb_type = br.ue()
            if is_p and mb_type < 5:
                assert mb_type == 0, f"subset: P mb_type {mb_type}"
                db_infos.append(self.decode_p16(br, mbx, mb_row, ctx, qp))
            else:
                it = mb_type - 5 if is_p else mb_type
                assert 1 <= it <= 24, f"subset: mb_type {mb_type}"
                self.decode_i16(br, mbx, mb_row, it, qp, ctx)
                db_infos.append({"intra": 1, "mv": (0, 0), "nz": 0})
            mbx += 1
            if is_p and mbx < mbw and br.more_rbsp_data():
                skip_left = br.ue()
        if dbf == 2 and db_infos:
            deblock_segment_py(self.y, self.cb, self.cr, mb_row,
                               first_mb % mbw, db_infos, qp, chroma_qp(qp))
        return mb_row, mbx - first_mb % mbw

    def decode_skip(self, mbx, mby):
        x0, y0 = mbx * 16, mby * 16
        cx0, cy0 = mbx * 8, mby * 8
        self.y[y0:y0 + 16, x0:x0 + 16] = self.ref_y[y0:y0 + 16, x0:x0 + 16]
        if self.cb is None:
            return
        self.cb[cy0:cy0 + 8, cx0:cx0 + 8] = self.ref_cb[cy0:cy0 + 8,
                                                        cx0:cx0 + 8]
        self.cr[cy0:cy0 + 8, cx0:cx0 + 8] = self.ref_cr[cy0:cy0 + 8,
                                                        cx0:cx0 + 8]

    # coded_block_pattern me(v) inverse for Inter prediction: the six
    # Table 9-4 entries our encoder can emit (cbp_luma in {0,15},
    # cbp_chroma in {0,1,2}).
    INTER_CBP_FROM_CODENUM = {0: 0, 1: 16, 6: 32, 11: 15, 12: 47, 19: 31}

    def decode_p16(self, br, mbx, mby, ctx, qp):
        mvdx, mvdy = br.se(), br.se()
        cbp_cn = br.ue()
        assert cbp_cn in self.INTER_CBP_FROM_CODENUM, \
            f"subset: inter cbp codeNum {cbp_cn}"
        cbp = self.INTER_CBP_FROM_CODENUM[cbp_cn]
        if self.cb is None:
            # monochrome plane: the encoder only ever emits cbp == 0
            # (residual MBs fall back to I16x16)
            assert cbp == 0, f"mono inter cbp {cbp}"
        cbp_luma, cbp_chroma = cbp & 15, cbp >> 4
        mvpx, mvpy = ctx["left_mv"] if (ctx["left_avail"] and
                                        ctx["left_inter"]) else (0, 0)
        mvx, mvy = mvdx + mvpx, mvdy + mvpy
        x0, y0 = mbx * 16, mby * 16
        cx0, cy0 = mbx * 8, mby * 8
        pred_y = luma_mc(self.ref_y, x0, y0, mvx, mvy)
        mono = self.cb is None
        pred_cb = None if mono else chroma_mc(self.ref_cb, cx0, cy0,
                                              mvx, mvy)
        pred_cr = None if mono else chroma_mc(self.ref_cr, cx0, cy0,
                                              mvx, mvy)
        new_luma_nc = np.zeros((4, 4), np.int32)
        new_cb_nc = np.zeros((2, 2), np.int32)
        new_cr_nc = np.zeros((2, 2), np.int32)
        if cbp:
            qp = qp + br.se()  # mb_qp_delta
        if cbp_luma:
            # 16 full 4x4 blocks, Z-order, no DC Hadamard for inter
            for blk in range(16):
                bx, by = blk_xy(blk)
                if bx > 0:
                    nC = int(new_luma_nc[by, bx - 1])
                elif ctx["left_avail"]:
                    nC = ctx["left_luma_nc"][by]
                else:
                    nC = 0
                zz, tc = residual_cavlc(br, 16, nC)
                new_luma_nc[by, bx] = tc
                coeffs = np.zeros(16, np.int64)
                for i in range(16):
                    coeffs[ZZ4[i]] = dequant_ac(zz[i], qp,
                                                coeff_class(ZZ4[i] >> 2,
                                                            ZZ4[i] & 3))
                rec = idct4(coeffs).reshape(4, 4)
                pred_y[by * 4:by * 4 + 4, bx * 4:bx * 4 + 4] += rec
        qpc = chroma_qp(qp + self.pps["chroma_qp_offset"])
        cdc = {"cb": [0] * 4, "cr": [0] * 4}
        if cbp_chroma >= 1:
            for comp in ("cb", "cr"):
                zz, _ = residual_cavlc(br, 4, -1)
                q = zz
                w0 = q[0] + q[1] + q[2] + q[3]
                w1 = q[0] - q[1] + q[2] - q[3]
                w2 = q[0] + q[1] - q[2] - q[3]
                w3 = q[0] - q[1] - q[2] + q[3]
                cdc[comp] = [dequant_chroma_dc(w, qpc)
                             for w in (w0, w1, w2, w3)]
        cac = {"cb": np.zeros((4, 16), np.int64),
               "cr": np.zeros((4, 16), np.int64)}
        if cbp_chroma == 2:
            for comp, newnc, leftnc in (("cb", new_cb_nc, ctx["left_cb_nc"]),
                                        ("cr", new_cr_nc,
                                         ctx["left_cr_nc"])):
                for sub in range(4):
                    cx, cy = sub & 1, sub >> 1
                    if cx > 0:
                        nC = int(newnc[cy, 0])
                    elif ctx["left_avail"]:
                        nC = leftnc[cy]
                    else:
                        nC = 0
                    zz, tc = residual_cavlc(br, 15, nC)
                    newnc[cy, cx] = tc
                    for i in range(1, 16):
                        cac[comp][sub][ZZ4[i]] = dequant_ac(
                            zz[i - 1], qpc,
                            coeff_class(ZZ4[i] >> 2, ZZ4[i] & 3))
        if cbp_chroma >= 1:
            for comp, pred in (("cb", pred_cb), ("cr", pred_cr)):
                for sub in range(4):
                    scx, scy = (sub & 1) * 4, (sub >> 1) * 4
                    coeffs = cac[comp][sub].copy()
                    coeffs[0] = cdc[comp][sub]
                    rec = idct4(coeffs).reshape(4, 4)
                    pred[scy:scy + 4, scx:scx + 4] += rec
        self.y[y0:y0 + 16, x0:x0 + 16] = np.clip(pred_y, 0, 255)
        if not mono:
            self.cb[cy0:cy0 + 8, cx0:cx0 + 8] = np.clip(pred_cb, 0, 255)
            self.cr[cy0:cy0 + 8, cx0:cx0 + 8] = np.clip(pred_cr, 0, 255)
        nz = 0
        for by in range(4):
            for bx in range(4):
                if new_luma_nc[by, bx]:
                    nz |= 1 << (by * 4 + bx)
        ctx.update(left_avail=True, left_inter=True, left_mv=(mvx, mvy),
                   left_luma_nc=[int(new_luma_nc[by, 3]) for by in range(4)],
                   left_cb_nc=[int(new_cb_nc[cy, 1]) for cy in range(2)],
                   left_cr_nc=[int(new_cr_nc[cy, 1]) for cy in range(2)])
        return {"intra": 0, "mv": (mvx, mvy), "nz": nz}

    def decode_i16(self, br, mbx, mby, i16_type, qp, ctx):
        t = i16_type - 1
        pred_mode = t % 4
        cbp_chroma = (t // 4) % 3
        cbp_luma = 15 if t >= 12 else 0
        mono = self.cb is None
        if mono:
            assert cbp_chroma == 0, "mono I16 with chroma cbp"
            chroma_mode = 0          # intra_chroma_pred_mode not present
        else:
            chroma_mode = br.ue()
        qp = qp + br.se()  # mb_qp_delta
        x0, y0 = mbx * 16, mby * 16
        cx0, cy0 = mbx * 8, mby * 8

        # luma prediction
        assert pred_mode in (1, 2), "subset: V/Plane luma pred not emitted"
        pred = np.zeros((16, 16), np.int32)
        if pred_mode == 1:  # H
            assert ctx["left_avail"]
            col = self.y[y0:y0 + 16, x0 - 1]
            pred[:] = col[:, None]
        else:  # DC
            if ctx["left_avail"]:
                dc = (int(self.y[y0:y0 + 16, x0 - 1].sum()) + 8) >> 4
            else:
                dc = 128
            pred[:] = dc

        # DC block
        nC = ctx["left_luma_nc"][0] if ctx["left_avail"] else 0
        new_luma_nc = np.zeros((4, 4), np.int32)  # per (by,bx) totals
        zz_dc, _tc = residual_cavlc(br, 16, nC)
        raster_dc = [0] * 16
        for i in range(16):
            raster_dc[ZZ4[i]] = zz_dc[i]
        ih = hadamard4_inv(raster_dc)
        dcrec = [dequant_luma_dc(c, qp) for c in ih]

        ac = np.zeros((16, 16), np.int64)  # [blk][raster pos]
        if cbp_luma:
            for blk in range(16):
                bx, by = blk_xy(blk)
                if bx > 0:
                    nC = int(new_luma_nc[by, bx - 1])
                elif ctx["left_avail"]:
                    nC = ctx["left_luma_nc"][by]
                else:
                    nC = 0
                zz, tc = residual_cavlc(br, 15, nC)
                new_luma_nc[by, bx] = tc
                for i in range(1, 16):
                    ac[blk][ZZ4[i]] = dequant_ac(zz[i - 1], qp,
                                                 coeff_class(ZZ4[i] >> 2,
                                                             ZZ4[i] & 3))
        for blk in range(16):
            bx, by = blk_xy(blk)
            coeffs = ac[blk].copy()
            coeffs[0] = dcrec[by * 4 + bx]
            rec = idct4(coeffs).reshape(4, 4)
            self.y[y0 + by * 4:y0 + by * 4 + 4, x0 + bx * 4:x0 + bx * 4 + 4] = \
                np.clip(rec + pred[by * 4:by * 4 + 4, bx * 4:bx * 4 + 4],
                        0, 255)

        # chroma
        new_cb_nc = np.zeros((2, 2), np.int32)
        new_cr_nc = np.zeros((2, 2), np.int32)
        if mono:
            ctx.update(left_avail=True, left_inter=False,
                       left_luma_nc=[int(new_luma_nc[by, 3])
                                     for by in range(4)])
            return
        qpc = chroma_qp(qp + self.pps["chroma_qp_offset"])
        assert chroma_mode in (0, 1), "subset: chroma V/Plane not emitted"
        cpred = {}
        for comp, plane in (("cb", self.cb), ("cr", self.cr)):
            p = np.zeros((8, 8), np.int32)
            if chroma_mode == 1:  # H
                assert ctx["left_avail"]
                p[:] = plane[cy0:cy0 + 8, cx0 - 1][:, None]
            else:
                for sub in range(4):
                    scx, scy = (sub & 1) * 4, (sub >> 1) * 4
                    if ctx["left_avail"]:
                        d = (int(plane[cy0 + scy:cy0 + scy + 4,
                                       cx0 - 1].sum()) + 2) >> 2
                    else:
                        d = 128
                    p[scy:scy + 4, scx:scx + 4] = d
            cpred[comp] = p

        cdc = {"cb": [0] * 4, "cr": [0] * 4}
        if cbp_chroma >= 1:
            for comp in ("cb", "cr"):
                zz, _ = residual_cavlc(br, 4, -1)
                q = zz  # raster 2x2 (c00,c01,c10,c11)
                w0 = q[0] + q[1] + q[2] + q[3]
                w1 = q[0] - q[1] + q[2] - q[3]
                w2 = q[0] + q[1] - q[2] - q[3]
                w3 = q[0] - q[1] - q[2] + q[3]
                cdc[comp] = [dequant_chroma_dc(w, qpc)
                             for w in (w0, w1, w2, w3)]
        cac = {"cb": np.zeros((4, 16), np.int64),
               "cr": np.zeros((4, 16), np.int64)}
        if cbp_chroma == 2:
            for comp, newnc, leftnc in (("cb", new_cb_nc, ctx["left_cb_nc"]),
                                        ("cr", new_cr_nc, ctx["left_cr_nc"])):
                for sub in range(4):
                    cx, cy = sub & 1, sub >> 1
                    if cx > 0:
                        nC = int(newnc[cy, 0])
                    elif ctx["left_avail"]:
                        nC = leftnc[cy]
                    else:
                        nC = 0
                    zz, tc = residual_cavlc(br, 15, nC)
                    newnc[cy, cx] = tc
                    for i in range(1, 16):
                        cac[comp][sub][ZZ4[i]] = dequant_ac(
                            zz[i - 1], qpc,
                            coeff_class(ZZ4[i] >> 2, ZZ4[i] & 3))
        for comp, plane in (("cb", self.cb), ("cr", self.cr)):
            for sub in range(4):
                scx, scy = (sub & 1) * 4, (sub >> 1) * 4
                coeffs = cac[comp][sub].copy()
                coeffs[0] = cdc[comp][sub]
                rec = idct4(coeffs).reshape(4, 4)
                plane[cy0 + scy:cy0 + scy + 4, cx0 + scx:cx0 + scx + 4] = \
                    np.clip(rec + cpred[comp][scy:scy + 4, scx:scx + 4],
                            0, 255)

        ctx.update(left_avail=True, left_inter=False,
                   left_luma_nc=[int(new_luma_nc[by, 3]) for by in range(4)],
                   left_cb_nc=[int(new_cb_nc[cy, 1]) for cy in range(2)],
                   left_cr_nc=[int(new_cr_nc[cy, 1]) for cy in range(2)])

    # ---- top level ---------------------------------------------------------
    def decode(self, data: bytes):
        """Decode an Annex-B stream; returns list of (y, cb, cr) uint8 frames
        (cropped to the SPS-declared size)."""
        mbs_done = 0
        total_mbs = None
        for nal in split_nals(data):
            nal_type = nal[0] & 0x1F
            br = BitReader(nal[1:])
            if nal_type == 7:
                self.parse_sps(br)
                total_mbs = self.sps["mbw"] * self.sps["mbh"] * (
                    3 if self.sps["separate"] else 1)
            elif nal_type == 8:
                self.parse_pps(br)
            elif nal_type in (1, 5):
                _, n_mbs = self.decode_slice(br, nal_type)
                mbs_done += n_mbs
                if mbs_done == total_mbs:
                    self.finish_frame()
                    mbs_done = 0
        return self.frames

    def finish_frame(self):
        s = self.sps
        cl, cr_, ct, cb_ = s["crop"]
        W, H = s["mbw"] * 16, s["mbh"] * 16
        w, h = W - cl - cr_, H - ct - cb_
        if s["separate"]:
            # Hi444: three full-resolution planes form the frame
            self.frames.append(tuple(
                p[ct:ct + h, cl:cl + w].astype(np.uint8)
                for p in self.planes))
            self.ref_planes = [p.copy() for p in self.planes]
            return
        self.frames.append((
            self.y[ct:ct + h, cl:cl + w].astype(np.uint8),
            self.cb[ct // 2:(ct + h) // 2, cl // 2:(cl + w) // 2]
                .astype(np.uint8),
            self.cr[ct // 2:(ct + h) // 2, cl // 2:(cl + w) // 2]
                .astype(np.uint8),
        ))
        self.ref_y = self.y.copy()
        self.ref_cb = self.cb.copy()
        self.ref_cr = self.cr.copy()
