/* Opus (CELT-class) decoder for the hipflux audio path — a direct port
 * of tests/opus_ref_decoder.py (itself the from-spec mirror of
 * native/cpu/opus/celt.cpp; see the conformance ledger in celt.h).
 *
 * RFC 6716 §4.1 range decoder + CWRS PVQ un-indexing (BigInt where
 * V(n,k) can exceed 2^53) + naive IMDCT with a precomputed cos basis
 * (960×1920 Float32 ≈ 7 MB, ~92 MFLOP/s at 50 fps — negligible).
 *
 * Works in the browser (window.OpusDecoder) and under node
 * (module.exports) so tests/test_opus_js.py can anchor it byte-for-byte
 * against the C++ encoder via the Python reference decoder.
 */
"use strict";

(function (root) {

const EC_SYM_BITS = 8, EC_SYM_MAX = 0xff;
const EC_CODE_BITS = 32;
const EC_CODE_TOP = 2147483648;            // 1 << 31
const EC_CODE_BOT = EC_CODE_TOP / 256;     // 1 << 23
const EC_CODE_EXTRA = (EC_CODE_BITS - 2) % EC_SYM_BITS + 1;  // 7
const EC_UINT_BITS = 8;

const FRAME = 960, OVERLAP = 120, NBANDS = 21;
const BAND_BINS = [0, 8, 16, 24, 32, 40, 48, 56, 64, 80, 96, 112, 128,
                   160, 192, 224, 272, 320, 384, 480, 624, 800];
const EMEANS = [6.4375, 6.25, 5.75, 5.3125, 5.0625, 4.8125, 4.5, 4.375,
                4.875, 4.6875, 4.5625, 4.4375, 4.875, 4.625, 4.3125,
                4.5, 4.375, 4.625, 4.75, 4.4375, 3.75];
const ALLOC = [
  [0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0],
  [90, 80, 75, 69, 63, 56, 49, 40, 34, 29, 20, 18, 10, 0, 0, 0, 0, 0,
   0, 0, 0],
  [110, 100, 90, 84, 78, 71, 65, 58, 51, 45, 39, 32, 26, 20, 12, 0, 0,
   0, 0, 0, 0],
  [118, 110, 103, 93, 86, 80, 75, 70, 65, 59, 53, 47, 40, 31, 23, 15,
   4, 0, 0, 0, 0],
  [126, 119, 112, 104, 95, 89, 83, 78, 72, 66, 60, 54, 47, 39, 32, 25,
   17, 12, 1, 0, 0],
  [134, 127, 120, 114, 103, 97, 91, 85, 78, 72, 66, 60, 54, 47, 41, 35,
   29, 23, 16, 10, 1],
  [144, 137, 130, 124, 113, 107, 101, 95, 88, 82, 76, 70, 64, 57, 51,
   45, 39, 33, 26, 15, 1],
  [152, 145, 138, 132, 123, 117, 111, 105, 98, 92, 86, 80, 74, 67, 61,
   55, 49, 43, 36, 20, 1],
  [162, 155, 148, 142, 133, 127, 121, 115, 108, 102, 96, 90, 84, 77,
   71, 65, 59, 53, 46, 30, 1],
  [172, 165, 158, 152, 143, 137, 131, 125, 118, 112, 106, 100, 94, 87,
   81, 75, 69, 63, 56, 45, 20],
  [200, 200, 200, 200, 200, 200, 200, 200, 200, 200, 200, 200, 200,
   200, 200, 200, 200, 200, 200, 200, 200],
];

function ilog(v) { let l = 0; while (v) { l++; v = Math.floor(v / 2); } return l; }

/* ---- range decoder (RFC 6716 §4.1.2–§4.1.4) ---- */

class RangeDecoder {
  constructor(buf) {                       // buf: Uint8Array
    this.buf = buf;
    this.storage = buf.length;
    this.offs = 0;
    this.endOffs = 0;
    this.endWindow = 0;                    // < 2^45, exact in a double
    this.nendBits = 0;
    this.nbitsTotal = EC_CODE_BITS + 1 -
      Math.floor((EC_CODE_BITS - EC_CODE_EXTRA) / EC_SYM_BITS) * EC_SYM_BITS;
    this.rng = 1 << EC_CODE_EXTRA;
    this.rem = this.readByte();
    this.val = this.rng - 1 - (this.rem >> (EC_SYM_BITS - EC_CODE_EXTRA));
    this.error = false;
    this.ext = 0;
    this.normalize();
  }
  readByte() {
    return this.offs < this.storage ? this.buf[this.offs++] : 0;
  }
  readByteFromEnd() {
    if (this.endOffs < this.storage) {
      this.endOffs++;
      return this.buf[this.storage - this.endOffs];
    }
    return 0;
  }
  normalize() {
    /* val < 2^31 so val*256 < 2^39 — stay in double arithmetic, never
       JS 32-bit shifts */
    while (this.rng <= EC_CODE_BOT) {
      this.nbitsTotal += EC_SYM_BITS;
      this.rng = (this.rng * 256) % 4294967296;
      let sym = this.rem;
      this.rem = this.readByte();
      sym = ((sym * 256 + this.rem) >>> (EC_SYM_BITS - EC_CODE_EXTRA));
      this.val = (this.val * 256 + (EC_SYM_MAX & ~sym)) % EC_CODE_TOP;
    }
  }
  decode(ft) {
    this.ext = Math.floor(this.rng / ft);
    const s = Math.floor(this.val / this.ext);
    return ft - Math.min(s + 1, ft);
  }
  decUpdate(fl, fh, ft) {
    const s = this.ext * (ft - fh);
    this.val -= s;
    this.rng = fl > 0 ? this.ext * (fh - fl) : this.rng - s;
    this.normalize();
  }
  decBitLogp(logp) {
    const r = this.rng, d = this.val;
    const s = Math.floor(r / Math.pow(2, logp));
    const ret = d < s ? 1 : 0;
    if (!ret) this.val = d - s;
    this.rng = ret ? s : r - s;
    this.normalize();
    return ret;
  }
  decUint(ftIn) {
    let ft = ftIn - 1;
    const ftb = ilog(ft);
    if (ftb > EC_UINT_BITS) {
      const lo = ftb - EC_UINT_BITS;
      const fth = Math.floor(ft / Math.pow(2, lo)) + 1;
      const s = this.decode(fth);
      this.decUpdate(s, s + 1, fth);
      const t = s * Math.pow(2, lo) + this.decBits(lo);
      if (t <= ft) return t;
      this.error = true;
      return ft;
    }
    const s = this.decode(ft + 1);
    this.decUpdate(s, s + 1, ft + 1);
    return s;
  }
  decBits(bits) {
    let window = this.endWindow, available = this.nendBits;
    while (available < bits) {
      window += this.readByteFromEnd() * Math.pow(2, available);
      available += EC_SYM_BITS;
    }
    const p = Math.pow(2, bits);
    const ret = window % p;
    this.endWindow = Math.floor(window / p);
    this.nendBits = available - bits;
    this.nbitsTotal += bits;
    return ret;
  }
  tell() { return this.nbitsTotal - ilog(this.rng); }
}

/* ---- PVQ (CWRS) — BigInt: V(n,k) reaches the 2^60 split bound ---- */

const vCache = new Map();

function pvqV(n, k) {                      // -> BigInt
  if (k === 0) return 1n;
  if (n === 0) return 0n;
  const key = n * 1024 + k;
  const hit = vCache.get(key);
  if (hit !== undefined) return hit;
  let cur = [1n];
  for (let i = 0; i < k; i++) cur.push(2n);
  for (let d = 2; d <= n; d++) {
    const prev = cur;
    cur = [1n];
    for (let i = 1; i <= k; i++) cur.push(prev[i] + cur[i - 1] + prev[i - 1]);
  }
  vCache.set(key, cur[k]);
  return cur[k];
}

function pvqUnindex(idxIn, n, k) {
  let idx = idxIn;                         // BigInt
  const y = new Array(n).fill(0);
  let kleft = k;
  for (let i = 0; i < n && kleft > 0; i++) {
    const dims = n - 1 - i;
    let a = 0;
    for (;;) {
      const cnt = pvqV(dims, kleft - a);
      const width = a === 0 ? cnt : 2n * cnt;
      if (idx < width) break;
      idx -= width;
      a++;
    }
    let neg = false;
    if (a > 0) {
      const cnt = pvqV(dims, kleft - a);
      if (idx >= cnt) { idx -= cnt; neg = true; }
    }
    y[i] = neg ? -a : a;
    kleft -= a;
  }
  return y;
}

const SPLIT_BOUND = 1n << 60n;

function decodeBandPvq(dec, n, k) {
  if (k === 0 || n === 0) return new Array(n).fill(0);
  if (n > 2 && pvqV(n, k) >= SPLIT_BOUND) {
    const h = Math.floor(n / 2);
    const kl = dec.decUint(k + 1);
    return decodeBandPvq(dec, h, kl)
      .concat(decodeBandPvq(dec, n - h, k - kl));
  }
  const total = pvqV(n, k);
  let idx;
  if (total > (1n << 30n)) {
    const hi = dec.decUint(Number(total >> 30n) + 1);
    const lo = dec.decBits(30);
    idx = (BigInt(hi) << 30n) | BigInt(lo);
  } else {
    idx = BigInt(dec.decUint(Number(total)));
  }
  return pvqUnindex(idx, n, k);
}

/* ---- IMDCT: precomputed cos basis, lazily built ---- */

let IMDCT_BASIS = null;                    // Float32Array[(2N)*N]
let WINDOW = null;                         // Float64Array[2N]

function buildTables() {
  const n = FRAME;
  WINDOW = new Float64Array(2 * n);
  const z = (n - OVERLAP) / 2;
  for (let i = 0; i < OVERLAP; i++) {
    const t = Math.sin(0.5 * Math.PI * (i + 0.5) / OVERLAP);
    const r = Math.sin(0.5 * Math.PI * t * t);
    WINDOW[z + i] = r;
    WINDOW[2 * n - 1 - z - i] = r;
  }
  for (let i = z + OVERLAP; i < 2 * n - z - OVERLAP; i++) WINDOW[i] = 1;
  IMDCT_BASIS = new Float32Array(2 * n * n);
  const c = Math.PI / n;
  for (let j = 0; j < 2 * n; j++) {
    const a = c * (j + 0.5 + n / 2);
    for (let kk = 0; kk < n; kk++)
      IMDCT_BASIS[j * n + kk] = Math.cos(a * (kk + 0.5));
  }
}

/* ---- packet decoder ---- */

class OpusDecoder {
  constructor() {
    this.prevTail = new Float64Array(FRAME);
  }

  /* pkt: Uint8Array (TOC + payload). Returns Float32Array of FRAME
     mono samples at 48 kHz (one frame of algorithmic delay). */
  decodePacket(pkt) {
    if (!IMDCT_BASIS) buildTables();
    const toc = pkt[0];
    if (toc >> 3 !== 31 || (toc & 0x7) !== 0)
      throw new Error("unsupported Opus TOC " + toc);
    const dec = new RangeDecoder(pkt.subarray(1));

    if (dec.decBitLogp(15) !== 0) throw new Error("silence flag");
    if (dec.decBitLogp(1) !== 0) throw new Error("postfilter");
    if (dec.decBitLogp(3) !== 0) throw new Error("transient");
    if (dec.decBitLogp(3) !== 1) throw new Error("intra expected");

    const energy = new Float64Array(NBANDS);
    for (let b = 0; b < NBANDS; b++)
      energy[b] = EMEANS[b] + (dec.decUint(64) - 16);

    const q = dec.decUint(11);
    const fineBits = new Array(NBANDS);
    for (let b = 0; b < NBANDS; b++) {
      const nb = BAND_BINS[b + 1] - BAND_BINS[b];
      const frac = Math.floor(ALLOC[q][b] * nb / 4);
      fineBits[b] = Math.max(0, Math.min(7, Math.floor(frac / 160)));
    }
    for (let b = 0; b < NBANDS; b++) {
      if (fineBits[b] <= 0) continue;
      const fq = dec.decBits(fineBits[b]);
      energy[b] += (fq + 0.5) / Math.pow(2, fineBits[b]) - 0.5;
    }

    const bins = new Float64Array(FRAME);
    for (let b = 0; b < NBANDS; b++) {
      const n0 = BAND_BINS[b], nb = BAND_BINS[b + 1] - n0;
      if (!dec.decBitLogp(1)) continue;
      const k = dec.decUint(256);
      const y = decodeBandPvq(dec, nb, k);
      let ss = 0;
      for (let i = 0; i < nb; i++) ss += y[i] * y[i];
      const norm = Math.sqrt(ss) || 1;
      const g = Math.pow(2, energy[b]);
      for (let i = 0; i < nb; i++) bins[n0 + i] = y[i] / norm * g;
    }

    /* windowed IMDCT + overlap-add */
    const n = FRAME;
    const out = new Float32Array(n);
    const tail = new Float64Array(n);
    for (let j = 0; j < 2 * n; j++) {
      let acc = 0;
      const row = j * n;
      /* bins above BAND_BINS[21]=800 are always zero */
      for (let kk = 0; kk < 800; kk++) acc += IMDCT_BASIS[row + kk] * bins[kk];
      const v = acc * WINDOW[j];
      if (j < n) out[j] = this.prevTail[j] + v;
      else tail[j - n] = v;
    }
    this.prevTail = tail;
    return out;
  }
}

const api = { OpusDecoder, RangeDecoder, pvqV, pvqUnindex, decodeBandPvq,
              FRAME, OVERLAP };
if (typeof module !== "undefined" && module.exports) module.exports = api;
else root.SkOpus = api;

})(typeof self !== "undefined" ? self : this);
