// Baseline JFIF encoder (ITU-T T.81 sequential DCT, standard Annex-K tables).
// CPU reference implementation; the HIP path reuses the header/entropy code
// via jpeg_entropy_from_blocks so both paths emit identical bitstreams for
// identical quantized coefficients.
#include "jpeg_enc.h"

#include <algorithm>
#include <cmath>
#include <cstring>

namespace hipflux {
namespace {

// ---- spec constants (T.81 Annex K) ----------------------------------------
const uint8_t kZigzag[64] = {
    0,  1,  8,  16, 9,  2,  3,  10, 17, 24, 32, 25, 18, 11, 4,  5,
    12, 19, 26, 33, 40, 48, 41, 34, 27, 20, 13, 6,  7,  14, 21, 28,
    35, 42, 49, 56, 57, 50, 43, 36, 29, 22, 15, 23, 30, 37, 44, 51,
    58, 59, 52, 45, 38, 31, 39, 46, 53, 60, 61, 54, 47, 55, 62, 63};

const uint8_t kBaseQY[64] = {  // natural order
    16, 11, 10, 16, 24,  40,  51,  61,  12, 12, 14, 19, 26,  58,  60,  55,
    14, 13, 16, 24, 40,  57,  69,  56,  14, 17, 22, 29, 51,  87,  80,  62,
    18, 22, 37, 56, 68,  109, 103, 77,  24, 35, 55, 64, 81,  104, 113, 92,
    49, 64, 78, 87, 103, 121, 120, 101, 72, 92, 95, 98, 112, 100, 103, 99};

const uint8_t kBaseQC[64] = {
    17, 18, 24, 47, 99, 99, 99, 99, 18, 21, 26, 66, 99, 99, 99, 99,
    24, 26, 56, 99, 99, 99, 99, 99, 47, 66, 99, 99, 99, 99, 99, 99,
    99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99,
    99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99};

// Standard Huffman table specs: (bits[1..16], values)
const uint8_t kDcLumaBits[17] = {0, 0, 1, 5, 1, 1, 1, 1, 1, 1, 0, 0, 0, 0, 0, 0, 0};
const uint8_t kDcLumaVals[12] = {0, 1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11};
const uint8_t kDcChromaBits[17] = {0, 0, 3, 1, 1, 1, 1, 1, 1, 1, 1, 1, 0, 0, 0, 0, 0};
const uint8_t kDcChromaVals[12] = {0, 1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11};
const uint8_t kAcLumaBits[17] = {0, 0, 2, 1, 3, 3, 2, 4, 3, 5, 5, 4, 4, 0, 0, 1, 0x7d};
const uint8_t kAcLumaVals[162] = {
    0x01, 0x02, 0x03, 0x00, 0x04, 0x11, 0x05, 0x12, 0x21, 0x31, 0x41, 0x06,
    0x13, 0x51, 0x61, 0x07, 0x22, 0x71, 0x14, 0x32, 0x81, 0x91, 0xa1, 0x08,
    0x23, 0x42, 0xb1, 0xc1, 0x15, 0x52, 0xd1, 0xf0, 0x24, 0x33, 0x62, 0x72,
    0x82, 0x09, 0x0a, 0x16, 0x17, 0x18, 0x19, 0x1a, 0x25, 0x26, 0x27, 0x28,
    0x29, 0x2a, 0x34, 0x35, 0x36, 0x37, 0x38, 0x39, 0x3a, 0x43, 0x44, 0x45,
    0x46, 0x47, 0x48, 0x49, 0x4a, 0x53, 0x54, 0x55, 0x56, 0x57, 0x58, 0x59,
    0x5a, 0x63, 0x64, 0x65, 0x66, 0x67, 0x68, 0x69, 0x6a, 0x73, 0x74, 0x75,
    0x76, 0x77, 0x78, 0x79, 0x7a, 0x83, 0x84, 0x85, 0x86, 0x87, 0x88, 0x89,
    0x8a, 0x92, 0x93, 0x94, 0x95, 0x96, 0x97, 0x98, 0x99, 0x9a, 0xa2, 0xa3,
    0xa4, 0xa5, 0xa6, 0xa7, 0xa8, 0xa9, 0xaa, 0xb2, 0xb3, 0xb4, 0xb5, 0xb6,
    0xb7, 0xb8, 0xb9, 0xba, 0xc2, 0xc3, 0xc4, 0xc5, 0xc6, 0xc7, 0xc8, 0xc9,
    0xca, 0xd2, 0xd3, 0xd4, 0xd5, 0xd6, 0xd7, 0xd8, 0xd9, 0xda, 0xe1, 0xe2,
    0xe3, 0xe4, 0xe5, 0xe6, 0xe7, 0xe8, 0xe9, 0xea, 0xf1, 0xf2, 0xf3, 0xf4,
    0xf5, 0xf6, 0xf7, 0xf8, 0xf9, 0xfa};
const uint8_t kAcChromaBits[17] = {0, 0, 2, 1, 2, 4, 4, 3, 4, 7, 5, 4, 4, 0, 1, 2, 0x77};
const uint8_t kAcChromaVals[162] = {
    0x00, 0x01, 0x02, 0x03, 0x11, 0x04, 0x05, 0x21, 0x31, 0x06, 0x12, 0x41,
    0x51, 0x07, 0x61, 0x71, 0x13, 0x22, 0x32, 0x81, 0x08, 0x14, 0x42, 0x91,
    0xa1, 0xb1, 0xc1, 0x09, 0x23, 0x33, 0x52, 0xf0, 0x15, 0x62, 0x72, 0xd1,
    0x0a, 0x16, 0x24, 0x34, 0xe1, 0x25, 0xf1, 0x17, 0x18, 0x19, 0x1a, 0x26,
    0x27, 0x28, 0x29, 0x2a, 0x35, 0x36, 0x37, 0x38, 0x39, 0x3a, 0x43, 0x44,
    0x45, 0x46, 0x47, 0x48, 0x49, 0x4a, 0x53, 0x54, 0x55, 0x56, 0x57, 0x58,
    0x59, 0x5a, 0x63, 0x64, 0x65, 0x66, 0x67, 0x68, 0x69, 0x6a, 0x73, 0x74,
    0x75, 0x76, 0x77, 0x78, 0x79, 0x7a, 0x82, 0x83, 0x84, 0x85, 0x86, 0x87,
    0x88, 0x89, 0x8a, 0x92, 0x93, 0x94, 0x95, 0x96, 0x97, 0x98, 0x99, 0x9a,
    0xa2, 0xa3, 0xa4, 0xa5, 0xa6, 0xa7, 0xa8, 0xa9, 0xaa, 0xb2, 0xb3, 0xb4,
    0xb5, 0xb6, 0xb7, 0xb8, 0xb9, 0xba, 0xc2, 0xc3, 0xc4, 0xc5, 0xc6, 0xc7,
    0xc8, 0xc9, 0xca, 0xd2, 0xd3, 0xd4, 0xd5, 0xd6, 0xd7, 0xd8, 0xd9, 0xda,
    0xe2, 0xe3, 0xe4, 0xe5, 0xe6, 0xe7, 0xe8, 0xe9, 0xea, 0xf2, 0xf3, 0xf4,
    0xf5, 0xf6, 0xf7, 0xf8, 0xf9, 0xfa};

struct HuffTable {
  uint16_t code[256];
  uint8_t size[256];
};

// canonical code assignment (T.81 C.2)
HuffTable build_huff(const uint8_t bits[17], const uint8_t* vals, int nvals) {
  HuffTable t{};
  int code = 0, k = 0;
  for (int len = 1; len <= 16; ++len) {
    for (int i = 0; i < bits[len]; ++i) {
      t.code[vals[k]] = static_cast<uint16_t>(code);
      t.size[vals[k]] = static_cast<uint8_t>(len);
      ++code;
      ++k;
    }
    code <<= 1;
  }
  (void)nvals;
  return t;
}

const HuffTable& dc_luma() { static HuffTable t = build_huff(kDcLumaBits, kDcLumaVals, 12); return t; }
const HuffTable& dc_chroma() { static HuffTable t = build_huff(kDcChromaBits, kDcChromaVals, 12); return t; }
const HuffTable& ac_luma() { static HuffTable t = build_huff(kAcLumaBits, kAcLumaVals, 162); return t; }
const HuffTable& ac_chroma() { static HuffTable t = build_huff(kAcChromaBits, kAcChromaVals, 162); return t; }

class BitWriter {
 public:
  explicit BitWriter(std::vector<uint8_t>& out) : out_(out) {}
  void put(uint32_t bits, int nbits) {
    acc_ = (acc_ << nbits) | (bits & ((1u << nbits) - 1));
    nacc_ += nbits;
    while (nacc_ >= 8) {
      uint8_t b = static_cast<uint8_t>(acc_ >> (nacc_ - 8));
      out_.push_back(b);
      if (b == 0xFF) out_.push_back(0x00);  // byte stuffing
      nacc_ -= 8;
    }
  }
  void flush() {
    if (nacc_ > 0) put(0x7F, 8 - nacc_);  // pad with 1s
  }

 private:
  std::vector<uint8_t>& out_;
  uint64_t acc_ = 0;
  int nacc_ = 0;
};

inline int bit_length(int v) {
  int n = 0;
  while (v) { ++n; v >>= 1; }
  return n;
}

// Encode one quantized 8x8 block (natural order). Returns new DC predictor.
int encode_block(BitWriter& bw, const int16_t* blk, int dc_pred,
                 const HuffTable& dc, const HuffTable& ac) {
  int diff = blk[0] - dc_pred;
  int mag = diff < 0 ? -diff : diff;
  int nbits = bit_length(mag);
  bw.put(dc.code[nbits], dc.size[nbits]);
  if (nbits) bw.put(diff < 0 ? diff + ((1 << nbits) - 1) : diff, nbits);

  int run = 0;
  for (int i = 1; i < 64; ++i) {
    int v = blk[kZigzag[i]];
    if (v == 0) {
      ++run;
      continue;
    }
    while (run >= 16) {
      bw.put(ac.code[0xF0], ac.size[0xF0]);  // ZRL
      run -= 16;
    }
    int m = v < 0 ? -v : v;
    int nb = bit_length(m);
    int sym = (run << 4) | nb;
    bw.put(ac.code[sym], ac.size[sym]);
    bw.put(v < 0 ? v + ((1 << nb) - 1) : v, nb);
    run = 0;
  }
  if (run) bw.put(ac.code[0x00], ac.size[0x00]);  // EOB
  return blk[0];
}

// ---- headers ---------------------------------------------------------------
void put16(std::vector<uint8_t>& o, int v) {
  o.push_back(static_cast<uint8_t>(v >> 8));
  o.push_back(static_cast<uint8_t>(v));
}

void write_headers(std::vector<uint8_t>& o, int width, int height,
                   const uint8_t qy[64], const uint8_t qc[64], bool fullcolor,
                   int restart_interval = 0) {
  // SOI
  o.push_back(0xFF); o.push_back(0xD8);
  // APP0 JFIF
  o.push_back(0xFF); o.push_back(0xE0);
  put16(o, 16);
  const char jfif[] = "JFIF";
  o.insert(o.end(), jfif, jfif + 5);
  o.push_back(1); o.push_back(1);   // version
  o.push_back(0);                    // aspect units
  put16(o, 1); put16(o, 1);          // aspect
  o.push_back(0); o.push_back(0);    // no thumbnail
  // DQT (two tables, zigzag order)
  o.push_back(0xFF); o.push_back(0xDB);
  put16(o, 2 + 2 * 65);
  o.push_back(0x00);
  for (int i = 0; i < 64; ++i) o.push_back(qy[kZigzag[i]]);
  o.push_back(0x01);
  for (int i = 0; i < 64; ++i) o.push_back(qc[kZigzag[i]]);
  // SOF0
  o.push_back(0xFF); o.push_back(0xC0);
  put16(o, 8 + 3 * 3);
  o.push_back(8);
  put16(o, height); put16(o, width);
  o.push_back(3);
  o.push_back(1); o.push_back(fullcolor ? 0x11 : 0x22); o.push_back(0);  // Y
  o.push_back(2); o.push_back(0x11); o.push_back(1);                     // Cb
  o.push_back(3); o.push_back(0x11); o.push_back(1);                     // Cr
  // DHT (4 tables)
  auto dht = [&o](uint8_t cls_id, const uint8_t bits[17], const uint8_t* vals) {
    int n = 0;
    for (int i = 1; i <= 16; ++i) n += bits[i];
    o.push_back(0xFF); o.push_back(0xC4);
    put16(o, 2 + 1 + 16 + n);
    o.push_back(cls_id);
    for (int i = 1; i <= 16; ++i) o.push_back(bits[i]);
    o.insert(o.end(), vals, vals + n);
  };
  dht(0x00, kDcLumaBits, kDcLumaVals);
  dht(0x10, kAcLumaBits, kAcLumaVals);
  dht(0x01, kDcChromaBits, kDcChromaVals);
  dht(0x11, kAcChromaBits, kAcChromaVals);
  if (restart_interval > 0) {             // DRI
    o.push_back(0xFF); o.push_back(0xDD);
    put16(o, 4);
    put16(o, restart_interval);
  }
  // SOS
  o.push_back(0xFF); o.push_back(0xDA);
  put16(o, 6 + 2 * 3);
  o.push_back(3);
  o.push_back(1); o.push_back(0x00);
  o.push_back(2); o.push_back(0x11);
  o.push_back(3); o.push_back(0x11);
  o.push_back(0); o.push_back(63); o.push_back(0);
}

// ---- forward DCT (separable, float) ----------------------------------------
struct DctLut {
  float m[8][8];  // m[u][x] = c(u)/2 * cos((2x+1) u pi / 16)
  DctLut() {
    for (int u = 0; u < 8; ++u) {
      double cu = (u == 0) ? std::sqrt(0.5) : 1.0;
      for (int x = 0; x < 8; ++x)
        m[u][x] = static_cast<float>(0.5 * cu *
                                     std::cos((2 * x + 1) * u * M_PI / 16.0));
    }
  }
};

void fdct8x8_quant(const float* in /*64, centered*/, const uint8_t* q,
                   int16_t* out /*64 natural order*/) {
  static const DctLut lut;
  float tmp[64];
  // rows: tmp[y][u] = sum_x in[y][x] * m[u][x]
  for (int y = 0; y < 8; ++y)
    for (int u = 0; u < 8; ++u) {
      float s = 0.f;
      for (int x = 0; x < 8; ++x) s += in[y * 8 + x] * lut.m[u][x];
      tmp[y * 8 + u] = s;
    }
  // cols + quant
  for (int v = 0; v < 8; ++v)
    for (int u = 0; u < 8; ++u) {
      float s = 0.f;
      for (int y = 0; y < 8; ++y) s += tmp[y * 8 + u] * lut.m[v][y];
      float qv = s / q[v * 8 + u];
      out[v * 8 + u] = static_cast<int16_t>(std::lrintf(qv));
    }
}

// BT.601 full-range RGB -> YCbCr (JFIF)
inline void rgb_to_ycbcr(float r, float g, float b, float& y, float& cb, float& cr) {
  y = 0.299f * r + 0.587f * g + 0.114f * b;
  cb = -0.168736f * r - 0.331264f * g + 0.5f * b + 128.f;
  cr = 0.5f * r - 0.418688f * g - 0.081312f * b + 128.f;
}

// Round to the 8-bit sample grid — matches the HIP path, which stores
// Y/Cb/Cr planes as uint8 before the DCT (so CPU and GPU bitstreams agree).
inline float to_u8_sample(float v) {
  v = std::lrintf(std::min(std::max(v, 0.f), 255.f));
  return v;
}

}  // namespace

void jpeg_quality_tables(int quality, uint8_t qy[64], uint8_t qc[64]) {
  quality = std::clamp(quality, 1, 100);
  int scale = quality < 50 ? 5000 / quality : 200 - 2 * quality;
  for (int i = 0; i < 64; ++i) {
    int y = (kBaseQY[i] * scale + 50) / 100;
    int c = (kBaseQC[i] * scale + 50) / 100;
    qy[i] = static_cast<uint8_t>(std::clamp(y, 1, 255));
    qc[i] = static_cast<uint8_t>(std::clamp(c, 1, 255));
  }
}

void jpeg_encode_bgrx(const uint8_t* bgrx, int stride, int width, int height,
                      int quality, bool fullcolor, std::vector<uint8_t>& out,
                      bool restart_rows) {
  uint8_t qy[64], qc[64];
  jpeg_quality_tables(quality, qy, qc);
  int mcu_w = fullcolor ? 8 : 16;
  int mcux_hdr = (width + mcu_w - 1) / mcu_w;
  write_headers(out, width, height, qy, qc, fullcolor,
                restart_rows ? mcux_hdr : 0);
  BitWriter bw(out);
  int rst = 0;

  int dcY = 0, dcCb = 0, dcCr = 0;
  int16_t blk[64];

  auto sample = [&](int x, int y_, float& r, float& g, float& b) {
    x = std::min(x, width - 1);
    y_ = std::min(y_, height - 1);
    const uint8_t* p = bgrx + y_ * stride + x * 4;
    b = p[0]; g = p[1]; r = p[2];
  };

  if (!fullcolor) {
    int mcux = (width + 15) / 16, mcuy = (height + 15) / 16;
    float Y[256], Cb[256], Cr[256];  // 16x16 samples
    float plane[64];
    for (int my = 0; my < mcuy; ++my) {
      if (restart_rows && my > 0) {
        bw.flush();
        out.push_back(0xFF);
        out.push_back(static_cast<uint8_t>(0xD0 + (rst++ & 7)));
        dcY = dcCb = dcCr = 0;
      }
      for (int mx = 0; mx < mcux; ++mx) {
        for (int yy = 0; yy < 16; ++yy)
          for (int xx = 0; xx < 16; ++xx) {
            float r, g, b;
            sample(mx * 16 + xx, my * 16 + yy, r, g, b);
            rgb_to_ycbcr(r, g, b, Y[yy * 16 + xx], Cb[yy * 16 + xx],
                         Cr[yy * 16 + xx]);
            Y[yy * 16 + xx] = to_u8_sample(Y[yy * 16 + xx]);
          }
        // 4 luma blocks
        for (int by = 0; by < 2; ++by)
          for (int bx = 0; bx < 2; ++bx) {
            for (int yy = 0; yy < 8; ++yy)
              for (int xx = 0; xx < 8; ++xx)
                plane[yy * 8 + xx] =
                    Y[(by * 8 + yy) * 16 + bx * 8 + xx] - 128.f;
            fdct8x8_quant(plane, qy, blk);
            dcY = encode_block(bw, blk, dcY, dc_luma(), ac_luma());
          }
        // subsampled chroma (2x2 average)
        for (int c = 0; c < 2; ++c) {
          const float* src = c == 0 ? Cb : Cr;
          for (int yy = 0; yy < 8; ++yy)
            for (int xx = 0; xx < 8; ++xx) {
              float s = src[(2 * yy) * 16 + 2 * xx] +
                        src[(2 * yy) * 16 + 2 * xx + 1] +
                        src[(2 * yy + 1) * 16 + 2 * xx] +
                        src[(2 * yy + 1) * 16 + 2 * xx + 1];
              plane[yy * 8 + xx] = to_u8_sample(s * 0.25f) - 128.f;
            }
          fdct8x8_quant(plane, qc, blk);
          if (c == 0)
            dcCb = encode_block(bw, blk, dcCb, dc_chroma(), ac_chroma());
          else
            dcCr = encode_block(bw, blk, dcCr, dc_chroma(), ac_chroma());
        }
      }
    }
  } else {
    int mcux = (width + 7) / 8, mcuy = (height + 7) / 8;
    float plane[3][64];
    for (int my = 0; my < mcuy; ++my) {
      if (restart_rows && my > 0) {
        bw.flush();
        out.push_back(0xFF);
        out.push_back(static_cast<uint8_t>(0xD0 + (rst++ & 7)));
        dcY = dcCb = dcCr = 0;
      }
      for (int mx = 0; mx < mcux; ++mx) {
        for (int yy = 0; yy < 8; ++yy)
          for (int xx = 0; xx < 8; ++xx) {
            float r, g, b, y_, cb, cr;
            sample(mx * 8 + xx, my * 8 + yy, r, g, b);
            rgb_to_ycbcr(r, g, b, y_, cb, cr);
            plane[0][yy * 8 + xx] = to_u8_sample(y_) - 128.f;
            plane[1][yy * 8 + xx] = to_u8_sample(cb) - 128.f;
            plane[2][yy * 8 + xx] = to_u8_sample(cr) - 128.f;
          }
        fdct8x8_quant(plane[0], qy, blk);
        dcY = encode_block(bw, blk, dcY, dc_luma(), ac_luma());
        fdct8x8_quant(plane[1], qc, blk);
        dcCb = encode_block(bw, blk, dcCb, dc_chroma(), ac_chroma());
        fdct8x8_quant(plane[2], qc, blk);
        dcCr = encode_block(bw, blk, dcCr, dc_chroma(), ac_chroma());
      }
    }
  }
  bw.flush();
  out.push_back(0xFF);
  out.push_back(0xD9);  // EOI
}

void jpeg_entropy_from_blocks(const int16_t* blocks, int mcu_count_x,
                              int mcu_count_y, int width, int height,
                              int quality, bool fullcolor,
                              std::vector<uint8_t>& out,
                              bool restart_rows) {
  uint8_t qy[64], qc[64];
  jpeg_quality_tables(quality, qy, qc);
  write_headers(out, width, height, qy, qc, fullcolor,
                restart_rows ? mcu_count_x : 0);
  const int per_mcu = fullcolor ? 3 : 6;
  int dcY = 0, dcCb = 0, dcCr = 0;
  auto encode_mcu = [&](BitWriter& bw, int m) {
    const int16_t* mcu = blocks + static_cast<size_t>(m) * per_mcu * 64;
    if (!fullcolor) {
      for (int b = 0; b < 4; ++b)
        dcY = encode_block(bw, mcu + b * 64, dcY, dc_luma(), ac_luma());
      dcCb = encode_block(bw, mcu + 4 * 64, dcCb, dc_chroma(), ac_chroma());
      dcCr = encode_block(bw, mcu + 5 * 64, dcCr, dc_chroma(), ac_chroma());
    } else {
      dcY = encode_block(bw, mcu + 0 * 64, dcY, dc_luma(), ac_luma());
      dcCb = encode_block(bw, mcu + 1 * 64, dcCb, dc_chroma(), ac_chroma());
      dcCr = encode_block(bw, mcu + 2 * 64, dcCr, dc_chroma(), ac_chroma());
    }
  };
  if (!restart_rows) {
    BitWriter bw(out);
    for (int m = 0; m < mcu_count_x * mcu_count_y; ++m) encode_mcu(bw, m);
    bw.flush();
  } else {
    int rst = 0;
    for (int my = 0; my < mcu_count_y; ++my) {
      if (my > 0) {
        // byte-aligned RSTn between restart intervals; DC preds reset
        out.push_back(0xFF);
        out.push_back(static_cast<uint8_t>(0xD0 + (rst & 7)));
        ++rst;
        dcY = dcCb = dcCr = 0;
      }
      BitWriter bw(out);
      for (int mx = 0; mx < mcu_count_x; ++mx)
        encode_mcu(bw, my * mcu_count_x + mx);
      bw.flush();   // 1-fill to byte boundary at interval end
    }
  }
  out.push_back(0xFF);
  out.push_back(0xD9);
}

// public header writer for the GPU entropy path
void jpeg_write_headers(std::vector<uint8_t>& out, int width, int height,
                        int quality, bool fullcolor,
                        int restart_interval) {
  uint8_t qy[64], qc[64];
  jpeg_quality_tables(quality, qy, qc);
  write_headers(out, width, height, qy, qc, fullcolor, restart_interval);
}

// append one restart interval's bit run: big-endian u32 words from the
// GPU scatter -> bytes with 1-fill padding and 0xFF00 stuffing
void jpeg_append_row_bits(const uint32_t* words, int bits,
                          std::vector<uint8_t>& out) {
  int nbytes = (bits + 7) / 8;
  out.reserve(out.size() + nbytes + nbytes / 64 + 4);
  for (int i = 0; i < nbytes; ++i) {
    uint8_t b = static_cast<uint8_t>(words[i / 4] >> (24 - 8 * (i % 4)));
    if (i == nbytes - 1 && (bits & 7))
      b |= static_cast<uint8_t>(0xFF >> (bits & 7));  // 1-fill padding
    out.push_back(b);
    if (b == 0xFF) out.push_back(0x00);               // byte stuffing
  }
}

// export the Huffman code tables for the GPU entropy kernel: packed
// code | (size << 16), indexed by symbol
void jpeg_export_huff(uint32_t dcl[12], uint32_t acl[256], uint32_t dcc[12],
                      uint32_t acc[256]) {
  const HuffTable& a = dc_luma();
  const HuffTable& b = ac_luma();
  const HuffTable& c = dc_chroma();
  const HuffTable& d = ac_chroma();
  for (int i = 0; i < 12; ++i) {
    dcl[i] = a.code[i] | (uint32_t(a.size[i]) << 16);
    dcc[i] = c.code[i] | (uint32_t(c.size[i]) << 16);
  }
  for (int i = 0; i < 256; ++i) {
    acl[i] = b.code[i] | (uint32_t(b.size[i]) << 16);
    acc[i] = d.code[i] | (uint32_t(d.size[i]) << 16);
  }
}

}  // namespace hipflux
