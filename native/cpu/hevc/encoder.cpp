// HEVC stripe encoder — CPU reference (see encoder.h for the design).
#include "encoder.h"

#include <algorithm>
#include <climits>
#include <cstring>

#include "cabac.h"
#include "entropy.h"
#include "headers.h"
#include "intra.h"
#include "tables.h"
#include "transform.h"

namespace hipflux {
namespace hevc {

namespace {

inline int ceil_div(int a, int b) { return (a + b - 1) / b; }

inline uint8_t clip8(int v) {
  return static_cast<uint8_t>(v < 0 ? 0 : v > 255 ? 255 : v);
}

// One transform block's coded data.
template <int N>
struct TbData {
  int16_t level[N * N];
  bool cbf = false;
};

}  // namespace

// ---- encoder impl --------------------------------------------------------

struct StripeEncoder::Impl {
  int w, h;            // visible
  int cw, ch;          // coded (16-aligned)
  int ctb_w, ctb_h;    // CTUs per row / rows
  int slices_per_row;
  int addr_bits;
  std::vector<uint8_t> ry, rcb, rcr;        // recon planes (coded dims)
  std::vector<uint8_t> sy, scb, scr;        // padded source planes
  std::vector<uint8_t> luma_mode;           // per-CTU chosen mode
  std::vector<uint8_t> ps_cache;            // VPS/SPS/PPS bytes
  int ypitch, cpitch;

  Impl(int width, int height, int spr)
      : w(width), h(height), cw((width + 15) & ~15), ch((height + 15) & ~15),
        ctb_w(cw / 16), ctb_h(ch / 16),
        slices_per_row(std::max(1, std::min(spr, cw / 16))) {
    int n_ctb = ctb_w * ctb_h;
    addr_bits = 1;
    while ((1 << addr_bits) < n_ctb) ++addr_bits;
    ypitch = cw;
    cpitch = cw / 2;
    ry.resize(static_cast<size_t>(ypitch) * ch);
    rcb.resize(static_cast<size_t>(cpitch) * ch / 2);
    rcr.resize(static_cast<size_t>(cpitch) * ch / 2);
    sy = ry;
    scb = rcb;
    scr = rcr;
    luma_mode.resize(n_ctb);
    write_vps_nal(ps_cache, level_idc_for(cw, ch));
    write_sps_nal(ps_cache, cw, ch, w, h);
    write_pps_nal(ps_cache);
  }

  // pad source into coded-size planes with edge replication
  void load_source(const uint8_t* y, int ypitch_in, const uint8_t* cb,
                   const uint8_t* cr, int cpitch_in) {
    for (int r = 0; r < ch; ++r) {
      int sr = std::min(r, h - 1);
      std::memcpy(&sy[static_cast<size_t>(r) * ypitch],
                  y + static_cast<size_t>(sr) * ypitch_in, w);
      for (int x = w; x < cw; ++x)
        sy[static_cast<size_t>(r) * ypitch + x] = sy[r * ypitch + w - 1];
    }
    int cwh = cw / 2, chh = ch / 2, wh = (w + 1) / 2, hh = (h + 1) / 2;
    for (int r = 0; r < chh; ++r) {
      int sr = std::min(r, hh - 1);
      std::memcpy(&scb[static_cast<size_t>(r) * cpitch],
                  cb + static_cast<size_t>(sr) * cpitch_in, wh);
      std::memcpy(&scr[static_cast<size_t>(r) * cpitch],
                  cr + static_cast<size_t>(sr) * cpitch_in, wh);
      for (int x = wh; x < cwh; ++x) {
        scb[static_cast<size_t>(r) * cpitch + x] = scb[r * cpitch + wh - 1];
        scr[static_cast<size_t>(r) * cpitch + x] = scr[r * cpitch + wh - 1];
      }
    }
  }

  // Encode one 16x16 CTU: choose mode, transform, reconstruct, and emit
  // CABAC bins. left_avail reflects the slice-segment boundary.
  void encode_ctu(CabacEncoder& cab, ContextBank& bank, int cx, int cy,
                  bool left_avail, int qp, uint8_t left_mode) {
    const int x0 = cx * 16, y0 = cy * 16;
    // --- luma mode decision + transform
    Avail av;  // one slice per CTU row: only left neighbors exist
    av.left = left_avail;
    av.corner = false;
    av.top = av.top_right = av.below_left = false;

    uint8_t ref[65], reff[65];
    build_refs<16>(ry.data(), ypitch, x0, y0, av, ref);
    filter_refs<16>(ref, reff);

    static const int kModes[4] = {kDc, kHor, kVer, kPlanar};
    uint8_t pred[4][256];
    int best = 0;
    long best_sad = LONG_MAX;
    for (int m = 0; m < 4; ++m) {
      predict<16>(kModes[m], kModes[m] == kPlanar ? reff : ref, 0, pred[m]);
      long sad = 0;
      for (int yy = 0; yy < 16; ++yy)
        for (int xx = 0; xx < 16; ++xx) {
          int d = sy[static_cast<size_t>(y0 + yy) * ypitch + x0 + xx] -
                  pred[m][yy * 16 + xx];
          sad += d < 0 ? -d : d;
        }
      if (sad < best_sad) {
        best_sad = sad;
        best = m;
      }
    }
    const int mode = kModes[best];
    luma_mode[cy * ctb_w + cx] = static_cast<uint8_t>(mode);

    int16_t res[256], coef[256];
    TbData<16> yb;
    for (int yy = 0; yy < 16; ++yy)
      for (int xx = 0; xx < 16; ++xx)
        res[yy * 16 + xx] =
            static_cast<int16_t>(
                sy[static_cast<size_t>(y0 + yy) * ypitch + x0 + xx] -
                pred[best][yy * 16 + xx]);
    fwd_transform<16>(res, 16, coef);
    yb.cbf = quantize<16>(coef, qp, yb.level) != 0;
    if (yb.cbf) {
      int16_t deq[256], rres[256];
      dequantize<16>(yb.level, qp, deq);
      inv_transform<16>(deq, rres, 16);
      for (int yy = 0; yy < 16; ++yy)
        for (int xx = 0; xx < 16; ++xx)
          ry[static_cast<size_t>(y0 + yy) * ypitch + x0 + xx] =
              clip8(pred[best][yy * 16 + xx] + rres[yy * 16 + xx]);
    } else {
      for (int yy = 0; yy < 16; ++yy)
        std::memcpy(&ry[static_cast<size_t>(y0 + yy) * ypitch + x0],
                    &pred[best][yy * 16], 16);
    }

    // --- chroma (DM mode), 8x8 TBs
    const int qpc = chroma_qp(qp);
    TbData<8> cbb, crb;
    uint8_t cpred[2][64];
    const int cx0 = x0 / 2, cy0 = y0 / 2;
    for (int pl = 0; pl < 2; ++pl) {
      uint8_t* rp = pl ? rcr.data() : rcb.data();
      const uint8_t* sp = pl ? scr.data() : scb.data();
      TbData<8>& tb = pl ? crb : cbb;
      uint8_t cref[33];
      build_refs<8>(rp, cpitch, cx0, cy0, av, cref);
      predict<8>(mode, cref, 1, cpred[pl]);
      int16_t cres[64], ccoef[64];
      for (int yy = 0; yy < 8; ++yy)
        for (int xx = 0; xx < 8; ++xx)
          cres[yy * 8 + xx] = static_cast<int16_t>(
              sp[static_cast<size_t>(cy0 + yy) * cpitch + cx0 + xx] -
              cpred[pl][yy * 8 + xx]);
      fwd_transform<8>(cres, 8, ccoef);
      tb.cbf = quantize<8>(ccoef, qpc, tb.level) != 0;
      if (tb.cbf) {
        int16_t deq[64], rres[64];
        dequantize<8>(tb.level, qpc, deq);
        inv_transform<8>(deq, rres, 8);
        for (int yy = 0; yy < 8; ++yy)
          for (int xx = 0; xx < 8; ++xx)
            rp[static_cast<size_t>(cy0 + yy) * cpitch + cx0 + xx] =
                clip8(cpred[pl][yy * 8 + xx] + rres[yy * 8 + xx]);
      } else {
        for (int yy = 0; yy < 8; ++yy)
          std::memcpy(&rp[static_cast<size_t>(cy0 + yy) * cpitch + cx0],
                      &cpred[pl][yy * 8], 8);
      }
    }

    // --- CABAC bins for the CTU (shared emission path, entropy.h)
    code_ctu_syntax(cab, bank, mode, left_avail ? left_mode : -1, yb.cbf,
                    cbb.cbf, crb.cbf, yb.level, cbb.level, crb.level);
  }
};

StripeEncoder::StripeEncoder(int width, int height, int slices_per_row)
    : impl_(new Impl(width, height, slices_per_row)),
      width_(width),
      height_(height) {}

StripeEncoder::~StripeEncoder() = default;

const uint8_t* StripeEncoder::recon_y() const { return impl_->ry.data(); }
const uint8_t* StripeEncoder::recon_cb() const { return impl_->rcb.data(); }
const uint8_t* StripeEncoder::recon_cr() const { return impl_->rcr.data(); }
int StripeEncoder::recon_ypitch() const { return impl_->ypitch; }
int StripeEncoder::recon_cpitch() const { return impl_->cpitch; }

void StripeEncoder::encode_frame(const uint8_t* y, int ypitch,
                                 const uint8_t* cb, const uint8_t* cr,
                                 int cpitch, int qp,
                                 std::vector<uint8_t>& out,
                                 EncodeStats* stats) {
  Impl& im = *impl_;
  qp = std::clamp(qp, 0, 51);
  im.load_source(y, ypitch, cb, cr, cpitch);
  out.insert(out.end(), im.ps_cache.begin(), im.ps_cache.end());

  // one slice segment per CTU row (split into slices_per_row segments)
  for (int cy = 0; cy < im.ctb_h; ++cy) {
    int per = ceil_div(im.ctb_w, im.slices_per_row);
    for (int s0 = 0; s0 < im.ctb_w; s0 += per) {
      int s1 = std::min(s0 + per, im.ctb_w);
      BitWriter bw;
      bool first = (cy == 0 && s0 == 0);
      write_slice_header(bw, first, cy * im.ctb_w + s0, im.addr_bits, qp);
      std::vector<uint8_t> cabac_bytes;
      CabacEncoder cab(cabac_bytes);
      ContextBank bank;
      bank.init(qp);
      uint8_t left_mode = kDc;
      for (int cx = s0; cx < s1; ++cx) {
        im.encode_ctu(cab, bank, cx, cy, cx > s0, qp, left_mode);
        left_mode = im.luma_mode[cy * im.ctb_w + cx];
        cab.encode_terminate(cx == s1 - 1 ? 1 : 0);
      }
      auto tail = cab.finish();
      for (uint8_t b : cabac_bytes) bw.u(b, 8);
      bw.u(tail.bits, tail.nbits);
      // rbsp_slice_segment_trailing_bits: stop bit + alignment
      bw.rbsp_trailing();
      emit_nal(bw, out, kNalIdr);
    }
  }
  if (stats) {
    stats->frame_qp = qp;
    stats->ctu_count = im.ctb_w * im.ctb_h;
    stats->bytes = out.size();
  }
}

}  // namespace hevc
}  // namespace hipflux
