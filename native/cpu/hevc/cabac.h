// HEVC CABAC encoder engine (§9.3). The byte-emission scheme follows the
// standard carry-buffered arithmetic encoder (HM TEncBinCABAC shape); its
// correctness contract is the spec DECODER (§9.3.4.3) — the pair is
// round-trip fuzzed in tests/test_hevc.py against tests/hevc_ref_decoder.py.
#pragma once

#include <cstdint>
#include <vector>

#include "tables.h"

namespace hipflux {
namespace hevc {

// One context: 6-bit probability state + MPS bit, packed.
struct Ctx {
  uint8_t state = 0;  // pStateIdx
  uint8_t mps = 0;    // valMps

  // §9.3.2.2 initialization from an 8-bit initValue at SliceQpY.
  void init(uint8_t init_value, int qp) {
    int slope = (init_value >> 4) * 5 - 45;
    int offset = ((init_value & 15) << 3) - 16;
    int pre = ((slope * (qp < 0 ? 0 : qp > 51 ? 51 : qp)) >> 4) + offset;
    pre = pre < 1 ? 1 : pre > 126 ? 126 : pre;
    if (pre <= 63) {
      state = static_cast<uint8_t>(63 - pre);
      mps = 0;
    } else {
      state = static_cast<uint8_t>(pre - 64);
      mps = 1;
    }
  }
};

class CabacEncoder {
 public:
  explicit CabacEncoder(std::vector<uint8_t>& out) : out_(out) { reset(); }

  void reset() {
    low_ = 0;
    range_ = 510;
    bits_left_ = 23;
    buffered_byte_ = 0xFF;
    num_buffered_ = 0;
  }

  void encode_bin(Ctx& c, int bin) {
    uint32_t lps = kRangeTabLps[c.state][(range_ >> 6) & 3];
    range_ -= lps;
    if (bin != c.mps) {
      int n = kRenormTable[lps >> 3];
      low_ = (low_ + range_) << n;
      range_ = lps << n;
      if (c.state == 0) c.mps ^= 1;
      c.state = kTransIdxLps[c.state];
      bits_left_ -= n;
      test_write();
    } else {
      c.state = kTransIdxMps[c.state];
      if (range_ >= 256) return;
      low_ <<= 1;
      range_ <<= 1;
      --bits_left_;
      test_write();
    }
  }

  void encode_bypass(int bin) {
    low_ <<= 1;
    if (bin) low_ += range_;
    --bits_left_;
    test_write();
  }

  void encode_bypass_bins(uint32_t bins, int n) {
    for (int i = n - 1; i >= 0; --i) encode_bypass((bins >> i) & 1);
  }

  // end_of_slice_segment_flag etc. (§9.3.4.3.5 encoder dual).
  void encode_terminate(int bin) {
    range_ -= 2;
    if (bin) {
      low_ = (low_ + range_) << 7;
      range_ = 2 << 7;
      bits_left_ -= 7;
    } else if (range_ >= 256) {
      return;
    } else {
      low_ <<= 1;
      range_ <<= 1;
      --bits_left_;
    }
    test_write();
  }

  // Flush after the final terminate bin. Returns the tail bits that have
  // not formed a whole byte: the caller appends them (then the RBSP stop
  // bit + alignment) through its BitWriter so the NAL stays byte-exact.
  struct Tail {
    uint32_t bits;
    int nbits;
  };
  Tail finish() {
    Tail t{0, 0};
    if ((low_ >> (32 - bits_left_)) != 0) {
      out_.push_back(static_cast<uint8_t>(buffered_byte_ + 1));
      while (num_buffered_ > 1) {
        out_.push_back(0x00);
        --num_buffered_;
      }
      low_ -= 1u << (32 - bits_left_);
    } else {
      if (num_buffered_ > 0)
        out_.push_back(static_cast<uint8_t>(buffered_byte_));
      while (num_buffered_ > 1) {
        out_.push_back(0xFF);
        --num_buffered_;
      }
    }
    int nbits = 24 - bits_left_;   // in [?]: whole bytes first
    uint32_t val = low_ >> 8;
    while (nbits >= 8) {
      out_.push_back(static_cast<uint8_t>(val >> (nbits - 8)));
      nbits -= 8;
    }
    t.bits = val & ((1u << nbits) - 1);
    t.nbits = nbits;
    return t;
  }

 private:
  void test_write() {
    if (bits_left_ >= 12) return;
    uint32_t lead = low_ >> (24 - bits_left_);
    bits_left_ += 8;
    low_ &= 0xFFFFFFFFu >> bits_left_;
    if (lead == 0xFF) {
      ++num_buffered_;
    } else if (num_buffered_ > 0) {
      uint32_t carry = lead >> 8;
      out_.push_back(static_cast<uint8_t>(buffered_byte_ + carry));
      uint8_t fill = static_cast<uint8_t>((0xFF + carry) & 0xFF);
      while (num_buffered_ > 1) {
        out_.push_back(fill);
        --num_buffered_;
      }
      buffered_byte_ = lead & 0xFF;
    } else {
      num_buffered_ = 1;
      buffered_byte_ = lead & 0xFF;
    }
  }

  std::vector<uint8_t>& out_;
  uint32_t low_ = 0;
  uint32_t range_ = 510;
  int bits_left_ = 23;
  uint32_t buffered_byte_ = 0xFF;
  int num_buffered_ = 0;
};

// Full I-slice context bank, initialized per §9.3.2.2 (initType 0).
struct ContextBank {
  Ctx ctx[kNumContexts];

  void init(int qp) {
    for (int i = 0; i < 3; ++i)
      ctx[kCtxSplitCu + i].init(kInitSplitCuFlag[i], qp);
    ctx[kCtxPrevIntraLuma].init(kInitPrevIntraLumaPredFlag, qp);
    ctx[kCtxIntraChroma].init(kInitIntraChromaPredMode, qp);
    for (int i = 0; i < 2; ++i)
      ctx[kCtxCbfLuma + i].init(kInitCbfLuma[i], qp);
    for (int i = 0; i < 4; ++i)
      ctx[kCtxCbfChroma + i].init(kInitCbfChroma[i], qp);
    for (int i = 0; i < 18; ++i) {
      ctx[kCtxLastSigX + i].init(kInitLastSigXPrefix[i], qp);
      ctx[kCtxLastSigY + i].init(kInitLastSigYPrefix[i], qp);
    }
    for (int i = 0; i < 4; ++i)
      ctx[kCtxCodedSubBlock + i].init(kInitCodedSubBlockFlag[i], qp);
    for (int i = 0; i < 42; ++i)
      ctx[kCtxSigCoeff + i].init(kInitSigCoeffFlag[i], qp);
    for (int i = 0; i < 24; ++i)
      ctx[kCtxGreater1 + i].init(kInitGreater1Flag[i], qp);
    for (int i = 0; i < 6; ++i)
      ctx[kCtxGreater2 + i].init(kInitGreater2Flag[i], qp);
  }
};

}  // namespace hevc
}  // namespace hipflux
