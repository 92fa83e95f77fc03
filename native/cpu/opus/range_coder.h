// RFC 6716 §4.1 range coder (the CELT/Opus entropy coder): 32-bit
// renormalizing range encoder with carry buffering at the FRONT of the
// buffer and raw bits packed from the BACK. The decoder mirror lives in
// tests/opus_ref_decoder.py; the pair is round-trip fuzzed bin-by-bin
// (tests/test_opus.py) exactly like the H.264/HEVC entropy engines.
#pragma once

#include <cstdint>
#include <vector>

namespace hipflux {
namespace opus {

constexpr uint32_t kEcSymBits = 8;
constexpr uint32_t kEcSymMax = 0xFF;
constexpr uint32_t kEcCodeBits = 32;
constexpr uint32_t kEcCodeTop = 1u << 31;
constexpr uint32_t kEcCodeBot = kEcCodeTop >> kEcSymBits;   // 1 << 23
constexpr uint32_t kEcCodeShift = kEcCodeBits - kEcSymBits - 1;  // 23
constexpr int kEcUintBits = 8;

inline int ec_ilog(uint32_t v) {
  int l = 0;
  while (v) {
    ++l;
    v >>= 1;
  }
  return l;
}

class RangeEncoder {
 public:
  explicit RangeEncoder(size_t capacity) : buf_(capacity, 0) {
    reset();
  }

  void reset() {
    offs_ = 0;
    end_offs_ = 0;
    end_window_ = 0;
    nend_bits_ = 0;
    nbits_total_ = kEcCodeBits + 1;
    val_ = 0;
    rng_ = kEcCodeTop;
    rem_ = -1;
    ext_ = 0;
    error_ = false;
  }

  // encode symbol with cumulative freqs [fl, fh) of total ft
  void encode(uint32_t fl, uint32_t fh, uint32_t ft) {
    uint32_t r = rng_ / ft;
    if (fl > 0) {
      val_ += rng_ - r * (ft - fl);
      rng_ = r * (fh - fl);
    } else {
      rng_ -= r * (ft - fh);
    }
    normalize();
  }

  // binary symbol with probability of 0 being (2^logp - 1) / 2^logp
  void enc_bit_logp(int bit, unsigned logp) {
    uint32_t r = rng_ >> logp;
    uint32_t s = rng_ - r;
    if (bit) {
      val_ += s;
      rng_ = r;
    } else {
      rng_ = s;
    }
    normalize();
  }

  // icdf table: icdf[s] = ft - cumfreq(s+1), ft = 1 << ftb
  void enc_icdf(int s, const uint8_t* icdf, unsigned ftb) {
    uint32_t r = rng_ >> ftb;
    if (s > 0) {
      val_ += rng_ - r * icdf[s - 1];
      rng_ = r * (icdf[s - 1] - icdf[s]);
    } else {
      rng_ -= r * icdf[s];
    }
    normalize();
  }

  // integer in [0, ft): range-coded high part + raw low bits (§4.1.5)
  void enc_uint(uint32_t fl, uint32_t ft) {
    --ft;
    int ftb = ec_ilog(ft);
    if (ftb > kEcUintBits) {
      ftb -= kEcUintBits;
      uint32_t fth = (ft >> ftb) + 1;
      encode(fl >> ftb, (fl >> ftb) + 1, fth);
      enc_bits(fl & ((1u << ftb) - 1), ftb);
    } else {
      encode(fl, fl + 1, ft + 1);
    }
  }

  // raw bits packed from the end of the buffer
  void enc_bits(uint32_t fl, unsigned bits) {
    end_window_ |= static_cast<uint64_t>(fl) << nend_bits_;
    nend_bits_ += bits;
    while (nend_bits_ >= kEcSymBits) {
      if (end_offs_ + offs_ >= buf_.size()) {
        error_ = true;
        return;
      }
      buf_[buf_.size() - 1 - end_offs_] =
          static_cast<uint8_t>(end_window_ & kEcSymMax);
      ++end_offs_;
      end_window_ >>= kEcSymBits;
      nend_bits_ -= kEcSymBits;
    }
    nbits_total_ += bits;
  }

  // bits consumed so far in 1/8-bit units (§4.1.6.1: the log2(rng)
  // estimate refines to 3 fractional bits, then subtracts)
  int tell_frac() const {
    int nbits = nbits_total_ << 3;
    int l = ec_ilog(rng_);
    uint32_t r = rng_ >> (l - 16);
    for (int i = 0; i < 3; ++i) {
      r = (r * r) >> 15;
      int b = static_cast<int>(r >> 16);
      l = (l << 1) | b;
      r >>= b;
    }
    return nbits - l;
  }
  int tell() const { return nbits_total_ - ec_ilog(rng_); }

  // finalize; returns the stream length in bytes
  size_t done() {
    int l = kEcCodeBits - ec_ilog(rng_);
    uint32_t msk = (kEcCodeTop - 1) >> l;
    uint32_t end = (val_ + msk) & ~msk;
    if ((end | msk) >= val_ + rng_) {
      ++l;
      msk >>= 1;
      end = (val_ + msk) & ~msk;
    }
    while (l > 0) {
      carry_out(static_cast<int>(end >> kEcCodeShift));
      end = (end << kEcSymBits) & (kEcCodeTop - 1);
      l -= kEcSymBits;
    }
    if (rem_ >= 0 || ext_ > 0) carry_out(0);
    // flush raw-bit window (libopus ec_enc_done semantics)
    uint64_t window = end_window_;
    int used = nend_bits_;
    while (used >= static_cast<int>(kEcSymBits)) {
      if (end_offs_ >= buf_.size()) {
        error_ = true;
        break;
      }
      buf_[buf_.size() - 1 - end_offs_] =
          static_cast<uint8_t>(window & kEcSymMax);
      ++end_offs_;
      window >>= kEcSymBits;
      used -= kEcSymBits;
    }
    if (used > 0 && !error_) {
      if (end_offs_ >= buf_.size()) {
        error_ = true;
      } else {
        buf_[buf_.size() - end_offs_ - 1] |=
            static_cast<uint8_t>(window);
        ++end_offs_;   // the partial byte is part of the stream
      }
    }
    n_bytes_ = offs_ + end_offs_;
    return n_bytes_;
  }

  // final stream (call after done()); length = max(front, capacity) layout
  std::vector<uint8_t> stream() const {
    // the Opus payload is the WHOLE capacity buffer conceptually; the
    // minimal conformant payload keeps front bytes, zero fill, back bytes
    std::vector<uint8_t> out(buf_.size(), 0);
    for (size_t i = 0; i < offs_; ++i) out[i] = buf_[i];
    for (size_t i = 0; i < end_offs_; ++i)
      out[out.size() - 1 - i] = buf_[buf_.size() - 1 - i];
    return out;
  }

  bool error() const { return error_; }
  size_t range_bytes() const { return offs_; }

 private:
  void carry_out(int c) {
    if (c != static_cast<int>(kEcSymMax)) {
      int carry = c >> kEcSymBits;
      if (rem_ >= 0) put_byte(static_cast<uint8_t>(rem_ + carry));
      for (; ext_ > 0; --ext_)
        put_byte(static_cast<uint8_t>((kEcSymMax + carry) & kEcSymMax));
      rem_ = c & kEcSymMax;
    } else {
      ++ext_;
    }
  }

  void put_byte(uint8_t b) {
    if (offs_ + end_offs_ >= buf_.size()) {
      error_ = true;
      return;
    }
    buf_[offs_++] = b;
  }

  void normalize() {
    while (rng_ <= kEcCodeBot) {
      carry_out(static_cast<int>(val_ >> kEcCodeShift));
      val_ = (val_ << kEcSymBits) & (kEcCodeTop - 1);
      rng_ <<= kEcSymBits;
      nbits_total_ += kEcSymBits;
    }
  }

  std::vector<uint8_t> buf_;
  size_t offs_ = 0;
  size_t end_offs_ = 0;
  uint64_t end_window_ = 0;
  int nend_bits_ = 0;
  int nbits_total_ = 0;
  uint32_t val_ = 0;
  uint32_t rng_ = 0;
  int rem_ = -1;
  int ext_ = 0;
  bool error_ = false;
  size_t n_bytes_ = 0;
};

}  // namespace opus
}  // namespace hipflux
