// H.264 RBSP bit writer: MSB-first bits, Exp-Golomb, NAL wrapping with
// emulation prevention (00 00 0x -> 00 00 03 0x).
#pragma once

#include <cassert>
#include <cstddef>
#include <cstdint>
#include <vector>

namespace hipflux {
namespace h264 {

class BitWriter {
 public:
  void clear() {
    buf_.clear();
    acc_ = 0;
    nacc_ = 0;
  }

  // 64-bit accumulator: append up to 32 bits per call, flush whole bytes.
  void u(uint32_t value, int nbits) {
    assert(nbits >= 0 && nbits <= 32);
    if (nbits == 0) return;
    acc_ = (acc_ << nbits) |
           (static_cast<uint64_t>(value) &
            ((nbits == 32 ? 0ull : (1ull << nbits)) - 1ull));
    nacc_ += nbits;
    while (nacc_ >= 8) {
      buf_.push_back(static_cast<uint8_t>(acc_ >> (nacc_ - 8)));
      nacc_ -= 8;
    }
  }

  void ue(uint32_t v) {
    // Exp-Golomb: codeNum v -> [v+1 in binary] with leading zeros
    uint32_t cw = v + 1;
    int len = 0;
    for (uint32_t t = cw; t > 1; t >>= 1) ++len;
    u(0, len);
    u(cw, len + 1);
  }

  void se(int32_t v) {
    uint32_t cn = v > 0 ? 2 * static_cast<uint32_t>(v) - 1
                        : 2 * static_cast<uint32_t>(-v);
    ue(cn);
  }

  void put_bit(int b) { u(b & 1, 1); }

  // Bulk-append whole bytes (writer must be byte-aligned).
  void append_bytes(const uint8_t* data, size_t n) {
    assert(nacc_ == 0 && "append_bytes requires byte alignment");
    buf_.insert(buf_.end(), data, data + n);
  }

  void rbsp_trailing() {
    put_bit(1);
    if (nacc_ != 0) u(0, 8 - nacc_);
  }

  size_t bit_count() const { return buf_.size() * 8 + nacc_; }
  const std::vector<uint8_t>& bytes() const { return buf_; }

  // Append this RBSP as a NAL unit (with start code + emulation prevention).
  void emit_nal(std::vector<uint8_t>& out, int nal_ref_idc, int nal_type,
                bool long_startcode = true) const {
    assert(nacc_ == 0 && "call rbsp_trailing() first");
    if (long_startcode) out.push_back(0);
    out.push_back(0);
    out.push_back(0);
    out.push_back(1);
    out.push_back(static_cast<uint8_t>((nal_ref_idc << 5) | nal_type));
    int zeros = 0;
    for (uint8_t b : buf_) {
      if (zeros >= 2 && b <= 3) {
        out.push_back(3);
        zeros = 0;
      }
      out.push_back(b);
      zeros = (b == 0) ? zeros + 1 : 0;
    }
  }

 private:
  std::vector<uint8_t> buf_;
  uint64_t acc_ = 0;
  int nacc_ = 0;
};

}  // namespace h264
}  // namespace hipflux
