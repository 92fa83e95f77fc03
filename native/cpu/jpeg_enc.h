// Baseline JFIF (sequential DCT, Huffman) encoder — CPU reference path.
// Each stripe is encoded as a complete standalone JFIF image, matching the
// reference's striped-MJPEG wire contract (SURVEY.md §2.3 "jpeg" encoder;
// client runs one decoder per stripe row).
#pragma once

#include <cstdint>
#include <vector>

namespace hipflux {

// Encode a BGRX region as a baseline JPEG. 4:2:0 by default, 4:4:4 when
// fullcolor. quality in [1,100] (IJG scaling). Appends to `out`.
void jpeg_encode_bgrx(const uint8_t* bgrx, int stride, int width, int height,
                      int quality, bool fullcolor, std::vector<uint8_t>& out);

// Quant table for a given quality (natural order), IJG scaling — shared by
// the CPU and HIP paths so their bitstreams match.
void jpeg_quality_tables(int quality, uint8_t qy[64], uint8_t qc[64]);

// Entropy-encode pre-quantized coefficient blocks (natural order, int16) into
// a JFIF bitstream. Used by the HIP path: the GPU produces quantized blocks,
// the CPU packs Huffman bits. Blocks are laid out in MCU scan order:
//   4:2:0 -> per MCU: Y00 Y01 Y10 Y11 Cb Cr  (6 blocks of 64)
//   4:4:4 -> per MCU: Y Cb Cr                (3 blocks of 64)
void jpeg_entropy_from_blocks(const int16_t* blocks, int mcu_count_x,
                              int mcu_count_y, int width, int height,
                              int quality, bool fullcolor,
                              std::vector<uint8_t>& out);

}  // namespace hipflux
