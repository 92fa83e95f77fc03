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
                      int quality, bool fullcolor, std::vector<uint8_t>& out,
                      bool restart_rows = false);

// Quant table for a given quality (natural order), IJG scaling — shared by
// the CPU and HIP paths so their bitstreams match.
void jpeg_quality_tables(int quality, uint8_t qy[64], uint8_t qc[64]);

// Entropy-encode pre-quantized coefficient blocks (natural order, int16) into
// a JFIF bitstream. Used by the HIP path: the GPU produces quantized blocks,
// the CPU packs Huffman bits. Blocks are laid out in MCU scan order:
//   4:2:0 -> per MCU: Y00 Y01 Y10 Y11 Cb Cr  (6 blocks of 64)
//   4:4:4 -> per MCU: Y Cb Cr                (3 blocks of 64)
// restart_rows: emit DRI + an RSTn marker between MCU rows (DC
// predictors reset per row) — the GPU entropy kernel's framing, kept
// bit-identical here for byte-equality tests.
void jpeg_entropy_from_blocks(const int16_t* blocks, int mcu_count_x,
                              int mcu_count_y, int width, int height,
                              int quality, bool fullcolor,
                              std::vector<uint8_t>& out,
                              bool restart_rows = false);

// JFIF headers only (SOI..SOS), for the GPU entropy assembly.
void jpeg_write_headers(std::vector<uint8_t>& out, int width, int height,
                        int quality, bool fullcolor,
                        int restart_interval);

// One restart interval's GPU bit run -> padded, stuffed scan bytes.
void jpeg_append_row_bits(const uint32_t* words, int bits,
                          std::vector<uint8_t>& out);

// Huffman code tables as code | (size << 16) for the GPU kernel.
void jpeg_export_huff(uint32_t dcl[12], uint32_t acl[256], uint32_t dcc[12],
                      uint32_t acc[256]);

}  // namespace hipflux
