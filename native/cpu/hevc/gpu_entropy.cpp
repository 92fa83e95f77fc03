// HEVC bitstream assembly from GPU buffers (see gpu_entropy.h).
#include "gpu_entropy.h"

#include "../../hip/hevc_gpu_layout.h"
#include "cabac.h"
#include "entropy.h"
#include "headers.h"

namespace hipflux {
namespace hevc {

void assemble_hevc_slice_nal(const uint8_t* cabac_bytes, int n_bytes,
                             int tail_bits, int tail_nbits, bool first_slice,
                             int slice_addr, int addr_bits, int qp,
                             std::vector<uint8_t>& out) {
  BitWriter bw;
  write_slice_header(bw, first_slice, slice_addr, addr_bits, qp);
  bw.append_bytes(cabac_bytes, static_cast<size_t>(n_bytes));
  bw.u(static_cast<uint32_t>(tail_bits), tail_nbits);
  bw.rbsp_trailing();
  emit_nal(bw, out, kNalIdr);
}

void encode_hevc_job_nal(const int16_t* levels, const int* meta, int ctbw,
                         int ctu_row, int ctu_x0, int seg_w, int qp,
                         bool first_slice, int slice_addr, int addr_bits,
                         std::vector<uint8_t>& out) {
  std::vector<uint8_t> cabac_bytes;
  CabacEncoder cab(cabac_bytes);
  ContextBank bank;
  bank.init(qp);
  int left_mode = -1;
  for (int ci = 0; ci < seg_w; ++ci) {
    const size_t mb = static_cast<size_t>(ctu_row) * ctbw + ctu_x0 + ci;
    const int mode = meta[mb * hevcgpu::kHevcMetaPerCtu + 0];
    const int cbf = meta[mb * hevcgpu::kHevcMetaPerCtu + 1];
    const int16_t* lv = levels + mb * hevcgpu::kHevcLevelsPerCtu;
    code_ctu_syntax(cab, bank, mode, left_mode, cbf & 1, cbf & 2, cbf & 4,
                    lv, lv + 256, lv + 320);
    left_mode = mode;
    cab.encode_terminate(ci == seg_w - 1 ? 1 : 0);
  }
  auto tail = cab.finish();
  assemble_hevc_slice_nal(cabac_bytes.data(),
                          static_cast<int>(cabac_bytes.size()), tail.bits,
                          tail.nbits, first_slice, slice_addr, addr_bits, qp,
                          out);
}

void write_hevc_stripe_headers(int coded_w, int coded_h, int vis_w,
                               int vis_h, std::vector<uint8_t>& out) {
  write_vps_nal(out, level_idc_for(coded_w, coded_h));
  write_sps_nal(out, coded_w, coded_h, vis_w, vis_h);
  write_pps_nal(out);
}

}  // namespace hevc
}  // namespace hipflux
