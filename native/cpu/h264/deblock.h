// H.264 in-loop deblocking filter (§8.7) for the one-slice-per-MB-row
// design: disable_deblocking_filter_idc = 2 filters WITHIN each slice
// only, so no cross-row (or cross-segment) dependency exists — each
// 16-px row deblocks independently, which is exactly the seam the HIP
// row kernels need. Shared scalar reference for k_h264_deblock
// (native/hip/h264_deblock.hip); the from-spec Python decoder
// re-implements it independently and recon equality is asserted.
#pragma once

#include <cstdint>
#include <cstdlib>

namespace hipflux {
namespace h264 {

// Table 8-16 (alpha / beta thresholds by indexA/indexB).
inline constexpr uint8_t kDbAlpha[52] = {
    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
    4, 4, 5, 6, 7, 8, 9, 10, 12, 13, 15, 17, 20, 22, 25, 28,
    32, 36, 40, 45, 50, 56, 63, 71, 80, 90, 101, 113, 127, 144,
    162, 182, 203, 226, 255, 255};
inline constexpr uint8_t kDbBeta[52] = {
    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
    2, 2, 2, 3, 3, 3, 3, 4, 4, 4, 6, 6, 7, 7, 8, 8,
    9, 9, 10, 10, 11, 11, 12, 12, 13, 13, 14, 14, 15, 15,
    16, 16, 17, 17, 18, 18};
// Table 8-17 (tC0 by indexA and bS 1..3).
inline constexpr uint8_t kDbTc0[52][3] = {
    {0,0,0},{0,0,0},{0,0,0},{0,0,0},{0,0,0},{0,0,0},{0,0,0},{0,0,0},
    {0,0,0},{0,0,0},{0,0,0},{0,0,0},{0,0,0},{0,0,0},{0,0,0},{0,0,0},
    {0,0,0},{0,0,1},{0,0,1},{0,0,1},{0,0,1},{0,1,1},{0,1,1},{1,1,1},
    {1,1,1},{1,1,1},{1,1,1},{1,1,2},{1,1,2},{1,1,2},{1,1,2},{1,2,3},
    {1,2,3},{2,2,3},{2,2,4},{2,3,4},{2,3,4},{3,3,5},{3,4,6},{3,4,6},
    {4,5,7},{4,5,8},{5,6,9},{6,7,10},{6,8,11},{7,9,12},{8,10,13},
    {9,12,15},{10,13,17},{11,16,20},{13,18,23},{14,20,25}};

// Per-MB inputs the filter needs (mirrors the GPU meta).
struct DbMb {
  uint8_t intra = 0;          // 1 = intra-coded MB
  int16_t mvx = 0, mvy = 0;   // quarter-pel
  uint16_t nz = 0;            // bit b = luma 4x4 block b (raster) has coeffs
};

inline int db_clip3(int lo, int hi, int v) {
  return v < lo ? lo : v > hi ? hi : v;
}
inline uint8_t db_clip8(int v) {
  return static_cast<uint8_t>(v < 0 ? 0 : v > 255 ? 255 : v);
}

// Filter one line of a luma edge (p3..p0 | q0..q3 across the edge).
// `px[i]`/`qx[i]` index away from the edge.
template <typename GetP, typename GetQ, typename SetP, typename SetQ>
inline void db_filter_luma_line(int bs, int alpha, int beta, int tc0,
                                GetP P, GetQ Q, SetP SP, SetQ SQ) {
  int p0 = P(0), p1 = P(1), p2 = P(2), p3 = P(3);
  int q0 = Q(0), q1 = Q(1), q2 = Q(2), q3 = Q(3);
  if (std::abs(p0 - q0) >= alpha || std::abs(p1 - p0) >= beta ||
      std::abs(q1 - q0) >= beta)
    return;
  const bool ap = std::abs(p2 - p0) < beta;
  const bool aq = std::abs(q2 - q0) < beta;
  if (bs < 4) {
    const int tc = tc0 + (ap ? 1 : 0) + (aq ? 1 : 0);
    int delta = db_clip3(-tc, tc, ((q0 - p0) * 4 + (p1 - q1) + 4) >> 3);
    SP(0, db_clip8(p0 + delta));
    SQ(0, db_clip8(q0 - delta));
    if (ap)
      SP(1, static_cast<uint8_t>(
                p1 + db_clip3(-tc0, tc0,
                              (p2 + ((p0 + q0 + 1) >> 1) - 2 * p1) >> 1)));
    if (aq)
      SQ(1, static_cast<uint8_t>(
                q1 + db_clip3(-tc0, tc0,
                              (q2 + ((p0 + q0 + 1) >> 1) - 2 * q1) >> 1)));
  } else {
    const bool small = std::abs(p0 - q0) < ((alpha >> 2) + 2);
    if (ap && small) {
      SP(0, static_cast<uint8_t>(
                (p2 + 2 * p1 + 2 * p0 + 2 * q0 + q1 + 4) >> 3));
      SP(1, static_cast<uint8_t>((p2 + p1 + p0 + q0 + 2) >> 2));
      SP(2, static_cast<uint8_t>(
                (2 * p3 + 3 * p2 + p1 + p0 + q0 + 4) >> 3));
    } else {
      SP(0, static_cast<uint8_t>((2 * p1 + p0 + q1 + 2) >> 2));
    }
    if (aq && small) {
      SQ(0, static_cast<uint8_t>(
                (q2 + 2 * q1 + 2 * q0 + 2 * p0 + p1 + 4) >> 3));
      SQ(1, static_cast<uint8_t>((q2 + q1 + q0 + p0 + 2) >> 2));
      SQ(2, static_cast<uint8_t>(
                (2 * q3 + 3 * q2 + q1 + q0 + p0 + 4) >> 3));
    } else {
      SQ(0, static_cast<uint8_t>((2 * q1 + q0 + p1 + 2) >> 2));
    }
  }
}

// Chroma line: only p0/q0 move.
template <typename GetP, typename GetQ, typename SetP, typename SetQ>
inline void db_filter_chroma_line(int bs, int alpha, int beta, int tc0,
                                  GetP P, GetQ Q, SetP SP, SetQ SQ) {
  int p0 = P(0), p1 = P(1);
  int q0 = Q(0), q1 = Q(1);
  if (std::abs(p0 - q0) >= alpha || std::abs(p1 - p0) >= beta ||
      std::abs(q1 - q0) >= beta)
    return;
  if (bs < 4) {
    const int tc = tc0 + 1;
    int delta = db_clip3(-tc, tc, ((q0 - p0) * 4 + (p1 - q1) + 4) >> 3);
    SP(0, db_clip8(p0 + delta));
    SQ(0, db_clip8(q0 - delta));
  } else {
    SP(0, static_cast<uint8_t>((2 * p1 + p0 + q1 + 2) >> 2));
    SQ(0, static_cast<uint8_t>((2 * q1 + q0 + p1 + 2) >> 2));
  }
}

// bS for a vertical luma edge column between 4x4 blocks p (left) and q
// (right). blk indices raster 0..15 within each MB.
inline int db_bs(const DbMb& mp, int pblk, const DbMb& mq, int qblk,
                 bool mb_edge) {
  if (mp.intra || mq.intra) return mb_edge ? 4 : 3;
  if (((mp.nz >> pblk) & 1) || ((mq.nz >> qblk) & 1)) return 2;
  if (std::abs(mp.mvx - mq.mvx) >= 4 || std::abs(mp.mvy - mq.mvy) >= 4)
    return 1;
  return 0;
}

// Deblock one slice segment's luma+chroma (MB columns [mbx0, mbx0+segw)
// of MB row `mb_row`). Top edges never filter (the row above is another
// slice); the segment's left boundary edge never filters either
// (disable_deblocking_filter_idc == 2 skips slice-boundary edges).
inline void deblock_segment(uint8_t* Y, int ypitch, uint8_t* Cb,
                            uint8_t* Cr, int cpitch, int mb_row, int mbx0,
                            int segw, const DbMb* mbs, int qp, int qpc) {
  const int iA = db_clip3(0, 51, qp), iB = iA;
  const int alpha = kDbAlpha[iA], beta = kDbBeta[iB];
  const int iAc = db_clip3(0, 51, qpc);
  const int alpha_c = kDbAlpha[iAc], beta_c = kDbBeta[iAc];
  const int y0 = mb_row * 16, cy0 = mb_row * 8;

  for (int m = 0; m < segw; ++m) {
    const int mbx = mbx0 + m;
    const DbMb& cur = mbs[m];
    const DbMb* left = m > 0 ? &mbs[m - 1] : nullptr;
    const int x0 = mbx * 16;

    // ---- vertical luma edges (local x = 0, 4, 8, 12), top to bottom
    for (int e = 0; e < 4; ++e) {
      const int ex = x0 + e * 4;
      const bool mb_edge = e == 0;
      if (mb_edge && left == nullptr) continue;  // slice/segment boundary
      for (int yy = 0; yy < 16; ++yy) {
        const int pblk = (yy >> 2) * 4 + (mb_edge ? 3 : e - 1);
        const int qblk = (yy >> 2) * 4 + e;
        const int bs = db_bs(mb_edge ? *left : cur, pblk, cur, qblk,
                             mb_edge);
        if (bs == 0) continue;
        const int tc0 = bs < 4 ? kDbTc0[iA][bs - 1] : 0;
        uint8_t* row = Y + static_cast<size_t>(y0 + yy) * ypitch;
        db_filter_luma_line(
            bs, alpha, beta, tc0,
            [&](int i) -> int { return row[ex - 1 - i]; },
            [&](int i) -> int { return row[ex + i]; },
            [&](int i, uint8_t v) { row[ex - 1 - i] = v; },
            [&](int i, uint8_t v) { row[ex + i] = v; });
      }
    }
    // ---- horizontal luma edges (local y = 4, 8, 12; y = 0 is the
    // slice boundary and never filters)
    for (int e = 1; e < 4; ++e) {
      const int ey = y0 + e * 4;
      for (int xx = 0; xx < 16; ++xx) {
        const int pblk = (e - 1) * 4 + (xx >> 2);
        const int qblk = e * 4 + (xx >> 2);
        const int bs = db_bs(cur, pblk, cur, qblk, false);
        if (bs == 0) continue;
        const int tc0 = bs < 4 ? kDbTc0[iA][bs - 1] : 0;
        uint8_t* col = Y + x0 + xx;
        db_filter_luma_line(
            bs, alpha, beta, tc0,
            [&](int i) -> int {
              return col[static_cast<size_t>(ey - 1 - i) * ypitch];
            },
            [&](int i) -> int {
              return col[static_cast<size_t>(ey + i) * ypitch];
            },
            [&](int i, uint8_t v) {
              col[static_cast<size_t>(ey - 1 - i) * ypitch] = v;
            },
            [&](int i, uint8_t v) {
              col[static_cast<size_t>(ey + i) * ypitch] = v;
            });
      }
    }
    // ---- chroma edges: vertical at luma x offsets 0 and 8 (chroma
    // 0, 4); horizontal at luma y offset 8 (chroma 4). bS comes from
    // the corresponding luma edge position. Monochrome callers
    // (ChromaArrayType == 0 / Hi444 separate planes) pass null chroma.
    if (Cb == nullptr) continue;
    for (int pl = 0; pl < 2; ++pl) {
      uint8_t* C = pl ? Cr : Cb;
      const int cx0 = mbx * 8;
      for (int e = 0; e < 2; ++e) {               // vertical
        const bool mb_edge = e == 0;
        if (mb_edge && left == nullptr) continue;
        const int ex = cx0 + e * 4;
        for (int yy = 0; yy < 8; ++yy) {
          const int lum_y = yy * 2;
          const int pblk = (lum_y >> 2) * 4 + (mb_edge ? 3 : 1);
          const int qblk = (lum_y >> 2) * 4 + (mb_edge ? 0 : 2);
          const int bs = db_bs(mb_edge ? *left : cur, pblk, cur, qblk,
                               mb_edge);
          if (bs == 0) continue;
          const int tc0 = bs < 4 ? kDbTc0[iAc][bs - 1] : 0;
          uint8_t* row = C + static_cast<size_t>(cy0 + yy) * cpitch;
          db_filter_chroma_line(
              bs, alpha_c, beta_c, tc0,
              [&](int i) -> int { return row[ex - 1 - i]; },
              [&](int i) -> int { return row[ex + i]; },
              [&](int i, uint8_t v) { row[ex - 1 - i] = v; },
              [&](int i, uint8_t v) { row[ex + i] = v; });
        }
      }
      {                                           // horizontal (chroma y=4)
        const int ey = cy0 + 4;
        for (int xx = 0; xx < 8; ++xx) {
          const int lum_x = xx * 2;
          const int pblk = 1 * 4 + (lum_x >> 2);
          const int qblk = 2 * 4 + (lum_x >> 2);
          const int bs = db_bs(cur, pblk, cur, qblk, false);
          if (bs == 0) continue;
          const int tc0 = bs < 4 ? kDbTc0[iAc][bs - 1] : 0;
          uint8_t* col = C + cx0 + xx;
          db_filter_chroma_line(
              bs, alpha_c, beta_c, tc0,
              [&](int i) -> int {
                return col[static_cast<size_t>(ey - 1 - i) * cpitch];
              },
              [&](int i) -> int {
                return col[static_cast<size_t>(ey + i) * cpitch];
              },
              [&](int i, uint8_t v) {
                col[static_cast<size_t>(ey - 1 - i) * cpitch] = v;
              },
              [&](int i, uint8_t v) {
                col[static_cast<size_t>(ey + i) * cpitch] = v;
              });
        }
      }
    }
  }
}

}  // namespace h264
}  // namespace hipflux
