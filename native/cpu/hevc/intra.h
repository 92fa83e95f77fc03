// HEVC intra prediction (§8.4.4.2) for the mode subset this encoder
// emits: Planar (0), DC (1), Horizontal (10), Vertical (26).
// Reference-sample substitution and the [1 2 1] smoothing filter follow
// the spec so reconstruction matches the from-spec decoder bit-exactly.
#pragma once

#include <cstdint>
#include <cstring>

namespace hipflux {
namespace hevc {

enum IntraMode { kPlanar = 0, kDc = 1, kHor = 10, kVer = 26 };

// Neighbor availability for one NxN block at (x0, y0) inside a slice whose
// first pixel row is slice_y0 (slices own whole CTU rows or row segments;
// availability is same-slice only).
struct Avail {
  bool left = false;        // column x0-1, rows y0 .. y0+N-1
  bool below_left = false;  // column x0-1, rows y0+N .. y0+2N-1
  bool top = false;         // row y0-1, cols x0 .. x0+N-1
  bool top_right = false;   // row y0-1, cols x0+N .. x0+2N-1
  bool corner = false;      // (x0-1, y0-1)
  bool any() const { return left || below_left || top || top_right || corner; }
};

// Build the 4N+1 reference array: ref[0] = corner, ref[1..2N] = top row
// left-to-right, ref[2N+1..4N] = left column top-to-bottom.
// Substitution per §8.4.4.2.2: scan from the deepest-left sample upward
// then across the top; fill gaps from the previous available sample.
template <int N>
inline void build_refs(const uint8_t* plane, int pitch, int x0, int y0,
                       const Avail& av, uint8_t* ref) {
  const int total = 4 * N + 1;
  bool have[4 * 16 + 1];
  // order for substitution: index 0 = bottom-most left sample
  // (x0-1, y0+2N-1), rising to corner at index 2N, then top row to
  // (x0+2N-1, y0-1) at index 4N.
  uint8_t lin[4 * 16 + 1];
  for (int i = 0; i < total; ++i) have[i] = false;
  auto put = [&](int idx, bool ok, int px, int py) {
    if (ok) {
      lin[idx] = plane[static_cast<size_t>(py) * pitch + px];
      have[idx] = true;
    }
  };
  for (int i = 0; i < N; ++i) {  // below-left, bottom to top
    put(i, av.below_left, x0 - 1, y0 + 2 * N - 1 - i);
  }
  for (int i = 0; i < N; ++i) {  // left, bottom to top
    put(N + i, av.left, x0 - 1, y0 + N - 1 - i);
  }
  put(2 * N, av.corner, x0 - 1, y0 - 1);
  for (int i = 0; i < N; ++i) {  // top
    put(2 * N + 1 + i, av.top, x0 + i, y0 - 1);
  }
  for (int i = 0; i < N; ++i) {  // top-right
    put(3 * N + 1 + i, av.top_right, x0 + N + i, y0 - 1);
  }
  if (!av.any()) {
    std::memset(ref, 128, total);
    return;
  }
  if (!have[0]) {
    for (int i = 1; i < total; ++i)
      if (have[i]) {
        lin[0] = lin[i];
        break;
      }
  }
  for (int i = 1; i < total; ++i)
    if (!have[i]) lin[i] = lin[i - 1];
  // repack: ref[0]=corner, ref[1..2N]=top..topright, ref[2N+1..4N]=left
  // top-to-bottom then below-left.
  ref[0] = lin[2 * N];
  for (int i = 0; i < 2 * N; ++i) ref[1 + i] = lin[2 * N + 1 + i];
  for (int i = 0; i < 2 * N; ++i) ref[2 * N + 1 + i] = lin[2 * N - 1 - i];
}

// §8.4.4.2.3 [1 2 1] smoothing (luma only; applied for Planar at N >= 8).
template <int N>
inline void filter_refs(const uint8_t* ref, uint8_t* out) {
  const int total = 4 * N + 1;
  // linear order for filtering: left-bottom .. corner .. top-right
  uint8_t lin[4 * 16 + 1], flt[4 * 16 + 1];
  for (int i = 0; i < 2 * N; ++i) lin[i] = ref[4 * N - i];
  lin[2 * N] = ref[0];
  for (int i = 0; i < 2 * N; ++i) lin[2 * N + 1 + i] = ref[1 + i];
  flt[0] = lin[0];
  flt[total - 1] = lin[total - 1];
  for (int i = 1; i < total - 1; ++i)
    flt[i] = static_cast<uint8_t>((lin[i - 1] + 2 * lin[i] + lin[i + 1] + 2) >>
                                  2);
  for (int i = 0; i < 2 * N; ++i) out[4 * N - i] = flt[i];
  out[0] = flt[2 * N];
  for (int i = 0; i < 2 * N; ++i) out[1 + i] = flt[2 * N + 1 + i];
}

// Predict NxN into pred (stride N). ref layout as built above.
// cidx 0 = luma (DC edge filter + V/H edge adjust apply at N < 32).
template <int N>
inline void predict(int mode, const uint8_t* ref, int cidx, uint8_t* pred) {
  const uint8_t* top = ref + 1;             // top[x], x in 0..2N-1
  const uint8_t* left = ref + 2 * N + 1;    // left[y], y in 0..2N-1
  const uint8_t corner = ref[0];
  const int log2n = N == 4 ? 2 : N == 8 ? 3 : 4;
  auto clip8 = [](int v) {
    return static_cast<uint8_t>(v < 0 ? 0 : v > 255 ? 255 : v);
  };
  if (mode == kPlanar) {
    for (int y = 0; y < N; ++y)
      for (int x = 0; x < N; ++x)
        pred[y * N + x] = static_cast<uint8_t>(
            ((N - 1 - x) * left[y] + (x + 1) * top[N] +
             (N - 1 - y) * top[x] + (y + 1) * left[N] + N) >> (log2n + 1));
  } else if (mode == kDc) {
    int sum = N;
    for (int i = 0; i < N; ++i) sum += top[i] + left[i];
    int dc = sum >> (log2n + 1);
    for (int i = 0; i < N * N; ++i) pred[i] = static_cast<uint8_t>(dc);
    if (cidx == 0 && N < 32) {
      pred[0] = static_cast<uint8_t>((left[0] + 2 * dc + top[0] + 2) >> 2);
      for (int x = 1; x < N; ++x)
        pred[x] = static_cast<uint8_t>((top[x] + 3 * dc + 2) >> 2);
      for (int y = 1; y < N; ++y)
        pred[y * N] = static_cast<uint8_t>((left[y] + 3 * dc + 2) >> 2);
    }
  } else if (mode == kVer) {
    for (int y = 0; y < N; ++y)
      for (int x = 0; x < N; ++x) pred[y * N + x] = top[x];
    if (cidx == 0 && N < 32)
      for (int y = 0; y < N; ++y)
        pred[y * N] = clip8(top[0] + ((left[y] - corner) >> 1));
  } else {  // kHor
    for (int y = 0; y < N; ++y)
      for (int x = 0; x < N; ++x) pred[y * N + x] = left[y];
    if (cidx == 0 && N < 32)
      for (int x = 0; x < N; ++x)
        pred[x] = clip8(left[0] + ((top[x] - corner) >> 1));
  }
}

}  // namespace hevc
}  // namespace hipflux
