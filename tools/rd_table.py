"""Rate-distortion evidence: encode mixed-content frames at several QPs
on the GPU pipeline, decode with the from-spec decoder, report PSNR to
the source and bits/frame. Run on a GPU box:
    python tools/rd_table.py > profiles/rd_v27.txt
"""
import math
import sys
import os

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "tests"))

import numpy as np
import hipflux
from hipflux import _native
from h264_ref_decoder import Decoder


def make_frames(w, h, n):
    rng = np.random.default_rng(7)
    base = np.zeros((h, w, 4), np.uint8)
    base[:, :, 0] = np.linspace(0, 255, w, dtype=np.uint8)[None, :]
    base[:, :, 1] = np.linspace(0, 255, h, dtype=np.uint8)[:, None]
    base[:, :, 2] = 96
    base[h // 3:h // 3 + 24, :] = rng.integers(0, 256, (24, w, 4),
                                               dtype=np.uint8)
    out = []
    for i in range(n):
        f = base.copy()
        x = (8 + i * 12) % max(1, w - 64)
        f[24:88, x:x + 64, 0] = 250
        f[24:88, x:x + 64, 2] = 20
        out.append(np.ascontiguousarray(f))
    return out


def psnr(a, b):
    mse = ((a.astype(np.int64) - b.astype(np.int64)) ** 2).mean()
    return 10 * math.log10(255 * 255 / max(mse, 1e-12))


def main():
    w, h, n = 640, 384, 12
    frames = make_frames(w, h, n)
    print(f"# GPU H.264 rate-distortion ({w}x{h}, {n} frames, "
          f"gradient+noise-band+moving-box content)")
    print("qp  kbits/frame  luma PSNR dB (last frame vs source)")
    for qp in (16, 22, 28, 34, 40):
        out = _native._pipeline_encode("gpu", frames, w, h, qp, 64, 1)
        bits = sum(len(d) * 8 for fr in out for d, *_ in fr) / n
        # reassemble stripes and decode
        rows = {}
        for fr in out:
            for d, y, hh, k in fr:
                rows.setdefault(y, b"")
                rows[y] += bytes(d)
        ys = []
        for y in sorted(rows):
            ys.append(Decoder().decode(rows[y])[n - 1][0])
        rec = np.concatenate(ys, 0)[:h]
        src_y, _, _ = hipflux.bgrx_to_yuv420(frames[-1].tobytes(), w, h)
        sy = np.frombuffer(src_y, np.uint8).reshape(h, w)
        print(f"{qp:2d}  {bits/1000.0:10.1f}  {psnr(rec, sy):.2f}")


if __name__ == "__main__":
    main()
