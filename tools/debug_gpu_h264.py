"""Localize GPU-vs-CPU H.264 pipeline divergence at MB granularity.

Run on a GPU box:  python tools/debug_gpu_h264.py
Encodes one IDR (and optionally a P frame) with both pipelines, decodes
with the from-spec reference decoder, prints per-plane PSNR, a per-MB
luma-diff grid, and the first MB whose parsed header (mb_type /
chroma_mode) differs.
"""
import sys
import os

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "tests"))

import numpy as np
from hipflux import _native  # noqa: E402
import h264_ref_decoder as R  # noqa: E402


class TraceDecoder(R.Decoder):
    def __init__(self):
        super().__init__()
        self.mbs = []  # (mbx, mby, kind, detail)

    def decode_i16(self, br, mbx, mby, i16_type, qp, ctx):
        t = i16_type - 1
        self.mbs.append((mbx, mby, "I16",
                         dict(pred=t % 4, cbp_c=(t // 4) % 3,
                              cbp_l=15 if t >= 12 else 0)))
        return super().decode_i16(br, mbx, mby, i16_type, qp, ctx)

    def decode_skip(self, mbx, mby):
        self.mbs.append((mbx, mby, "SKIP", {}))
        return super().decode_skip(mbx, mby)

    def decode_p16(self, br, mbx, mby, ctx, qp):
        self.mbs.append((mbx, mby, "P16", {}))
        return super().decode_p16(br, mbx, mby, ctx, qp)


def psnr(a, b):
    d = a.astype(np.float64) - b.astype(np.float64)
    mse = (d * d).mean()
    return 99.0 if mse == 0 else 10 * np.log10(255.0 * 255.0 / mse)


def run(w=256, h=128, n=1, qp=28, stripe=64):
    rng = np.random.default_rng(7)
    base = rng.integers(0, 256, (h, w, 4), dtype=np.uint8)
    base[:, :, 3] = 255
    frames = []
    for i in range(n):
        f = base.copy()
        if i:
            f[: h // 2] = np.roll(f[: h // 2], 2 * i, axis=1)
        frames.append(np.ascontiguousarray(f))
    gpu = _native._pipeline_encode("gpu", frames, w, h, qp, stripe, 1)
    cpu = _native._pipeline_encode("cpu", frames, w, h, qp, stripe, 1)

    def rows_of(out):
        rows = {}
        for fr in out:
            for data, y, _, _ in fr:
                rows.setdefault(y, bytearray()).extend(bytes(data))
        return rows

    grows, crows = rows_of(gpu), rows_of(cpu)
    for y in sorted(grows):
        dg, dc = TraceDecoder(), TraceDecoder()
        g_frames = dg.decode(bytes(grows[y]))
        c_frames = dc.decode(bytes(crows[y]))
        for fi, ((gy, gcb, gcr), (cy, ccb, ccr)) in enumerate(
                zip(g_frames, c_frames)):
            ps = [psnr(gy, cy), psnr(gcb, ccb), psnr(gcr, ccr)]
            print(f"stripe y={y} frame={fi}: PSNR Y={ps[0]:.1f} "
                  f"Cb={ps[1]:.1f} Cr={ps[2]:.1f}")
            if min(ps) < 55:
                mbw, mbh = gy.shape[1] // 16, gy.shape[0] // 16
                grid = []
                for mby in range(mbh):
                    line = []
                    for mbx in range(mbw):
                        dl = int(np.abs(
                            gy[mby*16:mby*16+16, mbx*16:mbx*16+16].astype(int)
                            - cy[mby*16:mby*16+16, mbx*16:mbx*16+16]
                            .astype(int)).max())
                        db = int(np.abs(
                            gcb[mby*8:mby*8+8, mbx*8:mbx*8+8].astype(int)
                            - ccb[mby*8:mby*8+8, mbx*8:mbx*8+8]
                            .astype(int)).max())
                        line.append(f"{dl:3d}/{db:3d}")
                    grid.append(" ".join(line))
                print("  per-MB maxdiff Y/Cb:")
                for ln in grid:
                    print("   ", ln)
        # header diff
        ng = len(dg.mbs)
        for i, (a, b) in enumerate(zip(dg.mbs, dc.mbs)):
            if a != b:
                print(f"  first header diff at item {i}: gpu={a} cpu={b}")
                break
        else:
            if ng != len(dc.mbs):
                print(f"  mb count differs: gpu={ng} cpu={len(dc.mbs)}")
            else:
                print(f"  headers identical ({ng} mbs)")


def run_recon_check(w=256, h=128, qp=28, stripe=64):
    """Kernel recon vs decoder recon of the kernel's own stream."""
    rng = np.random.default_rng(7)
    f = rng.integers(0, 256, (h, w, 4), dtype=np.uint8)
    f[:, :, 3] = 255
    out, dump = _native._pipeline_encode(
        "gpu", [np.ascontiguousarray(f)], w, h, qp, stripe, 1, True)
    yp, cp = dump["ypitch"], dump["cpitch"]
    ky = np.frombuffer(dump["y"], np.uint8).reshape(-1, yp)[:h, :w]
    kcb = np.frombuffer(dump["cb"], np.uint8).reshape(-1, cp)[:h//2, :w//2]
    kcr = np.frombuffer(dump["cr"], np.uint8).reshape(-1, cp)[:h//2, :w//2]
    rows = {}
    for fr in out:
        for data, y, _, _ in fr:
            rows.setdefault(y, bytearray()).extend(bytes(data))
    for y0 in sorted(rows):
        dec = R.Decoder().decode(bytes(rows[y0]))
        dy, dcb, dcr = dec[0]
        sh = dy.shape[0]
        sky = ky[y0:y0+sh]
        skb = kcb[y0//2:y0//2+sh//2]
        skr = kcr[y0//2:y0//2+sh//2]
        dpy = np.abs(sky.astype(int) - dy.astype(int))
        dpb = np.abs(skb.astype(int) - dcb.astype(int))
        dpr = np.abs(skr.astype(int) - dcr.astype(int))
        print(f"stripe y={y0}: kernel-vs-decoder maxdiff "
              f"Y={dpy.max()} Cb={dpb.max()} Cr={dpr.max()}")
        if dpy.max() > 0:
            ys, xs = np.nonzero(dpy)
            i = np.argmin(ys * 10000 + xs)
            py_, px_ = int(ys[i]), int(xs[i])
            print(f"  first Y diff at ({py_},{px_}) mb=({py_//16},{px_//16})"
                  f" blkpos=({py_%16},{px_%16})"
                  f" kernel={sky[py_,px_]} decoder={dy[py_,px_]}")
            mby, mbx = py_ // 16, px_ // 16
            print("  kernel MB:")
            print(sky[mby*16:mby*16+16, mbx*16:mbx*16+16])
            print("  decoder MB:")
            print(dy[mby*16:mby*16+16, mbx*16:mbx*16+16])
            break


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "recon":
        run_recon_check()
    else:
        n = int(sys.argv[1]) if len(sys.argv) > 1 else 1
        run(n=n)
