import sys, os
import sys, os
sys.path.insert(0, os.getcwd())
sys.path.insert(0, "tests")
import numpy as np
from hipflux import _native

def make_frames(w, h, n, seed=42):
    rng = np.random.default_rng(seed)
    base = np.zeros((h, w, 4), np.uint8)
    base[:, :, 0] = np.linspace(0, 255, w, dtype=np.uint8)[None, :]
    base[:, :, 1] = np.linspace(0, 255, h, dtype=np.uint8)[:, None]
    base[:, :, 2] = 80
    base[:, :, 3] = 255
    band = min(16, h // 4)
    base[h // 2:h // 2 + band, :] = rng.integers(0, 256, (band, w, 4), dtype=np.uint8)
    frames = []
    for i in range(n):
        f = base.copy()
        x = (16 + i * 24) % max(1, w - 48)
        f[16:48, x:x + 48, 0] = 255
        f[16:48, x:x + 48, 2] = 0
        frames.append(np.ascontiguousarray(f))
    return frames

os.makedirs("gpurun_out", exist_ok=True)
w, h, qp = 320, 192, 18
frames = make_frames(w, h, 1)
for kind in ("gpu", "cpu"):
    out = _native._pipeline_encode(kind, frames, w, h, qp, 64, 2)
    rows = {}
    for fr in out:
        for d, y, hh, k in fr:
            rows.setdefault(y, b"")
            rows[y] += bytes(d)
    for y, s in rows.items():
        open(f"gpurun_out/hevc_{kind}_y{y}.bin", "wb").write(s)
print("dumped")
