"""Tile-parallel collective layer (BASELINE config 5 communication
pattern): RCCL-over-xGMI boundary all-gather on GPU, gloo fallback on
CPU. Multi-process CPU coverage runs here; the RCCL path initializes and
round-trips on a single GPU in test_gpu_tile_comm.py, and the driver's
8-GPU scaling run exercises bench.py --mode tile across devices."""

import json
import os
import socket
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def test_tile_mode_two_rank_cpu_exchange():
    """bench.py --mode tile with world 2 on CPU: bands encode per rank and
    the boundary rows all-gather over gloo every step."""
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", str(free_port()),
           os.path.join(ROOT, "bench.py"), "--mode", "tile", "--cpu",
           "--steps", "3", "--warmup", "1", "--width", "320",
           "--height", "192"]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=240,
                       cwd=ROOT)
    assert r.returncode == 0, r.stderr[-2000:]
    line = next(l for l in r.stdout.splitlines() if l.startswith("{"))
    d = json.loads(line)
    assert d["config"]["parallelism"] == "tile2"
    assert d["scaling"] == "strong"
    ex = d["config"]["tile_boundary_exchange"]
    assert ex is not None
    assert ex["bytes"] > 0
    assert ex["schedule"] == "gloo-fallback"
    assert ex["p50_ms"] >= 0


def test_tile_mode_band_partition_covers_frame():
    """The 16-aligned band split in bench.py covers every row exactly
    once for any world size."""
    height = 2160
    for world in (1, 2, 3, 4, 8):
        rows16 = (height + 15) // 16
        per, extra = rows16 // world, rows16 % world
        covered = []
        for rank in range(world):
            band_rows = per + (1 if rank < extra else 0)
            band0 = (per * rank + min(rank, extra)) * 16
            enc_h = min(band_rows * 16, height - band0)
            covered.append((band0, band0 + enc_h))
        assert covered[0][0] == 0
        assert covered[-1][1] == height
        for (a0, a1), (b0, b1) in zip(covered, covered[1:]):
            assert a1 == b0, (world, covered)
